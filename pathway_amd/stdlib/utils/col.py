"""Column helpers (reference stdlib/utils/col.py)."""
from __future__ import annotations

from typing import Any


def flatten_column(column, origin_id: str = "origin_id"):
    table = column.table
    return table.flatten(column, origin_id=origin_id)


def unpack_col(column, *unpacked_columns: Any, schema=None):
    """Unpack a tuple column into named columns."""
    table = column.table
    names = []
    for c in unpacked_columns:
        names.append(c if isinstance(c, str) else c.name)
    if schema is not None:
        names = schema.column_names()
    kwargs = {n: column[i] for i, n in enumerate(names)}
    return table.select(**kwargs)


def multiapply_all_rows(*cols, fun, result_col_names):
    """Apply `fun` to ALL rows of the columns at once; returns a table on
    the original universe with the result columns (reference
    col.py:apply_all_rows/multiapply_all_rows — whole-table transform,
    meant for infrequent runs on small tables)."""
    from pathway_amd.engine.nodes_recompute import RecomputeNode
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table

    tab = cols[0].table
    names = [c.name for c in cols]

    def fn(in_rows, in_keys):
        rows, keys = in_rows[0], in_keys[0]
        ins = [[r[n] for r in rows] for n in names]
        outs = fun(*ins)
        if len(result_col_names) == 1 and not isinstance(outs, tuple):
            outs = (outs,)
        return [
            (
                k,
                {
                    rn: outs[ci][i]
                    for ci, rn in enumerate(result_col_names)
                },
            )
            for i, k in enumerate(keys)
        ]

    out_dtypes = {rn: dt.ANY for rn in result_col_names}
    node = RecomputeNode(
        [tab._node], fn, list(result_col_names), out_dtypes, get_device()
    )
    return Table(node, out_dtypes, tab._universe)


def apply_all_rows(*cols, fun, result_col_name):
    """Single-result-column variant of multiapply_all_rows."""
    name = (
        result_col_name
        if isinstance(result_col_name, str)
        else result_col_name.name
    )
    return multiapply_all_rows(
        *cols, fun=lambda *a: (fun(*a),), result_col_names=[name]
    )


def groupby_reduce_majority(column_group, column_val):
    """Per group, the most frequent value of column_val (reference
    col.py:groupby_reduce_majority)."""
    from collections import Counter

    import pathway_amd.internals.common as common
    import pathway_amd.reducers as reducers
    from pathway_amd.internals import dtype as dt

    tab = column_group.table
    gname = column_group.name

    g = tab.groupby(column_group).reduce(
        column_group, _vals=reducers.tuple(column_val)
    )
    return g.select(
        g[gname],
        majority=common.apply_with_type(
            lambda vs: Counter(v for v in vs if v is not None).most_common(1)[
                0
            ][0]
            if vs
            else None,
            dt.ANY,
            g._vals,
        ),
    )
