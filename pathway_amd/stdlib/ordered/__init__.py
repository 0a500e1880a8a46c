"""Ordered ops (reference stdlib/ordered): diff."""
from __future__ import annotations

from typing import Any


def diff(table, timestamp: Any, *values: Any, instance: Any = None):
    """Per-instance difference vs previous row by timestamp order.

    Implemented on the RecomputeNode host path (order-dependent op).
    """
    from pathway_amd.engine.nodes_recompute import RecomputeNode
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals import expression as ex
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    ts = table._resolve(timestamp)
    if not isinstance(ts, ex.ColumnReference):
        raise NotImplementedError("diff timestamp must be a column")
    vnames = []
    for v in values:
        rv = table._resolve(v)
        if not isinstance(rv, ex.ColumnReference):
            raise NotImplementedError("diff values must be columns")
        vnames.append(rv.name)
    iname = None
    if instance is not None:
        ri = table._resolve(instance)
        iname = ri.name
    out_columns = [f"diff_{n}" for n in vnames]

    def fn(in_rows, in_keys):
        rows, keys = in_rows[0], in_keys[0]
        groups: dict[Any, list] = {}
        for row, key in zip(rows, keys):
            g = row[iname] if iname else None
            groups.setdefault(g, []).append((row[ts.name], row, key))
        out = []
        for g, rl in groups.items():
            rl.sort(key=lambda x: x[0])
            prev = None
            for t, row, key in rl:
                vals = {}
                for n in vnames:
                    vals[f"diff_{n}"] = (
                        row[n] - prev[n] if prev is not None else None
                    )
                out.append((key, vals))
                prev = row
        return out

    out_dtypes = {
        f"diff_{n}": dt.Optional(dt.unoptionalize(table._dtypes[n])) for n in vnames
    }
    node = RecomputeNode([table._node], fn, out_columns, out_dtypes, get_device())
    return Table(node, out_dtypes, table._universe)


__all__ = ["diff"]
