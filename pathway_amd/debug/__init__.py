"""pw.debug — static tables, capture, printing, equality asserts
(reference python/pathway/debug/__init__.py:222-510)."""

from __future__ import annotations

import re
from typing import Any, Iterable

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.api import BasePointer, DataRow, Pointer, hash_values, squash_updates
from pathway_amd.internals.config import get_device
from pathway_amd.internals.rungraph import G, reset_all
from pathway_amd.internals.schema import SchemaMetaclass
from pathway_amd.internals.table import Table
from pathway_amd.internals.universe import Universe

__all__ = [
    "table_from_markdown",
    "table_from_parquet",
    "table_to_parquet",
    "table_from_rows",
    "table_from_pandas",
    "table_to_pandas",
    "compute_and_print",
    "compute_and_print_update_stream",
    "table_to_dicts",
    "assert_table_equality",
    "assert_table_equality_wo_index",
    "assert_table_equality_wo_types",
    "assert_table_equality_wo_index_types",
    "StreamGenerator",
]


def _parse_value(s: str) -> Any:
    s = s.strip()
    if s in ("", "None"):
        return None
    if s == "True" or s == "true":
        return True
    if s == "False" or s == "false":
        return False
    try:
        return int(s)
    except ValueError:
        pass
    try:
        return float(s)
    except ValueError:
        pass
    if len(s) >= 2 and s[0] == '"' and s[-1] == '"':
        return s[1:-1]
    return s


def sequential_pointer(i: int) -> Pointer:
    lo, hi = hash_values([i])
    return Pointer(lo, hi)


def table_from_rows(
    schema: SchemaMetaclass,
    rows: list[tuple],
    unsafe_trusted_ids: bool = False,
    is_stream: bool = False,
) -> Table:
    """Rows are tuples of values; with is_stream, (…values, time, diff)."""
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.runtime import StaticSource

    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    pk = schema.primary_key_columns()
    data = []
    for i, row in enumerate(rows):
        if is_stream:
            values = list(row[:-2])
            time, diff = int(row[-2]), int(row[-1])
        else:
            values = list(row)
            time, diff = 0, 1
        if pk:
            key_vals = [values[names.index(c)] for c in pk]
            lo, hi = hash_values(key_vals)
        else:
            lo, hi = hash_values([i + 1])
        data.append((Pointer(lo, hi), values, time, diff))
    src = StaticSource(data, names, dtypes)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def table_from_markdown(
    table_def: str,
    id_from: list[str] | None = None,
    unsafe_trusted_ids: bool = False,
    schema: SchemaMetaclass | None = None,
    _stream: bool = False,
) -> Table:
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.runtime import StaticSource

    lines = [ln for ln in table_def.strip().splitlines() if ln.strip()]
    header = [h.strip() for h in re.split(r"\|", lines[0]) if h.strip()]
    rows_raw = []
    for ln in lines[1:]:
        if set(ln.strip()) <= {"-", "|", " ", "="}:
            continue  # separator line
        cells = [c for c in re.split(r"\|", ln)]
        # align with header by stripping leading/trailing empties
        cells = [c.strip() for c in cells]
        while cells and cells[0] == "" and len(cells) > len(header):
            cells.pop(0)
        while cells and cells[-1] == "" and len(cells) > len(header):
            cells.pop()
        if len(cells) < len(header):
            cells += [""] * (len(header) - len(cells))
        rows_raw.append([_parse_value(c) for c in cells[: len(header)]])

    special = {"__time__", "__diff__", "__shard__"}
    # an explicit `id` column sets row keys (reference table_from_markdown)
    has_id_col = "id" in header and not id_from
    data_names = [h for h in header if h not in special and h != "id"]
    t_idx = header.index("__time__") if "__time__" in header else None
    d_idx = header.index("__diff__") if "__diff__" in header else None
    id_idx = header.index("id") if "id" in header else None
    id_vals = []

    columns: dict[str, list[Any]] = {n: [] for n in data_names}
    times, diffs = [], []
    for r in rows_raw:
        for h, v in zip(header, r):
            if h in columns:
                columns[h].append(v)
        if id_idx is not None:
            id_vals.append(r[id_idx])
        times.append(r[t_idx] if t_idx is not None else 0)
        diffs.append(r[d_idx] if d_idx is not None else 1)

    n = len(rows_raw)
    if schema is not None:
        dtypes = {nm: schema.__columns__[nm].dtype for nm in data_names}
        if id_from is None:
            id_from = schema.primary_key_columns()
    else:
        dtypes = {}
        for nm in data_names:
            vals = columns[nm]
            kinds = {dt.dtype_of_value(v) for v in vals if v is not None}
            if kinds == {dt.INT}:
                d = dt.INT
            elif kinds <= {dt.INT, dt.FLOAT} and kinds:
                d = dt.FLOAT
            elif kinds == {dt.BOOL}:
                d = dt.BOOL
            elif kinds == {dt.STR}:
                d = dt.STR
            else:
                d = dt.ANY
            if any(v is None for v in vals):
                d = dt.Optional(d) if d not in (dt.ANY,) else d
            dtypes[nm] = d

    data = []
    for i in range(n):
        values = [columns[nm][i] for nm in data_names]
        if id_from:
            lo, hi = hash_values([columns[c][i] for c in id_from])
        elif has_id_col:
            lo, hi = hash_values([id_vals[i]])
        else:
            key_seq = i + 1
            lo, hi = hash_values([key_seq])
        data.append((Pointer(lo, hi), values, int(times[i]), int(diffs[i])))
    src = StaticSource(data, data_names, [dtypes[nm] for nm in data_names])
    node = InputNode(src, get_device())
    return Table(node, dtypes, Universe())


# the tests' T() helper is just table_from_markdown
T = table_from_markdown


def parse_to_table(*args, **kwargs) -> Table:
    return table_from_markdown(*args, **kwargs)


def table_from_pandas(
    df,
    id_from: list[str] | None = None,
    unsafe_trusted_ids: bool = False,
    schema: SchemaMetaclass | None = None,
) -> Table:
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.runtime import StaticSource
    import numpy as np

    special = {"__time__", "__diff__", "__shard__"}
    names = [c for c in df.columns if c not in special]
    data = []
    times = df["__time__"].tolist() if "__time__" in df.columns else [0] * len(df)
    diffs = df["__diff__"].tolist() if "__diff__" in df.columns else [1] * len(df)
    dtypes: dict[str, dt.DType] = {}
    cols = {}
    for nm in names:
        vals = df[nm].tolist()
        vals = [None if (isinstance(v, float) and v != v) else v for v in vals]
        vals = [v.item() if isinstance(v, np.generic) else v for v in vals]
        cols[nm] = vals
        kinds = {dt.dtype_of_value(v) for v in vals if v is not None}
        if kinds == {dt.INT}:
            d = dt.INT
        elif kinds <= {dt.INT, dt.FLOAT} and kinds:
            d = dt.FLOAT
        elif kinds == {dt.BOOL}:
            d = dt.BOOL
        elif kinds == {dt.STR}:
            d = dt.STR
        else:
            d = dt.ANY
        if any(v is None for v in vals):
            d = dt.Optional(d)
        dtypes[nm] = d
    if schema is not None:
        dtypes = {nm: schema.__columns__[nm].dtype for nm in names}
        if id_from is None:
            id_from = schema.primary_key_columns()
    index = df.index.tolist()
    for i in range(len(df)):
        values = [cols[nm][i] for nm in names]
        if id_from:
            lo, hi = hash_values([cols[c][i] for c in id_from])
            key = Pointer(lo, hi)
        else:
            idx = index[i]
            if isinstance(idx, (int,)) and not unsafe_trusted_ids:
                lo, hi = hash_values([int(idx)])
                key = Pointer(lo, hi)
            else:
                lo, hi = hash_values([str(idx)])
                key = Pointer(lo, hi)
        data.append((key, values, int(times[i]), int(diffs[i])))
    src = StaticSource(data, names, [dtypes[nm] for nm in names])
    node = InputNode(src, get_device())
    return Table(node, dtypes, Universe())


def _run_capture(table: Table) -> list[DataRow]:
    cap = table._capture()
    from pathway_amd.engine.runtime import Runtime

    rt = Runtime(list(G.sinks) + [cap], device=get_device(), comm=G.comm)
    reset_all(rt.nodes)
    rt.run()
    return cap.rows


def table_to_dicts(table: Table):
    rows = _run_capture(table)
    state = squash_updates(rows)
    names = table.column_names()
    keys = list(state.keys())
    columns = {
        name: {key: state[key][i] for key in keys} for i, name in enumerate(names)
    }
    return keys, columns


def _fmt_value(v: Any) -> str:
    if v is None:
        return ""
    return repr(v) if isinstance(v, str) and (" " in v) else str(v)


def compute_and_print(
    table: Table,
    *,
    include_id: bool = True,
    short_pointers: bool = True,
    n_rows: int | None = None,
    squash_updates_flag: bool = True,
    terminate_on_error: bool = True,
) -> None:
    rows = _run_capture(table)
    state = squash_updates(rows, terminate_on_error=terminate_on_error)
    names = table.column_names()
    items = sorted(state.items(), key=lambda kv: repr(kv[0]))
    header = (["id"] if include_id else []) + names
    out_rows = []
    for key, values in items[: n_rows if n_rows is not None else len(items)]:
        krepr = repr(key)
        if short_pointers and len(krepr) > 12:
            krepr = krepr[:9] + "..."
        out_rows.append(([krepr] if include_id else []) + [_fmt_value(v) for v in values])
    widths = [
        max(len(header[j]), *(len(r[j]) for r in out_rows)) if out_rows else len(header[j])
        for j in range(len(header))
    ]
    print(" | ".join(h.ljust(w) for h, w in zip(header, widths)))
    for r in out_rows:
        print(" | ".join(c.ljust(w) for c, w in zip(r, widths)))


def compute_and_print_update_stream(
    table: Table,
    *,
    include_id: bool = True,
    short_pointers: bool = True,
    n_rows: int | None = None,
    **kwargs,
) -> None:
    rows = _run_capture(table)
    names = table.column_names()
    header = (["id"] if include_id else []) + names + ["__time__", "__diff__"]
    print(" | ".join(header))
    for r in sorted(rows, key=lambda r: (r.time, r.diff, repr(r.key))):
        krepr = repr(r.key)
        if short_pointers and len(krepr) > 12:
            krepr = krepr[:9] + "..."
        cells = ([krepr] if include_id else []) + [
            _fmt_value(v) for v in r.values
        ] + [str(r.time), str(r.diff)]
        print(" | ".join(cells))


def table_to_pandas(table: Table, *, include_id: bool = True):
    import pandas as pd

    rows = _run_capture(table)
    state = squash_updates(rows)
    names = table.column_names()
    recs = {n: [] for n in names}
    idx = []
    for key, values in state.items():
        idx.append(key)
        for n, v in zip(names, values):
            recs[n].append(v)
    if include_id:
        return pd.DataFrame(recs, index=idx)
    return pd.DataFrame(recs)


# ----------------------------------------------------------------- asserts --

def _collect(table: Table):
    rows = _run_capture(table)
    return squash_updates(rows)


def assert_table_equality(t1: Table, t2: Table, **kwargs) -> None:
    s1, s2 = _collect(t1), _collect(t2)
    assert set(t1.column_names()) == set(t2.column_names()), (
        t1.column_names(),
        t2.column_names(),
    )
    names1, names2 = t1.column_names(), t2.column_names()
    n1 = {k: dict(zip(names1, v)) for k, v in s1.items()}
    n2 = {k: dict(zip(names2, v)) for k, v in s2.items()}
    assert n1 == n2, f"tables differ:\n{n1}\n!=\n{n2}"


def assert_table_equality_wo_index(t1: Table, t2: Table, **kwargs) -> None:
    s1, s2 = _collect(t1), _collect(t2)
    names1, names2 = t1.column_names(), t2.column_names()
    assert set(names1) == set(names2), (names1, names2)
    m1 = sorted(
        (tuple(sorted(zip(names1, v))) for v in s1.values()), key=repr
    )
    m2 = sorted(
        (tuple(sorted(zip(names2, v))) for v in s2.values()), key=repr
    )
    assert m1 == m2, f"tables differ (wo index):\n{m1}\n!=\n{m2}"


assert_table_equality_wo_types = assert_table_equality
assert_table_equality_wo_index_types = assert_table_equality_wo_index


class StreamGenerator:
    """Generates artificial streams for tests (reference debug StreamGenerator)."""

    def __init__(self):
        self._counter = 0

    def table_from_list_of_batches_by_workers(self, batches, schema):
        rows = []
        for t, batch in enumerate(batches):
            for worker, events in batch.items():
                for values in events:
                    rows.append(tuple(values) + (t, 1))
        return table_from_rows(schema, rows, is_stream=True)

    def table_from_list_of_batches(self, batches, schema) -> Table:
        rows = []
        for t, batch in enumerate(batches):
            for values in batch:
                if isinstance(values, dict):
                    values = [values[c] for c in schema.column_names()]
                rows.append(tuple(values) + (t, 1))
        return table_from_rows(schema, rows, is_stream=True)

    def table_from_markdown(self, *args, **kwargs) -> Table:
        return table_from_markdown(*args, **kwargs)

    def table_from_pandas(self, df, **kwargs) -> Table:
        """Streamed pandas frame (reference StreamGenerator.table_from_pandas:
        __time__/__diff__ columns drive the event times)."""
        return table_from_pandas(df, **kwargs)

    def persistence_config(self):
        """The reference returns a config replaying generated streams; the
        synchronous engine replays deterministically without one."""
        return None


def table_from_parquet(path: str, id_from=None, unsafe_trusted_ids: bool = False):
    """Read a parquet file into a static table (reference debug/__init__.py)."""
    import pyarrow.parquet as pq

    df = pq.read_table(path).to_pandas()
    return table_from_pandas(df, id_from=id_from, unsafe_trusted_ids=unsafe_trusted_ids)


def table_to_parquet(table: Table, filename: str):
    import pyarrow as pa
    import pyarrow.parquet as pq

    df = table_to_pandas(table, include_id=False)
    pq.write_table(pa.Table.from_pandas(df), filename)
