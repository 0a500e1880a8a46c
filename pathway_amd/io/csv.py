"""pw.io.csv (reference io/csv)."""
from __future__ import annotations

import csv as _csv
import os
from typing import Any

from pathway_amd.internals import dtype as dt


def read(
    path: str,
    *,
    schema=None,
    mode: str = "streaming",
    csv_settings: Any = None,
    autocommit_duration_ms: int | None = 1500,
    name: str | None = None,
    **kwargs: Any,
):
    if mode in ("streaming", "streaming_with_deletions"):
        from pathway_amd.io import fs as io_fs

        return io_fs.read(
            path, format="csv", schema=schema, mode=mode, name=name,
            **kwargs,
        )
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.schema import schema_from_csv

    from pathway_amd.io._utils import expand_paths

    files = expand_paths(path)
    delimiter = ","
    quotechar = '"'
    if csv_settings is not None:
        delimiter = getattr(csv_settings, "delimiter", ",") or ","
        quotechar = getattr(csv_settings, "quote", '"') or '"'
    if schema is None:
        schema = schema_from_csv(files[0], delimiter=delimiter, quote=quotechar)
    names = schema.column_names()

    def convert(n, v):
        d = dt.unoptionalize(schema.__columns__[n].dtype)
        if v is None or v == "":
            return None
        if d == dt.INT:
            return int(v)
        if d == dt.FLOAT:
            return float(v)
        if d == dt.BOOL:
            return v.lower() in ("true", "1")
        return v

    rows = []
    from pathway_amd.ops import native_io

    for f in files:
        if native_io.available() and quotechar == '"' and len(delimiter) == 1:
            # native scan (libpwio mmap state machine — the data plane the
            # reference runs in Rust data_format::dsv)
            header, recs = native_io.read_csv(f, delimiter=delimiter)
            idx = [header.index(n) if n in header else None for n in names]
            for rec in recs:
                rows.append(
                    tuple(
                        convert(n, rec[i] if i is not None else None)
                        for n, i in zip(names, idx)
                    )
                )
        else:
            with open(f, newline="") as fh:
                reader = _csv.DictReader(fh, delimiter=delimiter, quotechar=quotechar)
                for rec in reader:
                    rows.append(tuple(convert(n, rec.get(n)) for n in names))
    return table_from_rows(schema, rows)


class CsvWriter:
    def __init__(self, filename: str, column_names: list[str]):
        self.filename = filename
        self.column_names = column_names
        import os

        fresh = not (os.path.exists(filename) and os.path.getsize(filename) > 0)
        self._fh = open(filename, "a", newline="")
        self._writer = _csv.writer(self._fh)
        if fresh:
            self._writer.writerow(column_names + ["time", "diff"])

    def __call__(self, batch):
        names = list(batch.columns.keys())
        for key, values, time, diff in batch.rows():
            self._writer.writerow(list(values) + [time, diff])
        self._fh.flush()

    def flush(self, time):
        self._fh.flush()


def write(table, filename: str, *, name: str | None = None, **kwargs):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    writer = CsvWriter(filename, table.column_names())
    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
