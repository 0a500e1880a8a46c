"""pw.io.jsonlines (reference io/jsonlines)."""
from __future__ import annotations

import json
import os
from typing import Any

from pathway_amd.internals import dtype as dt


def read(
    path: str,
    *,
    schema=None,
    mode: str = "streaming",
    json_field_paths: dict | None = None,
    autocommit_duration_ms: int | None = 1500,
    name: str | None = None,
    **kwargs: Any,
):
    if mode in ("streaming", "streaming_with_deletions"):
        # live mode: the fs poller tails the directory (new/modified
        # files append, deletions retract)
        from pathway_amd.io import fs as io_fs

        return io_fs.read(
            path, format="json", schema=schema, mode=mode, name=name,
            **kwargs,
        )
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.json import Json

    from pathway_amd.io._utils import expand_paths

    files = expand_paths(path)
    records = []
    for f in files:
        with open(f) as fh:
            for line in fh:
                line = line.strip()
                if line:
                    records.append(json.loads(line))
    if schema is None:
        from pathway_amd.internals.schema import schema_from_types

        keys: dict[str, Any] = {}
        for rec in records:
            for k, v in rec.items():
                keys.setdefault(k, type(v) if v is not None else str)
        schema = schema_from_types(**keys)
    names = schema.column_names()
    rows = []
    for rec in records:
        row = []
        for n in names:
            v = rec.get(n)
            d = dt.unoptionalize(schema.__columns__[n].dtype)
            if isinstance(v, (dict, list)) and d == dt.JSON:
                v = Json(v)
            row.append(v)
        rows.append(tuple(row))
    return table_from_rows(schema, rows)


class JsonlWriter:
    def __init__(self, filename: str, column_names: list[str]):
        self.filename = filename
        self.column_names = column_names
        self._fh = open(filename, "w")

    def __call__(self, batch):
        names = list(batch.columns.keys())
        for key, values, time, diff in batch.rows():
            rec = dict(zip(names, [_jsonable(v) for v in values]))
            rec["time"] = time
            rec["diff"] = diff
            self._fh.write(json.dumps(rec) + "\n")
        self._fh.flush()

    def flush(self, time):
        self._fh.flush()


def _jsonable(v):
    from pathway_amd.internals.json import Json
    from pathway_amd.internals.api import BasePointer

    if isinstance(v, Json):
        return v.value
    if isinstance(v, BasePointer):
        return repr(v)
    if isinstance(v, bytes):
        import base64

        return base64.b64encode(v).decode()
    if isinstance(v, tuple):
        return list(v)
    return v


def write(table, filename: str, *, name: str | None = None, **kwargs):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    writer = JsonlWriter(filename, table.column_names())
    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
