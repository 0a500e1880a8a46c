"""pw.io.fs (reference io/fs) — static mode reads; streaming mode lands with
the connector-runtime phase."""
from __future__ import annotations

import json
import os
from typing import Any

from pathway_amd.internals import dtype as dt


def read(
    path: str,
    *,
    format: str = "plaintext",
    schema=None,
    mode: str = "streaming",
    with_metadata: bool = False,
    autocommit_duration_ms: int | None = 1500,
    name: str | None = None,
    refresh_interval: float = 0.5,
    _max_polls: int | None = None,
    **kwargs: Any,
):
    if mode in ("streaming", "streaming_with_deletions"):
        return _read_streaming(
            path,
            format=format,
            schema=schema,
            with_metadata=with_metadata,
            name=name,
            refresh_interval=refresh_interval,
            max_polls=_max_polls,
        )
    from pathway_amd.io import csv as io_csv, jsonlines as io_jsonlines, plaintext as io_plaintext

    if format in ("csv",):
        return io_csv.read(path, schema=schema, mode=mode, name=name, **kwargs)
    if format in ("json", "jsonlines"):
        return io_jsonlines.read(path, schema=schema, mode=mode, name=name, **kwargs)
    if format in ("plaintext", "plaintext_by_file"):
        return io_plaintext.read(path, mode=mode, name=name, **kwargs)
    if format == "binary":
        return _read_binary(path, with_metadata=with_metadata)
    raise ValueError(f"unknown format {format!r}")


def _read_streaming(
    path: str,
    *,
    format: str,
    schema,
    with_metadata: bool,
    name: str | None,
    refresh_interval: float,
    max_polls: int | None,
):
    """Live directory/file polling (reference posix_like scanner)."""
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import (
        FilePollReader,
        StreamingSource,
        spawn_reader,
    )
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    if schema is None:
        if format in ("plaintext", "plaintext_by_file"):
            schema = schema_from_types(data=str)
        elif format == "binary":
            schema = schema_from_types(data=bytes)
        else:
            raise ValueError("streaming csv/json reads need schema=")
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    if with_metadata:
        names = names + ["_metadata"]
        dtypes = dtypes + [dt.JSON]
    src = StreamingSource(names, dtypes, name=name)
    reader = FilePollReader(
        src, path, format if format != "plaintext_by_file" else "binary",
        schema, "streaming", with_metadata,
        refresh_interval=refresh_interval, max_polls=max_polls,
    )
    spawn_reader(reader.run, src, sharded=True)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def _read_binary(path: str, with_metadata: bool = False):
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.schema import schema_from_types

    from pathway_amd.io._utils import expand_paths

    files = expand_paths(path)
    rows = []
    for f in files:
        with open(f, "rb") as fh:
            data = fh.read()
        if with_metadata:
            from pathway_amd.internals.json import Json

            meta = Json({"path": f, "size": len(data), "seen_at": 0,
                         "modified_at": int(os.path.getmtime(f)),
                         "owner": "unknown"})
            rows.append((data, meta))
        else:
            rows.append((data,))
    if with_metadata:
        schema = schema_from_types(data=bytes, _metadata=dt.JSON)
    else:
        schema = schema_from_types(data=bytes)
    return table_from_rows(schema, rows)


def write(table, filename: str, *, format: str = "csv", name: str | None = None, **kwargs):
    from pathway_amd.io import csv as io_csv, jsonlines as io_jsonlines

    if format == "csv":
        return io_csv.write(table, filename, name=name)
    if format in ("json", "jsonlines"):
        return io_jsonlines.write(table, filename, name=name)
    raise ValueError(f"unknown format {format!r}")
