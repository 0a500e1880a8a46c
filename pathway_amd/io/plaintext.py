"""pw.io.plaintext (reference io/plaintext)."""
from __future__ import annotations

import os
from typing import Any


def read(path: str, *, mode: str = "streaming", name: str | None = None, **kwargs: Any):
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.schema import schema_from_types

    from pathway_amd.io._utils import expand_paths

    files = expand_paths(path)
    rows = []
    for f in files:
        with open(f) as fh:
            for line in fh:
                rows.append((line.rstrip("\n"),))
    schema = schema_from_types(data=str)
    return table_from_rows(schema, rows)
