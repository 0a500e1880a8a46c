"""pw.io.sqlite (reference sqlite.rs:1698) — stdlib sqlite3 backed."""
from __future__ import annotations

import sqlite3
from typing import Any


def read(path: str, table_name: str, schema, *, mode: str = "static", autocommit_duration_ms=1500, name=None, **kwargs):
    from pathway_amd.debug import table_from_rows

    con = sqlite3.connect(path)
    names = schema.column_names()
    cur = con.execute(f"SELECT {', '.join(names)} FROM {table_name}")
    rows = [tuple(r) for r in cur.fetchall()]
    con.close()
    return table_from_rows(schema, rows)


def write(table, path: str, table_name: str, *, name: str | None = None, **kwargs):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    names = table.column_names()
    con = sqlite3.connect(path, check_same_thread=False)
    cols = ", ".join(names)
    qs = ", ".join("?" * (len(names) + 2))
    con.execute(
        f"CREATE TABLE IF NOT EXISTS {table_name} ({cols}, time INTEGER, diff INTEGER)"
    )

    def writer(batch):
        for key, values, time, diff in batch.rows():
            con.execute(
                f"INSERT INTO {table_name} VALUES ({qs})",
                [_plain(v) for v in values] + [time, diff],
            )
        con.commit()

    def _plain(v):
        from pathway_amd.internals.json import Json

        if isinstance(v, Json):
            return v.dumps()
        if isinstance(v, (tuple, list)):
            import json

            return json.dumps(list(v), default=str)
        return v

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
