"""pw.io.clickhouse — ClickHouse connector over the HTTP interface.

Reference: src/connectors/data_storage/clickhouse.rs (947 LoC over
clickhouse-rs).  Uses the standard HTTP interface: INSERT ... FORMAT
JSONEachRow posts, SELECT ... FORMAT JSON reads — works against any
ClickHouse server; tested against the capturing fake service.
"""

from __future__ import annotations

import json
import urllib.parse
from typing import Any

from pathway_amd.io import _rest


def _url(settings: dict, query: str) -> str:
    host = settings.get("host", "127.0.0.1")
    port = settings.get("port", 8123)
    base = host if "://" in str(host) else f"http://{host}:{port}"
    params = {"query": query}
    if settings.get("database"):
        params["database"] = settings["database"]
    if settings.get("user"):
        params["user"] = settings["user"]
    if settings.get("password"):
        params["password"] = settings["password"]
    return f"{base}/?{urllib.parse.urlencode(params)}"


def write(
    table,
    settings: dict,
    table_name: str,
    *,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    names = table.column_names()

    def writer(batch):
        lines = []
        for _key, values, time, diff in batch.rows():
            rec = dict(zip(names, values))
            rec["time"] = time
            rec["diff"] = diff
            lines.append(json.dumps(rec, default=str))
        if not lines:
            return
        _rest.request(
            "POST",
            _url(settings, f"INSERT INTO {table_name} FORMAT JSONEachRow"),
            raw_body=("\n".join(lines) + "\n").encode(),
            content_type="application/x-ndjson",
        )

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
