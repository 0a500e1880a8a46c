"""pw.io.questdb — QuestDB sink over InfluxDB line protocol (ILP/TCP).

Reference: src/connectors/data_storage/questdb.rs (questdb-rs ILP
sender).  Emits standard ILP lines ``table,sym=.. field=.. ts`` over a
raw TCP socket; tested against an in-process line-capturing server.
"""

from __future__ import annotations

import socket
import time as _time
from typing import Any


def _escape_tag(s: str) -> str:
    return s.replace(" ", "\\ ").replace(",", "\\,").replace("=", "\\=")


def _field_value(v: Any) -> str:
    if isinstance(v, bool):
        return "t" if v else "f"
    if isinstance(v, int):
        return f"{v}i"
    if isinstance(v, float):
        return repr(v)
    s = str(v).replace('"', '\\"')
    return f'"{s}"'


def write(
    table,
    connection_string_or_host: str | dict,
    table_name: str,
    *,
    designated_timestamp_policy: str | None = None,
    at_column: str | None = None,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    if isinstance(connection_string_or_host, dict):
        host = connection_string_or_host.get("host", "127.0.0.1")
        port = int(connection_string_or_host.get("port", 9009))
    else:
        host, _, port_s = connection_string_or_host.partition(":")
        port = int(port_s or 9009)
    names = table.column_names()

    def writer(batch):
        lines = []
        for _key, values, time, diff in batch.rows():
            fields = [
                f"{_escape_tag(n)}={_field_value(v)}"
                for n, v in zip(names, values)
                if v is not None
            ]
            fields.append(f"time={time}i")
            fields.append(f"diff={diff}i")
            ts = _time.time_ns()
            lines.append(f"{_escape_tag(table_name)} {','.join(fields)} {ts}")
        if not lines:
            return
        with socket.create_connection((host, port), timeout=10) as s:
            s.sendall(("\n".join(lines) + "\n").encode())

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
