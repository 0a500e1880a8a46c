"""pw.io.pubsub — Google Cloud Pub/Sub connector over the REST API.

Reference: src/connectors/data_storage (pubsub writer).  write():
topics.publish with base64 payloads.  read(): subscriptions.pull +
acknowledge polling.
"""

from __future__ import annotations

import base64
import json as _json
import time as _time
from typing import Any

from pathway_amd.io import _rest

DEFAULT_BASE = "https://pubsub.googleapis.com/v1"


def _headers(credentials: Any) -> dict:
    token = getattr(credentials, "token", None) or (
        credentials if isinstance(credentials, str) else None
    )
    return {"Authorization": f"Bearer {token}"} if token else {}


class PubSubReader:
    def __init__(self, source, base_url: str, project_id: str,
                 subscription: str, parse, headers: dict, *,
                 max_polls: int | None = None):
        self.source = source
        self.url = (f"{base_url}/projects/{project_id}"
                    f"/subscriptions/{subscription}")
        self.parse = parse
        self.headers = headers
        self.max_polls = max_polls

    def run(self) -> None:
        try:
            polls = 0
            while True:
                out = _rest.request(
                    "POST", f"{self.url}:pull",
                    body={"maxMessages": 1000}, headers=self.headers,
                ) or {}
                ack_ids = []
                for rm in out.get("receivedMessages", []):
                    msg = rm.get("message", {})
                    data = base64.b64decode(msg.get("data", ""))
                    for values, diff in self.parse(data):
                        self.source.emit(values, diff=diff)
                    ack_ids.append(rm.get("ackId"))
                if ack_ids:
                    _rest.request(
                        "POST", f"{self.url}:acknowledge",
                        body={"ackIds": ack_ids}, headers=self.headers,
                    )
                polls += 1
                if self.max_polls is not None and polls >= self.max_polls:
                    return
                if not ack_ids:
                    _time.sleep(0.2)
        except Exception as e:
            self.source.fail(e)
        finally:
            self.source.finish()


def read(
    project_id: str,
    subscription: str,
    *,
    schema=None,
    format: str = "raw",
    credentials: Any = None,
    base_url: str = DEFAULT_BASE,
    name: str | None = None,
    _max_polls: int | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    if schema is None:
        schema = schema_from_types(data=bytes if format == "raw" else str)
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]

    def parse(payload: bytes):
        if format == "raw":
            return [([payload], 1)]
        if format == "plaintext":
            return [([payload.decode("utf-8", "replace")], 1)]
        if format == "json":
            rec = _json.loads(payload)
            return [([rec.get(n) for n in names], 1)]
        raise ValueError(f"unsupported pubsub format {format!r}")

    src = StreamingSource(names, dtypes, name=name)
    reader = PubSubReader(src, base_url, project_id, subscription, parse,
                          _headers(credentials), max_polls=_max_polls)
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    project_id: str,
    topic_id: str,
    *,
    credentials: Any = None,
    base_url: str = DEFAULT_BASE,
    format: str = "json",
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    url = f"{base_url}/projects/{project_id}/topics/{topic_id}:publish"
    headers = _headers(credentials)
    names = table.column_names()

    def writer(batch):
        messages = []
        for _key, values, time, diff in batch.rows():
            if format == "json":
                rec = dict(zip(names, values))
                rec["time"] = time
                rec["diff"] = diff
                payload = _json.dumps(rec, default=str).encode()
            else:
                v = values[0]
                payload = v if isinstance(v, bytes) else str(v).encode()
            messages.append({"data": base64.b64encode(payload).decode()})
        if messages:
            _rest.request("POST", url, body={"messages": messages},
                          headers=headers)

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
