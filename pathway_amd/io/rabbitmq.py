"""pw.io.rabbitmq — RabbitMQ connector over the pure-python AMQP client.

Reference: python/pathway/io/rabbitmq + src/connectors/data_storage
(amqprs reader/writer).  read() consumes a queue (no-ack) in raw/
plaintext/json formats; write() declares the queue and publishes one
message per row.  Both speak real AMQP 0-9-1 frames
(io/_amqp_protocol.py) and are tested against the in-process fake
broker.
"""

from __future__ import annotations

import json as _json
import urllib.parse
from typing import Any

from pathway_amd.io._amqp_protocol import AmqpClient


def _client(uri: str) -> AmqpClient:
    u = urllib.parse.urlparse(uri if "://" in uri else f"amqp://{uri}")
    return AmqpClient(
        host=u.hostname or "127.0.0.1",
        port=u.port or 5672,
        user=u.username or "guest",
        password=u.password or "guest",
        vhost=u.path.lstrip("/") or "/",
    )


class RabbitReader:
    def __init__(self, source, uri: str, queue: str, parse, *,
                 max_messages: int | None = None):
        self.source = source
        self.uri = uri
        self.queue = queue
        self.parse = parse
        self.max_messages = max_messages

    def run(self) -> None:
        client = None
        try:
            client = _client(self.uri)
            client.queue_declare(self.queue)
            client.consume(self.queue)
            seen = 0
            while True:
                _rk, body = client.next_delivery()
                for values, diff in self.parse(body):
                    self.source.emit(values, diff=diff)
                seen += 1
                if self.max_messages is not None and seen >= self.max_messages:
                    return
        except Exception as e:
            self.source.fail(e)
        finally:
            if client is not None:
                client.close()
            self.source.finish()


def read(
    uri: str,
    queue_name: str,
    *,
    schema=None,
    format: str = "raw",
    mode: str = "streaming",
    autocommit_duration_ms: int | None = 1500,
    name: str | None = None,
    _max_messages: int | None = None,
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

    def parse(body: bytes):
        if format == "raw":
            return [([body], 1)]
        if format == "plaintext":
            return [([body.decode("utf-8", "replace")], 1)]
        if format == "json":
            rec = _json.loads(body)
            return [([rec.get(n) for n in names], 1)]
        raise ValueError(f"unsupported rabbitmq format {format!r}")

    src = StreamingSource(names, dtypes, name=name)
    reader = RabbitReader(src, uri, queue_name, parse,
                          max_messages=_max_messages)
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    uri: str,
    routing_key: str,
    *,
    exchange: str = "",
    format: str = "json",
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    client = _client(uri)
    client.queue_declare(routing_key)
    names = table.column_names()

    def writer(batch):
        for _key, values, time, diff in batch.rows():
            if format == "json":
                rec = dict(zip(names, values))
                rec["time"] = time
                rec["diff"] = diff
                body = _json.dumps(rec, default=str).encode()
            else:
                v = values[0]
                body = v if isinstance(v, bytes) else str(v).encode()
            client.publish(routing_key, body, exchange=exchange)

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
