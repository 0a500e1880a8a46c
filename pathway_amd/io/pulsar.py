"""pw.io.pulsar — Pulsar connector over the WebSocket API.

Reference: python/pathway/io/pulsar + src/connectors/data_storage
(pulsar.rs, 1,836 LoC over the binary protocol).  Pulsar's WebSocket
proxy speaks JSON frames with base64 payloads:

  producer:  ws://host:8080/ws/v2/producer/persistent/<tenant>/<ns>/<topic>
             send {"payload": b64, ...} -> recv {"result": "ok", ...}
  consumer:  .../ws/v2/consumer/persistent/<t>/<ns>/<topic>/<subscription>
             recv {"messageId", "payload": b64, ...} -> send {"messageId"}

This build implements that protocol with aiohttp (in-image); the fake
broker in tests/fakes/fake_pulsar.py serves the same frames.
"""

from __future__ import annotations

import asyncio
import base64
import json as _json
import threading
from typing import Any


def _topic_path(topic: str) -> str:
    # accept "persistent://tenant/ns/topic" or bare "topic"
    if topic.startswith("persistent://"):
        return "persistent/" + topic[len("persistent://"):]
    if topic.count("/") >= 2:
        return f"persistent/{topic}"
    return f"persistent/public/default/{topic}"


class PulsarWsProducer:
    """Synchronous facade over an aiohttp websocket producer."""

    def __init__(self, service_url: str, topic: str):
        import aiohttp

        base = service_url.rstrip("/").replace("pulsar://", "http://")
        self.url = f"{base}/ws/v2/producer/{_topic_path(topic)}"
        self.loop = asyncio.new_event_loop()
        self.thread = threading.Thread(target=self.loop.run_forever, daemon=True)
        self.thread.start()

        async def connect():
            session = aiohttp.ClientSession()
            ws = await session.ws_connect(self.url)
            return session, ws

        self.session, self.ws = asyncio.run_coroutine_threadsafe(
            connect(), self.loop
        ).result(30)

    def send(self, payload: bytes, properties: dict | None = None) -> None:
        async def go():
            await self.ws.send_str(_json.dumps({
                "payload": base64.b64encode(payload).decode(),
                "properties": properties or {},
            }))
            ack = await self.ws.receive_json()
            if ack.get("result") not in ("ok", None):
                raise RuntimeError(f"pulsar send failed: {ack}")

        asyncio.run_coroutine_threadsafe(go(), self.loop).result(30)

    def close(self) -> None:
        async def go():
            await self.ws.close()
            await self.session.close()

        try:
            asyncio.run_coroutine_threadsafe(go(), self.loop).result(10)
        except Exception:
            pass
        self.loop.call_soon_threadsafe(self.loop.stop)


class PulsarReader:
    def __init__(self, source, service_url: str, topic: str,
                 subscription: str, parse, *, max_messages: int | None = None):
        self.source = source
        self.service_url = service_url
        self.topic = topic
        self.subscription = subscription
        self.parse = parse
        self.max_messages = max_messages

    def run(self) -> None:
        try:
            asyncio.run(self._run())
        except Exception as e:
            self.source.fail(e)
        finally:
            self.source.finish()

    async def _run(self) -> None:
        import aiohttp

        base = self.service_url.rstrip("/").replace("pulsar://", "http://")
        url = (f"{base}/ws/v2/consumer/{_topic_path(self.topic)}/"
               f"{self.subscription}")
        seen = 0
        async with aiohttp.ClientSession() as session:
            async with session.ws_connect(url) as ws:
                while True:
                    msg = await ws.receive_json()
                    payload = base64.b64decode(msg.get("payload", ""))
                    for values, diff in self.parse(payload):
                        self.source.emit(values, diff=diff)
                    await ws.send_str(_json.dumps(
                        {"messageId": msg.get("messageId")}
                    ))
                    seen += 1
                    if self.max_messages is not None and seen >= self.max_messages:
                        return


def read(
    service_url: str,
    topic: str,
    *,
    consumer_name: str = "pathway",
    subscription: str | None = None,
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

    def parse(payload: bytes):
        if format == "raw":
            return [([payload], 1)]
        if format == "plaintext":
            return [([payload.decode("utf-8", "replace")], 1)]
        if format == "json":
            rec = _json.loads(payload)
            return [([rec.get(n) for n in names], 1)]
        raise ValueError(f"unsupported pulsar format {format!r}")

    src = StreamingSource(names, dtypes, name=name)
    reader = PulsarReader(
        src, service_url, topic, subscription or consumer_name, parse,
        max_messages=_max_messages,
    )
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    service_url: str,
    topic: str,
    *,
    format: str = "json",
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    producer = PulsarWsProducer(service_url, topic)
    names = table.column_names()

    def writer(batch):
        for _key, values, time, diff in batch.rows():
            if format == "json":
                rec = dict(zip(names, values))
                rec["time"] = time
                rec["diff"] = diff
                payload = _json.dumps(rec, default=str).encode()
            else:
                v = values[0]
                payload = v if isinstance(v, bytes) else str(v).encode()
            producer.send(payload)

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
