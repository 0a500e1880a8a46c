"""pw.io.mqtt — MQTT connector over the 3.1.1 binary protocol.

Reference: python/pathway/io/mqtt + src/connectors/data_storage (rumqttc).
Implements MQTT 3.1.1 directly over TCP: CONNECT/CONNACK, PUBLISH (QoS 0),
SUBSCRIBE/SUBACK, PINGREQ/PINGRESP, DISCONNECT — no client library.
Tested against the in-process fake broker (tests/fakes/fake_mqtt.py).
"""

from __future__ import annotations

import json as _json
import socket
import struct
import threading
import urllib.parse
from typing import Any

CONNECT, CONNACK, PUBLISH, SUBSCRIBE, SUBACK = 1, 2, 3, 8, 9
PINGREQ, PINGRESP, DISCONNECT = 12, 13, 14


class MqttError(RuntimeError):
    pass


def _encode_remaining(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n % 128
        n //= 128
        out.append(b | 0x80 if n else b)
        if not n:
            return bytes(out)


class MqttClient:
    def __init__(self, uri: str = "mqtt://127.0.0.1:1883", *,
                 client_id: str = "pathway-amd", timeout: float = 30.0):
        u = urllib.parse.urlparse(uri if "://" in uri else f"mqtt://{uri}")
        self.sock = socket.create_connection(
            (u.hostname or "127.0.0.1", u.port or 1883), timeout=timeout
        )
        self.lock = threading.Lock()
        self._packet_id = 0
        var = b"\x00\x04MQTT\x04\x02\x00\x3c"  # proto, level 4, clean session, keepalive 60
        payload = struct.pack(">H", len(client_id)) + client_id.encode()
        self._send_packet(CONNECT, 0, var + payload)
        ptype, _flags, data = self._recv_packet()
        if ptype != CONNACK or data[1] != 0:
            raise MqttError(f"CONNACK failed: {data!r}")

    def _send_packet(self, ptype: int, flags: int, body: bytes) -> None:
        with self.lock:
            self.sock.sendall(
                bytes([(ptype << 4) | flags]) + _encode_remaining(len(body)) + body
            )

    def _recv_exact(self, n: int) -> bytes:
        buf = b""
        while len(buf) < n:
            chunk = self.sock.recv(n - len(buf))
            if not chunk:
                raise MqttError("broker closed connection")
            buf += chunk
        return buf

    def _recv_packet(self) -> tuple[int, int, bytes]:
        head = self._recv_exact(1)[0]
        mult, rem = 1, 0
        while True:
            b = self._recv_exact(1)[0]
            rem += (b & 0x7F) * mult
            if not b & 0x80:
                break
            mult *= 128
        data = self._recv_exact(rem) if rem else b""
        return head >> 4, head & 0x0F, data

    def publish(self, topic: str, payload: bytes) -> None:
        tb = topic.encode()
        self._send_packet(PUBLISH, 0, struct.pack(">H", len(tb)) + tb + payload)

    def subscribe(self, topic_filter: str) -> None:
        self._packet_id += 1
        tb = topic_filter.encode()
        body = struct.pack(">H", self._packet_id)
        body += struct.pack(">H", len(tb)) + tb + b"\x00"  # QoS 0
        self._send_packet(SUBSCRIBE, 2, body)
        ptype, _f, _d = self._recv_packet()
        if ptype != SUBACK:
            raise MqttError(f"expected SUBACK, got {ptype}")

    def next_message(self) -> tuple[str, bytes]:
        """Blocking read of the next PUBLISH; answers PINGREQ/RESP."""
        while True:
            ptype, flags, data = self._recv_packet()
            if ptype == PUBLISH:
                (tlen,) = struct.unpack_from(">H", data, 0)
                topic = data[2 : 2 + tlen].decode()
                i = 2 + tlen
                qos = (flags >> 1) & 3
                if qos:
                    i += 2  # packet id
                return topic, data[i:]
            if ptype == PINGREQ:
                self._send_packet(PINGRESP, 0, b"")
            # PINGRESP etc: ignore

    def close(self) -> None:
        try:
            self._send_packet(DISCONNECT, 0, b"")
        except OSError:
            pass
        import socket as _socket

        try:
            self.sock.shutdown(_socket.SHUT_WR)
            self.sock.settimeout(0.25)
            while self.sock.recv(65536):
                pass
        except OSError:
            pass
        try:
            self.sock.close()
        except OSError:
            pass


class MqttReader:
    def __init__(self, source, uri: str, topic: str, parse, *,
                 max_messages: int | None = None):
        self.source = source
        self.uri = uri
        self.topic = topic
        self.parse = parse
        self.max_messages = max_messages

    def run(self) -> None:
        client = None
        try:
            client = MqttClient(self.uri, client_id=f"pw-r-{id(self):x}")
            client.subscribe(self.topic)
            seen = 0
            while True:
                _topic, payload = client.next_message()
                for values, diff in self.parse(payload):
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
    topic: str,
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

    def parse(payload: bytes):
        if format == "raw":
            return [([payload], 1)]
        if format == "plaintext":
            return [([payload.decode("utf-8", "replace")], 1)]
        if format == "json":
            rec = _json.loads(payload)
            return [([rec.get(n) for n in names], 1)]
        raise ValueError(f"unsupported mqtt format {format!r}")

    src = StreamingSource(names, dtypes, name=name)
    reader = MqttReader(src, uri, topic, parse, max_messages=_max_messages)
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    uri: str,
    topic: str,
    *,
    format: str = "json",
    qos: int = 0,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    client = MqttClient(uri, client_id=f"pw-w-{id(table):x}")
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
            client.publish(topic, payload)

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
