"""pw.io.nats — NATS connector over the text wire protocol.

Reference: python/pathway/io/nats + src/connectors/data_storage (nats.rs
over async-nats).  Implements the NATS client protocol directly
(INFO/CONNECT/PUB/SUB/MSG/PING/PONG over TCP) — no client library.
Tested against the in-process fake server (tests/fakes/fake_nats.py).
"""

from __future__ import annotations

import json as _json
import socket
import threading
import time as _time
import urllib.parse
from typing import Any, Callable


class NatsError(RuntimeError):
    pass


class NatsClient:
    def __init__(self, uri: str = "nats://127.0.0.1:4222", timeout: float = 30.0):
        u = urllib.parse.urlparse(uri)
        self.sock = socket.create_connection(
            (u.hostname or "127.0.0.1", u.port or 4222), timeout=timeout
        )
        self.buf = b""
        self.lock = threading.Lock()
        self._sid = 0
        self._handlers: dict[str, Callable[[bytes], None]] = {}
        line = self._read_line()
        if not line.startswith(b"INFO "):
            raise NatsError(f"expected INFO, got {line[:40]!r}")
        self._send(
            b"CONNECT "
            + _json.dumps({"verbose": False, "pedantic": False,
                           "name": "pathway-amd", "lang": "python",
                           "version": "1"}).encode()
            + b"\r\n"
        )

    def _send(self, data: bytes) -> None:
        with self.lock:
            self.sock.sendall(data)

    def _read_line(self) -> bytes:
        while b"\r\n" not in self.buf:
            chunk = self.sock.recv(65536)
            if not chunk:
                raise NatsError("server closed connection")
            self.buf += chunk
        line, self.buf = self.buf.split(b"\r\n", 1)
        return line

    def _read_exact(self, n: int) -> bytes:
        while len(self.buf) < n:
            chunk = self.sock.recv(65536)
            if not chunk:
                raise NatsError("server closed connection")
            self.buf += chunk
        out, self.buf = self.buf[:n], self.buf[n:]
        return out

    def publish(self, subject: str, payload: bytes) -> None:
        self._send(
            f"PUB {subject} {len(payload)}\r\n".encode() + payload + b"\r\n"
        )

    def subscribe(self, subject: str) -> int:
        self._sid += 1
        self._send(f"SUB {subject} {self._sid}\r\n".encode())
        return self._sid

    def next_message(self) -> tuple[str, bytes] | None:
        """Blocking read of the next MSG; answers PING transparently."""
        while True:
            line = self._read_line()
            if line.startswith(b"MSG "):
                parts = line.decode().split(" ")
                subject = parts[1]
                nbytes = int(parts[-1])
                payload = self._read_exact(nbytes)
                self._read_exact(2)  # trailing CRLF
                return subject, payload
            if line == b"PING":
                self._send(b"PONG\r\n")
            elif line.startswith(b"-ERR"):
                raise NatsError(line.decode())
            # +OK / PONG / INFO: ignore

    def close(self) -> None:
        # orderly shutdown: if the server sent anything we never read
        # (+OK in verbose mode, INFO updates), closing with unread data
        # RSTs the connection and can destroy our still-buffered
        # outbound frames server-side — half-close and drain first
        try:
            self.sock.shutdown(socket.SHUT_WR)
            self.sock.settimeout(0.25)
            while self.sock.recv(65536):
                pass
        except OSError:
            pass
        try:
            self.sock.close()
        except OSError:
            pass


class NatsReader:
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
            client = NatsClient(self.uri)
            client.subscribe(self.topic)
            seen = 0
            while True:
                msg = client.next_message()
                if msg is None:
                    return
                _subject, payload = msg
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
        schema = schema_from_types(
            data=bytes if format == "raw" else str
        )
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
        raise ValueError(f"unsupported nats format {format!r}")

    src = StreamingSource(names, dtypes, name=name)
    reader = NatsReader(src, uri, topic, parse, max_messages=_max_messages)
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
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    client = NatsClient(uri)
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
