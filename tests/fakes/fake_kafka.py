"""In-process fake Kafka broker.

Speaks the real wire protocol subset the pathway_amd client uses
(Metadata v0, Produce v3, Fetch v4, ListOffsets v1, RecordBatch magic 2)
over a localhost TCP socket, backed by in-memory per-partition logs.
Topics auto-create with a configurable partition count.
"""

from __future__ import annotations

import socket
import socketserver
import struct
import threading
import time

from pathway_amd.io._kafka_protocol import (
    API_FETCH,
    API_LIST_OFFSETS,
    API_METADATA,
    API_PRODUCE,
    Reader,
    Writer,
    decode_record_batches,
    encode_record_batch,
)


class _Partition:
    def __init__(self):
        #: (offset, key, value, ts_ms)
        self.records: list[tuple[int, bytes | None, bytes | None, int]] = []
        self.lock = threading.Lock()

    @property
    def next_offset(self) -> int:
        return self.records[-1][0] + 1 if self.records else 0

    def append(self, recs: list[tuple[bytes | None, bytes | None]]) -> int:
        with self.lock:
            base = self.next_offset
            ts = int(time.time() * 1000)
            for i, (k, v) in enumerate(recs):
                self.records.append((base + i, k, v, ts))
            return base

    def read_from(self, offset: int, max_records: int = 10000):
        with self.lock:
            return [r for r in self.records if r[0] >= offset][:max_records]


class FakeKafkaBroker:
    def __init__(self, num_partitions: int = 2):
        self.num_partitions = num_partitions
        self.topics: dict[str, dict[int, _Partition]] = {}
        self.node_id = 0
        broker = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                sock: socket.socket = self.request
                try:
                    while True:
                        head = self._recv_exact(sock, 4)
                        if head is None:
                            return
                        (size,) = struct.unpack(">i", head)
                        frame = self._recv_exact(sock, size)
                        if frame is None:
                            return
                        resp = broker.handle_frame(frame)
                        sock.sendall(struct.pack(">i", len(resp)) + resp)
                except (ConnectionResetError, BrokenPipeError, OSError):
                    return

            @staticmethod
            def _recv_exact(sock, n):
                buf = b""
                while len(buf) < n:
                    try:
                        chunk = sock.recv(n - len(buf))
                    except OSError:
                        return None
                    if not chunk:
                        return None
                    buf += chunk
                return buf

        class Server(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self.server = Server(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    # -- lifecycle --

    @property
    def port(self) -> int:
        return self.server.server_address[1]

    @property
    def bootstrap(self) -> str:
        return f"127.0.0.1:{self.port}"

    def start(self) -> "FakeKafkaBroker":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()

    # -- topic store --

    def topic(self, name: str) -> dict[int, _Partition]:
        t = self.topics.get(name)
        if t is None:
            t = {i: _Partition() for i in range(self.num_partitions)}
            self.topics[name] = t
        return t

    def seed(self, topic: str, partition: int,
             records: list[tuple[bytes | None, bytes | None]]) -> int:
        """Test helper: append records directly."""
        return self.topic(topic)[partition].append(records)

    def all_values(self, topic: str) -> list[bytes]:
        out = []
        for p in self.topic(topic).values():
            out.extend(v for _, _, v, _ in p.records)
        return out

    # -- protocol --

    def handle_frame(self, frame: bytes) -> bytes:
        r = Reader(frame)
        api_key = r.i16()
        api_version = r.i16()
        corr = r.i32()
        r.string()  # client_id
        body = Reader(frame[r.i :])
        w = Writer()
        w.i32(corr)
        if api_key == API_METADATA:
            self._metadata(body, w)
        elif api_key == API_PRODUCE:
            self._produce(body, w)
        elif api_key == API_FETCH:
            self._fetch(body, w)
        elif api_key == API_LIST_OFFSETS:
            self._list_offsets(body, w)
        else:
            raise ValueError(f"fake broker: unsupported api key {api_key}")
        return w.data()

    def _metadata(self, body: Reader, w: Writer) -> None:
        n = body.i32()
        names = [body.string() for _ in range(n)]
        if not names:
            names = list(self.topics.keys())
        w.i32(1)  # one broker
        w.i32(self.node_id)
        w.string("127.0.0.1")
        w.i32(self.port)
        w.i32(len(names))
        for t in names:
            parts = self.topic(t)
            w.i16(0)
            w.string(t)
            w.i32(len(parts))
            for pid in sorted(parts):
                w.i16(0)
                w.i32(pid)
                w.i32(self.node_id)  # leader
                w.i32(1)
                w.i32(self.node_id)  # replicas
                w.i32(1)
                w.i32(self.node_id)  # isr
        return

    def _produce(self, body: Reader, w: Writer) -> None:
        body.string()  # transactional id
        body.i16()  # acks
        body.i32()  # timeout
        ntopics = body.i32()
        results = []
        for _ in range(ntopics):
            topic = body.string()
            nparts = body.i32()
            for _ in range(nparts):
                pid = body.i32()
                record_set = body.bytes_() or b""
                recs = decode_record_batches(record_set)
                base = self.topic(topic)[pid].append(
                    [(k, v) for _, k, v, _ in recs]
                )
                results.append((topic, pid, base))
        w.i32(len({t for t, _, _ in results}))
        by_topic: dict[str, list[tuple[int, int]]] = {}
        for t, pid, base in results:
            by_topic.setdefault(t, []).append((pid, base))
        for t, parts in by_topic.items():
            w.string(t)
            w.i32(len(parts))
            for pid, base in parts:
                w.i32(pid)
                w.i16(0)
                w.i64(base)
                w.i64(-1)  # log append time
        w.i32(0)  # throttle

    def _fetch(self, body: Reader, w: Writer) -> None:
        body.i32()  # replica
        body.i32()  # max_wait
        body.i32()  # min_bytes
        body.i32()  # max_bytes
        body.i8()  # isolation
        ntopics = body.i32()
        w.i32(0)  # throttle
        w.i32(ntopics)
        for _ in range(ntopics):
            topic = body.string()
            nparts = body.i32()
            w.string(topic)
            w.i32(nparts)
            for _ in range(nparts):
                pid = body.i32()
                offset = body.i64()
                body.i32()  # partition max bytes
                part = self.topic(topic)[pid]
                recs = part.read_from(offset)
                w.i32(pid)
                w.i16(0)
                w.i64(part.next_offset)  # high watermark
                w.i64(part.next_offset)  # last stable
                w.i32(0)  # aborted txns
                if recs:
                    base = recs[0][0]
                    batch = encode_record_batch(
                        base, [(k, v) for _, k, v, _ in recs],
                        timestamp_ms=recs[0][3],
                    )
                    w.bytes_(batch)
                else:
                    w.bytes_(b"")

    def _list_offsets(self, body: Reader, w: Writer) -> None:
        body.i32()  # replica
        ntopics = body.i32()
        w.i32(ntopics)
        for _ in range(ntopics):
            topic = body.string()
            nparts = body.i32()
            w.string(topic)
            w.i32(nparts)
            for _ in range(nparts):
                pid = body.i32()
                ts = body.i64()
                part = self.topic(topic)[pid]
                if ts == -2:
                    off = part.records[0][0] if part.records else 0
                elif ts == -1:
                    off = part.next_offset
                else:
                    matching = [o for o, _, _, t in part.records if t >= ts]
                    off = matching[0] if matching else part.next_offset
                w.i32(pid)
                w.i16(0)
                w.i64(ts)
                w.i64(off)
