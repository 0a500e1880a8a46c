"""Pure-python AMQP 0-9-1 client (RabbitMQ wire protocol subset).

Replaces the reference's amqprs dependency (src/connectors/data_storage
rabbitmq reader/writer) with a from-scratch implementation of the frame
protocol: connection/channel handshake (PLAIN auth), queue.declare,
basic.publish (method + content header + body frames) and
basic.consume/deliver.  Exercised against tests/fakes/fake_rabbitmq.py.
"""

from __future__ import annotations

import socket
import struct
import threading
from typing import Any

FRAME_METHOD, FRAME_HEADER, FRAME_BODY, FRAME_HEARTBEAT = 1, 2, 3, 8
FRAME_END = 0xCE


class AmqpError(RuntimeError):
    pass


def shortstr(s: str) -> bytes:
    b = s.encode()
    return bytes([len(b)]) + b


def longstr(b: bytes) -> bytes:
    return struct.pack(">I", len(b)) + b


class _R:
    def __init__(self, data: bytes):
        self.d = data
        self.i = 0

    def u8(self):
        v = self.d[self.i]
        self.i += 1
        return v

    def u16(self):
        (v,) = struct.unpack_from(">H", self.d, self.i)
        self.i += 2
        return v

    def u32(self):
        (v,) = struct.unpack_from(">I", self.d, self.i)
        self.i += 4
        return v

    def u64(self):
        (v,) = struct.unpack_from(">Q", self.d, self.i)
        self.i += 8
        return v

    def sstr(self) -> str:
        n = self.u8()
        v = self.d[self.i : self.i + n].decode()
        self.i += n
        return v

    def lstr(self) -> bytes:
        n = self.u32()
        v = self.d[self.i : self.i + n]
        self.i += n
        return v


class AmqpClient:
    def __init__(self, host: str = "127.0.0.1", port: int = 5672, *,
                 user: str = "guest", password: str = "guest",
                 vhost: str = "/", timeout: float = 30.0):
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self.lock = threading.Lock()
        self._pending_deliver: list[tuple[str, bytes]] = []
        self.sock.sendall(b"AMQP\x00\x00\x09\x01")
        cls, mth, _ = self._expect_method()
        if (cls, mth) != (10, 10):
            raise AmqpError(f"expected connection.start, got {cls}.{mth}")
        resp = b"\x00" + user.encode() + b"\x00" + password.encode()
        args = struct.pack(">I", 0)  # empty client-properties table
        args += shortstr("PLAIN") + longstr(resp) + shortstr("en_US")
        self._send_method(0, 10, 11, args)  # start-ok
        cls, mth, body = self._expect_method()
        if (cls, mth) == (10, 30):  # tune
            r = _R(body)
            chmax, fmax, hb = r.u16(), r.u32(), r.u16()
            self._send_method(0, 10, 31,
                              struct.pack(">HIH", chmax or 1, fmax or 131072, 0))
            self._send_method(0, 10, 40, shortstr(vhost) + b"\x00\x00")  # open
            cls, mth, _ = self._expect_method()
            if (cls, mth) != (10, 41):
                raise AmqpError("connection.open failed")
        self._send_method(1, 20, 10, shortstr(""))  # channel.open
        cls, mth, _ = self._expect_method()
        if (cls, mth) != (20, 11):
            raise AmqpError("channel.open failed")

    # -- framing --

    def _recv_exact(self, n: int) -> bytes:
        buf = b""
        while len(buf) < n:
            chunk = self.sock.recv(n - len(buf))
            if not chunk:
                raise AmqpError("broker closed connection")
            buf += chunk
        return buf

    def _read_frame(self) -> tuple[int, int, bytes]:
        head = self._recv_exact(7)
        ftype, channel, size = struct.unpack(">BHI", head)
        payload = self._recv_exact(size)
        end = self._recv_exact(1)
        if end[0] != FRAME_END:
            raise AmqpError("bad frame end")
        return ftype, channel, payload

    def _send_frame(self, ftype: int, channel: int, payload: bytes) -> None:
        self.sock.sendall(
            struct.pack(">BHI", ftype, channel, len(payload)) + payload
            + bytes([FRAME_END])
        )

    def _send_method(self, channel: int, cls: int, mth: int, args: bytes) -> None:
        self._send_frame(FRAME_METHOD, channel,
                         struct.pack(">HH", cls, mth) + args)

    def _expect_method(self) -> tuple[int, int, bytes]:
        while True:
            ftype, _ch, payload = self._read_frame()
            if ftype == FRAME_HEARTBEAT:
                continue
            if ftype != FRAME_METHOD:
                raise AmqpError(f"unexpected frame type {ftype}")
            cls, mth = struct.unpack_from(">HH", payload, 0)
            return cls, mth, payload[4:]

    # -- operations --

    def queue_declare(self, queue: str) -> None:
        args = b"\x00\x00" + shortstr(queue) + b"\x00" + struct.pack(">I", 0)
        self._send_method(1, 50, 10, args)
        cls, mth, _ = self._expect_method()
        if (cls, mth) != (50, 11):
            raise AmqpError("queue.declare failed")

    def publish(self, routing_key: str, body: bytes, exchange: str = "") -> None:
        with self.lock:
            args = b"\x00\x00" + shortstr(exchange) + shortstr(routing_key) + b"\x00"
            self._send_method(1, 60, 40, args)
            header = struct.pack(">HHQH", 60, 0, len(body), 0)
            self._send_frame(FRAME_HEADER, 1, header)
            self._send_frame(FRAME_BODY, 1, body)

    def consume(self, queue: str) -> None:
        args = (b"\x00\x00" + shortstr(queue) + shortstr("ctag")
                + b"\x02" + struct.pack(">I", 0))  # no-ack
        self._send_method(1, 60, 20, args)
        cls, mth, _ = self._expect_method()
        if (cls, mth) != (60, 21):
            raise AmqpError("basic.consume failed")

    def next_delivery(self) -> tuple[str, bytes]:
        """Blocking read of the next basic.deliver -> (routing_key, body)."""
        while True:
            cls, mth, args = self._expect_method()
            if (cls, mth) != (60, 60):
                continue
            r = _R(args)
            r.sstr()  # consumer tag
            r.u64()  # delivery tag
            r.u8()  # redelivered
            r.sstr()  # exchange
            rk = r.sstr()
            ftype, _ch, header = self._read_frame()
            if ftype != FRAME_HEADER:
                raise AmqpError("expected content header")
            (_cls, _w, body_size, _flags) = struct.unpack_from(">HHQH", header, 0)
            body = b""
            while len(body) < body_size:
                ftype, _ch, chunk = self._read_frame()
                if ftype != FRAME_BODY:
                    raise AmqpError("expected body frame")
                body += chunk
            return rk, body

    def close(self) -> None:
        # half-close + drain: closing with unread inbound data RSTs the
        # connection and can destroy still-buffered outbound frames
        # server-side (see io/nats.py close)
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
