"""In-process fake RabbitMQ broker (AMQP 0-9-1 subset)."""

from __future__ import annotations

import socketserver
import struct
import threading

from pathway_amd.io._amqp_protocol import FRAME_BODY, FRAME_END, FRAME_HEADER, FRAME_METHOD, _R, longstr, shortstr


class FakeRabbit:
    def __init__(self):
        self.lock = threading.Lock()
        #: queue -> list of (send_fn) subscribers
        self.consumers: dict[str, list] = {}
        self.published: list[tuple[str, bytes]] = []
        #: queue -> backlog for messages before any consumer
        self.backlog: dict[str, list[bytes]] = {}
        broker = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                sock = self.request
                self.sendlock = threading.Lock()
                try:
                    if broker._recv_exact(sock, 8) != b"AMQP\x00\x00\x09\x01":
                        return
                    self._method(sock, 0, 10, 10,
                                 b"\x00\x09" + struct.pack(">I", 0)
                                 + longstr(b"PLAIN") + longstr(b"en_US"))
                    my_queues = []
                    publish_state = {}
                    while True:
                        fr = broker._read_frame(sock)
                        if fr is None:
                            return
                        ftype, ch, payload = fr
                        if ftype == FRAME_METHOD:
                            cls, mth = struct.unpack_from(">HH", payload, 0)
                            args = payload[4:]
                            if (cls, mth) == (10, 11):  # start-ok
                                self._method(sock, 0, 10, 30,
                                             struct.pack(">HIH", 1, 131072, 0))
                            elif (cls, mth) == (10, 31):
                                pass  # tune-ok
                            elif (cls, mth) == (10, 40):
                                self._method(sock, 0, 10, 41, shortstr(""))
                            elif (cls, mth) == (20, 10):
                                self._method(sock, ch, 20, 11, longstr(b""))
                            elif (cls, mth) == (50, 10):
                                r = _R(args)
                                r.u16()
                                q = r.sstr()
                                with broker.lock:
                                    broker.backlog.setdefault(q, [])
                                self._method(sock, ch, 50, 11,
                                             shortstr(q)
                                             + struct.pack(">II", 0, 0))
                            elif (cls, mth) == (60, 40):  # publish
                                r = _R(args)
                                r.u16()
                                r.sstr()  # exchange
                                rk = r.sstr()
                                publish_state["rk"] = rk
                                publish_state["body"] = b""
                                publish_state["size"] = None
                            elif (cls, mth) == (60, 20):  # consume
                                r = _R(args)
                                r.u16()
                                q = r.sstr()
                                def deliver(body, rk=None, s=sock, me=self):
                                    frame = (shortstr("ctag")
                                             + struct.pack(">QB", 1, 0)
                                             + shortstr("")
                                             + shortstr(rk or q))
                                    with me.sendlock:
                                        me._method(s, 1, 60, 60, frame)
                                        hdr = struct.pack(
                                            ">HHQH", 60, 0, len(body), 0)
                                        broker._send_frame(s, FRAME_HEADER, 1, hdr)
                                        broker._send_frame(s, FRAME_BODY, 1, body)
                                with broker.lock:
                                    broker.consumers.setdefault(q, []).append(deliver)
                                    pending = broker.backlog.get(q, [])[:]
                                    broker.backlog[q] = []
                                self._method(sock, ch, 60, 21, shortstr("ctag"))
                                for b in pending:
                                    deliver(b)
                                my_queues.append(q)
                        elif ftype == FRAME_HEADER:
                            (_c, _w, size, _f) = struct.unpack_from(">HHQH", payload, 0)
                            publish_state["size"] = size
                            if size == 0:
                                broker._route(publish_state)
                        elif ftype == FRAME_BODY:
                            publish_state["body"] += payload
                            if len(publish_state["body"]) >= (publish_state["size"] or 0):
                                broker._route(publish_state)
                except (ConnectionResetError, BrokenPipeError, OSError):
                    return

            @staticmethod
            def _method(sock, ch, cls, mth, args):
                payload = struct.pack(">HH", cls, mth) + args
                FakeRabbit._send_frame(sock, FRAME_METHOD, ch, payload)

        class Server(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self.server = Server(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    def _route(self, st):
        rk, body = st.get("rk", ""), st.get("body", b"")
        with self.lock:
            self.published.append((rk, body))
            subs = list(self.consumers.get(rk, []))
            if not subs:
                self.backlog.setdefault(rk, []).append(body)
        for d in subs:
            try:
                d(body, rk)
            except OSError:
                pass
        st["body"] = b""
        st["size"] = None

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

    @classmethod
    def _read_frame(cls, sock):
        head = cls._recv_exact(sock, 7)
        if head is None:
            return None
        ftype, ch, size = struct.unpack(">BHI", head)
        payload = cls._recv_exact(sock, size)
        end = cls._recv_exact(sock, 1)
        if payload is None or end is None or end[0] != FRAME_END:
            return None
        return ftype, ch, payload

    @staticmethod
    def _send_frame(sock, ftype, ch, payload):
        sock.sendall(struct.pack(">BHI", ftype, ch, len(payload)) + payload
                     + bytes([FRAME_END]))

    @property
    def port(self) -> int:
        return self.server.server_address[1]

    def start(self) -> "FakeRabbit":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
