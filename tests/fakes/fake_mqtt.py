"""In-process fake MQTT 3.1.1 broker (QoS 0 subset)."""

from __future__ import annotations

import socketserver
import struct
import threading

CONNECT, CONNACK, PUBLISH, SUBSCRIBE, SUBACK = 1, 2, 3, 8, 9
PINGREQ, PINGRESP, DISCONNECT = 12, 13, 14


def _encode_remaining(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n % 128
        n //= 128
        out.append(b | 0x80 if n else b)
        if not n:
            return bytes(out)


class FakeMqtt:
    def __init__(self):
        self.lock = threading.Lock()
        #: topic filter -> list of (socket, lock)
        self.subs: dict[str, list[tuple]] = {}
        self.published: list[tuple[str, bytes]] = []
        broker = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                sock = self.request
                try:
                    while True:
                        pkt = self._recv_packet(sock)
                        if pkt is None:
                            return
                        ptype, flags, data = pkt
                        if ptype == CONNECT:
                            self._send(sock, CONNACK, 0, b"\x00\x00")
                        elif ptype == SUBSCRIBE:
                            (pid,) = struct.unpack_from(">H", data, 0)
                            i = 2
                            codes = b""
                            while i < len(data):
                                (tlen,) = struct.unpack_from(">H", data, i)
                                topic = data[i + 2 : i + 2 + tlen].decode()
                                i += 2 + tlen + 1
                                with broker.lock:
                                    broker.subs.setdefault(topic, []).append(
                                        (sock, threading.Lock())
                                    )
                                codes += b"\x00"
                            self._send(sock, SUBACK, 0,
                                       struct.pack(">H", pid) + codes)
                        elif ptype == PUBLISH:
                            (tlen,) = struct.unpack_from(">H", data, 0)
                            topic = data[2 : 2 + tlen].decode()
                            payload = data[2 + tlen :]
                            with broker.lock:
                                broker.published.append((topic, payload))
                                targets = list(broker.subs.get(topic, []))
                            tb = topic.encode()
                            body = struct.pack(">H", len(tb)) + tb + payload
                            for s, slock in targets:
                                try:
                                    with slock:
                                        s.sendall(
                                            bytes([PUBLISH << 4])
                                            + _encode_remaining(len(body)) + body
                                        )
                                except OSError:
                                    pass
                        elif ptype == PINGREQ:
                            self._send(sock, PINGRESP, 0, b"")
                        elif ptype == DISCONNECT:
                            return
                except (ConnectionResetError, BrokenPipeError, OSError):
                    return
                finally:
                    with broker.lock:
                        for t in broker.subs:
                            broker.subs[t] = [
                                e for e in broker.subs[t] if e[0] is not sock
                            ]

            @staticmethod
            def _send(sock, ptype, flags, body):
                sock.sendall(bytes([(ptype << 4) | flags])
                             + _encode_remaining(len(body)) + body)

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

            def _recv_packet(self, sock):
                head = self._recv_exact(sock, 1)
                if head is None:
                    return None
                mult, rem = 1, 0
                while True:
                    b = self._recv_exact(sock, 1)
                    if b is None:
                        return None
                    rem += (b[0] & 0x7F) * mult
                    if not b[0] & 0x80:
                        break
                    mult *= 128
                data = self._recv_exact(sock, rem) if rem else b""
                if data is None:
                    return None
                return head[0] >> 4, head[0] & 0x0F, data

        class Server(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self.server = Server(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    @property
    def uri(self) -> str:
        return f"mqtt://127.0.0.1:{self.server.server_address[1]}"

    def start(self) -> "FakeMqtt":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
