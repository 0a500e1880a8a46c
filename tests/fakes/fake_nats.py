"""In-process fake NATS server (text protocol subset)."""

from __future__ import annotations

import json
import socketserver
import threading


class FakeNats:
    def __init__(self):
        self.lock = threading.Lock()
        #: subject -> list of (handler socket, sid)
        self.subs: dict[str, list[tuple]] = {}
        self.published: list[tuple[str, bytes]] = []
        server_self = self

        class Handler(socketserver.StreamRequestHandler):
            # buffered writer: raw SocketIO.write is a single send() that
            # may write PARTIALLY under load, desyncing the subscriber's
            # frame stream (observed as a lost MSG); BufferedWriter.flush
            # loops until the frame is fully on the wire
            wbufsize = -1

            def handle(self):
                self.wfile.write(
                    b"INFO " + json.dumps({"server_id": "fake", "version": "2"}).encode() + b"\r\n"
                )
                self.wfile.flush()
                my_subs = []
                try:
                    while True:
                        line = self.rfile.readline()
                        if not line:
                            break
                        line = line.rstrip(b"\r\n")
                        if line.upper().startswith(b"CONNECT"):
                            # +OK only in verbose mode (NATS protocol).
                            # An unconditional +OK sits unread in a
                            # publish-only client's receive buffer, so
                            # its close() turns into a TCP RST that can
                            # DESTROY still-buffered PUB frames server-
                            # side (observed: second publish lost).
                            try:
                                opts = json.loads(line[7:].strip() or b"{}")
                            except ValueError:
                                opts = {}
                            if opts.get("verbose"):
                                self.wfile.write(b"+OK\r\n")
                                self.wfile.flush()
                        elif line.upper() == b"PING":
                            self.wfile.write(b"PONG\r\n")
                            self.wfile.flush()
                        elif line.upper().startswith(b"SUB "):
                            _, subject, sid = line.decode().split(" ")
                            with server_self.lock:
                                server_self.subs.setdefault(subject, []).append(
                                    (self.wfile, sid, threading.Lock())
                                )
                            my_subs.append(subject)
                        elif line.upper().startswith(b"PUB "):
                            parts = line.decode().split(" ")
                            subject = parts[1]
                            nbytes = int(parts[-1])
                            payload = self.rfile.read(nbytes)
                            self.rfile.read(2)
                            with server_self.lock:
                                server_self.published.append((subject, payload))
                                for wf, sid, wlock in server_self.subs.get(subject, []):
                                    try:
                                        with wlock:
                                            wf.write(
                                                f"MSG {subject} {sid} {nbytes}\r\n".encode()
                                                + payload + b"\r\n"
                                            )
                                            wf.flush()
                                    except (OSError, ValueError):
                                        pass
                except (ConnectionResetError, BrokenPipeError, OSError):
                    pass
                finally:
                    with server_self.lock:
                        for s in my_subs:
                            server_self.subs[s] = [
                                e for e in server_self.subs.get(s, [])
                                if e[0] is not self.wfile
                            ]

        class Server(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self.server = Server(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    @property
    def uri(self) -> str:
        return f"nats://127.0.0.1:{self.server.server_address[1]}"

    def start(self) -> "FakeNats":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
