"""Generic in-process capturing HTTP server for REST-API connector tests.

Records every request (method, path, headers, body) and replies with a
configurable JSON body — the service-specific connectors (elasticsearch,
clickhouse, bigquery, slack, dynamodb, pubsub, vector sinks ...) are
tested against the real HTTP request shapes they emit.
"""

from __future__ import annotations

import json
import threading
from dataclasses import dataclass, field
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


@dataclass
class CapturedRequest:
    method: str
    path: str
    headers: dict
    body: bytes

    def json(self):
        return json.loads(self.body)

    def ndjson(self):
        return [json.loads(l) for l in self.body.decode().splitlines() if l.strip()]


@dataclass
class FakeHTTPService:
    #: path-prefix -> (status, json-able reply); "" matches everything
    replies: dict = field(default_factory=dict)
    requests: list = field(default_factory=list)

    def __post_init__(self):
        svc = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, *a):
                pass

            def _handle(self):
                n = int(self.headers.get("Content-Length", 0))
                body = self.rfile.read(n) if n else b""
                req = CapturedRequest(self.command, self.path,
                                      dict(self.headers), body)
                svc.requests.append(req)
                status, reply = 200, {"ok": True}
                for prefix, (st, rep) in svc.replies.items():
                    if self.path.startswith(prefix):
                        status, reply = st, rep
                        break
                out = json.dumps(reply() if callable(reply) else reply).encode()
                self.send_response(status)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(out)))
                self.end_headers()
                self.wfile.write(out)

            do_GET = do_POST = do_PUT = do_DELETE = do_PATCH = _handle

        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.server.server_address[1]}"

    def start(self) -> "FakeHTTPService":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
