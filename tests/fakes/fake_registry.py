"""In-process Confluent schema-registry fake (HTTP)."""

from __future__ import annotations

import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class FakeSchemaRegistry:
    def __init__(self):
        self.schemas: dict[int, str] = {}
        self.subjects: dict[str, list[int]] = {}
        self._next_id = 1
        reg = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _json(self, code: int, obj):
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/vnd.schemaregistry.v1+json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                parts = self.path.strip("/").split("/")
                if parts[0] == "schemas" and parts[1] == "ids":
                    sid = int(parts[2])
                    if sid not in reg.schemas:
                        return self._json(404, {"error_code": 40403})
                    return self._json(200, {"schema": reg.schemas[sid]})
                if parts[0] == "subjects" and len(parts) == 4 and parts[2] == "versions":
                    subj = parts[1]
                    versions = reg.subjects.get(subj)
                    if not versions:
                        return self._json(404, {"error_code": 40401})
                    v = len(versions) if parts[3] == "latest" else int(parts[3])
                    sid = versions[v - 1]
                    return self._json(
                        200,
                        {"subject": subj, "version": v, "id": sid,
                         "schema": reg.schemas[sid]},
                    )
                return self._json(404, {"error_code": 404})

            def do_POST(self):
                n = int(self.headers.get("Content-Length", 0))
                body = json.loads(self.rfile.read(n))
                parts = self.path.strip("/").split("/")
                if parts[0] == "subjects" and parts[-1] == "versions":
                    subj = parts[1]
                    schema = body["schema"]
                    for sid, s in reg.schemas.items():
                        if s == schema:
                            if sid not in reg.subjects.setdefault(subj, []):
                                reg.subjects[subj].append(sid)
                            return self._json(200, {"id": sid})
                    sid = reg._next_id
                    reg._next_id += 1
                    reg.schemas[sid] = schema
                    reg.subjects.setdefault(subj, []).append(sid)
                    return self._json(200, {"id": sid})
                return self._json(404, {"error_code": 404})

        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.server.server_address[1]}"

    def start(self) -> "FakeSchemaRegistry":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
