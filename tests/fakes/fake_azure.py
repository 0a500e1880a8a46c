"""In-process fake Azure Blob endpoint (List Blobs XML + blob CRUD)."""

from __future__ import annotations

import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from xml.sax.saxutils import escape


class FakeAzureBlob:
    def __init__(self):
        #: (container, name) -> bytes
        self.blobs: dict[tuple[str, str], bytes] = {}
        self.lock = threading.Lock()
        store = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, *a):
                pass

            def _split(self):
                p = urllib.parse.urlparse(self.path)
                parts = urllib.parse.unquote(p.path).lstrip("/").split("/", 1)
                return parts[0], (parts[1] if len(parts) > 1 else ""), dict(
                    urllib.parse.parse_qsl(p.query)
                )

            def _reply(self, code, body=b"", ctype="application/octet-stream"):
                self.send_response(code)
                self.send_header("Content-Type", ctype)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_PUT(self):
                container, name, _ = self._split()
                n = int(self.headers.get("Content-Length", 0))
                data = self.rfile.read(n)
                with store.lock:
                    store.blobs[(container, name)] = data
                self._reply(201)

            def do_GET(self):
                container, name, q = self._split()
                if q.get("comp") == "list":
                    prefix = q.get("prefix", "")
                    with store.lock:
                        items = sorted(
                            n for (c, n) in store.blobs
                            if c == container and n.startswith(prefix)
                        )
                    blobs = "".join(
                        f"<Blob><Name>{escape(n)}</Name><Properties>"
                        f"<Etag>{hash(store.blobs[(container, n)]) & 0xffffffff:x}</Etag>"
                        f"<Content-Length>{len(store.blobs[(container, n)])}</Content-Length>"
                        f"</Properties></Blob>"
                        for n in items
                    )
                    xml = (f"<?xml version=\"1.0\"?><EnumerationResults>"
                           f"<Blobs>{blobs}</Blobs></EnumerationResults>")
                    return self._reply(200, xml.encode(), "application/xml")
                with store.lock:
                    data = store.blobs.get((container, name))
                if data is None:
                    return self._reply(404)
                self._reply(200, data)

            def do_DELETE(self):
                container, name, _ = self._split()
                with store.lock:
                    store.blobs.pop((container, name), None)
                self._reply(202)

        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.server.server_address[1]}"

    def start(self) -> "FakeAzureBlob":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
