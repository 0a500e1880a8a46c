"""In-process fake S3 endpoint (path-style HTTP, ListObjectsV2 XML)."""

from __future__ import annotations

import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from xml.sax.saxutils import escape


class FakeS3:
    def __init__(self):
        #: (bucket, key) -> bytes
        self.objects: dict[tuple[str, str], bytes] = {}
        self.lock = threading.Lock()
        store = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, *a):
                pass

            def _split(self):
                parsed = urllib.parse.urlparse(self.path)
                parts = urllib.parse.unquote(parsed.path).lstrip("/").split("/", 1)
                bucket = parts[0]
                key = parts[1] if len(parts) > 1 else ""
                query = dict(urllib.parse.parse_qsl(parsed.query))
                return bucket, key, query

            def _reply(self, code: int, body: bytes = b"",
                       content_type: str = "application/octet-stream"):
                self.send_response(code)
                self.send_header("Content-Type", content_type)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                if self.command != "HEAD":
                    self.wfile.write(body)

            def do_PUT(self):
                bucket, key, _ = self._split()
                n = int(self.headers.get("Content-Length", 0))
                data = self.rfile.read(n)
                src = self.headers.get("x-amz-copy-source")
                with store.lock:
                    if src:
                        sb, _, sk = src.lstrip("/").partition("/")
                        if (sb, sk) not in store.objects:
                            return self._reply(404, b"<Error/>", "application/xml")
                        store.objects[(bucket, key)] = store.objects[(sb, sk)]
                    else:
                        store.objects[(bucket, key)] = data
                self._reply(200)

            def do_GET(self):
                bucket, key, query = self._split()
                if not key:  # list
                    prefix = query.get("prefix", "")
                    with store.lock:
                        keys = sorted(
                            k for (b, k) in store.objects
                            if b == bucket and k.startswith(prefix)
                        )
                    contents = "".join(
                        f"<Contents><Key>{escape(k)}</Key>"
                        f"<Size>{len(store.objects[(bucket, k)])}</Size>"
                        f"<ETag>\"{hash(store.objects[(bucket, k)]) & 0xffffffff:x}\"</ETag>"
                        f"<LastModified>2024-01-01T00:00:00Z</LastModified>"
                        f"</Contents>"
                        for k in keys
                    )
                    xml = (
                        "<?xml version=\"1.0\"?><ListBucketResult>"
                        f"<Name>{escape(bucket)}</Name><KeyCount>{len(keys)}</KeyCount>"
                        f"{contents}</ListBucketResult>"
                    )
                    return self._reply(200, xml.encode(), "application/xml")
                with store.lock:
                    data = store.objects.get((bucket, key))
                if data is None:
                    return self._reply(404, b"<Error><Code>NoSuchKey</Code></Error>",
                                       "application/xml")
                self._reply(200, data)

            def do_HEAD(self):
                bucket, key, _ = self._split()
                with store.lock:
                    data = store.objects.get((bucket, key))
                if data is None:
                    return self._reply(404)
                self._reply(200, data)

            def do_DELETE(self):
                bucket, key, _ = self._split()
                with store.lock:
                    store.objects.pop((bucket, key), None)
                self._reply(204)

        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    @property
    def endpoint(self) -> str:
        return f"http://127.0.0.1:{self.server.server_address[1]}"

    def start(self) -> "FakeS3":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
