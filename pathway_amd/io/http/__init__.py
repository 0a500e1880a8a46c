"""pw.io.http: polling reads, HTTP writes, and the REST connector
(reference io/http + io/http/_server.py PathwayWebserver)."""
from __future__ import annotations

import json
import threading
from typing import Any, Callable

from pathway_amd.internals import dtype as dt


class PathwayWebserver:
    """Shared webserver for rest_connector endpoints (reference
    _server.py:496-875)."""

    def __init__(self, host: str, port: int, with_cors: bool = False, **kwargs):
        self.host = host
        self.port = port
        self.routes: dict[str, dict] = {}
        self._started = False

    def _register(self, route: str, record: dict):
        self.routes[route] = record


def rest_connector(
    host: str | None = None,
    port: int | None = None,
    *,
    webserver: PathwayWebserver | None = None,
    route: str = "/",
    schema=None,
    methods=("POST",),
    autocommit_duration_ms: int | None = 1500,
    keep_queries: bool = False,
    delete_completed_queries: bool = False,
    request_validator: Callable | None = None,
    documentation=None,
):
    """Returns (query_table, response_writer) — reference io/http
    rest_connector contract."""
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.runtime import PushSource
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    if webserver is None:
        webserver = PathwayWebserver(host or "127.0.0.1", port or 8080)
    if schema is None:
        schema = schema_from_types(query=str)
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    src = PushSource(names, dtypes)
    node = InputNode(src, get_device())
    table = Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())
    record = {
        "schema": schema,
        "source": src,
        "capture": None,
        "validator": request_validator,
    }
    webserver._register(route, record)
    G.services.append(webserver)

    def response_writer(result_table):
        from pathway_amd.engine.runtime import CaptureNode

        record["capture"] = CaptureNode(result_table._node, get_device())
        G.add_sink(record["capture"])

    return table, response_writer


def serve_forever(webservers, rt) -> None:
    """Run registered webservers against a live Runtime (pw.run serving
    mode)."""
    from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

    from pathway_amd.internals.api import Pointer, hash_values

    lock = threading.Lock()
    clock = [10_000]

    def handle(ws: PathwayWebserver, route: str, payload: dict):
        rec = ws.routes.get(route)
        if rec is None:
            return {"error": "unknown route"}, 404
        if rec["validator"] is not None:
            err = rec["validator"](payload)
            if err is not None:
                return {"error": str(err)}, 400
        names = rec["schema"].column_names()
        with lock:
            t = clock[0]
            clock[0] += 2
            lo, hi = hash_values([t, json.dumps(payload, sort_keys=True, default=str)])
            key = Pointer(lo, hi)
            rec["source"].push(key, [payload.get(n) for n in names], t)
            rt.run()
            cap = rec["capture"]
            result = None
            if cap is not None:
                for row in reversed(cap.rows):
                    if row.key == key and row.diff > 0:
                        result = row.values
                        break
        if result is None:
            return {"error": "no result"}, 500
        out = result[0] if len(result) == 1 else dict(zip(range(len(result)), result))
        if hasattr(out, "value"):
            out = out.value
        return out, 200

    servers = []
    for ws in webservers:
        def make_handler(ws=ws):
            class Handler(BaseHTTPRequestHandler):
                def do_POST(self):
                    ln = int(self.headers.get("Content-Length", "0") or 0)
                    body = self.rfile.read(ln) if ln else b"{}"
                    try:
                        payload = json.loads(body or b"{}")
                    except Exception:
                        payload = {}
                    out, code = handle(ws, self.path, payload)
                    data = json.dumps(out, default=str).encode()
                    self.send_response(code)
                    self.send_header("Content-Type", "application/json")
                    self.send_header("Content-Length", str(len(data)))
                    self.end_headers()
                    self.wfile.write(data)

                def log_message(self, *a):
                    pass

            return Handler

        httpd = ThreadingHTTPServer((ws.host, ws.port), make_handler())
        th = threading.Thread(target=httpd.serve_forever, daemon=True)
        th.start()
        servers.append(httpd)
        ws._httpd = httpd
    return servers


def read(
    url: str,
    *,
    schema=None,
    format: str = "json",
    mode: str = "streaming",
    refresh_interval_ms: int = 1000,
    name: str | None = None,
    n_polls: int | None = None,
    **kwargs: Any,
):
    """HTTP polling source (reference http reader, 476 LoC)."""
    import time as _time
    import urllib.request

    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    if schema is None:
        schema = schema_from_types(data=bytes)
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    src = StreamingSource(names, dtypes, name=name)

    def reader():
        polls = 0
        try:
            while n_polls is None or polls < n_polls:
                try:
                    with urllib.request.urlopen(url, timeout=10) as resp:
                        payload = resp.read()
                    if format == "json":
                        rec = json.loads(payload)
                        recs = rec if isinstance(rec, list) else [rec]
                        for r in recs:
                            src.emit([r.get(n) for n in names])
                    else:
                        src.emit([payload])
                except Exception:
                    pass
                polls += 1
                if mode == "static":
                    break
                _time.sleep(refresh_interval_ms / 1000)
        finally:
            src.finish()

    spawn_reader(reader, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(table, url: str, *, method: str = "POST", format: str = "json", name: str | None = None, headers: dict | None = None, **kwargs):
    import urllib.request

    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    names = table.column_names()

    def writer(batch):
        for key, values, time, diff in batch.rows():
            rec = dict(zip(names, values))
            rec.update({"time": time, "diff": diff})
            req = urllib.request.Request(
                url,
                data=json.dumps(rec, default=str).encode(),
                method=method,
                headers={"Content-Type": "application/json", **(headers or {})},
            )
            urllib.request.urlopen(req, timeout=10)

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node


class RetryPolicy:
    @classmethod
    def default(cls):
        return cls()
