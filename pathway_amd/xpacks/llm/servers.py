"""REST servers (reference xpacks/llm/servers.py:16-270 + io/http/_server.py).

QARestServer / DocumentStoreServer: HTTP requests become query-table rows
pushed into the running engine (one timestamp per request batch); answers
are read back from the capture of the result table — the synchronous-
engine analog of the reference's RestServerSubject + asyncio response
plumbing (_server.py:445-470).
"""
from __future__ import annotations

import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Any

from pathway_amd.internals import dtype as dt


class _QueryPort:
    """PushSource + result capture for one endpoint."""

    def __init__(self, schema, build_result):
        from pathway_amd.debug import table_from_rows
        from pathway_amd.engine.runtime import CaptureNode, PushSource
        from pathway_amd.engine.nodes import InputNode
        from pathway_amd.internals.config import get_device
        from pathway_amd.internals.table import Table
        from pathway_amd.internals.universe import Universe

        names = schema.column_names()
        dtypes = [schema.__columns__[n].dtype for n in names]
        self.schema = schema
        self.source = PushSource(names, dtypes)
        node = InputNode(self.source, get_device())
        self.query_table = Table(
            node, {n: d for n, d in zip(names, dtypes)}, Universe()
        )
        self.result_table = build_result(self.query_table)
        self.capture = CaptureNode(self.result_table._node, get_device())

    def push(self, payload: dict, time: int):
        from pathway_amd.internals.api import Pointer, hash_values

        names = self.schema.column_names()
        values = []
        for n in names:
            v = payload.get(n)
            d = dt.unoptionalize(self.schema.__columns__[n].dtype)
            if v is not None and d == dt.INT:
                v = int(v)
            values.append(v)
        lo, hi = hash_values([time, json.dumps(payload, sort_keys=True, default=str)])
        key = Pointer(lo, hi)
        self.source.push(key, values, time)
        return key

    def result_for(self, key):
        for row in reversed(self.capture.rows):
            if row.key == key and row.diff > 0:
                return row.values
        return None


class BaseRestServer:
    def __init__(self, host: str, port: int):
        self.host = host
        self.port = port
        self.ports: dict[str, _QueryPort] = {}
        self._time = [2]
        self._lock = threading.Lock()
        self._rt = None

    def add_endpoint(self, route: str, schema, build_result):
        self.ports[route] = _QueryPort(schema, build_result)

    def _ensure_runtime(self):
        if self._rt is None:
            from pathway_amd.engine.runtime import Runtime
            from pathway_amd.internals.config import get_device
            from pathway_amd.internals.rungraph import G, reset_all

            caps = [p.capture for p in self.ports.values()]
            self._rt = Runtime(list(G.sinks) + caps, device=get_device(), comm=G.comm)
            reset_all(self._rt.nodes)
            self._rt.run()  # ingest static/streamed docs first
        return self._rt

    def handle(self, route: str, payload: dict) -> Any:
        port = self.ports.get(route)
        if port is None:
            return {"error": f"unknown route {route}"}, 404
        with self._lock:
            rt = self._ensure_runtime()
            t = self._time[0]
            self._time[0] += 2
            key = port.push(payload or {}, t)
            rt.run()  # drain source times (incremental steps)
            vals = port.result_for(key)
        if vals is None:
            return {"error": "no result"}, 500
        out = vals[0]
        if hasattr(out, "value"):
            out = out.value
        return out, 200

    def run(self, threaded: bool = False, with_cache: bool = True, terminate_on_error: bool = False, **kwargs):
        server = self

        class Handler(BaseHTTPRequestHandler):
            def do_POST(self):
                ln = int(self.headers.get("Content-Length", "0") or 0)
                body = self.rfile.read(ln) if ln else b"{}"
                try:
                    payload = json.loads(body or b"{}")
                except Exception:
                    payload = {}
                out, code = server.handle(self.path, payload)
                data = json.dumps(out, default=str).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(data)))
                self.end_headers()
                self.wfile.write(data)

            def log_message(self, *a):
                pass

        httpd = ThreadingHTTPServer((self.host, self.port), Handler)
        self._httpd = httpd
        if threaded:
            th = threading.Thread(target=httpd.serve_forever, daemon=True)
            th.start()
            return th
        httpd.serve_forever()

    def shutdown(self):
        if getattr(self, "_httpd", None):
            self._httpd.shutdown()


class DocumentStoreServer(BaseRestServer):
    def __init__(self, host: str, port: int, document_store, **kwargs):
        super().__init__(host, port)
        from pathway_amd.xpacks.llm.document_store import DocumentStore

        self.add_endpoint(
            "/v1/retrieve", DocumentStore.RetrieveQuerySchema, document_store.retrieve_query
        )
        self.add_endpoint(
            "/v1/statistics", DocumentStore.StatisticsQuerySchema, document_store.statistics_query
        )
        self.add_endpoint(
            "/v1/inputs", DocumentStore.InputsQuerySchema, document_store.inputs_query
        )


class QARestServer(BaseRestServer):
    def __init__(self, host: str, port: int, rag_question_answerer, **kwargs):
        super().__init__(host, port)
        qa = rag_question_answerer
        self.add_endpoint(
            "/v1/pw_ai_answer", type(qa).AnswerQuerySchema, qa.answer_query
        )
        self.add_endpoint(
            "/v2/answer", type(qa).AnswerQuerySchema, qa.answer_query
        )
        store = qa._store()
        from pathway_amd.xpacks.llm.document_store import DocumentStore

        self.add_endpoint(
            "/v1/retrieve", DocumentStore.RetrieveQuerySchema, store.retrieve_query
        )
        self.add_endpoint(
            "/v2/list_documents", DocumentStore.InputsQuerySchema, store.inputs_query
        )
        self.add_endpoint(
            "/v1/statistics", DocumentStore.StatisticsQuerySchema, store.statistics_query
        )


class QASummaryRestServer(QARestServer):
    def __init__(self, host, port, qa, **kwargs):
        super().__init__(host, port, qa, **kwargs)
        self.add_endpoint(
            "/v1/pw_ai_summary", type(qa).SummarizeQuerySchema, qa.summarize_query
        )
