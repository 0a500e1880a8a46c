"""VectorStoreServer/Client (reference xpacks/llm/vector_store.py:31,356)."""
from __future__ import annotations

from typing import Any, Callable

from pathway_amd.internals.schema import Schema
from pathway_amd.xpacks.llm.document_store import DocumentStore


class VectorStoreServer:
    """DocumentStore + REST endpoints (server lands with the REST phase;
    the query tables work in-engine already)."""

    def __init__(
        self,
        *docs,
        embedder: Callable | None = None,
        parser: Callable | None = None,
        splitter: Callable | None = None,
        doc_post_processors=None,
        index_factory=None,
    ):
        from pathway_amd.stdlib.indexing.nearest_neighbors import BruteForceKnnFactory

        if index_factory is None:
            index_factory = BruteForceKnnFactory(embedder=embedder)
        elif embedder is not None and getattr(index_factory, "embedder", None) is None:
            index_factory.embedder = embedder
        self.document_store = DocumentStore(
            list(docs),
            retriever_factory=index_factory,
            parser=parser,
            splitter=splitter,
            doc_post_processors=doc_post_processors,
        )

    def retrieve_query(self, queries):
        return self.document_store.retrieve_query(queries)

    def statistics_query(self, queries):
        return self.document_store.statistics_query(queries)

    def inputs_query(self, queries):
        return self.document_store.inputs_query(queries)

    def run_server(self, host: str = "127.0.0.1", port: int = 8000, threaded: bool = False, with_cache: bool = True, **kwargs):
        from pathway_amd.xpacks.llm.servers import DocumentStoreServer

        srv = DocumentStoreServer(host, port, self.document_store)
        return srv.run(threaded=threaded, **kwargs)


class VectorStoreClient:
    def __init__(self, host: str = "127.0.0.1", port: int = 8000, url: str | None = None, timeout: int = 15, additional_headers: dict | None = None):
        self.url = url or f"http://{host}:{port}"
        self.timeout = timeout
        self.headers = additional_headers or {}

    def query(self, query: str, k: int = 3, metadata_filter: str | None = None, filepath_globpattern: str | None = None):
        import json
        import urllib.request

        data = json.dumps(
            {
                "query": query,
                "k": k,
                "metadata_filter": metadata_filter,
                "filepath_globpattern": filepath_globpattern,
            }
        ).encode()
        req = urllib.request.Request(
            self.url + "/v1/retrieve",
            data=data,
            headers={"Content-Type": "application/json", **self.headers},
        )
        with urllib.request.urlopen(req, timeout=self.timeout) as resp:
            return json.loads(resp.read())

    __call__ = query

    def get_vectorstore_statistics(self):
        import json
        import urllib.request

        req = urllib.request.Request(
            self.url + "/v1/statistics",
            data=b"{}",
            headers={"Content-Type": "application/json", **self.headers},
        )
        with urllib.request.urlopen(req, timeout=self.timeout) as resp:
            return json.loads(resp.read())

    def get_input_files(self, metadata_filter=None, filepath_globpattern=None):
        import json
        import urllib.request

        data = json.dumps(
            {
                "metadata_filter": metadata_filter,
                "filepath_globpattern": filepath_globpattern,
            }
        ).encode()
        req = urllib.request.Request(
            self.url + "/v1/inputs",
            data=data,
            headers={"Content-Type": "application/json", **self.headers},
        )
        with urllib.request.urlopen(req, timeout=self.timeout) as resp:
            return json.loads(resp.read())


class SlidesVectorStoreServer(VectorStoreServer):
    """VectorStoreServer preset for slide decks (reference
    vector_store.py SlidesVectorStoreServer): same serving surface with
    slide-parser defaults."""
