"""pw.io.chroma — Chroma sink over the v1 collections REST API
(reference src/connectors/data_storage/chroma.rs, 499 LoC)."""

from __future__ import annotations

from typing import Any

from pathway_amd.io import _rest
from pathway_amd.io._vector_sink import make_vector_writer


def write(
    table,
    url: str,
    collection_id: str,
    *,
    vector_column: str = "vector",
    document_column: str | None = None,
    name: str | None = None,
    **kwargs: Any,
):
    base = url.rstrip("/")
    coll = f"{base}/api/v1/collections/{collection_id}"

    def upsert(points):
        body = {
            "ids": [p["id"] for p in points],
            "embeddings": [p["vector"] for p in points],
            "metadatas": [p["metadata"] for p in points],
        }
        if document_column:
            body["documents"] = [
                p["metadata"].get(document_column) for p in points
            ]
        _rest.request("POST", f"{coll}/upsert", body=body)

    def delete(ids):
        _rest.request("POST", f"{coll}/delete", body={"ids": ids})

    return make_vector_writer(table, vector_column, upsert=upsert, delete=delete)
