"""pw.io.milvus — Milvus sink over the v2 vectordb REST API
(reference src/connectors/data_storage/milvus.rs)."""

from __future__ import annotations

from typing import Any

from pathway_amd.io import _rest
from pathway_amd.io._vector_sink import make_vector_writer


def write(
    table,
    url: str,
    collection_name: str,
    *,
    vector_column: str = "vector",
    token: str | None = None,
    db_name: str = "default",
    name: str | None = None,
    **kwargs: Any,
):
    base = url.rstrip("/")
    headers = {"Authorization": f"Bearer {token}"} if token else {}

    def upsert(points):
        _rest.request(
            "POST", f"{base}/v2/vectordb/entities/upsert",
            body={"collectionName": collection_name, "dbName": db_name,
                  "data": [
                      {"id": p["id"], "vector": p["vector"], **p["metadata"]}
                      for p in points
                  ]},
            headers=headers,
        )

    def delete(ids):
        idlist = ", ".join(f'"{i}"' for i in ids)
        _rest.request(
            "POST", f"{base}/v2/vectordb/entities/delete",
            body={"collectionName": collection_name, "dbName": db_name,
                  "filter": f"id in [{idlist}]"},
            headers=headers,
        )

    return make_vector_writer(table, vector_column, upsert=upsert, delete=delete)
