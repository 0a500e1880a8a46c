"""pw.io.pinecone — Pinecone sink over the vectors REST API
(reference src/connectors/data_storage/pinecone.rs, 746 LoC)."""

from __future__ import annotations

from typing import Any

from pathway_amd.io import _rest
from pathway_amd.io._vector_sink import make_vector_writer


def write(
    table,
    index_host: str,
    api_key: str | None = None,
    *,
    namespace: str = "",
    vector_column: str = "vector",
    name: str | None = None,
    **kwargs: Any,
):
    base = index_host.rstrip("/")
    if "://" not in base:
        base = f"https://{base}"
    headers = {"Api-Key": api_key} if api_key else {}

    def upsert(points):
        _rest.request(
            "POST", f"{base}/vectors/upsert",
            body={"vectors": [
                {"id": p["id"], "values": p["vector"], "metadata": p["metadata"]}
                for p in points
            ], "namespace": namespace},
            headers=headers,
        )

    def delete(ids):
        _rest.request(
            "POST", f"{base}/vectors/delete",
            body={"ids": ids, "namespace": namespace},
            headers=headers,
        )

    return make_vector_writer(table, vector_column, upsert=upsert, delete=delete)
