"""pw.io.weaviate — Weaviate sink over the v1 batch/objects REST API
(reference src/connectors/data_storage/weaviate.rs)."""

from __future__ import annotations

import hashlib
from typing import Any

from pathway_amd.io import _rest
from pathway_amd.io._vector_sink import make_vector_writer


def _uuid_of(rid: str) -> str:
    h = hashlib.md5(rid.encode()).hexdigest()
    return f"{h[:8]}-{h[8:12]}-{h[12:16]}-{h[16:20]}-{h[20:32]}"


def write(
    table,
    url: str,
    class_name: str,
    *,
    vector_column: str = "vector",
    api_key: str | None = None,
    name: str | None = None,
    **kwargs: Any,
):
    base = url.rstrip("/")
    headers = {"Authorization": f"Bearer {api_key}"} if api_key else {}

    def upsert(points):
        _rest.request(
            "POST", f"{base}/v1/batch/objects",
            body={"objects": [
                {"class": class_name, "id": _uuid_of(p["id"]),
                 "vector": p["vector"],
                 "properties": {**p["metadata"], "_pw_id": p["id"]}}
                for p in points
            ]},
            headers=headers,
        )

    def delete(ids):
        for i in ids:
            _rest.request(
                "DELETE", f"{base}/v1/objects/{class_name}/{_uuid_of(i)}",
                headers=headers,
            )

    return make_vector_writer(table, vector_column, upsert=upsert, delete=delete)
