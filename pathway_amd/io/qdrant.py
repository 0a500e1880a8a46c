"""pw.io.qdrant — Qdrant sink over the points REST API
(reference src/external_integration/qdrant_integration.rs + io sink)."""

from __future__ import annotations

import hashlib
from typing import Any

from pathway_amd.io import _rest
from pathway_amd.io._vector_sink import make_vector_writer


def _point_id(rid: str) -> str:
    """Qdrant point ids must be uuids or uints; derive a uuid from the key."""
    h = hashlib.md5(rid.encode()).hexdigest()
    return f"{h[:8]}-{h[8:12]}-{h[12:16]}-{h[16:20]}-{h[20:32]}"


def write(
    table,
    url: str,
    collection_name: str,
    *,
    vector_column: str = "vector",
    api_key: str | None = None,
    name: str | None = None,
    **kwargs: Any,
):
    base = url.rstrip("/")
    headers = {"api-key": api_key} if api_key else {}

    def upsert(points):
        _rest.request(
            "PUT", f"{base}/collections/{collection_name}/points",
            body={"points": [
                {"id": _point_id(p["id"]),
                 "vector": p["vector"],
                 "payload": {**p["metadata"], "_pw_id": p["id"]}}
                for p in points
            ]},
            headers=headers,
        )

    def delete(ids):
        _rest.request(
            "POST", f"{base}/collections/{collection_name}/points/delete",
            body={"points": [_point_id(i) for i in ids]},
            headers=headers,
        )

    return make_vector_writer(table, vector_column, upsert=upsert, delete=delete)
