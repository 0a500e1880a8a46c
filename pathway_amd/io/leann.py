"""pw.io.leann — LEANN vector sink (reference io/leann): REST upsert of
(id, vector, metadata) documents to a LEANN index server."""

from __future__ import annotations

from typing import Any

from pathway_amd.io import _rest
from pathway_amd.io._vector_sink import make_vector_writer


def write(
    table,
    url: str,
    index_name: str,
    *,
    vector_column: str = "vector",
    name: str | None = None,
    **kwargs: Any,
):
    base = url.rstrip("/")

    def upsert(points):
        _rest.request(
            "POST", f"{base}/indexes/{index_name}/documents",
            body={"documents": points},
        )

    def delete(ids):
        _rest.request(
            "POST", f"{base}/indexes/{index_name}/documents/delete",
            body={"ids": ids},
        )

    return make_vector_writer(table, vector_column, upsert=upsert, delete=delete)
