"""pw.io.pinecone (reference io/pinecone) — API-parity surface.

Requires the pinecone client library (offline image: raises at call time).
"""
from __future__ import annotations

from typing import Any

from pathway_amd.io._utils import require_client


def write(table, *args: Any, name: str | None = None, **kwargs: Any):
    require_client("pinecone", "pinecone")
    raise NotImplementedError("pw.io.pinecone.write: client library loaded but offline transport is unavailable in this environment")
