"""pw.io.qdrant (reference io/qdrant) — API-parity surface.

Requires the qdrant_client client library (offline image: raises at call time).
"""
from __future__ import annotations

from typing import Any

from pathway_amd.io._utils import require_client


def write(table, *args: Any, name: str | None = None, **kwargs: Any):
    require_client("qdrant", "qdrant_client")
    raise NotImplementedError("pw.io.qdrant.write: client library loaded but offline transport is unavailable in this environment")
