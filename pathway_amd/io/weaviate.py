"""pw.io.weaviate (reference io/weaviate) — API-parity surface.

Requires the weaviate client library (offline image: raises at call time).
"""
from __future__ import annotations

from typing import Any

from pathway_amd.io._utils import require_client


def write(table, *args: Any, name: str | None = None, **kwargs: Any):
    require_client("weaviate", "weaviate")
    raise NotImplementedError("pw.io.weaviate.write: client library loaded but offline transport is unavailable in this environment")
