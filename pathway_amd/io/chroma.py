"""pw.io.chroma (reference io/chroma) — API-parity surface.

Requires the chromadb client library (offline image: raises at call time).
"""
from __future__ import annotations

from typing import Any

from pathway_amd.io._utils import require_client


def write(table, *args: Any, name: str | None = None, **kwargs: Any):
    require_client("chroma", "chromadb")
    raise NotImplementedError("pw.io.chroma.write: client library loaded but offline transport is unavailable in this environment")
