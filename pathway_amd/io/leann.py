"""pw.io.leann (reference io/leann) — API-parity surface.

Requires the leann client library (offline image: raises at call time).
"""
from __future__ import annotations

from typing import Any

from pathway_amd.io._utils import require_client


def write(table, *args: Any, name: str | None = None, **kwargs: Any):
    require_client("leann", "leann")
    raise NotImplementedError("pw.io.leann.write: client library loaded but offline transport is unavailable in this environment")
