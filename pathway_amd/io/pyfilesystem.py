"""pw.io.pyfilesystem (reference io/pyfilesystem) — API-parity surface.

Requires the fs client library (offline image: raises at call time).
"""
from __future__ import annotations

from typing import Any

from pathway_amd.io._utils import require_client


def read(*args: Any, schema=None, mode: str = "streaming", name: str | None = None, autocommit_duration_ms: int | None = 1500, **kwargs: Any):
    require_client("pyfilesystem", "fs")
    raise NotImplementedError("pw.io.pyfilesystem.read: client library loaded but offline transport is unavailable in this environment")
