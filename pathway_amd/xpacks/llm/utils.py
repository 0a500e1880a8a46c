"""llm xpack helpers (reference xpacks/llm/utils.py behavior)."""
from __future__ import annotations

from typing import Any


def combine_metadatas(*metas: Any) -> dict:
    """Merge metadata dicts left-to-right (later wins)."""
    out: dict = {}
    for m in metas:
        if m:
            out.update(dict(m))
    return out


def _coerce_sync(fn):
    return fn
