"""Stateful ops (reference stdlib/stateful): deduplicate."""
from __future__ import annotations

from typing import Any


def deduplicate(
    table,
    *,
    value: Any,
    instance: Any = None,
    acceptor: Any = None,
    persistent_id: str | None = None,
    name: str | None = None,
):
    return table.deduplicate(
        value=value, instance=instance, acceptor=acceptor, name=name
    )


__all__ = ["deduplicate"]
