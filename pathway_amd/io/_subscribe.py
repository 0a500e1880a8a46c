"""pw.io.subscribe (reference io/_subscribe.py:17)."""
from __future__ import annotations

from typing import Any, Callable

from pathway_amd.engine.runtime import SubscribeNode
from pathway_amd.internals.config import get_device
from pathway_amd.internals.rungraph import G

#: on_change(key, row, time, is_addition) — sync and async variants
#: (reference internals/api OnChangeCallback / OnChangeCallbackAsync)
OnChangeCallback = Callable[..., None]
OnChangeCallbackAsync = Callable[..., Any]
#: on_end() — called when the stream finishes
OnFinishCallback = Callable[[], None]


def subscribe(
    table,
    on_change: Callable,
    on_end: Callable | None = None,
    on_time_end: Callable | None = None,
    *,
    skip_persisted_batch: bool = False,
    name: str | None = None,
    sort_by: Any = None,
):
    node = SubscribeNode(
        table._node, get_device(), on_change, on_time_end=on_time_end, on_end=on_end
    )
    G.add_sink(node)
    return node
