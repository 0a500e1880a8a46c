"""Input synchronization groups
(reference src/connectors/synchronization.rs:142,277 — cross-connector
watermark alignment by column with max_difference).

Sources registered in one group release only rows whose sync-column value
is within max_difference of the slowest member's watermark; the rest stay
buffered inside the source until the laggard catches up.
"""

from __future__ import annotations

import threading
from typing import Any


class InputSynchronizationGroup:
    def __init__(self, max_difference: Any, name: str | None = None):
        self.max_difference = max_difference
        self.name = name
        self.lock = threading.Lock()
        self.watermarks: dict[int, float] = {}  # member id -> max value seen

    def register(self, member_id: int) -> None:
        with self.lock:
            self.watermarks.setdefault(member_id, float("-inf"))

    def observe(self, member_id: int, value: float) -> None:
        with self.lock:
            cur = self.watermarks.get(member_id, float("-inf"))
            if value > cur:
                self.watermarks[member_id] = value

    def release_threshold(self) -> float:
        """Rows with sync value ≤ min watermark + max_difference may pass."""
        with self.lock:
            if not self.watermarks:
                return float("inf")
            slowest = min(self.watermarks.values())
        md = self.max_difference
        try:
            md = float(md)
        except (TypeError, ValueError):
            import pandas as pd

            md = float(pd.Timedelta(md).value)
        return slowest + md


def register_input_synchronization_group(
    *columns: Any, max_difference: Any, name: str | None = None
) -> InputSynchronizationGroup:
    """pw.io.register_input_synchronization_group(t1.t, t2.t, max_difference=...).

    Each column must belong to a table read from a streaming source; the
    group throttles those sources so their sync columns stay within
    max_difference of each other."""
    from pathway_amd.engine.streaming import StreamingSource
    from pathway_amd.engine.nodes import InputNode

    group = InputSynchronizationGroup(max_difference, name)
    for col in columns:
        table = col.table
        node = table._node
        if not isinstance(node, InputNode) or not isinstance(
            node.source, StreamingSource
        ):
            raise ValueError(
                "synchronization groups require streaming-source tables"
            )
        src = node.source
        member = id(src)
        group.register(member)
        idx = src.column_names.index(col.name)
        src.attach_sync_group(group, member, idx)
    return group
