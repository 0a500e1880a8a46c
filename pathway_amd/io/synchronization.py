"""Input synchronization groups
(reference src/connectors/synchronization.rs:142,277 — cross-connector
watermark alignment by column with max_difference).

Sources registered in one group release only rows whose sync-column value
is within max_difference of the slowest member's watermark; the rest stay
buffered inside the source until the laggard catches up.
"""

from __future__ import annotations

import dataclasses
import threading
import time as _time
from typing import Any


@dataclasses.dataclass(frozen=True)
class SynchronizedColumn:
    """Column spec within a synchronization group (reference
    io/_synchronization.py:20): `priority` gates admission against
    higher-priority members' watermarks; `idle_duration` (seconds or
    timedelta) temporarily excludes a silent source from the group's
    minimum so it cannot stall the others."""

    column: Any
    priority: int = 0
    idle_duration: Any = None


class InputSynchronizationGroup:
    def __init__(self, max_difference: Any, name: str | None = None):
        self.max_difference = max_difference
        self.name = name
        self.lock = threading.Lock()
        self.watermarks: dict[int, float] = {}  # member id -> max value seen
        self.priorities: dict[int, int] = {}
        self.idle_after: dict[int, float | None] = {}  # seconds or None
        self.last_seen: dict[int, float] = {}  # wall time of last observe

    def register(
        self, member_id: int, priority: int = 0,
        idle_duration: float | None = None,
    ) -> None:
        with self.lock:
            self.watermarks.setdefault(member_id, float("-inf"))
            self.priorities[member_id] = priority
            self.idle_after[member_id] = idle_duration
            self.last_seen[member_id] = _time.monotonic()

    def observe(self, member_id: int, value: float) -> None:
        with self.lock:
            self.last_seen[member_id] = _time.monotonic()
            cur = self.watermarks.get(member_id, float("-inf"))
            if value > cur:
                self.watermarks[member_id] = value

    def _active_members(self) -> list[int]:
        now = _time.monotonic()
        out = []
        for m, wm in self.watermarks.items():
            idle = self.idle_after.get(m)
            if (
                idle is not None
                and wm != float("inf")
                and now - self.last_seen.get(m, now) > idle
            ):
                continue  # idle source: excluded until it produces again
            out.append(m)
        return out or list(self.watermarks)

    def admits(self, member_id: int, value: float) -> bool:
        """Priority gate: a value passes only if it does not exceed the
        max watermark of all strictly-higher-priority active members
        (reference SynchronizedColumn priority semantics)."""
        with self.lock:
            mine = self.priorities.get(member_id, 0)
            higher = [
                self.watermarks[m]
                for m in self._active_members()
                if self.priorities.get(m, 0) > mine and m != member_id
            ]
        if not higher:
            return True
        return value <= max(higher)

    def release_threshold(self) -> float:
        """Rows with sync value ≤ min watermark + max_difference may pass."""
        with self.lock:
            active = self._active_members()
            if not active:
                return float("inf")
            slowest = min(self.watermarks[m] for m in active)
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
        priority = 0
        idle = None
        if isinstance(col, SynchronizedColumn):
            priority = col.priority
            idle = col.idle_duration
            if idle is not None and not isinstance(idle, (int, float)):
                idle = float(idle.total_seconds())
            col = col.column
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
        group.register(member, priority=priority, idle_duration=idle)
        idx = src.column_names.index(col.name)
        src.attach_sync_group(group, member, idx)
    return group
