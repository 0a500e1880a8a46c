"""export_table / import_table — cross-graph table handoff.

Reference: src/engine/graph.rs:616-646 (``ExportedTable`` trait:
failed/frontier/data_from_offset/subscribe) + python_api.rs:9387.  A
table exported from one graph run can be imported into another graph as
an input that replays the exported update stream (times and diffs
preserved) and then follows live appends while the exporting graph is
still producing.

MI355X-native formulation: the export side is a CaptureNode sink
(device batches consolidated, then host DataRows); the import side is a
replay source feeding a new graph — the engine-level analog of the
reference's subscribe/data_from_offset pull loop.
"""

from __future__ import annotations

import threading
from typing import Any

from pathway_amd.internals import dtype as dt


class ExportedTable:
    """Handle to an exported update stream (graph.rs:616-646)."""

    def __init__(self, column_names: list[str], dtypes: dict[str, Any]):
        self.column_names = column_names
        self.dtypes = dtypes
        self._rows: list = []  # DataRow
        self._lock = threading.Lock()
        self._failed = False
        self._frontier: int = 0

    # -- reference ExportedTable surface --

    def failed(self) -> bool:
        return self._failed

    def frontier(self) -> int:
        """Smallest time not yet finalized (reference frontier())."""
        with self._lock:
            return self._frontier

    def data_from_offset(self, offset: int) -> tuple[list, int]:
        """(rows[offset:], next_offset) — the reference's pull API."""
        with self._lock:
            rows = list(self._rows[offset:])
            return rows, offset + len(rows)

    def snapshot_at(self, time: int | None = None) -> dict:
        """Squashed state at `time` (or at the frontier)."""
        from pathway_amd.internals.api import squash_updates

        with self._lock:
            rows = [
                r for r in self._rows
                if time is None or r.time <= time
            ]
        return squash_updates(rows, terminate_on_error=False)

    # -- producer side --

    def _append(self, rows: list) -> None:
        with self._lock:
            self._rows.extend(rows)

    def _advance(self, time: int) -> None:
        with self._lock:
            self._frontier = max(self._frontier, time + 1)


def export_table(table) -> ExportedTable:
    """Register `table` for export; the handle fills as the graph runs
    (reference Scope::export_table)."""
    from pathway_amd.engine.batch import DeltaBatch  # noqa: F401
    from pathway_amd.engine.runtime import CaptureNode
    from pathway_amd.internals.api import DataRow
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    names = table.column_names()
    handle = ExportedTable(names, dict(table._dtypes))

    class _ExportNode(CaptureNode):
        def step(self, time, inputs):
            before = len(self.rows)
            super().step(time, inputs)
            new = self.rows[before:]
            if new:
                handle._append(new)
            handle._advance(time)
            return None

    node = _ExportNode(table._node, get_device(), column_names=names)
    G.add_sink(node)
    return handle


class _ImportSource:
    """Replays an ExportedTable's stream into a new graph."""

    def __init__(self, handle: ExportedTable):
        self.handle = handle
        self.offset = 0
        self._pending: list = []

    def next_time(self):
        from pathway_amd.engine.runtime import STREAM_READY

        rows, self.offset = self.handle.data_from_offset(self.offset)
        self._pending.extend(rows)
        if self._pending:
            return min(r.time for r in self._pending)
        return None  # exporting graph finished filling (snapshot import)

    def pull(self, time, device):
        import torch

        from pathway_amd.engine.batch import DeltaBatch
        from pathway_amd.engine.column import column_from_pylist

        take = [r for r in self._pending if r.time <= time]
        self._pending = [r for r in self._pending if r.time > time]
        if not take:
            return None
        keys = torch.tensor(
            [list(r.key.as_signed_pair()) for r in take], dtype=torch.int64,
            device=device,
        ).reshape(len(take), 2)
        diffs = torch.tensor(
            [r.diff for r in take], dtype=torch.int64, device=device
        )
        cols = {}
        for j, n in enumerate(self.handle.column_names):
            vals = [r.values[j] for r in take]
            cols[n] = column_from_pylist(
                vals, self.handle.dtypes.get(n, dt.ANY), device
            )
        return DeltaBatch(keys, cols, diffs, time)

    def reset(self):
        self.offset = 0
        self._pending = []


def import_table(handle: ExportedTable):
    """Import an exported table into the current graph
    (reference Scope::import_table)."""
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    node = InputNode(_ImportSource(handle), get_device())
    return Table(node, dict(handle.dtypes), Universe())
