"""Streaming sources: live readers feeding the engine
(reference src/connectors/mod.rs:660-1060 Connector::run — reader thread →
bounded channel → main-thread poller, ≤100k events/step, autocommit times).
"""

from __future__ import annotations

import queue
import threading
import time as _time
from typing import Any, Callable

import torch

from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.column import Column, column_from_pylist
from pathway_amd.engine.runtime import Source
from pathway_amd.internals import dtype as dt
from pathway_amd.internals.api import BasePointer, Pointer, hash_values

#: reference mod.rs:61 — max events ingested per engine step
MAX_EVENTS_PER_STEP = 100_000


class StreamingSource(Source):
    """Thread-fed source; the runtime assigns commit times when pulling."""

    def __init__(
        self,
        column_names: list[str],
        dtypes: list[dt.DType],
        name: str | None = None,
        maxsize: int | None = None,
    ):
        self.column_names = column_names
        self.dtypes = dtypes
        self.name = name
        # bounded queue = reader backpressure (reference max_backlog_size,
        # connectors/backlog.rs): emit() blocks when the engine is behind
        self.q: queue.Queue = queue.Queue(maxsize=maxsize or 0)
        self._finished = threading.Event()
        self._seq = 0
        self._sync_group = None
        self._sync_member = None
        self._sync_idx = None
        self._held: list = []
        # ingest/compute overlap: a stager thread
        # pre-converts queued rows into CPU column bundles while the
        # engine computes the previous timestamp; pull() then only stamps
        # the time and moves tensors to the device.
        self._staged: queue.Queue = queue.Queue(maxsize=2)
        self._stager: threading.Thread | None = None
        self._drain_lock = threading.Lock()
        self._inflight = False

    def _ensure_stager(self) -> None:
        if self._stager is not None or self._sync_group is not None:
            return
        t = threading.Thread(target=self._stage_loop, daemon=True)
        self._stager = t
        t.start()

    def _stage_loop(self) -> None:
        import torch as _torch

        while True:
            if self.q.empty():
                # order matters: observe finished FIRST — finish() is the
                # reader's last call, so an empty queue seen after it is
                # final; the reverse order can strand the last rows
                if self._finished.is_set():
                    if self.q.empty():
                        return
                else:
                    _time.sleep(0.002)
                    continue
            # drain under the lock so next_time's done-check never sees
            # rows vanish mid-transfer
            with self._drain_lock:
                rows = []
                try:
                    while len(rows) < MAX_EVENTS_PER_STEP:
                        rows.append(self.q.get_nowait())
                except queue.Empty:
                    pass
                if not rows:
                    continue
                self._inflight = True
            keys = _torch.tensor(
                [list(k.as_signed_pair()) for k, _, _ in rows],
                dtype=_torch.int64,
            ).reshape(len(rows), 2)
            diffs = _torch.tensor(
                [d for _, _, d in rows], dtype=_torch.int64
            )
            cols = {}
            for j, nme in enumerate(self.column_names):
                vals = [v[j] if j < len(v) else None for _, v, _ in rows]
                cols[nme] = column_from_pylist(vals, self.dtypes[j], "cpu")
            self._staged.put((keys, cols, diffs))  # blocks at depth 2
            with self._drain_lock:
                self._inflight = False

    # -- producer side (reader thread) --

    def emit(self, values: list[Any], key: BasePointer | None = None, diff: int = 1):
        """Enqueue one update; returns the row key so readers can later
        retract the same row (deletions must reuse the insert's key)."""
        if key is None:
            self._seq += 1
            lo, hi = hash_values([self.name or "stream", self._seq])
            key = Pointer(lo, hi)
        self.q.put((key, values, diff))
        return key

    def fail(self, exc: Exception) -> None:
        """Reader-thread error: re-raised on the engine thread at next pull."""
        self._error = exc

    def finish(self):
        self._finished.set()
        if self._sync_group is not None:
            # a finished member's frontier is +inf: stop throttling peers
            self._sync_group.observe(self._sync_member, float("inf"))

    def attach_sync_group(self, group, member_id, col_idx) -> None:
        self._sync_group = group
        self._sync_member = member_id
        self._sync_idx = col_idx

    # -- consumer side (engine loop) --

    def has_pending(self) -> bool:
        return (
            bool(self._held)
            or not self.q.empty()
            or not self._staged.empty()
        )

    def is_live(self) -> bool:
        return not self._finished.is_set() or self.has_pending()

    def next_time(self):
        from pathway_amd.engine.runtime import STREAM_READY, STREAM_WAITING

        if self._sync_group is None:
            self._ensure_stager()
            with self._drain_lock:
                if self._held or not self._staged.empty():
                    return STREAM_READY
                if (
                    self._finished.is_set()
                    and self.q.empty()
                    and not self._inflight
                ):
                    return None
            return STREAM_WAITING
        if self.has_pending():
            return STREAM_READY
        if self.is_live():
            return STREAM_WAITING
        return None

    def pull(self, time: int, device) -> DeltaBatch | None:
        err = getattr(self, "_error", None)
        if err is not None:
            self._error = None
            raise RuntimeError(f"connector reader failed: {err}") from err
        if self._sync_group is None and self._stager is not None:
            # staged path: the bundle was pre-built off-thread
            try:
                keys, cols, diffs = self._staged.get_nowait()
            except queue.Empty:
                return None
            dev = torch.device(device)
            if dev.type != "cpu":
                keys = keys.to(dev, non_blocking=True)
                diffs = diffs.to(dev, non_blocking=True)
                cols = {n: c.to_device(dev) for n, c in cols.items()}
            return DeltaBatch(keys, cols, diffs, time)
        rows = list(self._held)
        self._held = []
        try:
            while len(rows) < MAX_EVENTS_PER_STEP:
                rows.append(self.q.get_nowait())
        except queue.Empty:
            pass
        if self._sync_group is not None and rows:
            # advance our watermark, then hold rows past the group threshold
            idx = self._sync_idx
            for _, values, _ in rows:
                v = values[idx]
                if v is not None:
                    self._sync_group.observe(self._sync_member, float(v))
            thr = self._sync_group.release_threshold()
            ready = [r for r in rows if r[1][idx] is None or float(r[1][idx]) <= thr]
            self._held = [r for r in rows if not (r[1][idx] is None or float(r[1][idx]) <= thr)]
            rows = ready
        if not rows:
            return None
        keys = torch.tensor(
            [list(k.as_signed_pair()) for k, _, _ in rows],
            dtype=torch.int64,
            device=device,
        ).reshape(len(rows), 2)
        diffs = torch.tensor([d for _, _, d in rows], dtype=torch.int64, device=device)
        cols: dict[str, Column] = {}
        for j, name in enumerate(self.column_names):
            vals = [v[j] if j < len(v) else None for _, v, _ in rows]
            cols[name] = column_from_pylist(vals, self.dtypes[j], device)
        return DeltaBatch(keys, cols, diffs, time)

    def reset(self):
        pass


def spawn_reader(
    fn: Callable[[], None], source=None, sharded: bool = False
) -> threading.Thread | None:
    """Start a connector reader thread.

    Multi-worker: a single-stream reader (CDC/WAL tail, change stream,
    queue subscription) must run on exactly ONE worker or every rank
    re-ingests the same rows (reference: connectors run per-worker only
    when the source itself shards — kafka partitions, file path hashes).
    Pass `source` to elect worker 0 and finish the source on the others;
    `sharded=True` marks readers that do their own worker assignment."""
    if source is not None and not sharded:
        from pathway_amd import parallel as par

        comm = par.get_comm()
        if comm is not None and comm.world > 1 and comm.rank != 0:
            source.finish()
            return None
    th = threading.Thread(target=fn, daemon=True)
    th.start()
    return th


class FilePollReader:
    """Directory/file poller (reference posix_like reader + metadata
    tracking, connectors/metadata/file_like.rs): emits rows for new or
    modified files; object deletions retract previous rows when
    with_metadata tracking is on."""

    def __init__(
        self,
        source: StreamingSource,
        path: str,
        format: str,
        schema,
        mode: str,
        with_metadata: bool,
        refresh_interval: float = 0.5,
        max_polls: int | None = None,
    ):
        self.source = source
        self.path = path
        self.format = format
        self.schema = schema
        self.mode = mode
        self.with_metadata = with_metadata
        self.refresh_interval = refresh_interval
        self.max_polls = max_polls
        self.seen: dict[str, float] = {}
        #: rows emitted per path (for retraction on delete/modify)
        self.emitted: dict[str, list] = {}

    def _list_files(self) -> list[str]:
        import os

        if os.path.isdir(self.path):
            out = []
            for root, _, files in os.walk(self.path):
                for f in sorted(files):
                    out.append(os.path.join(root, f))
        else:
            import glob

            out = sorted(glob.glob(self.path)) or (
                [self.path] if os.path.exists(self.path) else []
            )
        return self._shard_paths(out)

    @staticmethod
    def _shard_paths(paths: list[str]) -> list[str]:
        """Multi-worker file assignment by path hash (reference
        connectors/data_storage/sharding.rs): each file is read by
        exactly one worker; salt-free crc32 so every rank computes the
        same split."""
        from pathway_amd import parallel as par

        comm = par.get_comm()
        if comm is None or comm.world <= 1:
            return paths
        import zlib

        return [
            p
            for p in paths
            if zlib.crc32(p.encode()) % comm.world == comm.rank
        ]

    def _retract_file(self, path: str) -> None:
        """Object deletion/modification: retract the rows previously
        emitted for this path (reference metadata/file_like.rs tracking).
        Each retraction reuses the row's original key."""
        for key, row in self.emitted.pop(path, []):
            self.source.emit(list(row), key=key, diff=-1)

    def _emit_file(self, path: str):
        import csv as _csv
        import json
        import os

        meta = None
        if self.with_metadata:
            from pathway_amd.internals.json import Json

            st = os.stat(path)
            meta = Json(
                {
                    "path": path,
                    "size": st.st_size,
                    "modified_at": int(st.st_mtime),
                    "seen_at": int(_time.time()),
                    "owner": "unknown",
                }
            )
        names = self.schema.column_names() if self.schema else None
        from pathway_amd.ops import native_io

        rows_out = self.emitted.setdefault(path, [])

        def emit_row(row):
            key = self.source.emit(row)
            rows_out.append((key, list(row)))

        if self.format == "plaintext":
            if native_io.available():
                lines = native_io.read_lines(path)
            else:
                with open(path) as fh:
                    lines = [l.rstrip("\n") for l in fh]
            for line in lines:
                row = [line]
                if self.with_metadata:
                    row.append(meta)
                emit_row(row)
        elif self.format == "binary":
            with open(path, "rb") as fh:
                row = [fh.read()]
            if self.with_metadata:
                row.append(meta)
            emit_row(row)
        elif self.format == "csv":
            if native_io.available():
                header, recs = native_io.read_csv(path)
                idx = [header.index(n) if n in header else None for n in names]
                for rec in recs:
                    row = [
                        _convert(rec[i] if i is not None else None, self.schema, n)
                        for n, i in zip(names, idx)
                    ]
                    if self.with_metadata:
                        row.append(meta)
                    emit_row(row)
            else:
                with open(path, newline="") as fh:
                    reader = _csv.DictReader(fh)
                    for rec in reader:
                        row = [_convert(rec.get(n), self.schema, n) for n in names]
                        if self.with_metadata:
                            row.append(meta)
                        emit_row(row)
        elif self.format in ("json", "jsonlines"):
            with open(path) as fh:
                for line in fh:
                    line = line.strip()
                    if not line:
                        continue
                    rec = json.loads(line)
                    row = [rec.get(n) for n in names]
                    if self.with_metadata:
                        row.append(meta)
                    emit_row(row)

    def run(self):
        import os

        polls = 0
        while True:
            current = set()
            for f in self._list_files():
                try:
                    mtime = os.path.getmtime(f)
                except OSError:
                    continue
                current.add(f)
                if self.seen.get(f) != mtime:
                    if f in self.seen:
                        # modified: retract the previous version first
                        self._retract_file(f)
                    self.seen[f] = mtime
                    self._emit_file(f)
            for gone in [p for p in self.seen if p not in current]:
                del self.seen[gone]
                self._retract_file(gone)
            polls += 1
            if self.mode == "static" or (
                self.max_polls is not None and polls >= self.max_polls
            ):
                break
            _time.sleep(self.refresh_interval)
        self.source.finish()


def _convert(v, schema, name):
    if v is None or v == "":
        return None
    d = dt.unoptionalize(schema.__columns__[name].dtype)
    if d == dt.INT:
        return int(v)
    if d == dt.FLOAT:
        return float(v)
    if d == dt.BOOL:
        return str(v).lower() in ("true", "1")
    return v
