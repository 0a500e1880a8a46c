"""Persistence engine: input snapshots + metadata + recovery
(reference src/persistence/{tracker,state,input_snapshot}.rs — SURVEY §5.4).

Format (filesystem backend):
  <root>/metadata.json              — {"threshold_time": T, "sources": {...},
                                       "version": 1}
  <root>/snapshots/<source>/<chunk> — length-prefixed zlib-compressed
                                       pickled event blocks, one block per
                                       (time, batch) — the LZ4-block-chunk
                                       analog of input_snapshot.rs:14-60
                                       (codec differs: zlib in round 1).

Consistency rule (reference tracker.rs:182-200): chunks are written first,
metadata's threshold_time advances after — a crash between the two replays
the last time from the source instead of losing it.
"""

from __future__ import annotations

import json
import os
import pickle
import struct
import zlib
from typing import Any

from pathway_amd.internals.api import BasePointer, Pointer


class FilesystemSnapshotBackend:
    def __init__(self, root: str):
        self.root = root
        os.makedirs(os.path.join(root, "snapshots"), exist_ok=True)

    def metadata_path(self) -> str:
        return os.path.join(self.root, "metadata.json")

    def load_metadata(self) -> dict | None:
        p = self.metadata_path()
        if not os.path.exists(p):
            return None
        with open(p) as f:
            return json.load(f)

    def save_metadata(self, meta: dict) -> None:
        tmp = self.metadata_path() + ".tmp"
        with open(tmp, "w") as f:
            json.dump(meta, f)
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, self.metadata_path())

    def chunk_dir(self, source: str) -> str:
        d = os.path.join(self.root, "snapshots", source)
        os.makedirs(d, exist_ok=True)
        return d


class SnapshotWriter:
    """Appends (time, rows) event blocks for one source."""

    def __init__(self, backend: FilesystemSnapshotBackend, source: str,
                 max_chunk_bytes: int = 64 * 1024 * 1024):
        self.backend = backend
        self.source = source
        self.max_chunk_bytes = max_chunk_bytes
        d = backend.chunk_dir(source)
        existing = sorted(int(x) for x in os.listdir(d) if x.isdigit())
        self.chunk_id = (existing[-1] + 1) if existing else 0
        self._fh = None
        self._written = 0

    def _file(self):
        if self._fh is None or self._written > self.max_chunk_bytes:
            if self._fh is not None:
                self._fh.close()
                self.chunk_id += 1
            path = os.path.join(self.backend.chunk_dir(self.source), str(self.chunk_id))
            self._fh = open(path, "ab")
            self._written = 0
        return self._fh

    def write_block(self, time: int, rows: list) -> None:
        payload = zlib.compress(pickle.dumps((time, rows), protocol=4), level=1)
        f = self._file()
        f.write(struct.pack("<Q", len(payload)))
        f.write(payload)
        f.flush()
        os.fsync(f.fileno())
        self._written += len(payload) + 8

    def close(self):
        if self._fh is not None:
            self._fh.close()
            self._fh = None


class SnapshotReader:
    def __init__(self, backend: FilesystemSnapshotBackend, source: str):
        self.backend = backend
        self.source = source

    def blocks(self, up_to_time: int):
        d = self.backend.chunk_dir(self.source)
        for cid in sorted(int(x) for x in os.listdir(d) if x.isdigit()):
            path = os.path.join(d, str(cid))
            with open(path, "rb") as f:
                while True:
                    hdr = f.read(8)
                    if len(hdr) < 8:
                        break
                    (ln,) = struct.unpack("<Q", hdr)
                    payload = f.read(ln)
                    if len(payload) < ln:
                        break  # truncated tail (crash mid-write): ignore
                    time, rows = pickle.loads(zlib.decompress(payload))
                    if time <= up_to_time:
                        yield time, rows


class PersistenceManager:
    """Wires snapshots into the Runtime (one worker)."""

    def __init__(self, config, worker: int = 0):
        backend = getattr(config, "backend", None)
        root = getattr(backend, "path", None) or "/tmp/pw_persist"
        self.backend = FilesystemSnapshotBackend(os.path.join(root, f"w{worker}"))
        self.writers: dict[str, SnapshotWriter] = {}
        mode = getattr(config, "persistence_mode", None)
        self.operator_persisting = str(mode).lower().endswith("operator_persisting")
        from pathway_amd.persistence.operator_snapshot import OperatorSnapshotStore

        self.op_store = OperatorSnapshotStore(self.backend.root)
        meta = self.backend.load_metadata()
        self.threshold_time: int = meta["threshold_time"] if meta else -1

    def writer(self, source: str) -> SnapshotWriter:
        if source not in self.writers:
            self.writers[source] = SnapshotWriter(self.backend, source)
        return self.writers[source]

    def record(self, source: str, time: int, batch) -> None:
        if batch is None or len(batch) == 0:
            return
        rows = batch.rows()  # (key, values, time, diff)
        ser = [
            ((k.lo, k.hi), values, diff) for k, values, _, diff in rows
        ]
        names = list(batch.columns.keys())
        self.writer(source).write_block(time, {"names": names, "rows": ser})

    def commit(self, time: int) -> None:
        if time > self.threshold_time:
            self.threshold_time = time
            self.backend.save_metadata({"threshold_time": time, "version": 1})

    def replay_blocks(self, source: str):
        return SnapshotReader(self.backend, source).blocks(self.threshold_time)

    def close(self):
        for w in self.writers.values():
            w.close()
