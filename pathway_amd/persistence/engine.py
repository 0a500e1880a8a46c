"""Persistence engine: input snapshots + metadata + recovery.

Reference: src/persistence/{tracker,state,input_snapshot}.rs, backends/
(SURVEY §5.4).  Round-2 format (replaces round-1 pickle+zlib — ADVICE r1
finding 5):

  * events — bincode-compatible binary codec (persistence/codec.py):
    Insert/Delete(Key, Vec<Value>) + AdvanceTime(Timestamp, offsets);
    data-only, nothing executable on recovery.
  * chunks — numbered per (worker, source); each block is one frame
    ``[u64 raw_len][u64 lz4_len][lz4 block bytes]`` (LZ4 block codec in
    the native pw_io library).  Rotation per input_snapshot.rs:286-380:
    a chunk closes at 100k entries or `max_chunk_bytes`.
  * metadata — JSON under keys ``metadata/<version>-<worker>-<rotation>``
    (state.rs:16-130); recovery threshold = min over workers of each
    worker's latest finalized time.
  * backends — filesystem / S3 / Azure blob / mock behind one BlobStore
    interface (persistence/backends.py), selected by the pw.persistence
    Backend kind.

Consistency rule (tracker.rs:182-200): blocks are written and flushed
first, the metadata threshold advances after — a crash between the two
replays the last time from the source instead of losing it.
"""

from __future__ import annotations

import json
import struct
from typing import Any, Iterator

from pathway_amd.persistence import codec
from pathway_amd.persistence.backends import BlobStore, make_store

#: reference input_snapshot.rs chunk rotation bounds
MAX_CHUNK_ENTRIES = 100_000
DEFAULT_MAX_CHUNK_BYTES = 64 * 1024 * 1024

FORMAT_VERSION = 2


def _lz4():
    from pathway_amd.ops import native_io

    return native_io


class SnapshotWriter:
    """Appends event blocks for one (worker, source) stream."""

    def __init__(self, store: BlobStore, prefix: str,
                 max_chunk_bytes: int = DEFAULT_MAX_CHUNK_BYTES):
        self.store = store
        self.prefix = prefix  # snapshots/<worker>/<source>
        self.max_chunk_bytes = max_chunk_bytes
        existing = [
            int(k.rsplit("/", 1)[-1])
            for k in store.list(prefix + "/")
            if k.rsplit("/", 1)[-1].isdigit()
        ]
        self.chunk_id = (max(existing) + 1) if existing else 0
        self._entries = 0
        self._bytes = 0

    def _chunk_key(self) -> str:
        return f"{self.prefix}/{self.chunk_id}"

    def _maybe_rotate(self) -> None:
        if self._entries >= MAX_CHUNK_ENTRIES or self._bytes >= self.max_chunk_bytes:
            self.store.finalize(self._chunk_key())
            self.chunk_id += 1
            self._entries = 0
            self._bytes = 0

    def write_block(self, payload: bytes, entries: int) -> None:
        nio = _lz4()
        comp = nio.lz4_compress(payload)
        frame = struct.pack("<QQ", len(payload), len(comp)) + comp
        self.store.append(self._chunk_key(), frame)
        self._entries += entries
        self._bytes += len(frame)
        self._maybe_rotate()

    def close(self) -> None:
        self.store.finalize(self._chunk_key())


class SnapshotReader:
    def __init__(self, store: BlobStore, prefix: str):
        self.store = store
        self.prefix = prefix

    def frames(self) -> Iterator[bytes]:
        nio = _lz4()
        keys = sorted(
            (k for k in self.store.list(self.prefix + "/")
             if k.rsplit("/", 1)[-1].isdigit()),
            key=lambda k: int(k.rsplit("/", 1)[-1]),
        )
        for key in keys:
            data = self.store.get(key)
            if data is None:
                continue
            i = 0
            n = len(data)
            while i + 16 <= n:
                raw_len, comp_len = struct.unpack_from("<QQ", data, i)
                i += 16
                if i + comp_len > n:
                    break  # truncated tail (crash mid-write)
                yield nio.lz4_decompress(data[i : i + comp_len], raw_len)
                i += comp_len


class PersistenceManager:
    """Wires snapshots into the Runtime (one worker).

    Shared root across workers: snapshots/<worker>/<source>/<chunk>,
    metadata/<version>-<worker>-<rotation>.
    """

    def __init__(self, config, worker: int = 0):
        backend_cfg = getattr(config, "backend", None)
        self.store = make_store(backend_cfg)
        self.worker = worker
        self.writers: dict[str, SnapshotWriter] = {}
        mode = getattr(config, "persistence_mode", None)
        self.operator_persisting = str(mode).lower().endswith("operator_persisting")
        self.snapshot_interval_ms = getattr(config, "snapshot_interval_ms", 0) or 0
        from pathway_amd.persistence.operator_snapshot import OperatorSnapshotStore

        self.op_store = OperatorSnapshotStore.over_store(self.store, worker)
        self.rotation = self._load_rotation()
        self.threshold_time: int = self._load_threshold()
        self._offsets_fn = None  # optional: source -> offset list provider

    # -- metadata (<version>-<worker>-<rotation> keys, state.rs:16-130) --

    def _meta_entries(self) -> dict[int, list[tuple[int, dict]]]:
        """worker -> [(rotation, meta), ...]"""
        out: dict[int, list[tuple[int, dict]]] = {}
        for key in self.store.list("metadata/"):
            name = key.rsplit("/", 1)[-1]
            parts = name.split("-")
            if len(parts) != 3:
                continue
            try:
                _ver, w, rot = (int(p) for p in parts)
            except ValueError:
                continue
            data = self.store.get(key)
            if not data:
                continue
            try:
                meta = json.loads(data)
            except ValueError:
                continue
            out.setdefault(w, []).append((rot, meta))
        return out

    def _load_rotation(self) -> int:
        entries = self._meta_entries().get(self.worker, [])
        return (max(r for r, _ in entries) + 1) if entries else 0

    def _load_threshold(self) -> int:
        """Min over workers of each worker's latest finalized time
        (state.rs:160-320 threshold computation)."""
        entries = self._meta_entries()
        if not entries:
            return -1
        per_worker = []
        for _w, lst in entries.items():
            lst.sort()
            per_worker.append(lst[-1][1].get("threshold_time", -1))
        return min(per_worker)

    # -- input snapshots --

    def writer(self, source: str) -> SnapshotWriter:
        if source not in self.writers:
            self.writers[source] = SnapshotWriter(
                self.store, f"snapshots/{self.worker}/{source}"
            )
        return self.writers[source]

    def record(self, source: str, time: int, batch) -> None:
        if batch is None or len(batch) == 0:
            return
        names = list(batch.columns.keys())
        out = bytearray()
        out += struct.pack("<I", 0)  # frame kind 0 = data block
        out += struct.pack("<QQ", time, len(names))
        for n in names:
            nb = n.encode()
            out += struct.pack("<Q", len(nb))
            out += nb
        entries = 0
        for key, values, _t, diff in batch.rows():
            kind = codec.E_INSERT if diff > 0 else codec.E_DELETE
            reps = abs(int(diff))
            ev = codec.encode_event(kind, key=key, values=list(values))
            for _ in range(reps):
                out += ev
                entries += 1
        self.writer(source).write_block(bytes(out), entries)

    def commit(self, time: int) -> None:
        if time <= self.threshold_time:
            return
        # AdvanceTime events with reader offsets (seek metadata)
        offsets = self._collect_offsets()
        for source, w in self.writers.items():
            ev = codec.encode_event(
                codec.E_ADVANCE_TIME, time=time,
                offsets=offsets.get(source, []),
            )
            w.write_block(struct.pack("<I", 1) + ev, 1)  # frame kind 1
        self.threshold_time = time
        key = f"metadata/{FORMAT_VERSION}-{self.worker}-{self.rotation}"
        self.store.put(
            key,
            json.dumps({
                "threshold_time": time,
                "worker": self.worker,
                "version": FORMAT_VERSION,
            }).encode(),
        )
        self.rotation += 1
        # keep only the last two rotations per worker
        mine = sorted(
            k for k in self.store.list("metadata/")
            if k.rsplit("/", 1)[-1].split("-")[:2]
            == [str(FORMAT_VERSION), str(self.worker)]
        )
        for k in mine[:-2]:
            self.store.delete(k)

    def _collect_offsets(self) -> dict[str, list[tuple[str, str]]]:
        if self._offsets_fn is None:
            return {}
        try:
            return self._offsets_fn() or {}
        except Exception:
            return {}

    def replay_blocks(self, source: str):
        """Yield (time, {"names": [...], "rows": [((lo,hi), values, diff)]})
        up to the recovery threshold."""
        if self.threshold_time < 0:
            return
        reader = SnapshotReader(self.store, f"snapshots/{self.worker}/{source}")
        for frame in reader.frames():
            if len(frame) < 4:
                continue
            (ftype,) = struct.unpack_from("<I", frame, 0)
            if ftype != 0:
                continue  # AdvanceTime marker frame
            if len(frame) < 20:
                continue
            time, nnames = struct.unpack_from("<QQ", frame, 4)
            i = 20
            names = []
            ok = True
            for _ in range(nnames):
                if i + 8 > len(frame):
                    ok = False
                    break
                (ln,) = struct.unpack_from("<Q", frame, i)
                i += 8
                if i + ln > len(frame):
                    ok = False
                    break
                names.append(frame[i : i + ln].decode())
                i += ln
            if not ok or time > self.threshold_time:
                continue
            rows = []
            while i < len(frame):
                kind, payload, i = codec.decode_event(frame, i)
                if kind in (codec.E_INSERT, codec.E_DELETE):
                    key, values = payload
                    rows.append(
                        ((key.lo, key.hi), values,
                         1 if kind == codec.E_INSERT else -1)
                    )
            yield time, {"names": names, "rows": rows}

    def source_offsets(self, source: str) -> dict:
        """Latest persisted offsets for a source (seek on recovery)."""
        reader = SnapshotReader(self.store, f"snapshots/{self.worker}/{source}")
        out: dict = {}
        for frame in reader.frames():
            if len(frame) < 4 or struct.unpack_from("<I", frame, 0)[0] != 1:
                continue
            try:
                _k, (t, offs), _ = codec.decode_event(frame, 4)
            except codec.CodecError:
                continue
            if t <= self.threshold_time:
                out = dict(offs)
        return out

    def close(self):
        for w in self.writers.values():
            w.close()
