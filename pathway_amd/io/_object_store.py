"""Shared object-store reader/writer used by pw.io.{s3,minio,azure}.

Mirrors the reference's posix_like scanner semantics over object stores
(src/connectors/data_storage: S3 scanner + metadata tracking): objects
under a prefix are discovered by listing, parsed per format, tracked by
ETag so modified objects are re-read (old rows retracted) and deleted
objects are retracted.
"""

from __future__ import annotations

import csv as _csv
import io
import json
import time as _time
from typing import Any, Protocol


class ObjectStore(Protocol):
    def list(self, prefix: str) -> list[tuple[str, str]]:
        """[(key, etag), ...]"""

    def get(self, key: str) -> bytes | None: ...
    def put(self, key: str, data: bytes) -> None: ...
    def delete(self, key: str) -> None: ...


def parse_object(data: bytes, format: str, schema) -> list[list[Any]]:
    """Parse one object's bytes into rows (list of value lists)."""
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.json import Json

    if format == "binary":
        return [[data]]
    if format == "plaintext":
        return [[line] for line in data.decode("utf-8", "replace").splitlines()]
    if format in ("json", "jsonlines"):
        names = schema.column_names()
        rows = []
        for line in data.decode().splitlines():
            line = line.strip()
            if not line:
                continue
            rec = json.loads(line)
            row = []
            for n in names:
                v = rec.get(n)
                d = dt.unoptionalize(schema.__columns__[n].dtype)
                if isinstance(v, (dict, list)) and d == dt.JSON:
                    v = Json(v)
                row.append(v)
            rows.append(row)
        return rows
    if format == "csv":
        names = schema.column_names()
        rows = []
        reader = _csv.DictReader(io.StringIO(data.decode()))
        for rec in reader:
            row = []
            for n in names:
                v = rec.get(n)
                d = dt.unoptionalize(schema.__columns__[n].dtype)
                if v is not None and v != "":
                    if d == dt.INT:
                        v = int(v)
                    elif d == dt.FLOAT:
                        v = float(v)
                    elif d == dt.BOOL:
                        v = v.lower() in ("true", "1")
                elif v == "":
                    v = None if d != dt.STR else v
                row.append(v)
            rows.append(row)
        return rows
    raise ValueError(f"unsupported object-store format {format!r}")


class ObjectStoreReader:
    """Reader-thread body polling an object store prefix."""

    def __init__(
        self,
        source,
        store: ObjectStore,
        prefix: str,
        format: str,
        schema,
        *,
        mode: str = "streaming",
        refresh_interval: float = 0.5,
        max_polls: int | None = None,
        with_metadata: bool = False,
    ):
        self.source = source
        self.store = store
        self.prefix = prefix
        self.format = format
        self.schema = schema
        self.mode = mode
        self.refresh_interval = refresh_interval
        self.max_polls = max_polls
        self.with_metadata = with_metadata
        #: key -> etag of the version whose rows are live
        self.seen: dict[str, str] = {}
        #: key -> emitted rows (for retraction)
        self.emitted: dict[str, list] = {}

    def run(self) -> None:
        try:
            polls = 0
            while True:
                self._poll_once()
                if self.mode == "static":
                    return
                polls += 1
                if self.max_polls is not None and polls >= self.max_polls:
                    return
                _time.sleep(self.refresh_interval)
        except Exception as e:
            self.source.fail(e)
        finally:
            self.source.finish()

    def _poll_once(self) -> None:
        from pathway_amd.internals.json import Json

        listed = dict(self.store.list(self.prefix))
        # deletions (retractions reuse each row's original key)
        for key in list(self.seen.keys()):
            if key not in listed:
                for rk, row in self.emitted.pop(key, []):
                    self.source.emit(list(row), key=rk, diff=-1)
                del self.seen[key]
        # new / modified
        for key, etag in sorted(listed.items()):
            if self.seen.get(key) == etag:
                continue
            if key in self.seen:  # modified: retract old rows first
                for rk, row in self.emitted.pop(key, []):
                    self.source.emit(list(row), key=rk, diff=-1)
            data = self.store.get(key)
            if data is None:
                continue
            rows = parse_object(data, self.format, self.schema)
            if self.with_metadata:
                meta = Json({"path": key, "size": len(data), "etag": etag,
                             "seen_at": int(_time.time())})
                rows = [r + [meta] for r in rows]
            out = self.emitted.setdefault(key, [])
            for r in rows:
                rk = self.source.emit(list(r))
                out.append((rk, r))
            self.seen[key] = etag


class ObjectStoreWriter:
    """Sink: one object per non-empty output batch, numbered
    ``<prefix><time>-<seq>.<ext>``."""

    def __init__(self, store: ObjectStore, prefix: str, format: str = "json"):
        self.store = store
        self.prefix = prefix
        self.format = format
        self.seq = 0

    def __call__(self, batch) -> None:
        names = list(batch.columns.keys())
        lines = []
        for _key, values, time, diff in batch.rows():
            rec = dict(zip(names, [_plain(v) for v in values]))
            rec["time"] = time
            rec["diff"] = diff
            lines.append(json.dumps(rec, default=str))
        if not lines:
            return
        ext = "jsonl"
        key = f"{self.prefix}{batch.time}-{self.seq}.{ext}"
        self.seq += 1
        self.store.put(key, ("\n".join(lines) + "\n").encode())


def _plain(v):
    from pathway_amd.internals.api import BasePointer
    from pathway_amd.internals.json import Json

    if isinstance(v, Json):
        return v.value
    if isinstance(v, BasePointer):
        return repr(v)
    if isinstance(v, tuple):
        return list(v)
    return v
