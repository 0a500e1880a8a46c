"""pw.io.iceberg — Apache Iceberg connector (local warehouse).

Reference: python/pathway/io/iceberg + src/connectors/data_storage/
data_lake (iceberg.rs, 1,427 LoC over iceberg-rust).  This build writes
the Iceberg v2 table layout directly:

  metadata/version-hint.text            current metadata version
  metadata/v<N>.metadata.json           table metadata + snapshot log
  metadata/snap-<id>.avro               manifest list  (avro, in-repo codec)
  metadata/manifest-<id>.avro           manifest file  (avro)
  data/part-*.parquet                   row data (pyarrow)

with field-id'd avro schemas for the manifest structures.  The reader
replays the snapshot log (appended data files per snapshot) and can
poll for new snapshots in streaming mode.
"""

from __future__ import annotations

import json
import os
import time as _time
import uuid
from typing import Any

from pathway_amd.io.formats import avro as _avro

MANIFEST_ENTRY_SCHEMA = {
    "type": "record",
    "name": "manifest_entry",
    "fields": [
        {"name": "status", "type": "int", "field-id": 0},
        {"name": "snapshot_id", "type": ["null", "long"], "field-id": 1},
        {"name": "data_file", "type": {
            "type": "record", "name": "r2",
            "fields": [
                {"name": "file_path", "type": "string", "field-id": 100},
                {"name": "file_format", "type": "string", "field-id": 101},
                {"name": "record_count", "type": "long", "field-id": 103},
                {"name": "file_size_in_bytes", "type": "long", "field-id": 104},
            ],
        }, "field-id": 2},
    ],
}

MANIFEST_FILE_SCHEMA = {
    "type": "record",
    "name": "manifest_file",
    "fields": [
        {"name": "manifest_path", "type": "string", "field-id": 500},
        {"name": "manifest_length", "type": "long", "field-id": 501},
        {"name": "partition_spec_id", "type": "int", "field-id": 502},
        {"name": "added_snapshot_id", "type": "long", "field-id": 503},
        {"name": "added_files_count", "type": "int", "field-id": 504},
        {"name": "added_rows_count", "type": "long", "field-id": 512},
    ],
}


def _iceberg_type(d) -> str:
    from pathway_amd.internals import dtype as dt

    return {dt.INT: "long", dt.FLOAT: "double", dt.BOOL: "boolean",
            dt.STR: "string", dt.BYTES: "binary"}.get(dt.unoptionalize(d), "string")


class IcebergTable:
    def __init__(self, root: str):
        self.root = root
        self.meta_dir = os.path.join(root, "metadata")
        self.data_dir = os.path.join(root, "data")

    # -- metadata plumbing --

    def current_version(self) -> int:
        hint = os.path.join(self.meta_dir, "version-hint.text")
        if not os.path.exists(hint):
            return 0
        with open(hint) as f:
            return int(f.read().strip() or 0)

    def load_metadata(self) -> dict | None:
        v = self.current_version()
        if v == 0:
            return None
        with open(os.path.join(self.meta_dir, f"v{v}.metadata.json")) as f:
            return json.load(f)

    def _write_metadata(self, meta: dict) -> None:
        v = self.current_version() + 1
        os.makedirs(self.meta_dir, exist_ok=True)
        path = os.path.join(self.meta_dir, f"v{v}.metadata.json")
        with open(path + ".tmp", "w") as f:
            json.dump(meta, f)
        os.rename(path + ".tmp", path)
        hint = os.path.join(self.meta_dir, "version-hint.text")
        with open(hint + ".tmp", "w") as f:
            f.write(str(v))
        os.rename(hint + ".tmp", hint)

    def ensure_table(self, schema) -> None:
        if self.load_metadata() is not None:
            return
        os.makedirs(self.data_dir, exist_ok=True)
        fields = []
        for i, n in enumerate(schema.column_names()):
            fields.append({"id": i + 1, "name": n, "required": False,
                           "type": _iceberg_type(schema.__columns__[n].dtype)})
        fields.append({"id": len(fields) + 1, "name": "time",
                       "required": True, "type": "long"})
        fields.append({"id": len(fields) + 1, "name": "diff",
                       "required": True, "type": "long"})
        self._write_metadata({
            "format-version": 2,
            "table-uuid": str(uuid.uuid4()),
            "location": self.root,
            "last-updated-ms": int(_time.time() * 1000),
            "last-column-id": len(fields),
            "schemas": [{"schema-id": 0, "type": "struct", "fields": fields}],
            "current-schema-id": 0,
            "partition-specs": [{"spec-id": 0, "fields": []}],
            "default-spec-id": 0,
            "snapshots": [],
            "snapshot-log": [],
            "current-snapshot-id": -1,
        })

    def append_snapshot(self, data_files: list[tuple[str, int, int]]) -> None:
        """data_files: [(path, record_count, size_bytes)]."""
        meta = self.load_metadata()
        snap_id = int(_time.time() * 1000) + len(meta["snapshots"])
        # manifest file
        mpath = os.path.join(self.meta_dir, f"manifest-{snap_id}.avro")
        with open(mpath, "wb") as f:
            w = _avro.ContainerWriter(f, MANIFEST_ENTRY_SCHEMA)
            for path, nrec, size in data_files:
                w.append({
                    "status": 1,  # ADDED
                    "snapshot_id": snap_id,
                    "data_file": {
                        "file_path": path,
                        "file_format": "PARQUET",
                        "record_count": nrec,
                        "file_size_in_bytes": size,
                    },
                })
            w.close()
        # manifest list
        lpath = os.path.join(self.meta_dir, f"snap-{snap_id}.avro")
        with open(lpath, "wb") as f:
            w = _avro.ContainerWriter(f, MANIFEST_FILE_SCHEMA)
            w.append({
                "manifest_path": mpath,
                "manifest_length": os.path.getsize(mpath),
                "partition_spec_id": 0,
                "added_snapshot_id": snap_id,
                "added_files_count": len(data_files),
                "added_rows_count": sum(n for _, n, _ in data_files),
            })
            w.close()
        meta["snapshots"].append({
            "snapshot-id": snap_id,
            "timestamp-ms": int(_time.time() * 1000),
            "manifest-list": lpath,
            "summary": {"operation": "append"},
        })
        meta["snapshot-log"].append({
            "snapshot-id": snap_id, "timestamp-ms": int(_time.time() * 1000)
        })
        meta["current-snapshot-id"] = snap_id
        self._write_metadata(meta)

    def snapshot_files(self, snap: dict) -> list[str]:
        with open(snap["manifest-list"], "rb") as f:
            manifests = list(_avro.read_container(f))
        out = []
        for m in manifests:
            with open(m["manifest_path"], "rb") as f:
                for entry in _avro.read_container(f):
                    if entry["status"] in (1, 0):  # ADDED / EXISTING
                        out.append(entry["data_file"]["file_path"])
        return out


class IcebergWriter:
    def __init__(self, root: str, column_names: list[str], schema):
        import pyarrow  # noqa: F401

        self.table = IcebergTable(root)
        self.table.ensure_table(schema)
        self.column_names = column_names
        self.seq = 0

    def __call__(self, batch) -> None:
        import pyarrow as pa
        import pyarrow.parquet as pq

        rows = list(batch.rows())
        if not rows:
            return
        names = list(batch.columns.keys())
        cols: dict[str, list] = {n: [] for n in names}
        cols["time"] = []
        cols["diff"] = []
        for _key, values, time, diff in rows:
            for n, v in zip(names, values):
                cols[n].append(v)
            cols["time"].append(time)
            cols["diff"].append(diff)
        fpath = os.path.join(
            self.table.data_dir, f"part-{batch.time:012d}-{self.seq:05d}.parquet"
        )
        self.seq += 1
        pq.write_table(pa.table(cols), fpath)
        self.table.append_snapshot(
            [(fpath, len(rows), os.path.getsize(fpath))]
        )

    def flush(self, time) -> None:
        pass


def write(table, catalog_uri: str | None = None, namespace: list | None = None,
          table_name: str | None = None, *, warehouse: str | None = None,
          name: str | None = None, **kwargs: Any):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G
    from pathway_amd.internals.schema import schema_from_types

    root = warehouse or catalog_uri
    if root is None:
        raise ValueError("pw.io.iceberg.write needs warehouse=<path>")
    if table_name:
        root = os.path.join(root, *(namespace or []), table_name)
    names = table.column_names()
    schema = schema_from_types(
        **{n: table._dtypes[n].typehint for n in names}
    )
    writer = IcebergWriter(root, names, schema)
    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node


class IcebergReader:
    def __init__(self, source, root: str, schema, *, mode: str = "streaming",
                 refresh_interval: float = 0.5, max_polls: int | None = None):
        self.source = source
        self.table = IcebergTable(root)
        self.schema = schema
        self.mode = mode
        self.refresh_interval = refresh_interval
        self.max_polls = max_polls
        self.seen_snapshots: set[int] = set()

    def run(self) -> None:
        try:
            polls = 0
            while True:
                self._poll()
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

    def _poll(self) -> None:
        import pyarrow.parquet as pq

        meta = self.table.load_metadata()
        if meta is None:
            return
        names = self.schema.column_names()
        for snap in meta["snapshots"]:
            sid = snap["snapshot-id"]
            if sid in self.seen_snapshots:
                continue
            self.seen_snapshots.add(sid)
            for fpath in self.table.snapshot_files(snap):
                data = pq.read_table(fpath).to_pydict()
                n = len(next(iter(data.values()))) if data else 0
                diffs = data.get("diff", [1] * n)
                for i in range(n):
                    row = [data.get(c, [None] * n)[i] for c in names]
                    self.source.emit(row, diff=int(diffs[i]))


def read(catalog_uri: str | None = None, namespace: list | None = None,
         table_name: str | None = None, *, warehouse: str | None = None,
         schema=None, mode: str = "streaming", refresh_interval: float = 0.5,
         name: str | None = None, _max_polls: int | None = None,
         **kwargs: Any):
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    root = warehouse or catalog_uri
    if root is None:
        raise ValueError("pw.io.iceberg.read needs warehouse=<path>")
    if table_name:
        root = os.path.join(root, *(namespace or []), table_name)
    if schema is None:
        raise ValueError("pw.io.iceberg.read requires a schema")
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    src = StreamingSource(names, dtypes, name=name)
    reader = IcebergReader(src, root, schema, mode=mode,
                           refresh_interval=refresh_interval,
                           max_polls=_max_polls)
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())
