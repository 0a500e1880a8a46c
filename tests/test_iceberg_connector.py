"""Iceberg connector: v2 metadata + avro manifest structures + parquet.

Reference behavior: src/connectors/data_storage/data_lake/iceberg.rs.
"""

import json
import os
import threading
import time

import pytest

import pathway_amd as pw
from pathway_amd.internals.rungraph import G
from pathway_amd.internals.schema import schema_from_types

pa = pytest.importorskip("pyarrow")


def test_iceberg_write_layout(tmp_path):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        """
    )
    root = str(tmp_path / "wh")
    pw.io.iceberg.write(t, warehouse=root, namespace=["ns"], table_name="tbl")
    pw.run()
    tdir = os.path.join(root, "ns", "tbl")
    hint = os.path.join(tdir, "metadata", "version-hint.text")
    assert os.path.exists(hint)
    v = int(open(hint).read())
    meta = json.load(open(os.path.join(tdir, "metadata", f"v{v}.metadata.json")))
    assert meta["format-version"] == 2
    assert meta["current-snapshot-id"] != -1
    snap = meta["snapshots"][-1]
    # manifest list + manifest are valid avro containers
    from pathway_amd.io.formats import avro

    with open(snap["manifest-list"], "rb") as f:
        [mf] = list(avro.read_container(f))
    assert mf["added_rows_count"] == 2
    with open(mf["manifest_path"], "rb") as f:
        entries = list(avro.read_container(f))
    assert entries[0]["data_file"]["file_format"] == "PARQUET"
    assert os.path.exists(entries[0]["data_file"]["file_path"])


def test_iceberg_roundtrip_and_streaming(tmp_path):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        """
    )
    root = str(tmp_path / "wh2")
    pw.io.iceberg.write(t, warehouse=root, table_name="t1")
    pw.run()

    G.clear()
    back = pw.io.iceberg.read(
        warehouse=root, table_name="t1",
        schema=schema_from_types(a=int, b=str), mode="static",
    )
    keys, cols = pw.debug.table_to_dicts(back)
    assert sorted(zip(cols["a"].values(), cols["b"].values())) == [
        (1, "x"), (2, "y")
    ]

    # streaming: a second snapshot appended later is picked up
    from pathway_amd.io.iceberg import IcebergWriter

    schema = schema_from_types(a=int, b=str)
    w = IcebergWriter(os.path.join(root, "t1"), ["a", "b"], schema)

    class FakeBatch:
        time = 2
        columns = {"a": None, "b": None}

        def rows(self):
            yield None, [3, "z"], 2, 1

    def later():
        time.sleep(0.3)
        w(FakeBatch())

    th = threading.Thread(target=later)
    th.start()
    G.clear()
    live = pw.io.iceberg.read(
        warehouse=root, table_name="t1",
        schema=schema_from_types(a=int, b=str), mode="streaming",
        refresh_interval=0.1, _max_polls=12,
    )
    res = live.groupby().reduce(n=pw.reducers.count(), s=pw.reducers.sum(pw.this.a))
    keys, cols = pw.debug.table_to_dicts(res)
    th.join()
    assert list(cols["n"].values()) == [3]
    assert list(cols["s"].values()) == [6]
