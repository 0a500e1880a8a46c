"""Delta Lake connector: transaction-log protocol + parquet round trips.

Reference behavior: src/connectors/data_storage/data_lake/delta.rs.
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


def test_delta_write_creates_valid_log(tmp_path):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        """
    )
    root = str(tmp_path / "dt")
    pw.io.deltalake.write(t, root)
    pw.run()
    log = sorted(os.listdir(os.path.join(root, "_delta_log")))
    assert log[0] == f"{0:020d}.json"
    with open(os.path.join(root, "_delta_log", log[0])) as f:
        actions = [json.loads(l) for l in f]
    assert any("protocol" in a for a in actions)
    meta = next(a["metaData"] for a in actions if "metaData" in a)
    fields = {f["name"]: f["type"] for f in json.loads(meta["schemaString"])["fields"]}
    assert fields["a"] == "long" and fields["b"] == "string"
    assert fields["diff"] == "long"
    # at least one add commit with a parquet file
    adds = []
    for lf in log[1:]:
        with open(os.path.join(root, "_delta_log", lf)) as f:
            adds += [json.loads(l) for l in f if '"add"' in l]
    assert adds
    assert os.path.exists(os.path.join(root, adds[0]["add"]["path"]))


def test_delta_roundtrip(tmp_path):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        3 | z
        """
    )
    root = str(tmp_path / "dt")
    pw.io.deltalake.write(t, root)
    pw.run()

    G.clear()
    back = pw.io.deltalake.read(
        root, schema=schema_from_types(a=int, b=str), mode="static"
    )
    keys, cols = pw.debug.table_to_dicts(back)
    assert sorted((cols["a"][k], cols["b"][k]) for k in keys) == [
        (1, "x"), (2, "y"), (3, "z")
    ]


def test_delta_streaming_new_commits(tmp_path):
    import pyarrow as pa_
    import pyarrow.parquet as pq

    root = str(tmp_path / "dt")
    from pathway_amd.io.deltalake import DeltaTableWriter

    schema = schema_from_types(a=int)
    w = DeltaTableWriter(root, ["a"], schema)

    class FakeBatch:
        def __init__(self, vals, t):
            self.time = t
            self.columns = {"a": None}
            self._vals = vals

        def rows(self):
            for v in self._vals:
                yield None, [v], self.time, 1

    w(FakeBatch([1, 2], 0))

    def later():
        time.sleep(0.3)
        w(FakeBatch([3], 2))

    th = threading.Thread(target=later)
    th.start()
    G.clear()
    t = pw.io.deltalake.read(
        root, schema=schema, mode="streaming", refresh_interval=0.1,
        _max_polls=12,
    )
    res = t.groupby().reduce(s=pw.reducers.sum(pw.this.a), c=pw.reducers.count())
    keys, cols = pw.debug.table_to_dicts(res)
    th.join()
    assert list(cols["s"].values()) == [6]
    assert list(cols["c"].values()) == [3]


def test_delta_remove_action_retracts(tmp_path):
    root = str(tmp_path / "dt")
    from pathway_amd.io.deltalake import DeltaTableWriter, _list_versions, _log_path

    schema = schema_from_types(a=int)
    w = DeltaTableWriter(root, ["a"], schema)

    class FakeBatch:
        def __init__(self, vals, t):
            self.time = t
            self.columns = {"a": None}
            self._vals = vals

        def rows(self):
            for v in self._vals:
                yield None, [v], self.time, 1

    w(FakeBatch([1, 2], 0))
    w(FakeBatch([10], 2))
    # find the first data file and write a remove action for it
    import json as _json

    with open(_log_path(root, 1)) as f:
        add = next(_json.loads(l)["add"] for l in f if '"add"' in l)
    w._commit([{"remove": {"path": add["path"], "dataChange": True,
                           "deletionTimestamp": int(time.time() * 1000)}}])

    G.clear()
    t = pw.io.deltalake.read(root, schema=schema, mode="static")
    keys, cols = pw.debug.table_to_dicts(t)
    assert sorted(cols["a"].values()) == [10]
