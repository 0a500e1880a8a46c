"""Streaming connector tests: live fs polling + python subjects."""

import os
import threading
import time

import pytest

import pathway_amd as pw
from pathway_amd.internals.schema import schema_from_types


@pytest.mark.timeout(120)
def test_fs_streaming_picks_up_new_files(tmp_path):
    d = tmp_path / "in"
    d.mkdir()
    (d / "one.txt").write_text("hello\nworld\n")

    def later():
        time.sleep(0.4)
        (d / "two.txt").write_text("second\n")

    th = threading.Thread(target=later)
    th.start()
    t = pw.io.fs.read(
        str(d), format="plaintext", mode="streaming",
        refresh_interval=0.1, _max_polls=10,
    )
    res = t.groupby().reduce(c=pw.reducers.count())
    keys, cols = pw.debug.table_to_dicts(res)
    th.join()
    assert list(cols["c"].values()) == [3]


@pytest.mark.timeout(120)
def test_python_connector_streaming():
    class Subject(pw.io.python.ConnectorSubject):
        def run(self):
            for i in range(5):
                self.next(v=i)
                time.sleep(0.01)

    schema = schema_from_types(v=int)
    t = pw.io.python.read(Subject(), schema=schema)
    res = t.groupby().reduce(s=pw.reducers.sum(pw.this.v), c=pw.reducers.count())
    keys, cols = pw.debug.table_to_dicts(res)
    assert list(cols["s"].values()) == [10]
    assert list(cols["c"].values()) == [5]
