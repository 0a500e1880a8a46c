"""Format round-trips and the demo module (reference io format tests +
pathway.demo)."""

import json
import os

import pytest

import pathway_amd as pw
from pathway_amd.debug import table_from_markdown as T, table_to_dicts


@pytest.fixture(autouse=True)
def _clean_graph():
    yield
    pw.internals.rungraph.G.clear()


def _col_sorted(table, name):
    _, cols = table_to_dicts(table)
    return sorted(cols[name].values())


def test_csv_write_read_roundtrip(tmp_path):
    t = T(
        """
        a | b
        1 | x
        2 | y
        """
    )
    out = str(tmp_path / "out.csv")
    pw.io.csv.write(t, out)
    pw.run(monitoring_level=pw.MonitoringLevel.NONE)
    pw.internals.rungraph.G.clear()

    class S(pw.Schema):
        a: int
        b: str

    back = pw.io.csv.read(out, schema=S, mode="static")
    assert _col_sorted(back, "a") == [1, 2]
    assert _col_sorted(back, "b") == ["x", "y"]


def test_csv_read_schema_inference(tmp_path):
    p = tmp_path / "in.csv"
    p.write_text("x,y\n1,1.5\n2,2.5\n")
    t = pw.io.csv.read(str(p), mode="static")
    assert _col_sorted(t, "x") == [1, 2]
    assert _col_sorted(t, "y") == [1.5, 2.5]


def test_jsonlines_write_read_roundtrip(tmp_path):
    t = T(
        """
        a | b
        5 | p
        6 | q
        """
    )
    out = str(tmp_path / "out.jsonl")
    pw.io.jsonlines.write(t, out)
    pw.run(monitoring_level=pw.MonitoringLevel.NONE)
    pw.internals.rungraph.G.clear()

    recs = [json.loads(l) for l in open(out) if l.strip()]
    assert sorted(r["a"] for r in recs) == [5, 6]
    assert all(r["diff"] == 1 for r in recs)

    class S(pw.Schema):
        a: int
        b: str

    back = pw.io.jsonlines.read(out, schema=S, mode="static")
    assert _col_sorted(back, "a") == [5, 6]


def test_plaintext_read(tmp_path):
    p = tmp_path / "f.txt"
    p.write_text("alpha\nbeta\n")
    t = pw.io.plaintext.read(str(p), mode="static")
    assert _col_sorted(t, "data") == ["alpha", "beta"]


def test_fs_read_binary(tmp_path):
    p = tmp_path / "blob.bin"
    p.write_bytes(b"\x00\x01payload")
    t = pw.io.fs.read(str(p), format="binary", mode="static")
    assert _col_sorted(t, "data") == [b"\x00\x01payload"]


def test_demo_range_stream():
    t = pw.demo.range_stream(nb_rows=5)
    total = t.reduce(s=pw.reducers.sum(pw.this.value), n=pw.reducers.count())
    _, cols = table_to_dicts(total)
    assert list(cols["s"].values()) == [0 + 1 + 2 + 3 + 4]
    assert list(cols["n"].values()) == [5]


def test_demo_noisy_linear_stream():
    t = pw.demo.noisy_linear_stream(nb_rows=10)
    r = t.reduce(n=pw.reducers.count())
    _, cols = table_to_dicts(r)
    assert list(cols["n"].values()) == [10]


def test_demo_replay_csv(tmp_path):
    p = tmp_path / "in.csv"
    p.write_text("v\n3\n4\n5\n")

    class S(pw.Schema):
        v: int

    t = pw.demo.replay_csv(str(p), schema=S, input_rate=1e6)
    r = t.reduce(s=pw.reducers.sum(pw.this.v))
    _, cols = table_to_dicts(r)
    assert list(cols["s"].values()) == [12]


def test_csv_write_appends_updates(tmp_path):
    # an update stream writes retractions with diff == -1
    t = T(
        """
        g | v | __time__ | __diff__
        a | 1 |    2     |    1
        a | 1 |    4     |   -1
        a | 2 |    4     |    1
        """,
        id_from=["g"],
    )
    out = str(tmp_path / "upd.csv")
    pw.io.csv.write(t, out)
    pw.run(monitoring_level=pw.MonitoringLevel.NONE)
    lines = [l for l in open(out).read().splitlines() if l]
    # header + 3 events
    assert len(lines) == 4
    diffs = sorted(l.rsplit(",", 1)[1] for l in lines[1:])
    assert diffs == ["-1", "1", "1"]


def test_native_csv_scanner_matches_python_csv(tmp_path):
    import csv as _csv
    import io as _io

    from pathway_amd.ops import native_io

    if not native_io.available():
        pytest.skip("libpwio unavailable")
    content = (
        'a,b,c\n'
        '1,"x,y",3\n'
        '2,"he said ""hi""",4\n'
        '5,"multi\nline",6\n'
        '7,plain,9\n'
    )
    p = tmp_path / "t.csv"
    p.write_text(content)
    header, rows = native_io.read_csv(str(p))
    ref = list(_csv.reader(_io.StringIO(content)))
    assert header == ref[0]
    assert rows == ref[1:]


def test_native_scanner_used_by_csv_read(tmp_path):
    p = tmp_path / "q.csv"
    p.write_text('name,qty\n"widget, large",2\nbolt,3\n')

    class S(pw.Schema):
        name: str
        qty: int

    t = pw.io.csv.read(str(p), schema=S, mode="static")
    assert _col_sorted(t, "name") == ["bolt", "widget, large"]
    assert _col_sorted(t, "qty") == [2, 3]


def test_native_read_lines(tmp_path):
    from pathway_amd.ops import native_io

    if not native_io.available():
        pytest.skip("libpwio unavailable")
    p = tmp_path / "l.txt"
    p.write_bytes(b"one\r\ntwo\nthree")
    assert native_io.read_lines(str(p)) == ["one", "two", "three"]


def test_static_reads_expand_globs(tmp_path):
    (tmp_path / "a.txt").write_text("one\n")
    (tmp_path / "b.txt").write_text("two\n")
    (tmp_path / "c.log").write_text("skip\n")
    t = pw.io.fs.read(str(tmp_path / "*.txt"), format="plaintext", mode="static")
    assert _col_sorted(t, "data") == ["one", "two"]
