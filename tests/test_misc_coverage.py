"""Coverage batch: timezone conversions, parquet round-trip,
parse_to_table, compute_and_print, intervals_over inner, exactly-once
shift, CLI replay surface."""

import datetime
import io
import sys

import pytest

import pathway_amd as pw
from pathway_amd.debug import (
    compute_and_print,
    parse_to_table,
    table_from_markdown as T,
    table_from_parquet,
    table_from_rows,
    table_to_dicts,
    table_to_parquet,
)
from pathway_amd.internals.schema import schema_from_types


@pytest.fixture(autouse=True)
def _clean():
    yield
    pw.internals.rungraph.G.clear()


def test_timezone_conversions():
    t = table_from_rows(schema_from_types(s=str), [("2023-03-25 12:00:00",)])
    d = t.select(naive=pw.this.s.dt.strptime("%Y-%m-%d %H:%M:%S"))
    res = d.select(
        utc=pw.this.naive.dt.to_utc("Europe/Warsaw"),
    )
    back = res.select(
        warsaw=pw.this.utc.dt.to_naive_in_timezone("Europe/Warsaw"),
    )
    _, cols = table_to_dicts(back)
    (w,) = cols["warsaw"].values()
    assert str(w).startswith("2023-03-25 12:00")


def test_parquet_roundtrip(tmp_path):
    t = T(
        """
        a | b
        1 | x
        2 | y
        """
    )
    p = str(tmp_path / "t.parquet")
    table_to_parquet(t, p)
    pw.internals.rungraph.G.clear()
    back = table_from_parquet(p)
    _, cols = table_to_dicts(back)
    assert sorted(cols["a"].values()) == [1, 2]
    assert sorted(cols["b"].values()) == ["x", "y"]


def test_parse_to_table_alias():
    t = parse_to_table(
        """
        v
        7
        """
    )
    _, cols = table_to_dicts(t)
    assert list(cols["v"].values()) == [7]


def test_compute_and_print_smoke(capsys):
    t = T(
        """
        a
        5
        """
    )
    compute_and_print(t)
    out = capsys.readouterr().out
    assert "5" in out and "a" in out


def test_intervals_over_inner():
    t = T(
        """
        t | v
        1 | 10
        4 | 20
        9 | 30
        """
    )
    at = T(
        """
        p
        4
        100
        """
    )
    res = t.windowby(
        t.t,
        window=pw.temporal.intervals_over(
            at=at.p, lower_bound=-3, upper_bound=3, is_outer=False
        ),
    ).reduce(p=pw.this._pw_window_location, s=pw.reducers.sum(pw.this.v))
    _, cols = table_to_dicts(res)
    got = sorted(zip(cols["p"].values(), cols["s"].values()))
    # inner: the at=100 point with no rows in range produces NO window
    assert got == [(4, 30)]


def test_exactly_once_shift():
    t = T(
        """
        t | v | __time__
        1 | 1 |    2
        6 | 1 |    8
        """
    )
    res = t.windowby(
        t.t,
        window=pw.temporal.tumbling(duration=5),
        behavior=pw.temporal.exactly_once_behavior(shift=2),
    ).reduce(start=pw.this._pw_window_start, n=pw.reducers.count())
    _, cols = table_to_dicts(res)
    # window [0,5) closes when watermark >= 5+2=7: t=6 isn't enough, so
    # nothing is emitted for it yet; [5,10) stays open too
    got = sorted(zip(cols["start"].values(), cols["n"].values()))
    assert got == []


def test_cli_replay_surface():
    from pathway_amd.cli import main as cli_main

    # `pathway_amd replay --help` exits 0 (argparse SystemExit)
    old = sys.argv
    sys.argv = ["pathway_amd", "replay", "--help"]
    try:
        with pytest.raises(SystemExit) as e:
            cli_main()
        assert e.value.code == 0
    finally:
        sys.argv = old


def test_bin_namespace_exists():
    t = T(
        """
        a
        5
        """
    )
    e = pw.this.a.bin
    assert e is not None


def test_window_with_datetime_durations():
    import pandas as pd

    t = table_from_rows(schema_from_types(s=str, v=int), [
        ("2023-01-01 00:00:10", 1),
        ("2023-01-01 00:00:50", 2),
        ("2023-01-01 00:02:10", 3),
    ])
    d = t.select(ts=pw.this.s.dt.strptime("%Y-%m-%d %H:%M:%S"), v=pw.this.v)
    res = d.windowby(
        d.ts, window=pw.temporal.tumbling(duration=pd.Timedelta(minutes=1))
    ).reduce(n=pw.reducers.count(), s=pw.reducers.sum(pw.this.v))
    _, cols = table_to_dicts(res)
    got = sorted(zip(cols["n"].values(), cols["s"].values()))
    assert got == [(1, 3), (2, 3)]


def test_join_with_instances():
    l = T(
        """
        k | inst | a
        1 |  x   | p
        1 |  y   | q
        """
    )
    r = T(
        """
        k | inst | b
        1 |  x   | u
        1 |  y   | v
        """
    )
    res = l.join(
        r, l.k == r.k, left_instance=l.inst, right_instance=r.inst
    ).select(pw.left.a, pw.right.b)
    _, cols = table_to_dicts(res)
    got = sorted(zip(cols["a"].values(), cols["b"].values()))
    # instance colocation: only same-instance pairs match
    assert got == [("p", "u"), ("q", "v")]


def test_flatten_with_extra_columns():
    t = T(
        """
        g
        a
        """
    )
    lt = t.select(pw.this.g, items=pw.make_tuple(1, 2, 3))
    flat = lt.flatten(pw.this.items)
    _, cols = table_to_dicts(flat)
    assert sorted(cols["items"].values()) == [1, 2, 3]
    assert set(cols["g"].values()) == {"a"}


def test_gated_connectors_raise_helpfully():
    # the require_client gate (still used by connectors whose service has
    # no offline-implementable transport) raises a descriptive error
    from pathway_amd.io._utils import MissingServiceDependency, require_client

    with pytest.raises(MissingServiceDependency) as e:
        require_client("definitely_not_installed_xyz", "stub")
    assert "client library" in str(e.value)


def test_sharepoint_surface_importable():
    from pathway_amd.xpacks.connectors import sharepoint

    assert hasattr(sharepoint, "read")


def test_schema_from_csv(tmp_path):
    from pathway_amd.internals.schema import schema_from_csv

    p = tmp_path / "s.csv"
    p.write_text("name,qty,price\nwidget,3,1.5\n")
    S = schema_from_csv(str(p))
    from pathway_amd.internals import dtype as dt

    assert S.__columns__["name"].dtype == dt.STR
    assert S.__columns__["qty"].dtype == dt.INT
    assert S.__columns__["price"].dtype == dt.FLOAT


def test_fs_static_with_metadata(tmp_path):
    d = tmp_path / "docs"
    d.mkdir()
    (d / "x.txt").write_bytes(b"abc")
    t = pw.io.fs.read(str(d), format="binary", mode="static", with_metadata=True)
    _, cols = table_to_dicts(t)
    (meta,) = cols["_metadata"].values()
    mv = meta.value if hasattr(meta, "value") else meta
    assert mv["path"].endswith("x.txt")
    assert mv["size"] == 3


def test_run_monitoring_in_out(capsys):
    t = T(
        """
        a
        1
        """
    )
    import tempfile, os as _os

    out = tempfile.mktemp(suffix=".csv")
    pw.io.csv.write(t, out)
    pw.run(monitoring_level=pw.MonitoringLevel.IN_OUT)
    assert _os.path.exists(out)
    _os.unlink(out)


def test_http_polling_read():
    import json as _json
    import threading
    from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

    class H(BaseHTTPRequestHandler):
        def do_GET(self):
            body = _json.dumps({"v": 42}).encode()
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    httpd = ThreadingHTTPServer(("127.0.0.1", 0), H)
    port = httpd.server_address[1]
    th = threading.Thread(target=httpd.serve_forever, daemon=True)
    th.start()
    try:
        t = pw.io.http.read(
            f"http://127.0.0.1:{port}/",
            schema=schema_from_types(v=int),
            refresh_interval_ms=50,
            n_polls=2,
        )
        r = t.reduce(n=pw.reducers.count(), last=pw.reducers.latest(pw.this.v))
        _, cols = table_to_dicts(r)
        assert list(cols["last"].values()) == [42]
        assert list(cols["n"].values())[0] >= 1
    finally:
        httpd.shutdown()


def test_serialize_value_canonical_invariants():
    from pathway_amd.internals.api import hash_values, serialize_value

    # distinct types with "equal-looking" payloads hash differently
    pairs = [
        (1, 1.0),
        (1, True),
        ("1", 1),
        (b"1", "1"),
    ]
    # lists and tuples intentionally serialize identically (both map to
    # the reference's Value::Tuple)
    from pathway_amd.internals.api import serialize_value as _sv

    assert _sv((1, 2)) == _sv([1, 2])
    for a, b in pairs:
        if type(a) is type(b):
            continue
        sa, sb = serialize_value(a), serialize_value(b)
        assert sa != sb, (a, b)
    # stability: same value, same bytes across calls
    for v in [None, True, 7, -3.25, "text", b"bytes", (1, "x"), 10**18]:
        assert serialize_value(v) == serialize_value(v)
    # hash_values is order sensitive
    assert hash_values([1, 2]) != hash_values([2, 1])


def test_update_cells_fuzz_vs_pandas():
    import random as _r

    import pandas as pd

    for seed in range(5):
        rng = _r.Random(17000 + seed)
        base_rows = [(k, rng.randint(0, 9), rng.randint(0, 9)) for k in range(6)]
        upd_keys = rng.sample(range(6), 3)
        upd_rows = [(k, rng.randint(100, 109)) for k in upd_keys]
        pw.internals.rungraph.G.clear()
        base_md = ["id | x | y"] + [f"{k} | {x} | {y}" for k, x, y in base_rows]
        upd_md = ["id | x"] + [f"{k} | {x}" for k, x in upd_rows]
        base = T("\n".join(base_md))
        upd = T("\n".join(upd_md))
        res = base.update_cells(upd)
        _, cols = table_to_dicts(res)
        got = sorted(zip(cols["x"].values(), cols["y"].values()))
        df = pd.DataFrame(base_rows, columns=["k", "x", "y"]).set_index("k")
        for k, x in upd_rows:
            df.loc[k, "x"] = x
        expected = sorted(zip(df["x"].tolist(), df["y"].tolist()))
        assert got == expected, f"seed {seed}: {got} vs {expected}"


def test_select_splat_this():
    t = T(
        """
        a | b
        1 | 2
        """
    )
    r = t.select(*pw.this, c=pw.this.a + pw.this.b)
    _, cols = table_to_dicts(r)
    assert sorted(cols) == ["a", "b", "c"]
    assert list(cols["c"].values()) == [3]


def test_join_select_splat_left():
    l = T(
        """
        k | a
        1 | p
        """
    )
    r = T(
        """
        k | b
        1 | u
        """
    )
    res = l.join(r, l.k == r.k).select(*pw.left, b=pw.right.b)
    _, cols = table_to_dicts(res)
    assert sorted(cols) == ["a", "b", "k"]


def test_select_splat_without():
    t = T(
        """
        a | b | c
        1 | 2 | 3
        """
    )
    r = t.select(*pw.this.without(pw.this.a))
    _, cols = table_to_dicts(r)
    assert sorted(cols) == ["b", "c"]


def test_yaml_loader_instantiates_objects():
    import io as _io

    yml = """
$chat: !pw.xpacks.llm.llms.EchoChat
  prefix: "A> "
answerer:
  llm: $chat
  topk: 3
"""
    out = pw.load_yaml(_io.StringIO(yml))
    assert type(out["answerer"]["llm"]).__name__ == "EchoChat"
    assert out["answerer"]["topk"] == 3


def test_compute_and_print_update_stream(capsys):
    from pathway_amd.debug import compute_and_print_update_stream

    t = T(
        """
        a | __time__ | __diff__
        1 |    2     |    1
        1 |    4     |   -1
        2 |    4     |    1
        """,
        id_from=["a"],
    )
    compute_and_print_update_stream(t)
    out = capsys.readouterr().out
    assert "-1" in out and "2" in out


def test_universe_solver_algebra():
    """Relational universe reasoner (reference universe_solver.py SAT
    encoding): consequences of union/intersection/difference promises
    are derivable; unrelated universes are not conflated."""
    from pathway_amd.internals.universe import (
        Universe,
        promise_are_pairwise_disjoint,
        promise_is_subset_of,
    )

    a, b, w = Universe(), Universe(), Universe()
    promise_is_subset_of(a, w)
    promise_is_subset_of(b, w)
    u = Universe.union_of(a, b)
    # parts ⊆ union; union ⊆ any common superset
    assert a.is_subset_of(u) and b.is_subset_of(u)
    assert u.is_subset_of(w)
    assert not w.is_subset_of(u)

    i = Universe.intersection_of(a, b)
    assert i.is_subset_of(a) and i.is_subset_of(b)
    x = Universe()
    promise_is_subset_of(x, a)
    promise_is_subset_of(x, b)
    # x ⊆ a and x ⊆ b  ->  x ⊆ a∩b
    assert x.is_subset_of(i)

    d = Universe.difference_of(a, b)
    assert d.is_subset_of(a)
    assert d.is_disjoint_from(b)
    y = Universe()
    promise_is_subset_of(y, b)
    assert d.is_disjoint_from(y)

    p, q = Universe(), Universe()
    promise_are_pairwise_disjoint(p, q)
    sp, sq = Universe(parent=p), Universe(parent=q)
    assert sp.is_disjoint_from(sq)
    # soundness: no invented facts
    assert not p.is_subset_of(q)
    assert not p.is_equal(q)


def test_universe_solver_table_setops():
    import pathway_amd as pw
    from pathway_amd.internals.rungraph import G

    G.clear()
    t = pw.debug.table_from_markdown("a\n1\n2\n3\n")
    evens = t.filter(pw.this.a % 2 == 0)
    odds = t.filter(pw.this.a % 2 == 1)
    u = evens.concat(odds)
    # concat's universe is the union: contained in the source superset
    assert u._universe.is_subset_of(t._universe)
    inter = evens.intersect(t)
    assert inter._universe.is_subset_of(evens._universe)
    assert inter._universe.is_subset_of(t._universe)
    diff = t.difference(evens)
    assert diff._universe.is_subset_of(t._universe)


def test_license_ed25519_and_entitlements():
    """License keys (reference license.rs): pure-python RFC 8032
    ed25519 sign/verify, entitlement parsing, MAX_WORKERS cap."""
    from pathway_amd.internals import license as lic

    # RFC 8032 test vector 1 (empty message)
    seed = bytes.fromhex(
        "9d61b19deffd5a60ba844af492ec2cc44449c5697b326919703bac031cae7f60"
    )
    pub = lic.ed25519_public_key(seed)
    assert pub.hex() == (
        "d75a980182b10ab7d54bfed3c964073a0ee172f3daa62325af021a68f707511a"
    )
    sig = lic.ed25519_sign(seed, b"")
    assert sig.hex() == (
        "e5564300c360ac729086e2cc806e828a84877f1eb8e5d974d873e06522490155"
        "5fb8821590a33bacc61e39701cf9b46bd25bf5f0595bbe24655141438e7a100b"
    )
    assert lic.ed25519_verify(pub, b"", sig)
    assert not lic.ed25519_verify(pub, b"x", sig)

    key = lic.issue_key(["unlimited-workers", "monitoring"])
    parsed = lic.parse_key(key)
    assert parsed.valid and parsed.has("unlimited-workers")
    assert parsed.max_workers() is None
    # free tier: cap 8
    assert lic.parse_key(None).max_workers() == 8
    assert lic.parse_key("garbage").max_workers() == 8
    # a tampered key falls back to free
    bad = key[:-6] + "AAAAAA"
    assert not lic.parse_key(bad).valid

    lic.check_worker_limit(8, None)
    with pytest.raises(RuntimeError):
        lic.check_worker_limit(9, None)
    lic.check_worker_limit(64, key)


def test_gradual_broadcast_apportioning():
    """Reference gradual_broadcast.rs:120-190: rows whose key < max_key *
    (value-lower)/(upper-lower) receive `upper`, the rest `lower`; the
    fraction at `upper` tracks the requested value over uniform keys."""
    from pathway_amd.internals.rungraph import G

    G.clear()
    n = 400
    rows = table_from_rows(
        schema_from_types(x=int), [(i,) for i in range(n)]
    )
    thr = table_from_rows(
        schema_from_types(lo=float, v=float, hi=float), [(0.0, 0.25, 1.0)]
    )
    res = rows._gradual_broadcast(thr, thr.lo, thr.v, thr.hi)
    _keys, cols = pw.debug.table_to_dicts(res)
    vals = list(cols["apx_value"].values())
    assert set(vals) <= {0.0, 1.0}
    frac = sum(1 for v in vals if v == 1.0) / n
    # hashes are uniform: the upper-share must track (v-lo)/(hi-lo)
    assert 0.15 < frac < 0.35
    # degenerate interval: everything gets the collapsed bound
    G.clear()
    rows2 = table_from_rows(schema_from_types(x=int), [(i,) for i in range(5)])
    thr2 = table_from_rows(
        schema_from_types(lo=float, v=float, hi=float), [(2.0, 2.0, 2.0)]
    )
    res2 = rows2._gradual_broadcast(thr2, thr2.lo, thr2.v, thr2.hi)
    _k2, cols2 = pw.debug.table_to_dicts(res2)
    assert set(cols2["apx_value"].values()) == {2.0}
