"""Misc stdlib + io coverage: pw.io.subscribe, statistical.interpolate,
utils.col/filtering, monitoring HTTP server (reference misc suites)."""

import json
import time
import urllib.request

import pytest

import pathway_amd as pw
from pathway_amd.debug import table_from_markdown as T, table_from_rows, table_to_dicts
from pathway_amd.internals.schema import schema_from_types


@pytest.fixture(autouse=True)
def _clean_graph():
    yield
    pw.internals.rungraph.G.clear()


def test_io_subscribe_callbacks():
    t = T(
        """
        a | __time__ | __diff__
        1 |    2     |    1
        2 |    4     |    1
        1 |    6     |   -1
        """,
        id_from=["a"],
    )
    events = []
    done = []
    pw.io.subscribe(
        t,
        on_change=lambda key, row, time, is_addition: events.append(
            (row["a"], time, is_addition)
        ),
        on_end=lambda: done.append(True),
    )
    pw.run(monitoring_level=pw.MonitoringLevel.NONE)
    assert (1, 2, True) in events and (2, 4, True) in events
    assert (1, 6, False) in events
    assert done == [True]


def test_statistical_interpolate():
    from pathway_amd.stdlib.statistical import interpolate

    t = T(
        """
        t  | v
        0  | 0.0
        10 | 100.0
        """
    )
    q = T(
        """
        t
        5
        """
    )
    all_t = pw.Table.concat_reindex(
        t.select(pw.this.t, v=pw.cast(float, pw.this.v)),
        q.select(pw.this.t, v=pw.declare_type(float, None)),
    )
    res = interpolate(all_t, pw.this.t, pw.this.v)
    _, cols = table_to_dicts(res)
    vals = sorted(v for v in cols["v"].values() if v is not None)
    assert 50.0 in vals  # linear midpoint filled in


def test_utils_col_unpack_col():
    from pathway_amd.stdlib.utils.col import unpack_col

    t = T(
        """
        a | b
        1 | 2
        """
    )
    packed = t.select(tup=pw.make_tuple(pw.this.a, pw.this.b))
    un = unpack_col(packed.tup, "x", "y")
    _, cols = table_to_dicts(un)
    assert list(cols["x"].values()) == [1]
    assert list(cols["y"].values()) == [2]


def test_utils_filtering_argmin_rows():
    from pathway_amd.stdlib.utils.filtering import argmin_rows

    t = T(
        """
        g | v | extra
        a | 3 | p
        a | 1 | q
        b | 7 | r
        """
    )
    res = argmin_rows(t, pw.this.g, what=pw.this.v)
    _, cols = table_to_dicts(res)
    got = sorted(zip(cols["g"].values(), cols["v"].values(), cols["extra"].values()))
    assert got == [("a", 1, "q"), ("b", 7, "r")]


def test_bucketing_helpers():
    import datetime

    from pathway_amd.stdlib.utils.bucketing import (
        truncate_to_hours,
        truncate_to_minutes,
    )

    d = datetime.datetime(2023, 3, 25, 12, 34, 56, 789)
    assert truncate_to_minutes(d) == datetime.datetime(2023, 3, 25, 12, 34)
    assert truncate_to_hours(d) == datetime.datetime(2023, 3, 25, 12)


def test_monitoring_status_endpoint():
    from pathway_amd.engine.monitoring import RunStats, start_http_server

    stats = RunStats()
    stats.record_step(2, 0.001, 10, 5)
    httpd = start_http_server(stats, port=0)
    try:
        port = httpd.server_address[1]
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/status") as r:
            st = json.loads(r.read())
        assert st["rows_ingested"] == 10 and st["rows_output"] == 5
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/metrics") as r:
            body = r.read().decode()
        assert "pathway_rows_ingested_total 10" in body
    finally:
        httpd.shutdown()


def test_deduplicate_with_acceptor():
    t = T(
        """
        g | v | __time__
        a | 1 |    2
        a | 3 |    4
        a | 2 |    6
        a | 7 |    8
        """
    )
    # accept a new value only if it is at least 2 bigger than the last kept
    res = t.deduplicate(
        value=pw.this.v,
        instance=pw.this.g,
        acceptor=lambda new, old: new >= old + 2,
    )
    _, cols = table_to_dicts(res)
    assert list(cols["v"].values()) == [7]  # 1 -> 3 -> (2 rejected) -> 7; last state


def test_ordered_diff():
    from pathway_amd.stdlib.ordered import diff

    t = T(
        """
        t | v
        1 | 10
        2 | 13
        3 | 11
        """
    )
    res = diff(t, t.t, t.v)
    _, cols = table_to_dicts(res)
    by_t = {}
    keys, _ = table_to_dicts(t)
    # diff returns per-row deltas vs the previous row in t-order
    vals = sorted(v for v in cols["diff_v"].values() if v is not None)
    assert vals == [-2, 3]


def test_async_transformer_class():
    from pathway_amd.stdlib.utils.async_transformer import AsyncTransformer

    class S(pw.Schema):
        doubled: int
        tag: str

    class MyT(AsyncTransformer):
        output_schema = S

        async def invoke(self, a) -> dict:
            import asyncio as _a

            await _a.sleep(0.001)
            return {"doubled": 2 * a, "tag": f"row{a}"}

    t = T(
        """
        a
        1
        2
        """
    )
    res = MyT(input_table=t).successful
    _, cols = table_to_dicts(res)
    assert sorted(cols["doubled"].values()) == [2, 4]
    assert sorted(cols["tag"].values()) == ["row1", "row2"]


def test_pandas_transformer():
    import pandas as pd

    from pathway_amd.stdlib.utils.pandas_transformer import pandas_transformer

    class Out(pw.Schema):
        total: int

    @pandas_transformer(output_schema=Out)
    def column_sum(df: pd.DataFrame) -> pd.DataFrame:
        return pd.DataFrame({"total": [int(df["v"].sum())]})

    t = T(
        """
        v
        3
        4
        5
        """
    )
    res = column_sum(t)
    _, cols = table_to_dicts(res)
    assert list(cols["total"].values()) == [12]


def test_mcp_server_surface():
    from pathway_amd.xpacks.llm.mcp_server import McpServable, McpServer

    assert hasattr(McpServer, "tool") or hasattr(McpServer, "serve") or McpServable


def test_export_import_table_cross_graph():
    """Cross-graph handoff (reference graph.rs:616-646 ExportedTable)."""
    import pathway_amd as pw
    from pathway_amd.internals.rungraph import G

    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        3 | x
        """
    )
    agg = t.groupby(pw.this.b).reduce(pw.this.b, s=pw.reducers.sum(pw.this.a))
    handle = pw.export_table(agg)
    pw.run()
    assert not handle.failed()
    assert handle.frontier() >= 1
    snap = handle.snapshot_at()
    assert sorted((v[0], v[1]) for v in snap.values()) == [("x", 4), ("y", 2)]

    # second graph: import and continue computing
    G.clear()
    back = pw.import_table(handle)
    doubled = back.select(pw.this.b, d=pw.this.s * 2)
    keys, cols = pw.debug.table_to_dicts(doubled)
    assert sorted(zip(cols["b"].values(), cols["d"].values())) == [
        ("x", 8), ("y", 4)
    ]


def test_export_import_preserves_retraction_stream():
    import pathway_amd as pw
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.rungraph import G
    from pathway_amd.internals.schema import schema_from_types

    G.clear()
    t = table_from_rows(
        schema_from_types(v=int), [(1, 0, 1), (2, 2, 1)], is_stream=True
    )
    total = t.reduce(s=pw.reducers.sum(pw.this.v))
    handle = pw.export_table(total)
    pw.run()
    rows, _ = handle.data_from_offset(0)
    # incremental stream: +1 at t=0, then -1/+3 at t=2
    seq = sorted((r.time, r.diff, r.values[0]) for r in rows)
    assert seq == [(0, 1, 1), (2, -1, 1), (2, 1, 3)]

    G.clear()
    back = pw.import_table(handle)
    keys, cols = pw.debug.table_to_dicts(back)
    assert list(cols["s"].values()) == [3]


def test_async_transformer_views_and_options():
    """Deepened AsyncTransformer (reference async_transformer.rs:297):
    concurrent invoke via the async UDF executor; successful/failed
    views split on per-row errors."""
    import asyncio

    import pathway_amd as pw
    from pathway_amd.internals.rungraph import G
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.stdlib.utils.async_transformer import AsyncTransformer

    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a
        1
        2
        3
        4
        """
    )
    calls = []

    class MyT(AsyncTransformer):
        output_schema = schema_from_types(doubled=int)

        async def invoke(self, a):
            calls.append(a)
            await asyncio.sleep(0.01)
            if a == 3:
                raise ValueError("boom")
            return {"doubled": a * 2}

    tr = MyT(input_table=t).with_options(capacity=4)
    ok = tr.successful
    bad = tr.failed
    keys, cols = pw.debug.table_to_dicts(ok)
    assert sorted(cols["doubled"].values()) == [2, 4, 8]
    G.clear()
    tr2 = MyT(input_table=pw.debug.table_from_markdown("a\n3\n5\n"))
    # failed rows carry Error values; count them without reading values
    nfail = tr2.failed.reduce(c=pw.reducers.count())
    _, cols2 = pw.debug.table_to_dicts(nfail)
    assert list(cols2["c"].values()) == [1]  # only a==3 failed


def test_viz_live_plot():
    """Native live plot (reference stdlib/viz): subscriber state + SVG
    renderer + HTTP endpoint."""
    import json
    import urllib.request

    import pathway_amd as pw
    from pathway_amd.internals.rungraph import G
    from pathway_amd.stdlib.viz import plot, render_svg

    svg = render_svg([1.0, 3.0, 2.0])
    assert svg.startswith("<svg") and "polyline" in svg

    G.clear()
    t = pw.debug.table_from_markdown(
        """
        v | label
        1 | a
        5 | b
        3 | c
        """
    )
    lp = plot(t, value_column="v", serve=True)
    pw.run()
    port = lp.server.server_address[1]
    with urllib.request.urlopen(f"http://127.0.0.1:{port}/data", timeout=10) as r:
        data = json.loads(r.read())
    assert data["columns"] == ["v", "label"]
    assert len(data["rows"]) == 3
    assert "polyline" in data["svg"]
    with urllib.request.urlopen(f"http://127.0.0.1:{port}/", timeout=10) as r:
        page = r.read().decode()
    assert "pathway table" in page
    lp.server.shutdown()


def test_detailed_metrics_and_dashboard(tmp_path):
    """set_monitoring_config(detailed_metrics_dir=...) -> metrics.db;
    the web dashboard serves /api/history from it (reference
    integration_tests/monitoring behavior)."""
    import sqlite3

    import pathway_amd as pw
    from pathway_amd.internals.config import pathway_config
    from pathway_amd.internals.rungraph import G

    mdir = str(tmp_path / "metrics")
    pw.set_monitoring_config(detailed_metrics_dir=mdir)
    try:
        G.clear()
        t = T(
            """
            a
            1
            2
            """
        )
        out = str(tmp_path / "o.csv")
        pw.io.csv.write(t, out)
        pw.run()
        db = sqlite3.connect(f"{mdir}/metrics.db")
        runs = db.execute("SELECT COUNT(*) FROM run_metrics").fetchone()[0]
        ops = db.execute(
            "SELECT COUNT(DISTINCT operator) FROM operator_metrics"
        ).fetchone()[0]
        db.close()
        assert runs >= 1 and ops >= 1

        from starlette.testclient import TestClient

        from pathway_amd.web_dashboard.dashboard import create_app

        app = create_app(detailed_metrics_dir=mdir)
        client = TestClient(app)
        hist = client.get("/api/history").json()
        assert hist["points"] and hist["operators"]
        page = client.get("/").text
        assert "operators" in page and "svg" in page
    finally:
        pw.set_monitoring_config(detailed_metrics_dir=None)


def test_telemetry_otlp_http_export():
    """Telemetry exports OTLP/HTTP JSON (resourceSpans/resourceMetrics)
    to the configured endpoint (reference telemetry.rs OTLP export)."""
    from pathway_amd.internals.telemetry import Telemetry
    from tests.fakes.fake_http import FakeHTTPService

    srv = FakeHTTPService().start()
    try:
        t = Telemetry(endpoint=srv.url, service_name="svc1")
        with t.span("graph_runner.run", worker=0):
            pass
        t.gauge("pathway.steps", 5.0, worker=0)
        t.flush_otlp()
        t.close()
        traces = [r for r in srv.requests if r.path == "/v1/traces"]
        metrics = [r for r in srv.requests if r.path == "/v1/metrics"]
        assert traces and metrics
        sp = traces[0].json()["resourceSpans"][0]
        res_attrs = {a["key"]: a["value"] for a in sp["resource"]["attributes"]}
        assert res_attrs["service.name"]["stringValue"] == "svc1"
        span = sp["scopeSpans"][0]["spans"][0]
        assert span["name"] == "graph_runner.run"
        assert int(span["endTimeUnixNano"]) >= int(span["startTimeUnixNano"])
        m = metrics[0].json()["resourceMetrics"][0]["scopeMetrics"][0]["metrics"][0]
        assert m["name"] == "pathway.steps"
        assert m["gauge"]["dataPoints"][0]["asDouble"] == 5.0
    finally:
        srv.stop()
