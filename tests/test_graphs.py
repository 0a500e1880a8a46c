"""Graph algorithms (reference stdlib/graphs tests)."""

import math

import pytest

import pathway_amd as pw
from pathway_amd.debug import table_from_markdown as T, table_to_dicts


@pytest.mark.timeout(120)
def test_pagerank_cycle():
    edges = T(
        """
        u | v
        a | b
        b | c
        c | a
        """
    )
    res = pw.graphs.pagerank(edges, steps=30)
    keys, cols = table_to_dicts(res)
    ranks = {cols["vertex"][k]: cols["rank"][k] for k in keys}
    # symmetric cycle: all ranks equal, total conserved around 3000
    assert len(ranks) == 3
    assert len(set(ranks.values())) == 1
    assert abs(sum(ranks.values()) - 3000) < 100


@pytest.mark.timeout(120)
def test_pagerank_sink_heavy():
    edges = T(
        """
        u | v
        a | c
        b | c
        c | a
        """
    )
    res = pw.graphs.pagerank(edges, steps=30)
    keys, cols = table_to_dicts(res)
    ranks = {cols["vertex"][k]: cols["rank"][k] for k in keys}
    assert ranks["c"] > ranks["a"] > ranks["b"]


@pytest.mark.timeout(120)
def test_bellman_ford():
    vertices = T(
        """
        vtx | is_source
        a   | True
        b   | False
        c   | False
        d   | False
        """
    )
    edges = T(
        """
        u | v | dist
        a | b | 1.0
        b | c | 2.0
        a | c | 10.0
        """
    )
    res = pw.graphs.bellman_ford(vertices, edges, iteration_limit=10)
    keys, cols = table_to_dicts(res)
    dists = {cols["vtx"][k]: cols["dist_from_source"][k] for k in keys}
    assert dists == {"a": 0.0, "b": 1.0, "c": 3.0, "d": math.inf}


def test_louvain_two_cliques():
    # two triangles joined by one weak edge → two communities
    import pathway_amd as pw
    from pathway_amd.debug import table_from_markdown as T, table_to_dicts
    from pathway_amd.stdlib.graphs import exact_modularity, louvain_level

    raw = T(
        """
        a | b
        1 | 2
        2 | 3
        1 | 3
        4 | 5
        5 | 6
        4 | 6
        3 | 4
        """
    )
    edges = raw.select(
        u=raw.pointer_from(pw.this.a), v=raw.pointer_from(pw.this.b)
    )
    cl = louvain_level(edges)
    _, cols = table_to_dicts(cl)
    groups = {}
    for i in cols["u"]:
        groups.setdefault(repr(cols["c"][i]), set()).add(repr(cols["u"][i]))
    assert len(groups) == 2
    assert sorted(len(g) for g in groups.values()) == [3, 3]

    q = exact_modularity(edges, cl)
    _, qc = table_to_dicts(q)
    (qv,) = qc["modularity"].values()
    assert qv > 0.3  # two-clique split has high modularity


def test_weighted_graph_contraction():
    import pathway_amd as pw
    from pathway_amd.debug import table_from_markdown as T, table_to_dicts
    from pathway_amd.stdlib.graphs import louvain_level
    from pathway_amd.stdlib.graphs.graph import Graph

    raw = T(
        """
        a | b
        1 | 2
        2 | 3
        4 | 5
        3 | 4
        """
    )
    edges = raw.select(u=raw.pointer_from(pw.this.a), v=raw.pointer_from(pw.this.b))
    cl = louvain_level(edges)
    g = Graph(None, edges)
    g2 = g.contracted_to_simple_graph(cl)
    _, cols = table_to_dicts(g2.E)
    # contracted graph has at most as many edges as communities allow,
    # and no self loops
    for i in cols["u"]:
        assert repr(cols["u"][i]) != repr(cols["v"][i])


def test_iterate_with_streaming_outer_updates():
    """Product-time semantics (reference dataflow.rs:5060-5190): each
    outer timestamp re-runs the fixpoint on the updated input; outputs
    at t=0 are retracted/updated by the t=2 delta."""
    import pathway_amd as pw
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.rungraph import G
    from pathway_amd.internals.schema import schema_from_types

    G.clear()
    # x arrives at t=0 with value 40, at t=2 a second row 3 arrives
    t = table_from_rows(
        schema_from_types(v=int),
        [(40, 0, 1), (3, 2, 1)],
        is_stream=True,
    )

    def logic(t):
        # halve values over 10 until all <= 10 (terminating fixpoint)
        over = t.filter(pw.this.v > 10).select(v=pw.this.v // 2)
        done = t.filter(pw.this.v <= 10)
        return done.concat(over)

    res = pw.iterate(logic, t=t)
    keys, cols = pw.debug.table_to_dicts(res)
    assert sorted(cols["v"].values()) == [3, 10]

    # capture the update stream: the t=2 row must arrive at outer time 2
    G.clear()
    t2 = table_from_rows(
        schema_from_types(v=int), [(40, 0, 1), (3, 2, 1)], is_stream=True
    )
    res2 = pw.iterate(logic, t=t2)
    from pathway_amd.debug import _run_capture

    rows = _run_capture(res2)
    times = sorted({r.time for r in rows})
    assert len(times) >= 2  # outputs at both outer timestamps
