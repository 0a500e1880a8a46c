"""Louvain community detection (reference stdlib/graphs/louvain_communities/
impl.py:18-385 — same semantics, different engine mapping).

The reference runs the local-move phase as a differential iterate with
randomized tie-breaking.  Here one Louvain *level* is an incremental
host recompute over the (small, relative to streams) edge set: on every
change of the weighted edge table the local-move loop reruns to a local
modularity optimum and the clustering diffs stay incremental downstream.
"""

from __future__ import annotations

import pathway_amd as pw
from pathway_amd.internals import dtype as dt
from pathway_amd.internals.api import hash_values, Pointer
from pathway_amd.internals.config import get_device
from pathway_amd.internals.table import Table
from pathway_amd.internals.universe import Universe


def _louvain_host(edges):
    """One Louvain level over weighted undirected edges [(u, v, w), ...]:
    deterministic sequential local moves until no move improves modularity
    (reference impl.py louvain_gain formula: gain - penalty * degree)."""
    from collections import defaultdict

    adj: dict = defaultdict(lambda: defaultdict(float))
    deg: dict = defaultdict(float)
    total_w = 0.0
    for u, v, w in edges:
        adj[u][v] += w
        adj[v][u] += w
        deg[u] += w
        deg[v] += w
        total_w += 2 * w
    if total_w == 0:
        return {u: u for u in deg}
    cluster = {u: u for u in deg}
    cdeg = dict(deg)  # total degree per cluster
    order = sorted(deg, key=repr)
    improved = True
    rounds = 0
    while improved and rounds < 100:
        improved = False
        rounds += 1
        for u in order:
            cu = cluster[u]
            # weight from u to each neighboring cluster
            wto: dict = defaultdict(float)
            for v, w in adj[u].items():
                if v != u:
                    wto[cluster[v]] += w
            cdeg[cu] -= deg[u]
            best_c, best_gain = cu, wto.get(cu, 0.0) - cdeg[cu] * deg[u] / total_w
            for c, w in sorted(wto.items(), key=lambda kv: repr(kv[0])):
                gain = w - cdeg[c] * deg[u] / total_w
                if gain > best_gain + 1e-12:
                    best_c, best_gain = c, gain
            cdeg[best_c] = cdeg.get(best_c, 0.0) + deg[u]
            if best_c != cu:
                cluster[u] = best_c
                improved = True
    return cluster


def louvain_level(edges: Table, weight=None) -> Table:
    """Clustering table {u: Pointer, c: Pointer} for one Louvain level.

    `edges` must have columns u, v (pointers) and optionally a weight
    column (default weight 1.0).
    """
    from pathway_amd.engine.nodes_recompute import RecomputeNode

    wname = None
    if weight is not None:
        wname = weight.name if hasattr(weight, "name") else str(weight)

    def fn(in_rows, in_keys):
        (erows,) = in_rows
        es = [
            (repr(r["u"]), repr(r["v"]), float(r[wname]) if wname else 1.0)
            for r in erows
        ]
        uobj = {}
        for r in erows:
            uobj[repr(r["u"])] = r["u"]
            uobj[repr(r["v"])] = r["v"]
        cluster = _louvain_host(es)
        out = []
        for u, c in cluster.items():
            lo, hi = hash_values(["louvain", u])
            out.append((Pointer(lo, hi), {"u": uobj[u], "c": uobj[c]}))
        return out

    out_dtypes = {"u": dt.POINTER, "c": dt.POINTER}
    node = RecomputeNode([edges._node], fn, ["u", "c"], out_dtypes, get_device())
    return Table(node, out_dtypes, Universe())


def louvain_communities(edges: Table, weight=None, levels: int = 1) -> Table:
    """Iterated Louvain levels (reference louvain_communities_fixed_iterations):
    after each level, edges are contracted onto clusters and the next level
    refines; returns the final vertex→community table."""
    assignment = louvain_level(edges, weight=weight)
    for _ in range(levels - 1):
        # contract: relabel edge endpoints by current community, drop self-loops
        wexpr = (
            edges[weight.name if hasattr(weight, "name") else str(weight)]
            if weight is not None
            else pw.cast(float, 1.0)
        )
        lifted = (
            edges.join(assignment, edges.u == assignment.u)
            .select(cu=assignment.c, v=edges.v, w=wexpr)
            .join(assignment, pw.left.v == assignment.u)
            .select(u=pw.left.cu, v=assignment.c, w=pw.left.w)
        )
        contracted = (
            lifted.filter(pw.this.u != pw.this.v)
            .groupby(pw.this.u, pw.this.v)
            .reduce(pw.this.u, pw.this.v, w=pw.reducers.sum(pw.this.w))
        )
        upper = louvain_level(contracted, weight=contracted.w)
        assignment = (
            assignment.join(upper, assignment.c == upper.u)
            .select(u=assignment.u, c=upper.c)
        )
    return assignment


def exact_modularity(edges: Table, clustering: Table, weight=None) -> Table:
    """Modularity Q of a clustering (reference impl.py:340-380); single-row
    table {modularity: float} for testing."""
    from pathway_amd.engine.nodes_recompute import RecomputeNode

    wname = None
    if weight is not None:
        wname = weight.name if hasattr(weight, "name") else str(weight)

    def fn(in_rows, in_keys):
        erows, crows = in_rows
        cl = {repr(r["u"]): repr(r["c"]) for r in crows}
        from collections import defaultdict

        internal: dict = defaultdict(float)
        cdeg: dict = defaultdict(float)
        total = 0.0
        for r in erows:
            u, v = repr(r["u"]), repr(r["v"])
            w = float(r[wname]) if wname else 1.0
            total += 2 * w
            cdeg[cl.get(u, u)] += w
            cdeg[cl.get(v, v)] += w
            if cl.get(u, u) == cl.get(v, v):
                internal[cl.get(u, u)] += 2 * w
        q = 0.0
        if total > 0:
            for c in cdeg:
                q += internal.get(c, 0.0) / total - (cdeg[c] / total) ** 2
        lo, hi = hash_values(["modularity"])
        return [(Pointer(lo, hi), {"modularity": q})]

    out_dtypes = {"modularity": dt.FLOAT}
    node = RecomputeNode(
        [edges._node, clustering._node], fn, ["modularity"], out_dtypes, get_device()
    )
    return Table(node, out_dtypes, Universe())
