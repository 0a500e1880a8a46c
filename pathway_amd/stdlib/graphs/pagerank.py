"""PageRank via pw.iterate (reference stdlib/graphs/pagerank.py behavior:
fixed-point arithmetic, ranks scaled ×1000, damping 85/100)."""

from __future__ import annotations

import pathway_amd.internals.common as common
import pathway_amd.reducers as reducers
from pathway_amd.internals import thisclass
from pathway_amd.internals.iterate import run_iterate

this = thisclass.this
left = thisclass.left
right = thisclass.right


def pagerank(edges, steps: int = 5, damping_numerator: int = 85, damping_denominator: int = 100):
    """edges: table with u, v columns (vertex ids of any hashable dtype).
    Returns a table (one row per vertex) with `vertex` and `rank` columns."""
    degs = edges.groupby(this.u).reduce(u=this.u, degree=reducers.count())
    e = edges.join(degs, edges.u == degs.u).select(
        u=this.u, v=left.v, degree=right.degree
    )
    us = edges.select(vtx=this.u)
    vs = edges.select(vtx=this.v)
    verts = us.concat_reindex(vs).groupby(this.vtx).reduce(vtx=this.vtx)
    ranks0 = verts.select(vtx=this.vtx, rank=1000)
    dn, dd = damping_numerator, damping_denominator
    base = (1000 * (dd - dn)) // dd

    def step(ranks, e, verts):
        flows = e.join(ranks, e.u == ranks.vtx).select(
            v=left.v, flow=right.rank // left.degree
        )
        inbound = flows.groupby(this.v).reduce(
            vtx=this.v, inflow=reducers.sum(this.flow)
        )
        new = verts.join_left(inbound, verts.vtx == inbound.vtx).select(
            vtx=left.vtx,
            rank=base + (common.coalesce(right.inflow, 0) * dn) // dd,
        )
        return new

    result = run_iterate(
        lambda ranks, e, verts: step(ranks, e, verts),
        iteration_limit=steps,
        ranks=ranks0,
        e=e,
        verts=verts,
    )
    return result.select(vertex=this.vtx, rank=this.rank)
