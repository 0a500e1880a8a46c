"""PageRank via pw.iterate (reference stdlib/graphs/pagerank.py)."""
from __future__ import annotations

import pathway_amd.reducers as reducers
from pathway_amd.internals import iterate as it
from pathway_amd.internals import thisclass

this = thisclass.this


def pagerank(edges, steps: int = 5, damping_numerator: int = 85, damping_denominator: int = 100):
    """edges: table with u, v pointer columns; returns table keyed like
    vertices with a `rank` int column (fixed-point arithmetic like the
    reference: ranks scaled by 1000)."""
    # out-degrees
    degs = edges.groupby(edges.u).reduce(vertex=this.u, degree=reducers.count())
    degs_by_v = degs.with_id_from(this.vertex)
    # vertices = union of u and v endpoints
    us = edges.select(vertex=this.u).with_id_from(this.vertex)
    vs = edges.select(vertex=this.v).with_id_from(this.vertex)
    vertices = us.update_rows(vs)
    ranks0 = vertices.select(vertex=this.vertex, rank=1000)

    def one_step(ranks):
        # rank flowing along edges: rank[u]/deg[u] summed per v
        keyed_ranks = ranks.with_id_from(this.vertex)
        e = edges.select(u=this.u, v=this.v)
        eu = e.with_columns(
            _pw_uk=e.pointer_from(this.u),
        )
        ru = keyed_ranks.ix(eu._pw_uk, context=eu)
        du = degs_by_v.ix(eu._pw_uk, context=eu)
        flows = eu.select(
            v=this.v,
        )
        flows = flows.with_universe_of(eu).with_columns(
            flow=ru.with_universe_of(eu).rank // du.with_universe_of(eu).degree
        )
        inbound = flows.groupby(flows.v).reduce(
            vertex=this.v, inflow=reducers.sum(this.flow)
        )
        inbound_keyed = inbound.with_id_from(this.vertex)
        base = ranks.select(vertex=this.vertex, rank=150)
        base_keyed = base.with_id_from(this.vertex)
        got = inbound_keyed.select(
            vertex=this.vertex, rank=150 + (this.inflow * 85) // 100
        )
        new_ranks = base_keyed.update_rows(got)
        return new_ranks.select(vertex=this.vertex, rank=this.rank)

    result = it.run_iterate(
        lambda ranks: one_step(ranks), iteration_limit=steps, ranks=ranks0
    )
    return result
