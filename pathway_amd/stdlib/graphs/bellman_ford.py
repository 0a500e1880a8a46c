"""Bellman-Ford shortest paths via pw.iterate (reference stdlib/graphs)."""
from __future__ import annotations

import math

import pathway_amd.reducers as reducers
from pathway_amd.internals import iterate as it
from pathway_amd.internals import thisclass

this = thisclass.this


def bellman_ford(vertices, edges, iteration_limit: int = 50):
    """vertices: table with is_source bool; edges: u, v, dist float.
    Returns dist_from_source per vertex (float, inf if unreachable)."""
    import pathway_amd.internals.common as common

    d0 = vertices.select(
        dist=common.if_else(this.is_source, 0.0, math.inf),
    )

    def step(state):
        e = edges.with_columns(_pw_uk=this.u)
        du = state.ix(e._pw_uk, context=e)
        cand = e.with_columns(reach=du.with_universe_of(e).dist + this.dist)
        best = cand.groupby(cand.v).reduce(
            _pw_v=this.v, best=reducers.min(this.reach)
        )
        best_keyed = best.with_id_from_expr(
            best._pw_v.to_column_expression()
            if hasattr(best._pw_v, "to_column_expression")
            else best._pw_v
        )
        # actually key by the vertex pointer itself
        improved = state.copy()
        from pathway_amd.engine.nodes_join import KeyedMergeNode
        # relax: new dist = min(old, best inbound)
        joined = state.join_left(best, state.id == best._pw_v).select(
            dist=common.coalesce(
                common.if_else(
                    thisclass.right.best.is_not_none()
                    & (thisclass.right.best < thisclass.left.dist),
                    thisclass.right.best,
                    thisclass.left.dist,
                ),
                thisclass.left.dist,
            ),
            _pw_vid=thisclass.left.id,
        )
        out = joined.with_id_from_expr(joined._pw_vid).without("_pw_vid")
        return out

    res = it.run_iterate(lambda state: step(state), iteration_limit=iteration_limit, state=d0)
    return res.select(dist_from_source=this.dist)
