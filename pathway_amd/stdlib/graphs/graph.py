"""Graph / WeightedGraph containers (reference stdlib/graphs/graph.py:77-150
behavior: vertex/edge tables + cluster contraction)."""
from __future__ import annotations

import pathway_amd as pw
from pathway_amd.stdlib.graphs.common import Edge, Vertex


class Graph:
    """Undirected unweighted (multi)graph: vertex table V + edge table E
    with pointer columns u, v."""

    def __init__(self, V, E):
        self.V = V
        self.E = E

    def contracted_to_simple_graph(self, clustering, **kwargs) -> "Graph":
        """Collapse vertices by cluster assignment; drop self-loops and
        parallel edges (reference graph.py _contract)."""
        E = (
            self.E.join(clustering, self.E.u == clustering.u)
            .select(u=clustering.c, v=pw.left.v)
            .join(clustering, pw.left.v == clustering.u)
            .select(u=pw.left.u, v=clustering.c)
            .filter(pw.this.u != pw.this.v)
            .groupby(pw.this.u, pw.this.v)
            .reduce(pw.this.u, pw.this.v)
        )
        V = clustering.groupby(clustering.c).reduce(id=clustering.c)
        return Graph(V, E)


class WeightedGraph(Graph):
    """Graph with per-edge weights (WE table: u, v, weight)."""

    def __init__(self, V, E, WE=None):
        super().__init__(V, E)
        self.WE = WE if WE is not None else E

    @classmethod
    def from_vertices_and_weighted_edges(cls, V, WE) -> "WeightedGraph":
        return cls(V, WE.select(u=WE.u, v=WE.v), WE)
