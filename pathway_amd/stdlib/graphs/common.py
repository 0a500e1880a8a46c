"""Graph schemas (reference stdlib/graphs/common.py)."""
from __future__ import annotations

import pathway_amd.internals.schema as schema
from pathway_amd.internals.api import Pointer


class Vertex(schema.Schema):
    pass


class Edge(schema.Schema):
    u: Pointer
    v: Pointer


class Weight(schema.Schema):
    """Weight extension of Vertex / Edge (reference common.py:23)."""

    weight: float


class Cluster(Vertex):
    pass


class Clustering(schema.Schema):
    """Vertex (id) -> cluster c membership (reference common.py:35)."""

    c: Pointer


class Graph:
    def __init__(self, V, E):
        self.V = V
        self.E = E
