"""Graph schemas (reference stdlib/graphs/common.py)."""
from __future__ import annotations

import pathway_amd.internals.schema as schema
from pathway_amd.internals.api import Pointer


class Vertex(schema.Schema):
    pass


class Edge(schema.Schema):
    u: Pointer
    v: Pointer


class Graph:
    def __init__(self, V, E):
        self.V = V
        self.E = E
