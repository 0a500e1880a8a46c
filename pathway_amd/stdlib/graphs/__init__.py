"""Graph algorithms on pw.iterate (reference stdlib/graphs/)."""
from pathway_amd.stdlib.graphs.common import Edge, Vertex
from pathway_amd.stdlib.graphs.pagerank import pagerank
from pathway_amd.stdlib.graphs.bellman_ford import bellman_ford

__all__ = ["pagerank", "bellman_ford", "Edge", "Vertex"]
