"""Graph algorithms on pw.iterate (reference stdlib/graphs/)."""
from pathway_amd.stdlib.graphs import graph
from pathway_amd.stdlib.graphs.common import (
    Cluster,
    Clustering,
    Edge,
    Vertex,
    Weight,
)
from pathway_amd.stdlib.graphs.graph import Graph, WeightedGraph
from pathway_amd.stdlib.graphs.pagerank import pagerank
from pathway_amd.stdlib.graphs.bellman_ford import bellman_ford
from pathway_amd.stdlib.graphs.louvain import (
    exact_modularity,
    louvain_communities,
    louvain_level,
)

__all__ = [
    "pagerank",
    "bellman_ford",
    "louvain_level",
    "louvain_communities",
    "exact_modularity",
    "Edge",
    "Vertex",
]
