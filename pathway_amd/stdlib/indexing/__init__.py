"""Vector/full-text indexing (reference stdlib/indexing).

Round-1: BruteForceKnn (torch matmul cosine/L2 top-k; HIP MFMA kernel on
gfx950) + DataIndex plumbing.  HNSW/Tantivy-parity classes arrive with the
index phase.
"""
from pathway_amd.stdlib.indexing.bm25 import TantivyBM25, TantivyBM25Factory
from pathway_amd.stdlib.indexing.data_index import DataIndex
from pathway_amd.stdlib.indexing.filters import eval_jmespath_filter
from pathway_amd.stdlib.indexing.hybrid_index import HybridIndex
from pathway_amd.stdlib.indexing.nearest_neighbors import (
    BruteForceKnn,
    BruteForceKnnFactory,
    DistanceType,
    LshKnn,
    LshKnnFactory,
    USearchKnn,
    USearchMetricKind,
    UsearchKnnFactory,
)

__all__ = [
    "DataIndex",
    "BruteForceKnn",
    "BruteForceKnnFactory",
    "DistanceType",
    "USearchKnn",
    "UsearchKnnFactory",
    "USearchMetricKind",
    "LshKnn",
    "LshKnnFactory",
    "TantivyBM25",
    "TantivyBM25Factory",
    "HybridIndex",
    "eval_jmespath_filter",
]
