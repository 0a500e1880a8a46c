"""Vector/full-text indexing (reference stdlib/indexing).

Round-1: BruteForceKnn (torch matmul cosine/L2 top-k; HIP MFMA kernel on
gfx950) + DataIndex plumbing.  HNSW/Tantivy-parity classes arrive with the
index phase.
"""
from pathway_amd.stdlib.indexing.bm25 import TantivyBM25, TantivyBM25Factory
from pathway_amd.stdlib.indexing.retrievers import (
    AbstractRetrieverFactory,
    BruteForceKnnMetricKind,
    InnerIndexFactory,
    default_brute_force_knn_document_index,
    default_full_text_document_index,
    default_lsh_knn_document_index,
    default_usearch_knn_document_index,
    default_vector_document_index,
)
from pathway_amd.stdlib.indexing.data_index import DataIndex
from pathway_amd.stdlib.indexing.nearest_neighbors import _BruteForceIndexBase as InnerIndex
from pathway_amd.stdlib.indexing.filters import eval_jmespath_filter
from pathway_amd.stdlib.indexing.hybrid_index import HybridIndex, HybridIndexFactory
from pathway_amd.stdlib.indexing.nearest_neighbors import (
    BruteForceKnn,
    DefaultKnnFactory,
    BruteForceKnnFactory,
    DistanceType,
    LshKnn,
    LshKnnFactory,
    USearchKnn,
    USearchMetricKind,
    UsearchKnnFactory,
)

__all__ = [
    "AbstractRetrieverFactory",
    "InnerIndexFactory",
    "InnerIndex",
    "BruteForceKnnMetricKind",
    "DefaultKnnFactory",
    "HybridIndexFactory",
    "default_vector_document_index",
    "default_lsh_knn_document_index",
    "default_usearch_knn_document_index",
    "default_brute_force_knn_document_index",
    "default_full_text_document_index",
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
