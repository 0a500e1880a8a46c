"""Vector/full-text indexing (reference stdlib/indexing).

Round-1: BruteForceKnn (torch matmul cosine/L2 top-k; HIP MFMA kernel on
gfx950) + DataIndex plumbing.  HNSW/Tantivy-parity classes arrive with the
index phase.
"""
from pathway_amd.stdlib.indexing.data_index import DataIndex
from pathway_amd.stdlib.indexing.nearest_neighbors import (
    BruteForceKnn,
    BruteForceKnnFactory,
    DistanceType,
)

__all__ = ["DataIndex", "BruteForceKnn", "BruteForceKnnFactory", "DistanceType"]
