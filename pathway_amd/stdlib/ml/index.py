"""Legacy KNNIndex (reference stdlib/ml/index.py:9) — index phase."""
from __future__ import annotations


class KNNIndex:
    def __init__(self, data_embedding, data, n_dimensions: int, n_or: int = 20, n_and: int = 10, bucket_length: float = 10.0, distance_type: str = "euclidean", metadata=None):
        self.data_embedding = data_embedding
        self.data = data
        self.n_dimensions = n_dimensions

    def get_nearest_items(self, query_embedding, k: int = 3, collapse_rows: bool = True, with_distances: bool = False, metadata_filter=None):
        raise NotImplementedError("lands with the index phase")

    def get_nearest_items_asof_now(self, query_embedding, k: int = 3, collapse_rows: bool = True, with_distances: bool = False, metadata_filter=None):
        raise NotImplementedError("lands with the index phase")
