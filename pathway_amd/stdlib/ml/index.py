"""Legacy KNNIndex (reference stdlib/ml/index.py:9 — LSH-based; here served
exactly by the GPU brute-force index, same API)."""
from __future__ import annotations

from typing import Any

import pathway_amd.internals.common as common
from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as ex
from pathway_amd.internals import thisclass

this = thisclass.this


class KNNIndex:
    def __init__(
        self,
        data_embedding: ex.ColumnReference,
        data,
        n_dimensions: int,
        n_or: int = 20,
        n_and: int = 10,
        bucket_length: float = 10.0,
        distance_type: str = "euclidean",
        metadata: ex.ColumnReference | None = None,
    ):
        from pathway_amd.stdlib.indexing.nearest_neighbors import (
            BruteForceKnn,
            DistanceType,
        )

        metric = (
            DistanceType.COS if distance_type == "cosine" else DistanceType.L2SQ
        )
        self.data = data
        self.inner = BruteForceKnn(data_embedding, metadata, metric)

    def get_nearest_items(
        self,
        query_embedding: ex.ColumnReference,
        k: int = 3,
        collapse_rows: bool = True,
        with_distances: bool = False,
        metadata_filter: Any = None,
    ):
        from pathway_amd.stdlib.indexing.data_index import DataIndex

        di = DataIndex(self.data, self.inner)
        res = di.query_as_of_now(
            query_embedding,
            number_of_matches=k,
            collapse_rows=collapse_rows,
            metadata_filter=metadata_filter,
            with_distances=True,
        )
        if not with_distances and collapse_rows:
            res = res.without("_pw_index_reply_score")
        return res

    def get_nearest_items_asof_now(self, *args, **kwargs):
        return self.get_nearest_items(*args, **kwargs)
