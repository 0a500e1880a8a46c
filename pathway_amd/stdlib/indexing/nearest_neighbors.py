"""KNN retrievers (reference stdlib/indexing/nearest_neighbors.py:65-574).

BruteForceKnn: queries x index GEMM + top-k — torch path everywhere,
HIP bf16 MFMA kernel on gfx950 (ops/knn kernels, index phase).
"""
from __future__ import annotations

import enum
from dataclasses import dataclass
from typing import Any


class DistanceType(enum.Enum):
    COS = "cos"
    L2SQ = "l2sq"


@dataclass
class BruteForceKnnFactory:
    dimensions: int | None = None
    reserved_space: int = 1000
    auxiliary_space: int = 100
    metric: DistanceType = DistanceType.COS
    embedder: Any = None

    def build_index(self, data_column, data_table, **kwargs):
        return BruteForceKnn(self.metric)


class BruteForceKnn:
    def __init__(self, metric: DistanceType = DistanceType.COS):
        self.metric = metric

    def query(self, data_table, query_column, k: int):
        raise NotImplementedError("lands with the index phase")
