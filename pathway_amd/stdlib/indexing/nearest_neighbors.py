"""KNN retrievers (reference stdlib/indexing/nearest_neighbors.py:65-574).

BruteForceKnn: GPU brute-force cosine/L2 — queries × index GEMM + the
hand-written pw_topk kernel, over the ExternalIndexNode.
USearchKnn: the approximate index (reference: usearch HNSW,
usearch_integration.rs:20-152).  The MI355X-native approximate engine is
GPU IVF-Flat (engine/ann.py IvfFlatState): k-means coarse quantizer +
nprobe candidate lists + exact rerank — same DataIndex API and as-of-now
semantics; recall/perf measured in profiles/ann_r02.md.
LshKnn: real random-hyperplane LSH with multi-table buckets
(engine/ann.py LshState; reference _lsh.py semantics).
"""

from __future__ import annotations

import enum
from dataclasses import dataclass
from typing import Any

from pathway_amd.internals import expression as ex
from pathway_amd.internals.config import get_device
from pathway_amd.internals.universe import Universe


class DistanceType(enum.Enum):
    COS = "cos"
    L2SQ = "l2sq"


class USearchMetricKind(enum.Enum):
    COS = "cos"
    L2SQ = "l2sq"
    IP = "ip"


class _BruteForceIndexBase:
    """InnerIndex implementation over ExternalIndexNode."""

    #: engine/ann.py state class selector
    index_kind = "flat"

    def __init__(
        self,
        data_column: ex.ColumnReference,
        metadata_column: ex.ColumnReference | None = None,
        metric: DistanceType = DistanceType.COS,
        embedder: Any = None,
        index_params: dict | None = None,
    ):
        self.data_column = data_column
        self.metadata_column = metadata_column
        self.metric = metric
        self.embedder = embedder
        self.index_params = index_params or {}

    def query_as_of_now(
        self,
        query_column: ex.ColumnReference,
        number_of_matches: int = 3,
        metadata_filter: Any = None,
    ):
        from pathway_amd.engine.nodes_index import ExternalIndexNode
        from pathway_amd.internals import dtype as dt
        from pathway_amd.internals.table import Table

        data_table = self.data_column.table
        query_table = query_column.table
        dcol = self.data_column
        qexpr = query_column
        if self.embedder is not None:
            dcol = self.embedder(self.data_column)
            qexpr = self.embedder(query_column)
        if isinstance(dcol, ex.ColumnReference):
            index_src = data_table
            vec_name = dcol.name
        else:
            index_src = data_table.with_columns(_pw_vec=dcol)
            vec_name = "_pw_vec"
        filter_col = None
        if self.metadata_column is not None:
            if vec_name not in index_src._dtypes:
                pass
            index_src = index_src.with_columns(_pw_meta=self.metadata_column)
            filter_col = "_pw_meta"
        kexpr = None
        if isinstance(number_of_matches, ex.ColumnExpression):
            kexpr = number_of_matches
            kval = 16
        else:
            kval = int(number_of_matches)
        node = ExternalIndexNode(
            index_src._node,
            query_table._node,
            vec_name,
            qexpr,
            kval,
            get_device(),
            metric=self.metric.value,
            filter_data_col=filter_col,
            query_filter_expr=metadata_filter,
            query_k_expr=kexpr,
            index_kind=self.index_kind,
            index_params=self.index_params,
        )
        dtypes = {
            "_pw_index_reply_ids": dt.List(dt.POINTER),
            "_pw_index_reply_scores": dt.List(dt.FLOAT),
        }
        return Table(node, dtypes, query_table._universe)

    # non-asof-now query: same node; reference updates results as the index
    # changes — landing with the streaming-index phase
    def query(self, query_column, number_of_matches: int = 3, metadata_filter=None):
        return self.query_as_of_now(query_column, number_of_matches, metadata_filter)


class BruteForceKnn(_BruteForceIndexBase):
    pass


class USearchKnn(_BruteForceIndexBase):
    """Approximate KNN with the usearch-factory API (reference
    usearch_integration.rs:20-152).  Engine: GPU IVF-Flat
    (engine/ann.py) — k-means lists + nprobe + exact rerank."""

    index_kind = "ivf"


class LshKnn(_BruteForceIndexBase):
    """Random-hyperplane LSH (reference _lsh.py): n_or tables x n_and
    bits, bucket candidates, exact rerank."""

    index_kind = "lsh"


@dataclass
class BruteForceKnnFactory:
    dimensions: int | None = None
    reserved_space: int = 1000
    auxiliary_space: int = 100
    metric: DistanceType = DistanceType.COS
    embedder: Any = None

    def build_index(self, data_column, metadata_column=None, **kwargs) -> BruteForceKnn:
        return BruteForceKnn(data_column, metadata_column, self.metric, self.embedder)


@dataclass
class UsearchKnnFactory:
    dimensions: int | None = None
    reserved_space: int = 400
    metric: USearchMetricKind = USearchMetricKind.COS
    connectivity: int = 0
    expansion_add: int = 0
    expansion_search: int = 0
    embedder: Any = None

    def build_index(self, data_column, metadata_column=None, **kwargs) -> USearchKnn:
        m = DistanceType.COS if self.metric in (USearchMetricKind.COS, USearchMetricKind.IP) else DistanceType.L2SQ
        params = {}
        if self.expansion_search:
            params["nprobe"] = max(1, int(self.expansion_search))
        return USearchKnn(data_column, metadata_column, m, self.embedder,
                          index_params=params)


@dataclass
class LshKnnFactory:
    dimensions: int | None = None
    n_or: int = 20
    n_and: int = 10
    bucket_length: float = 10.0
    distance_type: DistanceType = DistanceType.COS
    embedder: Any = None

    def build_index(self, data_column, metadata_column=None, **kwargs) -> LshKnn:
        return LshKnn(
            data_column, metadata_column, self.distance_type, self.embedder,
            index_params={"n_or": self.n_or, "n_and": self.n_and},
        )


class DefaultKnnFactory(BruteForceKnnFactory):
    """Good-default KNN factory (reference nearest_neighbors.py:574)."""
