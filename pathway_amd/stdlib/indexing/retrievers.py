"""Retriever-factory bases + default document-index constructors
(reference stdlib/indexing/retrievers.py:7-30,
vector_document_index.py:34-190, full_text_document_index.py:8-40).
"""

from __future__ import annotations

import enum
from typing import Any

from pathway_amd.stdlib.indexing.data_index import DataIndex


class AbstractRetrieverFactory:
    """Builds a DataIndex over (data_column, data_table)."""

    def build_index(self, data_column, data_table, metadata_column=None) -> DataIndex:
        raise NotImplementedError


class InnerIndexFactory(AbstractRetrieverFactory):
    """Factory whose product is an inner index wrapped in a DataIndex."""

    def build_inner_index(self, data_column, metadata_column=None):
        return self.build_index_inner(data_column, metadata_column)

    def build_index(self, data_column, data_table, metadata_column=None) -> DataIndex:
        inner = self.build_inner_index(data_column, metadata_column)
        return DataIndex(data_table, inner)


class BruteForceKnnMetricKind(enum.Enum):
    """Metric names of the brute-force KNN (reference
    brute_force_knn_integration.rs metric kinds)."""

    COS = "cos"
    L2SQ = "l2sq"


def default_vector_document_index(
    data_column,
    data_table,
    *,
    dimensions: int,
    embedder: Any = None,
    metadata_column: Any = None,
) -> DataIndex:
    """An arbitrary good-default vector index (reference
    vector_document_index.py:34)."""
    return default_brute_force_knn_document_index(
        data_column,
        data_table,
        dimensions=dimensions,
        embedder=embedder,
        metadata_column=metadata_column,
    )


def default_lsh_knn_document_index(
    data_column,
    data_table,
    *,
    dimensions: int,
    embedder: Any = None,
    metadata_column: Any = None,
) -> DataIndex:
    from pathway_amd.stdlib.indexing.nearest_neighbors import LshKnnFactory

    f = LshKnnFactory(dimensions=dimensions, embedder=embedder)
    return DataIndex(data_table, f.build_index(data_column, metadata_column))


def default_usearch_knn_document_index(
    data_column,
    data_table,
    *,
    dimensions: int,
    embedder: Any = None,
    metadata_column: Any = None,
) -> DataIndex:
    from pathway_amd.stdlib.indexing.nearest_neighbors import UsearchKnnFactory

    f = UsearchKnnFactory(dimensions=dimensions, embedder=embedder)
    return DataIndex(data_table, f.build_index(data_column, metadata_column))


def default_brute_force_knn_document_index(
    data_column,
    data_table,
    *,
    dimensions: int,
    embedder: Any = None,
    metadata_column: Any = None,
) -> DataIndex:
    from pathway_amd.stdlib.indexing.nearest_neighbors import BruteForceKnnFactory

    f = BruteForceKnnFactory(dimensions=dimensions, embedder=embedder)
    return DataIndex(data_table, f.build_index(data_column, metadata_column))


def default_full_text_document_index(
    data_column,
    data_table,
    *,
    metadata_column: Any = None,
) -> DataIndex:
    from pathway_amd.stdlib.indexing.bm25 import TantivyBM25

    return DataIndex(data_table, TantivyBM25(data_column, metadata_column))
