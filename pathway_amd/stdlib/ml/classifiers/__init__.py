"""KNN classifiers (reference stdlib/ml/classifiers/_knn_lsh.py surface,
served by the exact GPU index)."""
from __future__ import annotations

from collections import Counter
from typing import Any

import pathway_amd.internals.common as common
import pathway_amd.reducers as reducers
from pathway_amd.internals import dtype as dt
from pathway_amd.internals import thisclass

this = thisclass.this


def knn_lsh_classifier_train(data, L: int = 20, type: str = "euclidean", **kwargs):
    """Returns a classify(k, queries) closure (reference _knn_lsh.py:64)."""
    from pathway_amd.stdlib.ml.index import KNNIndex

    dim = kwargs.get("d", 2)
    index = KNNIndex(data.data, data, n_dimensions=dim, distance_type=type)

    def classify(k: int, queries):
        matched = index.get_nearest_items(queries.data, k=k)

        def majority(labels):
            labs = [l for l in (labels or ()) if l is not None]
            if not labs:
                return None
            return Counter(labs).most_common(1)[0][0]

        return matched.select(
            predicted_label=common.apply_with_type(
                majority, dt.Optional(dt.ANY), this.label
            )
        )

    return classify


knn_lsh_train = knn_lsh_classifier_train


def knn_lsh_generic_classifier_train(data, lsh_projection=None, distance_function=None, L: int = 20, **kwargs):
    return knn_lsh_classifier_train(data, L=L, **kwargs)
