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


# ---------------------------------------------------------------- LSH --
# Reference stdlib/ml/classifiers/_lsh.py + _clustering_via_lsh.py.


def _fingerprint_i32(arr) -> int:
    import zlib

    import numpy as np

    return int(
        np.int32(zlib.crc32(np.ascontiguousarray(arr).tobytes()) & 0x7FFFFFFF)
    )


def generate_euclidean_lsh_bucketer(d: int, M: int, L: int, A: float = 1.0, seed: int = 0):
    """Euclidean LSH: project on M*L random unit lines, bucket width A;
    each of the L bands fingerprints its M AND-ed bucket ids
    (reference _lsh.py:31-55)."""
    import numpy as np

    gen = np.random.default_rng(seed=seed)
    total = M * L
    lines = gen.standard_normal((d, total))
    lines = lines / np.linalg.norm(lines, axis=0)
    shift = gen.random(size=total) * A

    def bucketify(x):
        buckets = np.floor_divide(np.asarray(x) @ lines + shift, A).astype(int)
        return np.array(
            [_fingerprint_i32(band) for band in np.split(buckets, L)]
        )

    return bucketify


def generate_cosine_lsh_bucketer(d: int, M: int, L: int, seed: int = 0):
    """Cosine LSH: sign patterns against M*L random hyperplanes, each
    band packs its M signs into one integer (reference _lsh.py:58-80)."""
    import numpy as np

    gen = np.random.default_rng(seed=seed)
    planes = gen.standard_normal((d, M * L))

    def bucketify(x):
        signs = (np.asarray(x) @ planes >= 0).astype(int)
        powers = 2 ** np.arange(M).reshape(-1, 1)
        return np.hstack([band @ powers for band in np.split(signs, L)])

    return bucketify


def lsh(data, bucketer, origin_id: str = "origin_id", include_data: bool = True):
    """Apply the bucketer per row and flatten to (origin, band, bucketing)
    rows (reference _lsh.py:82-104)."""
    from pathway_amd.stdlib.utils.col import unpack_col

    flat = data.select(
        buckets=common.apply(
            lambda x: [(int(b), int(v)) for b, v in enumerate(bucketer(x))],
            data.data,
        )
    )
    flat = flat.flatten(this.buckets, origin_id=origin_id)
    out = flat.select(flat[origin_id]) + unpack_col(
        flat.buckets, this.band, this.bucketing
    )
    if include_data:
        out += out.select(data.ix(out[origin_id]).data)
    return out


def clustering_via_lsh(data, bucketer, k: int):
    """LSH-bucketed k-means clustering: bucket representatives are
    clustered (sklearn KMeans) and points take the majority label of
    their buckets (reference _clustering_via_lsh.py:31-81)."""
    from pathway_amd.stdlib.utils.col import (
        apply_all_rows,
        groupby_reduce_majority,
    )

    flat = lsh(data, bucketer, origin_id="data_id", include_data=True)
    reps = (
        flat.groupby(this.bucketing, this.band)
        .reduce(
            this.bucketing,
            this.band,
            sum=reducers.sum(this.data),
            count=reducers.count(),
        )
        .select(
            this.bucketing,
            this.band,
            data=common.apply(lambda s, c: s / c, this.sum, this.count),
            weight=this.count,
        )
    )

    def clustering(datas, weights):
        from sklearn.cluster import KMeans

        km = KMeans(n_clusters=k, init="k-means++", random_state=0, n_init=10)
        km.fit(list(datas), sample_weight=list(weights))
        return [int(l) for l in km.labels_]

    labels = apply_all_rows(
        reps.data, reps.weight, fun=clustering, result_col_name="label"
    )
    reps += labels
    votes = flat.join(
        reps,
        flat.bucketing == reps.bucketing,
        flat.band == reps.band,
    ).select(flat.data_id, reps.label)
    result = groupby_reduce_majority(votes.data_id, votes.label)
    result = result.select(result.data_id, label=result.majority)
    return result.with_id(result.data_id).select(this.label)
