"""LSH bucketers + clustering_via_lsh (reference _lsh.py /
_clustering_via_lsh.py)."""

import numpy as np

import pathway_amd as pw
from pathway_amd.debug import table_from_rows, table_to_dicts
from pathway_amd.internals.rungraph import G
from pathway_amd.internals.schema import schema_from_types
from pathway_amd.stdlib.ml.classifiers import (
    clustering_via_lsh,
    generate_cosine_lsh_bucketer,
    generate_euclidean_lsh_bucketer,
    lsh,
)


def _points():
    rng = np.random.default_rng(0)
    return np.vstack(
        [rng.normal(0, 0.1, (10, 4)), rng.normal(5, 0.1, (10, 4))]
    )


def test_euclidean_bucketer_locality():
    b = generate_euclidean_lsh_bucketer(4, 3, 5, A=2.0)
    pts = _points()
    near = sum(
        (b(pts[0]) == b(pts[i])).any() for i in range(1, 10)
    )
    far = sum((b(pts[0]) == b(pts[i])).any() for i in range(10, 20))
    assert near > far  # same-cluster points share bands more often


def test_cosine_bucketer_shapes():
    b = generate_cosine_lsh_bucketer(4, 4, 6)
    v = b(_points()[0])
    assert len(v) == 6
    assert all(0 <= int(x) < 16 for x in v)  # M=4 sign bits per band


def test_lsh_flatten_table():
    G.clear()
    pts = _points()
    t = table_from_rows(
        schema_from_types(data=np.ndarray), [(pts[i],) for i in range(4)]
    )
    b = generate_euclidean_lsh_bucketer(4, 2, 3, A=2.0)
    flat = lsh(t, b)
    _k, cols = table_to_dicts(flat)
    assert len(cols["band"]) == 4 * 3  # one row per (point, band)
    assert set(cols["band"].values()) == {0, 1, 2}
    assert all(isinstance(v, np.ndarray) for v in cols["data"].values())


def test_clustering_via_lsh_two_blobs():
    G.clear()
    pts = _points()
    t = table_from_rows(
        schema_from_types(data=np.ndarray), [(pts[i],) for i in range(20)]
    )
    b = generate_euclidean_lsh_bucketer(4, 3, 5, A=2.0)
    res = clustering_via_lsh(t, b, 2)
    keys, cols = table_to_dicts(res)
    labs = list(cols["label"].values())
    assert len(labs) == 20
    assert len(set(labs)) == 2
