"""ANN indexes (engine/ann.py): flat tombstones, IVF-Flat recall,
LSH buckets — CPU; the GPU recall/perf run is gpu-marked in test_gpu.py.

Reference parity: usearch_integration.rs (approximate index semantics),
_lsh.py (random-hyperplane LSH).
"""

import numpy as np
import pytest
import torch

from pathway_amd.engine.ann import FlatIndexState, IvfFlatState, LshState


def _mk(n, d, seed=0):
    g = torch.Generator().manual_seed(seed)
    vecs = torch.randn(n, d, generator=g)
    keys = torch.stack([
        torch.arange(1, n + 1, dtype=torch.int64),
        torch.zeros(n, dtype=torch.int64),
    ], dim=1)
    return keys, vecs


def _brute_ids(vecs_n, q, k):
    scores = torch.nn.functional.normalize(q, dim=1) @ vecs_n.T
    return torch.topk(scores, k, dim=1).indices


def test_flat_tombstone_delete_and_compact():
    keys, vecs = _mk(100, 16)
    st = FlatIndexState("cpu", "cos")
    st.update(keys, vecs, torch.ones(100, dtype=torch.int64))
    assert len(st) == 100
    # delete 30 rows — device-side mark, no compaction yet at 25%... 30% triggers
    st.update(keys[:30], vecs[:30], -torch.ones(30, dtype=torch.int64))
    assert len(st) == 70
    assert st.keys.shape[0] == 70  # compacted (30% > 25%)
    ids, scores, valid = st.search(vecs[30:33], 5)
    assert ids.shape == (3, 5, 2)
    # the query vector itself is its own nearest neighbor
    assert ids[0, 0, 0].item() == 31
    # deleted rows never appear
    got = set(ids[:, :, 0].reshape(-1).tolist())
    assert all(g > 30 for g in got)


def test_flat_delete_small_marks_only():
    keys, vecs = _mk(100, 8)
    st = FlatIndexState("cpu", "cos")
    st.update(keys, vecs, torch.ones(100, dtype=torch.int64))
    st.update(keys[:5], vecs[:5], -torch.ones(5, dtype=torch.int64))
    assert len(st) == 95
    assert st.keys.shape[0] == 100  # tombstoned, not compacted
    ids, _, _ = st.search(vecs[:5], 3)
    assert all(i.item() > 5 for i in ids[:, 0, 0])


def test_ivf_recall_vs_brute_force():
    # clustered data (what real embeddings look like): 64 centers + noise
    n, d, k = 6000, 32, 10
    g = torch.Generator().manual_seed(1)
    centers = torch.randn(64, d, generator=g) * 3.0
    assign = torch.randint(0, 64, (n,), generator=g)
    vecs = centers[assign] + 0.3 * torch.randn(n, d, generator=g)
    keys = torch.stack([
        torch.arange(1, n + 1, dtype=torch.int64),
        torch.zeros(n, dtype=torch.int64),
    ], dim=1)
    st = IvfFlatState("cpu", "cos", min_train=1000, nprobe=16)
    st.update(keys, vecs, torch.ones(n, dtype=torch.int64))
    assert st.centroids is not None  # trained
    gq = torch.Generator().manual_seed(9)
    qa = torch.randint(0, 64, (50,), generator=gq)
    q = centers[qa] + 0.3 * torch.randn(50, d, generator=gq)
    vecs_n = torch.nn.functional.normalize(vecs, dim=1)
    ref = _brute_ids(vecs_n, q, k)
    ids, scores, valid = st.search(q, k)
    got_rows = ids[:, :, 0] - 1  # key = row+1
    recall = 0.0
    for i in range(q.shape[0]):
        recall += len(set(ref[i].tolist()) & set(got_rows[i].tolist())) / k
    recall /= q.shape[0]
    assert recall >= 0.8, recall


def test_ivf_pending_tail_searched_before_rebuild():
    n, d = 5000, 16
    keys, vecs = _mk(n, d, seed=2)
    st = IvfFlatState("cpu", "cos", min_train=1000, rebuild_every=100000)
    st.update(keys, vecs, torch.ones(n, dtype=torch.int64))
    clustered = st.clustered
    # add 10 fresh vectors AFTER training — go to the un-clustered tail
    extra_keys = torch.stack([
        torch.arange(n + 1, n + 11, dtype=torch.int64),
        torch.zeros(10, dtype=torch.int64),
    ], dim=1)
    extra = torch.randn(10, d, generator=torch.Generator().manual_seed(3))
    st.update(extra_keys, extra, torch.ones(10, dtype=torch.int64))
    assert st.clustered == clustered  # no rebuild yet
    ids, _, _ = st.search(extra, 1)
    assert torch.equal(ids[:, 0, 0], extra_keys[:, 0])  # exact self-match


def test_ivf_delete_excluded():
    n, d = 4000, 16
    keys, vecs = _mk(n, d, seed=4)
    st = IvfFlatState("cpu", "cos", min_train=1000)
    st.update(keys, vecs, torch.ones(n, dtype=torch.int64))
    st.update(keys[:1], vecs[:1], -torch.ones(1, dtype=torch.int64))
    ids, _, _ = st.search(vecs[:1], 1)
    assert ids[0, 0, 0].item() != 1


def test_lsh_self_retrieval_and_delete():
    n, d = 2000, 24
    keys, vecs = _mk(n, d, seed=5)
    st = LshState("cpu", "cos", n_or=12, n_and=8)
    st.update(keys, vecs, torch.ones(n, dtype=torch.int64))
    q = vecs[:20]
    ids, scores, valid = st.search(q, 3)
    # a vector's own bucket always contains it -> self-retrieval is exact
    assert torch.equal(ids[:, 0, 0], keys[:20, 0])
    st.update(keys[:1], vecs[:1], -torch.ones(1, dtype=torch.int64))
    ids2, _, v2 = st.search(vecs[:1], 1)
    assert ids2[0, 0, 0].item() != 1 or not bool(v2[0, 0])


def test_usearch_and_lsh_retrievers_end_to_end():
    import pathway_amd as pw
    from pathway_amd.internals.rungraph import G
    from pathway_amd.stdlib.indexing.nearest_neighbors import (
        LshKnnFactory,
        UsearchKnnFactory,
    )

    for factory in (UsearchKnnFactory(), LshKnnFactory(n_or=16, n_and=6)):
        G.clear()
        docs = pw.debug.table_from_markdown(
            """
            doc
            alpha
            beta
            gamma
            """
        )

        def vec_of(s):
            g = torch.Generator().manual_seed(abs(hash(s)) % (2**31))
            return tuple(torch.randn(8, generator=g).tolist())

        docs = docs.with_columns(vector=pw.apply(vec_of, pw.this.doc))
        queries = pw.debug.table_from_markdown(
            """
            q
            beta
            """
        ).with_columns(vector=pw.apply(vec_of, pw.this.q))
        index = factory.build_index(docs.vector, metadata_column=None)
        res = index.query_as_of_now(queries.vector, number_of_matches=1)
        keys, cols = pw.debug.table_to_dicts(res)
        [ids] = list(cols["_pw_index_reply_ids"].values())
        assert len(ids) == 1
        # resolve back: the matched doc is 'beta' (exact self vector)
        dkeys, dcols = pw.debug.table_to_dicts(docs)
        assert dcols["doc"][ids[0]] == "beta"


def test_ivf_chunked_rerank_equivalence():
    """The budget-bounded rerank (RERANK_BUDGET_BYTES) must return
    exactly the unchunked answers."""
    import torch

    from pathway_amd.engine.ann import IvfFlatState

    torch.manual_seed(0)
    n, d = 20000, 16
    vecs = torch.randn(n, d)
    keys = torch.stack(
        [
            torch.arange(1, n + 1, dtype=torch.int64),
            torch.zeros(n, dtype=torch.int64),
        ],
        dim=1,
    )
    st = IvfFlatState("cpu", "cos", nprobe=8, min_train=1000)
    st.update(keys, vecs, torch.ones(n, dtype=torch.int64))
    q = torch.randn(64, d)
    ids1, sc1, v1 = st.search(q, 10)
    st.RERANK_BUDGET_BYTES = 1 << 14  # force many query chunks
    ids2, sc2, v2 = st.search(q, 10)
    assert torch.equal(ids1, ids2)
    assert torch.allclose(sc1, sc2)
    assert torch.equal(v1, v2)
