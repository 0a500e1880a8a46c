"""GPU-marked tests: HIP kernel numerics vs torch/python references, and
end-to-end engine runs on the MI355X."""

import pytest
import torch

gpu = pytest.mark.gpu

requires_cuda = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


@gpu
@requires_cuda
def test_hash_kernel_vs_reference():
    from pathway_amd import ops
    from pathway_amd.engine import hashing

    torch.manual_seed(7)
    for nwords in (1, 2, 3, 4, 6, 9):
        words = [
            torch.randint(-(2**62), 2**62, (4097,), dtype=torch.int64)
            for _ in range(nwords)
        ]
        # reference: torch CPU path (bit-exact with python xxh64 by test_hash)
        ref_lo = hashing.xxh64_words(words, seed=0)
        ref_hi = hashing.xxh64_words(words, seed=0x9E3779B185EBCA87 - (1 << 64))
        lo, hi = ops.hash128_words_gpu([w.cuda() for w in words])
        assert torch.equal(ref_lo, lo.cpu()), f"lo mismatch at {nwords} words"
        assert torch.equal(ref_hi, hi.cpu()), f"hi mismatch at {nwords} words"


@gpu
@requires_cuda
def test_value_hash_kernel():
    from pathway_amd import ops
    from pathway_amd.internals.api import MASK64, TAG_INT, hash128, serialize_value

    vals = torch.tensor([0, 1, -5, 2**40, 123456789], dtype=torch.int64)
    lo, hi = ops.value_hash_gpu(vals.cuda(), TAG_INT)
    for i, v in enumerate(vals.tolist()):
        elo, ehi = hash128(serialize_value(v))
        assert int(lo[i]) & MASK64 == elo
        assert int(hi[i]) & MASK64 == ehi


@gpu
@requires_cuda
def test_varlen_hash_kernel():
    from pathway_amd import ops
    from pathway_amd.internals.api import MASK64, TAG_STR, hash128, serialize_value

    strings = ["a", "hello", "x" * 55, "y" * 56, "z" * 57, "w" * 300, ""]
    data = b"".join(s.encode() for s in strings)
    offsets = [0]
    for s in strings:
        offsets.append(offsets[-1] + len(s.encode()))
    bt = torch.frombuffer(bytearray(data), dtype=torch.uint8).cuda()
    ot = torch.tensor(offsets, dtype=torch.int64).cuda()
    lo, hi = ops.varlen_hash_gpu(bt, ot, TAG_STR)
    for i, s in enumerate(strings):
        elo, ehi = hash128(serialize_value(s))
        assert int(lo[i]) & MASK64 == elo, f"lo mismatch for {s!r}"
        assert int(hi[i]) & MASK64 == ehi, f"hi mismatch for {s!r}"


@gpu
@requires_cuda
def test_wordcount_gpu_end_to_end():
    import os

    os.environ["PW_DEVICE"] = "cuda:0"
    from pathway_amd.internals.config import pathway_config

    pathway_config.device = "cuda:0"
    try:
        import pathway_amd as pw
        from pathway_amd.debug import table_from_markdown as T

        t = T(
            """
            word  | __time__ | __diff__
            apple | 0        | 1
            pear  | 0        | 1
            apple | 2        | 1
            pear  | 4        | -1
            """
        )
        res = t.groupby(pw.this.word).reduce(
            pw.this.word, c=pw.reducers.count()
        )
        keys, cols = pw.debug.table_to_dicts(res)
        counts = {cols["word"][k]: cols["c"][k] for k in keys}
        assert counts == {"apple": 2}
    finally:
        pathway_config.device = None


@gpu
@requires_cuda
def test_join_gpu():
    import os

    from pathway_amd.internals.config import pathway_config

    pathway_config.device = "cuda:0"
    try:
        import pathway_amd as pw
        from pathway_amd.debug import assert_table_equality_wo_index, table_from_markdown as T

        t1 = T(
            """
            a | k
            1 | x
            2 | y
            """
        )
        t2 = T(
            """
            b  | k
            10 | x
            """
        )
        res = t1.join(t2, t1.k == t2.k).select(t1.a, t2.b)
        expected = T(
            """
            a | b
            1 | 10
            """
        )
        assert_table_equality_wo_index(res, expected)
    finally:
        pathway_config.device = None


@gpu
@requires_cuda
def test_native_lib_is_used_on_gpu():
    """The GPU hash path must dispatch into libpwhip.so (no silent torch
    fallback)."""
    from pathway_amd import ops
    from pathway_amd.engine import hashing

    assert ops.lib_available()
    w = torch.arange(100, dtype=torch.int64).cuda()
    lo, hi = hashing.hash128_words([w, w])
    ref_lo, ref_hi = hashing.xxh64_words([w.cpu(), w.cpu()], 0), None
    assert torch.equal(lo.cpu(), ref_lo)


@gpu
@requires_cuda
def test_embedder_gpu_bf16():
    from pathway_amd.xpacks.llm._encoder import get_encoder

    enc = get_encoder(device="cuda:0")
    assert enc.dtype == torch.bfloat16
    out = enc.encode(["hello world", "hello world", "different text"])
    import numpy as np

    assert np.allclose(out[0], out[1])
    assert not np.allclose(out[0], out[2])
    assert abs(float(np.linalg.norm(out[0])) - 1.0) < 1e-2


@gpu
@requires_cuda
def test_knn_index_gpu():
    from pathway_amd.engine.nodes_index import VectorIndexState

    st = VectorIndexState(torch.device("cuda:0"), metric="cos")
    vecs = torch.eye(4, dtype=torch.float32).cuda()
    keys = torch.arange(8, dtype=torch.int64).reshape(4, 2).cuda()
    st.update(keys, vecs, torch.ones(4, dtype=torch.int64).cuda())
    q = torch.tensor([[0.0, 1.0, 0.05, 0.0]]).cuda()
    ids, scores, _ = st.search(q, 2)
    assert ids[0, 0].cpu().tolist() == [2, 3]  # second basis vector's key


@gpu
@requires_cuda
def test_document_store_gpu():
    from pathway_amd.internals.config import pathway_config

    pathway_config.device = "cuda:0"
    try:
        from pathway_amd.debug import table_from_rows, table_to_dicts
        from pathway_amd.internals.schema import schema_from_types
        from pathway_amd.xpacks.llm.document_store import DocumentStore

        schema = schema_from_types(data=bytes, _metadata=dict)
        docs = table_from_rows(
            schema,
            [
                (b"gpu streaming dataflow engine", {"path": "a.txt"}),
                (b"other thing entirely", {"path": "b.txt"}),
            ],
        )
        store = DocumentStore(docs)
        queries = table_from_rows(
            DocumentStore.RetrieveQuerySchema, [("streaming dataflow", 1, None, None)]
        )
        keys, cols = table_to_dicts(store.retrieve_query(queries))
        rv = cols["result"][keys[0]]
        rv = rv.value if hasattr(rv, "value") else rv
        assert len(rv) == 1 and "streaming" in rv[0]["text"]
    finally:
        pathway_config.device = None


@gpu
@requires_cuda
def test_varlen_hash_se_kernel():
    from pathway_amd import ops
    from pathway_amd.internals.api import MASK64, TAG_STR, hash128, serialize_value

    words = ["alpha", "beta", "x" * 80]
    wire = ("\n".join(words) + "\n").encode()
    buf = torch.frombuffer(bytearray(wire), dtype=torch.uint8).cuda()
    nl = (buf == 10).nonzero(as_tuple=True)[0]
    starts = torch.cat([torch.zeros(1, dtype=torch.int64).cuda(), nl[:-1] + 1])
    lo, hi = ops.varlen_hash_se_gpu(buf, starts, nl, TAG_STR)
    for i, w in enumerate(words):
        elo, ehi = hash128(serialize_value(w))
        assert int(lo[i]) & MASK64 == elo
        assert int(hi[i]) & MASK64 == ehi


@gpu
@requires_cuda
def test_gpu_cpu_pipeline_equivalence():
    """Same pipeline on cpu and cuda must produce identical results
    (determinism across devices — SURVEY §5.2 analog)."""
    from pathway_amd.internals.config import pathway_config

    def build_and_run():
        import pathway_amd as pw
        from pathway_amd.debug import table_from_markdown as T, table_to_dicts

        t = T(
            """
            g | v | t
            a | 1 | 1
            a | 4 | 3
            b | 2 | 2
            b | 3 | 9
            a | 7 | 12
            """
        )
        win = t.windowby(pw.this.t, window=pw.temporal.tumbling(duration=5)).reduce(
            start=pw.this._pw_window_start,
            s=pw.reducers.sum(pw.this.v),
            m=pw.reducers.max(pw.this.v),
        )
        other = T(
            """
            g | w
            a | 100
            b | 200
            """
        )
        j = t.join(other, t.g == other.g).select(pw.this.g, pw.this.v, pw.this.w)
        red = j.groupby(pw.this.g).reduce(
            pw.this.g, tot=pw.reducers.sum(pw.this.v + pw.this.w)
        )
        k1, c1 = table_to_dicts(win)
        k2, c2 = table_to_dicts(red)
        rows1 = sorted((c1["start"][k], c1["s"][k], c1["m"][k]) for k in k1)
        rows2 = sorted((c2["g"][k], c2["tot"][k]) for k in k2)
        return rows1, rows2

    pathway_config.device = "cpu"
    cpu_result = build_and_run()
    pathway_config.device = "cuda:0"
    try:
        gpu_result = build_and_run()
    finally:
        pathway_config.device = None
    assert cpu_result == gpu_result


@pytest.mark.gpu
def test_hash_agg_matches_sort_path():
    """pw_hash_agg preagg == plain torch sort+segsum reference, including
    negative diffs and heavy duplication."""
    from pathway_amd import ops

    dev = torch.device("cuda:0")
    g = torch.Generator(device="cpu").manual_seed(7)
    n, vocab = 1_000_000, 5000
    ids = torch.randint(0, vocab, (n,), generator=g).to(dev)
    # synthetic 128-bit keys from ids (distinct ids -> distinct keys)
    k0 = ids * 0x9E3779B185EBCA87 + 12345
    k1 = ids * 0xC2B2AE3D27D4EB4F + 999
    w = torch.where(
        torch.rand(n, generator=g).to(dev) < 0.2,
        torch.tensor(-1, device=dev),
        torch.tensor(1, device=dev),
    ).to(torch.int64)
    v = torch.randint(-50, 50, (n,), generator=g).to(dev)

    uk0, uk1, (aw, av), rep = ops.hash_agg_gpu(k0, k1, [w, v])
    # consolidate possible duplicate slots then sort by key
    order = torch.argsort(uk0)
    uk0, uk1 = uk0[order], uk1[order]
    aw, av, rep = aw[order], av[order], rep[order]

    # torch reference: aggregate by id
    ref_w = torch.zeros(vocab, dtype=torch.int64, device=dev)
    ref_w.index_add_(0, ids, w)
    ref_v = torch.zeros(vocab, dtype=torch.int64, device=dev)
    ref_v.index_add_(0, ids, v)
    touched = torch.zeros(vocab, dtype=torch.bool, device=dev)
    touched[ids] = True
    ref_ids = touched.nonzero(as_tuple=True)[0]
    ref_k0 = ref_ids * 0x9E3779B185EBCA87 + 12345
    ref_order = torch.argsort(ref_k0)
    ref_ids = ref_ids[ref_order]

    # merge duplicate hash-agg slots (rare) on host for comparison
    got = {}
    for a, b, x, y, r in zip(
        uk0.cpu().tolist(), uk1.cpu().tolist(), aw.cpu().tolist(),
        av.cpu().tolist(), rep.cpu().tolist(),
    ):
        key = (a, b)
        gw, gv = got.get(key, (0, 0))
        got[key] = (gw + x, gv + y)
        # representative must be a batch row of this key
        assert int(k0[r].item()) == a
    assert len(got) == int(ref_ids.numel())
    for i in ref_ids.cpu().tolist():
        key = (
            int((torch.tensor(i) * 0x9E3779B185EBCA87 + 12345).item()),
            int((torch.tensor(i) * 0xC2B2AE3D27D4EB4F + 999).item()),
        )
        assert got[key] == (
            int(ref_w[i].item()),
            int(ref_v[i].item()),
        )


@pytest.mark.gpu
def test_groupreduce_hashagg_vs_sort_pipeline():
    """Whole wordcount-style pipeline: hash-agg path == sort path output."""
    import os as _o
    import subprocess, sys, json

    code = """
import os, json, torch, sys
import pathway_amd as pw
from pathway_amd.debug import table_to_dicts
os.environ["PW_DEVICE"] = "cuda:0"
from pathway_amd.internals.config import get_device
from pathway_amd.debug import table_from_rows
from pathway_amd.internals.schema import schema_from_types
import random
random.seed(5)
rows = [(f"w{random.randint(0, 200)}",) for _ in range(20000)]
t = table_from_rows(schema_from_types(w=str), rows)
r = t.groupby(pw.this.w).reduce(pw.this.w, c=pw.reducers.count())
_, cols = table_to_dicts(r)
out = sorted(zip(cols["w"].values(), cols["c"].values()))
print(json.dumps(out))
"""
    outs = []
    for env_extra in ({"PW_HASHAGG": "1"}, {}):
        env = dict(_o.environ, **env_extra)
        res = subprocess.run(
            [sys.executable, "-c", code], capture_output=True, text=True, env=env
        )
        assert res.returncode == 0, res.stderr[-2000:]
        outs.append(res.stdout.strip().splitlines()[-1])
    assert json.loads(outs[0]) == json.loads(outs[1])


@pytest.mark.gpu
def test_seg_reduce_matches_reference():
    """pw_seg_reduce (fused run-starts + compaction + wave-segmented sums)
    vs plain torch on sorted keys, incl. negative weights."""
    from pathway_amd import ops

    dev = torch.device("cuda:0")
    g = torch.Generator(device="cpu").manual_seed(11)
    # n deliberately NOT a multiple of 64: the run tail can land on a
    # padding lane of the last wave (the bug class this test guards)
    n, vocab = 2_000_003, 37_000
    ids = torch.randint(0, vocab, (n,), generator=g).to(dev)
    k0 = (ids * 0x9E3779B185EBCA87 + 7).sort().values
    k1 = k0 * 3 + 1
    w = torch.where(
        torch.rand(n, generator=g).to(dev) < 0.3,
        torch.tensor(-1, device=dev),
        torch.tensor(2, device=dev),
    ).to(torch.int64)
    v = torch.randint(-9, 9, (n,), generator=g).to(dev)

    uk0, uk1, first, (aw, av) = ops.seg_reduce_gpu(k0, k1, [w, v])

    # torch reference
    starts = torch.ones(n, dtype=torch.bool, device=dev)
    starts[1:] = (k0[1:] != k0[:-1]) | (k1[1:] != k1[:-1])
    seg = torch.cumsum(starts.to(torch.int64), 0) - 1
    fidx = starts.nonzero(as_tuple=True)[0]
    nseg = int(fidx.numel())
    assert uk0.shape[0] == nseg
    assert torch.equal(uk0, k0.index_select(0, fidx))
    assert torch.equal(uk1, k1.index_select(0, fidx))
    assert torch.equal(first, fidx)
    for got, src in ((aw, w), (av, v)):
        ref = torch.zeros(nseg, dtype=torch.int64, device=dev)
        ref.index_add_(0, seg, src)
        assert torch.equal(got, ref)


@gpu
@requires_cuda
def test_partition_kernel_vs_argsort():
    from pathway_amd import ops

    torch.manual_seed(11)
    for n, world in [(0, 8), (1, 2), (1000, 3), (1 << 20, 8), (12345, 64)]:
        dest = torch.randint(0, world, (n,), dtype=torch.int64, device="cuda")
        perm, counts = ops.partition_gpu(dest, world)
        ref_counts = torch.bincount(dest.cpu(), minlength=world)
        assert torch.equal(counts.cpu(), ref_counts), (n, world)
        # perm is a permutation and groups rows by destination
        sorted_dest = dest.index_select(0, perm)
        assert torch.equal(
            sorted_dest.cpu(), torch.sort(dest.cpu()).values
        ), (n, world)
        assert torch.equal(
            torch.sort(perm.cpu()).values, torch.arange(n, dtype=torch.int64)
        ), (n, world)


@gpu
@requires_cuda
def test_device_tokenizer_matches_host():
    from pathway_amd.xpacks.llm._encoder import EncoderConfig, NativeEncoder

    enc = NativeEncoder(EncoderConfig(layers=1), device="cuda")
    texts = [
        "hello world",
        "",
        "One TWO three four five",
        "  padded   spaces  ",
        "tab\tseparated words",
        "x" * 30,
    ]
    ids_d, mask_d = enc._tokenize_device(texts, 512)
    # host reference on the same encoder geometry
    enc_cpu = NativeEncoder(EncoderConfig(layers=1), device="cpu")
    ids_h, mask_h = enc_cpu._tokenize_host(texts, 512)
    assert ids_d.shape == ids_h.shape
    assert torch.equal(ids_d.cpu(), ids_h)
    assert torch.equal(mask_d.cpu(), mask_h)


@gpu
@requires_cuda
def test_hipgraph_capture_matches_eager():
    import os

    from pathway_amd.xpacks.llm._encoder import EncoderConfig, NativeEncoder

    enc = NativeEncoder(EncoderConfig(layers=2), device="cuda")
    ids, mask = enc.tokenize(["alpha beta gamma", "delta"])
    eager = enc._forward_impl(ids, mask)
    graphed = enc._forward_graphed(ids, mask)
    assert torch.allclose(eager, graphed, atol=1e-3, rtol=1e-3)
    # replay with different data through the same graph
    ids2, mask2 = enc.tokenize(["zeta eta", "theta iota kappa"])
    eager2 = enc._forward_impl(ids2, mask2)
    graphed2 = enc._forward_graphed(ids2, mask2)
    assert torch.allclose(eager2, graphed2, atol=1e-3, rtol=1e-3)
    assert not torch.allclose(graphed, graphed2, atol=1e-3)


@gpu
@requires_cuda
def test_mfma_gemm_vs_torch():
    from pathway_amd import ops

    torch.manual_seed(3)
    for M, N, K in [(128, 128, 64), (257, 384, 384), (1000, 1536, 384),
                    (64, 128, 100), (512, 384, 1536)]:
        a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
        b = (torch.randn(K, N, device="cuda") * 0.5).to(torch.bfloat16)
        bias = torch.randn(N, device="cuda")
        ref = (a.float() @ b.float() + bias).to(torch.bfloat16).float()
        got = ops.gemm_bias_act_gpu(a, b.T.contiguous(), bias, act="none").float()
        err = (got - ref).abs().max().item()
        scale = ref.abs().max().item() + 1.0
        assert err / scale < 0.02, (M, N, K, err, scale)
        # fused GELU epilogue
        import torch.nn.functional as F

        ref_g = F.gelu(a.float() @ b.float() + bias)
        got_g = ops.gemm_bias_act_gpu(
            a, b.T.contiguous(), bias, act="gelu"
        ).float()
        errg = (got_g - ref_g).abs().max().item()
        assert errg / (ref_g.abs().max().item() + 1.0) < 0.02, (M, N, K, errg)


@gpu
@requires_cuda
def test_topk_kernel_vs_torch():
    from pathway_amd import ops

    torch.manual_seed(13)
    for nq, m, k in [(1, 100, 1), (7, 1000, 10), (64, 65536, 32), (3, 10, 8)]:
        scores = torch.randn(nq, m, device="cuda")
        vals, idx = ops.topk_gpu(scores, min(k, m))
        rv, ri = torch.topk(scores, min(k, m), dim=1)
        assert torch.allclose(vals, rv), (nq, m, k)
        # indices may differ on exact ties; values fully determine correctness
        gathered = scores.gather(1, idx)
        assert torch.allclose(gathered, rv), (nq, m, k)


@gpu
@requires_cuda
def test_ivf_gpu_recall_at_10():
    from pathway_amd.engine.ann import IvfFlatState

    torch.manual_seed(21)
    n, d, k = 200_000, 384, 10
    centers = (torch.randn(512, d, device="cuda") * 3.0)
    assign = torch.randint(0, 512, (n,), device="cuda")
    vecs = centers[assign] + 0.3 * torch.randn(n, d, device="cuda")
    keys = torch.stack([
        torch.arange(1, n + 1, dtype=torch.int64, device="cuda"),
        torch.zeros(n, dtype=torch.int64, device="cuda"),
    ], dim=1)
    st = IvfFlatState("cuda", "cos", nprobe=16)
    st.update(keys, vecs, torch.ones(n, dtype=torch.int64, device="cuda"))
    assert st.centroids is not None
    qa = torch.randint(0, 512, (100,), device="cuda")
    q = centers[qa] + 0.3 * torch.randn(100, d, device="cuda")
    vn = torch.nn.functional.normalize(vecs, dim=1)
    ref = torch.topk(torch.nn.functional.normalize(q, dim=1) @ vn.T, k, 1).indices
    ids, _, _ = st.search(q, k)
    got = (ids[:, :, 0] - 1).cpu()
    refc = ref.cpu()
    recall = sum(
        len(set(refc[i].tolist()) & set(got[i].tolist())) / k
        for i in range(100)
    ) / 100
    assert recall >= 0.9, recall


@gpu
@requires_cuda
def test_merge_consolidate_kernel():
    from pathway_amd import ops
    from pathway_amd.engine.state import lex_sort_words

    torch.manual_seed(31)
    for m, n, nacc in [(0, 10, 1), (10, 0, 2), (1000, 700, 3),
                       (100000, 60000, 2), (5, 5, 1)]:
        # unique sorted keys with guaranteed overlap
        space = max(m + n, 1) * 2
        ka = torch.randperm(space, device="cuda")[:m].sort().values
        kb_pool = torch.cat([
            ka[: m // 2],  # overlap half with A
            torch.randperm(space, device="cuda")[:n] + space,
        ])
        kb = kb_pool[torch.randperm(kb_pool.shape[0], device="cuda")[:n]]
        kb = torch.unique(kb).sort().values
        n_eff = kb.shape[0]
        a1 = torch.randint(-5, 5, (m,), dtype=torch.int64, device="cuda")
        b1 = torch.randint(-5, 5, (n_eff,), dtype=torch.int64, device="cuda")
        aw = [ka.to(torch.int64), a1]
        bw = [kb.to(torch.int64), b1]
        perm_a = lex_sort_words(aw)
        aw = [w.index_select(0, perm_a) for w in aw]
        perm_b = lex_sort_words(bw)
        bw = [w.index_select(0, perm_b) for w in bw]
        a_accs = [
            torch.randint(-3, 4, (m,), dtype=torch.int64, device="cuda")
            for _ in range(nacc)
        ]
        b_accs = [
            torch.randint(-3, 4, (n_eff,), dtype=torch.int64, device="cuda")
            for _ in range(nacc)
        ]
        ow, oa, rep = ops.merge_consolidate_gpu(aw, a_accs, bw, b_accs)

        # brute-force reference on host
        import collections

        acc = collections.defaultdict(lambda: [0] * nacc)
        src = {}
        for i in range(m):
            kk = (int(aw[0][i]), int(aw[1][i]))
            for c in range(nacc):
                acc[kk][c] += int(a_accs[c][i])
            src.setdefault(kk, i)
        for i in range(n_eff):
            kk = (int(bw[0][i]), int(bw[1][i]))
            for c in range(nacc):
                acc[kk][c] += int(b_accs[c][i])
            src.setdefault(kk, m + i)
        expect = sorted(
            (kk, v) for kk, v in acc.items() if v[0] != 0
        )
        got = sorted(
            (
                (int(ow[0][i]), int(ow[1][i])),
                [int(oa[c][i]) for c in range(nacc)],
            )
            for i in range(ow[0].shape[0])
        )
        assert [g[0] for g in got] == [e[0] for e in expect], (m, n, nacc)
        assert [g[1] for g in got] == [e[1] for e in expect], (m, n, nacc)
        # rep prefers the A row on matches
        for i in range(ow[0].shape[0]):
            kk = (int(ow[0][i]), int(ow[1][i]))
            assert int(rep[i]) == src[kk], (m, n, nacc)


@gpu
@requires_cuda
def test_groupreduce_fused_merge_matches_cpu():
    """End-to-end: GPU groupby (fused merge path) equals CPU groupby."""
    import os

    import pathway_amd as pw
    from pathway_amd.internals.rungraph import G

    torch.manual_seed(5)
    rows = "\n".join(
        f"w{i % 37} | {i % 11 - 5}" for i in range(4000)
    )

    def run(dev):
        os.environ["PW_DEVICE"] = dev
        from pathway_amd.internals import config as _c

        G.clear()
        t = pw.debug.table_from_markdown("word | v\n" + rows)
        res = t.groupby(pw.this.word).reduce(
            pw.this.word, s=pw.reducers.sum(pw.this.v), c=pw.reducers.count()
        )
        _, cols = pw.debug.table_to_dicts(res)
        return sorted(zip(cols["word"].values(), cols["s"].values(),
                          cols["c"].values()))

    gpu_out = run("cuda:0")
    cpu_out = run("cpu")
    os.environ["PW_DEVICE"] = "cuda:0"
    assert gpu_out == cpu_out


@gpu
@requires_cuda
def test_radix_sort_kernel_vs_torch():
    from pathway_amd import ops

    torch.manual_seed(41)
    for n in (1, 2, 1000, 1 << 20, 12345):
        keys = torch.randint(-(2**62), 2**62, (n,), dtype=torch.int64,
                             device="cuda")
        skeys, perm = ops.radix_sort64_gpu(keys)
        ref_vals, ref_perm = torch.sort(keys, stable=True)
        assert torch.equal(skeys, ref_vals), n
        assert torch.equal(keys.index_select(0, perm), skeys), n
        assert torch.equal(perm, ref_perm), n  # stability

    # heavy duplicates exercise stability hard
    keys = torch.randint(0, 7, (100000,), dtype=torch.int64, device="cuda")
    skeys, perm = ops.radix_sort64_gpu(keys)
    rv, rp = torch.sort(keys, stable=True)
    assert torch.equal(skeys, rv)
    assert torch.equal(perm, rp)


@gpu
@requires_cuda
def test_arrangement_fused_merge_matches_cpu():
    """Arrangement.merge via the fused 4-word kernel equals the host
    merge+consolidate on randomized weighted multisets."""
    from pathway_amd.engine.column import TensorColumn
    from pathway_amd.engine.state import Arrangement
    from pathway_amd.internals import dtype as dt_

    torch.manual_seed(19)

    def build(device):
        proto = {"v": TensorColumn(torch.zeros(0, dtype=torch.int64,
                                               device=device), dt_.INT)}
        arr = Arrangement(torch.device(device), proto)
        g = torch.Generator().manual_seed(99)
        for batch in range(6):
            n = 3000
            keys = torch.randint(0, 500, (n, 2), dtype=torch.int64,
                                 generator=g)
            v0 = torch.randint(0, 50, (n,), dtype=torch.int64, generator=g)
            v1 = v0 * 7 + 1
            w = torch.randint(-2, 3, (n,), dtype=torch.int64, generator=g)
            nz = w != 0
            keys, v0, v1, w = keys[nz], v0[nz], v1[nz], w[nz]
            cols = {"v": TensorColumn(v0.to(device), dt_.INT)}
            arr.merge(keys.to(device), (v0.to(device), v1.to(device)),
                      w.to(device), cols)
        return arr

    cpu = build("cpu")
    gpu_ = build("cuda:0")
    rows_cpu = sorted(zip(
        cpu.key_words[0].tolist(), cpu.key_words[1].tolist(),
        cpu.vhash_words[0].tolist(), cpu.weights.tolist(),
        cpu.columns["v"].tensor.tolist(),
    ))
    rows_gpu = sorted(zip(
        gpu_.key_words[0].cpu().tolist(), gpu_.key_words[1].cpu().tolist(),
        gpu_.vhash_words[0].cpu().tolist(), gpu_.weights.cpu().tolist(),
        gpu_.columns["v"].tensor.cpu().tolist(),
    ))
    assert rows_cpu == rows_gpu


@gpu
@requires_cuda
def test_scan_positions_kernel():
    from pathway_amd import ops

    torch.manual_seed(3)
    for n in (0, 1, 17, 1000, 4097, 1 << 20):
        buf = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
        got = ops.scan_positions_gpu(buf, 10)
        ref = (buf == 10).nonzero(as_tuple=True)[0]
        assert torch.equal(got, ref), n
    # unaligned base (slice off one byte) exercises the scalar path
    buf = torch.randint(0, 256, (100001,), dtype=torch.uint8, device="cuda")
    sub = buf[1:]
    got = ops.scan_positions_gpu(sub, 10)
    ref = (sub == 10).nonzero(as_tuple=True)[0]
    assert torch.equal(got, ref)


@gpu
@requires_cuda
def test_sort_repair_collisions():
    """lex_sort_words fast path with FORCED word0 collisions: the sync-free
    odd-even repair (k_sort_repair) must restore full lex order, including
    a 4-long equal-k0 run."""
    import os

    from pathway_amd.engine.state import lex_sort_words

    torch.manual_seed(11)
    n = 1 << 20
    k0 = torch.randint(-(1 << 62), 1 << 62, (n,), device="cuda")
    k1 = torch.randint(-(1 << 62), 1 << 62, (n,), device="cuda")
    # collide: 200 duplicate pairs + one run of 4 equal k0
    dup = torch.randint(0, n, (200,), device="cuda")
    k0[dup] = k0[(dup + 1) % n]
    k0[10:14] = k0[10]
    os.environ["PW_DEBUG_SORT"] = "1"
    try:
        perm = lex_sort_words([k0, k1])  # debug flag asserts lex order
    finally:
        os.environ.pop("PW_DEBUG_SORT", None)
    s0 = k0.index_select(0, perm)
    s1 = k1.index_select(0, perm)
    ok = (s0[1:] > s0[:-1]) | ((s0[1:] == s0[:-1]) & (s1[1:] >= s1[:-1]))
    assert bool(ok.all())


@gpu
@requires_cuda
def test_device_hash_table():
    from pathway_amd import ops

    torch.manual_seed(5)
    for m in (0, 1, 1000, 50_000):
        klo = torch.randperm(1 << 20, device="cuda")[:m] * 7919 + 13
        khi = torch.randint(-(1 << 62), 1 << 62, (m,), device="cuda")
        vals = torch.randint(0, 1 << 40, (m,), device="cuda")
        ht = ops.DeviceHashTable(klo, khi, vals)
        # every inserted key must probe back to its value
        got, found = ht.probe(klo, khi)
        assert bool(found.all())
        assert torch.equal(got, vals)
        # identity-valued table (vals=None) returns the insert index
        ht2 = ops.DeviceHashTable(klo, khi)
        got2, found2 = ht2.probe(klo, khi)
        assert bool(found2.all())
        assert torch.equal(got2, torch.arange(m, device="cuda"))
        # unknown keys miss with -1
        miss_lo = klo + 1 if m else torch.tensor([42], device="cuda")
        miss_hi = khi if m else torch.tensor([43], device="cuda")
        gotm, foundm = ht.probe(miss_lo, miss_hi)
        assert not bool(foundm.any())
        assert bool((gotm == -1).all())


@gpu
@requires_cuda
def test_gather_cols_kernel():
    from pathway_amd import ops

    torch.manual_seed(4)
    n = 1 << 20
    for ncols in (1, 3, 8, 11):
        cols = [
            torch.randint(-(1 << 60), 1 << 60, (n,), device="cuda")
            if i % 2 == 0
            else torch.randn(n, dtype=torch.float64, device="cuda")
            for i in range(ncols)
        ]
        idx = torch.randint(0, n, (n // 3,), device="cuda")
        got = ops.gather_cols_gpu(idx, cols)
        for g, c in zip(got, cols):
            assert g.dtype == c.dtype
            assert torch.equal(g, c.index_select(0, idx))
