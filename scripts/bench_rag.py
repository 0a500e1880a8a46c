"""KNN / embedder / RAG serving benchmarks (BASELINE configs 3 and 5 shapes).

Prints one JSON line per benchmark; run on 1 MI355X via gpurun.
"""
from __future__ import annotations

import json
import time

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def bench_knn(n_index=1_000_000, dim=384, n_queries=1024, k=10, iters=20):
    from pathway_amd.engine.nodes_index import VectorIndexState

    device = "cuda" if torch.cuda.is_available() else "cpu"
    st = VectorIndexState(torch.device(device), metric="cos")
    g = torch.Generator(device="cpu").manual_seed(0)
    vecs = torch.randn(n_index, dim, generator=g).to(device)
    keys = torch.randint(-2**62, 2**62, (n_index, 2), dtype=torch.int64, generator=g).to(device)
    diffs = torch.ones(n_index, dtype=torch.int64, device=device)
    st.update(keys, vecs, diffs)
    q = torch.randn(n_queries, dim, generator=g).to(device)
    # warmup
    for _ in range(3):
        st.search(q, k)
    if device != "cpu":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ids, scores, _ = st.search(q, k)
    if device != "cpu":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    qps = n_queries * iters / dt
    print(json.dumps({
        "bench": "knn_cosine_topk",
        "index_size": n_index, "dim": dim, "k": k,
        "batch_queries": n_queries,
        "qps": qps,
        "ms_per_query_batch": dt / iters * 1000,
        "device": device,
    }))
    return qps


def bench_embedder(batch=1024, iters=10, seq_len=64):
    from pathway_amd.xpacks.llm._encoder import get_encoder

    enc = get_encoder()
    texts = [" ".join(f"tok{i}_{j}" for j in range(seq_len)) for i in range(batch)]
    enc.encode(texts[:64])  # warmup
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        enc.encode(texts)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    tps = batch * iters / dt
    print(json.dumps({
        "bench": "embedder_forward_bge_small_class",
        "batch": batch, "seq_len": seq_len,
        "texts_per_s": tps,
        "ms_per_batch": dt / iters * 1000,
        "dtype": str(enc.dtype),
        "device": str(enc.device),
    }))
    return tps


def bench_rag_serving(n_docs=20000, n_queries=200, qps: float | None = None):
    """DocumentStore retrieve latency: docs indexed once, then per-query
    incremental steps (p50/p95 end-to-end in-engine latency)."""
    from pathway_amd.debug import table_from_rows
    from pathway_amd.engine.runtime import Runtime, CaptureNode, PushSource
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe
    from pathway_amd.internals.api import Pointer, hash_values
    from pathway_amd.internals import dtype as dt
    from pathway_amd.xpacks.llm.document_store import DocumentStore

    schema = schema_from_types(data=bytes, _metadata=dict)
    rng = np.random.default_rng(0)
    words = [f"w{i}" for i in range(5000)]
    docs_rows = []
    for i in range(n_docs):
        text = " ".join(rng.choice(words, size=24))
        docs_rows.append((text.encode(), {"path": f"doc{i}.txt"}))
    docs = table_from_rows(schema, docs_rows)
    store = DocumentStore(docs)
    qschema = DocumentStore.RetrieveQuerySchema
    names = qschema.column_names()
    dts = [qschema.__columns__[n].dtype for n in names]
    src = PushSource(names, dts)
    qnode = InputNode(src, get_device())
    qtable = Table(qnode, {n: d for n, d in zip(names, dts)}, Universe())
    result = store.retrieve_query(qtable)
    cap = CaptureNode(result._node, get_device())
    rt = Runtime([cap], device=get_device())
    t_ing0 = time.perf_counter()
    rt.run()  # ingest + embed + index docs
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    ingest_s = time.perf_counter() - t_ing0
    lat = []
    t = 100
    start = time.perf_counter()
    for i in range(n_queries):
        if qps:
            target = start + i / qps
            now = time.perf_counter()
            if now < target:
                time.sleep(target - now)
        qtext = " ".join(rng.choice(words, size=8))
        lo, hi = hash_values([i, "q"])
        s0 = time.perf_counter()
        src.push(Pointer(lo, hi), [qtext, 5, None, None], t)
        rt.run()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        lat.append(time.perf_counter() - s0)
        t += 2
    lat_ms = sorted(x * 1000 for x in lat)
    p50 = lat_ms[len(lat_ms) // 2]
    p95 = lat_ms[int(len(lat_ms) * 0.95)]
    print(json.dumps({
        "bench": "rag_retrieve_latency",
        "n_docs": n_docs, "n_queries": n_queries, "k": 5,
        "ingest_s": ingest_s,
        "p50_ms": p50, "p95_ms": p95,
        "fixed_qps": qps,
        "qps_serial": 1000.0 / p50,
        "device": str(get_device()),
    }))


def bench_knn_e2e(n_index=1_000_000, n_queries=1024, k=10, iters=10):
    """BASELINE config 3 shape: bge-small-class embedder bf16 + cosine
    top-k over a 1M-vector index — text in, neighbors out."""
    from pathway_amd.engine.nodes_index import VectorIndexState
    from pathway_amd.xpacks.llm._encoder import get_encoder

    device = "cuda" if torch.cuda.is_available() else "cpu"
    enc = get_encoder()
    dim = enc.cfg.dim
    st = VectorIndexState(torch.device(device), metric="cos")
    g = torch.Generator(device="cpu").manual_seed(0)
    vecs = torch.randn(n_index, dim, generator=g).to(device)
    keys = torch.randint(-2**62, 2**62, (n_index, 2), dtype=torch.int64, generator=g).to(device)
    st.update(keys, vecs, torch.ones(n_index, dtype=torch.int64, device=device))
    texts = [f"query about topic {i} with some more words appended" for i in range(n_queries)]
    # warmup
    q = torch.from_numpy(np.stack(enc.encode(texts[:64]))).to(device)
    st.search(q, k)
    if device != "cpu":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        emb = torch.from_numpy(np.stack(enc.encode(texts))).to(device)
        ids, scores, _ = st.search(emb, k)
    if device != "cpu":
        torch.cuda.synchronize()
    dt_s = time.perf_counter() - t0
    print(json.dumps({
        "bench": "knn_e2e_embed_plus_topk",
        "index_size": n_index, "dim": dim, "k": k,
        "batch_queries": n_queries,
        "qps": n_queries * iters / dt_s,
        "ms_per_batch": dt_s / iters * 1000,
        "dtype": "bf16 embedder + f32 scores",
        "device": device,
    }))


if __name__ == "__main__":
    import sys

    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("all", "knn"):
        bench_knn()
    if which in ("all", "embedder"):
        bench_embedder()
    if which in ("all", "rag"):
        bench_rag_serving()
    if which in ("all", "rag_qps"):
        bench_rag_serving(qps=40.0)
    if which in ("all", "knn_e2e"):
        bench_knn_e2e()
