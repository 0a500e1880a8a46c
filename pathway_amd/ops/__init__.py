"""pathway_amd.ops — native HIP/CDNA4 kernel bindings (ctypes, no shims).

On a GPU host the hash hot path MUST run through libpwhip.so; if the
library is missing there we raise instead of silently falling back to the
torch reference implementation (which only serves as the CPU path and the
numerics reference for the kernels).
"""

from __future__ import annotations

import ctypes
import os
from typing import Sequence

import torch

_THIS = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_THIS, "libpwhip.so")

_lib = None
_load_error: Exception | None = None


def _try_load():
    global _lib, _load_error
    if _lib is not None:
        return _lib
    try:
        if not os.path.exists(_LIB_PATH):
            from pathway_amd.ops import build as _build

            _build.build(verbose=False)
        _lib = ctypes.CDLL(_LIB_PATH)
        _lib.pw_hash128_words.restype = ctypes.c_int
        _lib.pw_value_hash.restype = ctypes.c_int
        _lib.pw_varlen_hash.restype = ctypes.c_int
        _lib.pw_run_starts.restype = ctypes.c_int
    except Exception as e:  # noqa: BLE001
        _load_error = e
        _lib = None
    return _lib


def lib_available() -> bool:
    return _try_load() is not None


def require_lib():
    lib = _try_load()
    if lib is None:
        raise RuntimeError(
            f"pathway_amd native HIP library missing on a GPU host: {_load_error}"
        )
    return lib


def _stream_ptr() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def hash128_words_gpu(words: Sequence[torch.Tensor]) -> tuple[torch.Tensor, torch.Tensor]:
    """Fused device 128-bit hash of rows of int64 words."""
    lib = require_lib()
    n = words[0].shape[0]
    lo = torch.empty(n, dtype=torch.int64, device=words[0].device)
    hi = torch.empty(n, dtype=torch.int64, device=words[0].device)
    arr = (ctypes.c_void_p * len(words))(
        *[ctypes.c_void_p(w.contiguous().data_ptr()) for w in words]
    )
    rc = lib.pw_hash128_words(
        arr,
        ctypes.c_int(len(words)),
        ctypes.c_int64(n),
        ctypes.c_void_p(lo.data_ptr()),
        ctypes.c_void_p(hi.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_hash128_words failed: hip error {rc}")
    return lo, hi


def value_hash_gpu(payload: torch.Tensor, tag: int) -> tuple[torch.Tensor, torch.Tensor]:
    lib = require_lib()
    payload = payload.contiguous()
    n = payload.shape[0]
    lo = torch.empty(n, dtype=torch.int64, device=payload.device)
    hi = torch.empty(n, dtype=torch.int64, device=payload.device)
    rc = lib.pw_value_hash(
        ctypes.c_void_p(payload.data_ptr()),
        ctypes.c_uint64(tag),
        ctypes.c_int64(n),
        ctypes.c_void_p(lo.data_ptr()),
        ctypes.c_void_p(hi.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_value_hash failed: hip error {rc}")
    return lo, hi


def varlen_hash_gpu(
    bytes_t: torch.Tensor, offsets: torch.Tensor, tag: int
) -> tuple[torch.Tensor, torch.Tensor]:
    """Device hash of varlen byte rows: offsets (n+1,) int64, bytes uint8."""
    lib = require_lib()
    n = offsets.shape[0] - 1
    lo = torch.empty(n, dtype=torch.int64, device=bytes_t.device)
    hi = torch.empty(n, dtype=torch.int64, device=bytes_t.device)
    rc = lib.pw_varlen_hash(
        ctypes.c_void_p(bytes_t.contiguous().data_ptr()),
        ctypes.c_void_p(offsets.contiguous().data_ptr()),
        ctypes.c_uint64(tag),
        ctypes.c_int64(n),
        ctypes.c_void_p(lo.data_ptr()),
        ctypes.c_void_p(hi.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_varlen_hash failed: hip error {rc}")
    return lo, hi


def run_starts_gpu(words: Sequence[torch.Tensor]) -> torch.Tensor:
    lib = require_lib()
    n = words[0].shape[0]
    starts = torch.empty(n, dtype=torch.bool, device=words[0].device)
    arr = (ctypes.c_void_p * len(words))(
        *[ctypes.c_void_p(w.contiguous().data_ptr()) for w in words]
    )
    rc = lib.pw_run_starts(
        arr,
        ctypes.c_int(len(words)),
        ctypes.c_int64(n),
        ctypes.c_void_p(starts.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_run_starts failed: hip error {rc}")
    return starts


def host_hash128_bytes(data: bytes) -> tuple[int, int]:
    lib = require_lib()
    lo = ctypes.c_uint64()
    hi = ctypes.c_uint64()
    lib.pw_host_hash128_bytes(data, ctypes.c_int64(len(data)), ctypes.byref(lo), ctypes.byref(hi))
    return lo.value, hi.value


def _ptr_arr(tensors):
    return (ctypes.c_void_p * len(tensors))(
        *[ctypes.c_void_p(t.data_ptr()) for t in tensors]
    )


def searchsorted_gpu(
    sorted_words: Sequence[torch.Tensor],
    query_words: Sequence[torch.Tensor],
    side: str = "left",
) -> torch.Tensor:
    lib = require_lib()
    m = sorted_words[0].shape[0]
    nq = query_words[0].shape[0]
    out = torch.empty(nq, dtype=torch.int64, device=query_words[0].device)
    s = [t.contiguous() for t in sorted_words]
    q = [t.contiguous() for t in query_words]
    rc = lib.pw_searchsorted(
        _ptr_arr(s),
        _ptr_arr(q),
        ctypes.c_int(len(s)),
        ctypes.c_int64(m),
        ctypes.c_int64(nq),
        ctypes.c_int(1 if side == "right" else 0),
        ctypes.c_void_p(out.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_searchsorted failed: hip error {rc}")
    return out


def lookup_gpu(
    sorted_words: Sequence[torch.Tensor], query_words: Sequence[torch.Tensor]
) -> tuple[torch.Tensor, torch.Tensor]:
    lib = require_lib()
    m = sorted_words[0].shape[0]
    nq = query_words[0].shape[0]
    pos = torch.empty(nq, dtype=torch.int64, device=query_words[0].device)
    found = torch.empty(nq, dtype=torch.bool, device=query_words[0].device)
    s = [t.contiguous() for t in sorted_words]
    q = [t.contiguous() for t in query_words]
    rc = lib.pw_lookup(
        _ptr_arr(s),
        _ptr_arr(q),
        ctypes.c_int(len(s)),
        ctypes.c_int64(m),
        ctypes.c_int64(nq),
        ctypes.c_void_p(pos.data_ptr()),
        ctypes.c_void_p(found.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_lookup failed: hip error {rc}")
    return pos, found


def key_range_gpu(
    sorted_words: Sequence[torch.Tensor], query_words: Sequence[torch.Tensor]
) -> tuple[torch.Tensor, torch.Tensor]:
    lib = require_lib()
    m = sorted_words[0].shape[0]
    nq = query_words[0].shape[0]
    lo = torch.empty(nq, dtype=torch.int64, device=query_words[0].device)
    hi = torch.empty(nq, dtype=torch.int64, device=query_words[0].device)
    s = [t.contiguous() for t in sorted_words]
    q = [t.contiguous() for t in query_words]
    rc = lib.pw_key_range(
        _ptr_arr(s),
        _ptr_arr(q),
        ctypes.c_int(len(s)),
        ctypes.c_int64(m),
        ctypes.c_int64(nq),
        ctypes.c_void_p(lo.data_ptr()),
        ctypes.c_void_p(hi.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_key_range failed: hip error {rc}")
    return lo, hi


def pool_hash_gpu(
    codes: torch.Tensor,
    pool_lo: torch.Tensor,
    pool_hi: torch.Tensor,
    none_lo: int,
    none_hi: int,
) -> tuple[torch.Tensor, torch.Tensor]:
    lib = require_lib()
    n = codes.shape[0]
    lo = torch.empty(n, dtype=torch.int64, device=codes.device)
    hi = torch.empty(n, dtype=torch.int64, device=codes.device)
    rc = lib.pw_pool_hash(
        ctypes.c_void_p(codes.contiguous().data_ptr()),
        ctypes.c_void_p(pool_lo.contiguous().data_ptr()),
        ctypes.c_void_p(pool_hi.contiguous().data_ptr()),
        ctypes.c_uint64(none_lo & ((1 << 64) - 1)),
        ctypes.c_uint64(none_hi & ((1 << 64) - 1)),
        ctypes.c_int64(n),
        ctypes.c_void_p(lo.data_ptr()),
        ctypes.c_void_p(hi.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_pool_hash failed: hip error {rc}")
    return lo, hi


def varlen_hash_se_gpu(
    bytes_t: torch.Tensor,
    starts: torch.Tensor,
    ends: torch.Tensor,
    tag: int,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Device hash of varlen rows with explicit [start, end) spans."""
    lib = require_lib()
    n = starts.shape[0]
    lo = torch.empty(n, dtype=torch.int64, device=bytes_t.device)
    hi = torch.empty(n, dtype=torch.int64, device=bytes_t.device)
    rc = lib.pw_varlen_hash_se(
        ctypes.c_void_p(bytes_t.contiguous().data_ptr()),
        ctypes.c_void_p(starts.contiguous().data_ptr()),
        ctypes.c_void_p(ends.contiguous().data_ptr()),
        ctypes.c_uint64(tag),
        ctypes.c_int64(n),
        ctypes.c_void_p(lo.data_ptr()),
        ctypes.c_void_p(hi.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_varlen_hash_se failed: hip error {rc}")
    return lo, hi


def hash_agg_gpu(
    k0: torch.Tensor,
    k1: torch.Tensor,
    contribs: Sequence[torch.Tensor],
    expected_uniques: int | None = None,
) -> tuple[torch.Tensor, torch.Tensor, list, torch.Tensor] | None:
    """Sort-free additive pre-aggregation (pw_hash_agg): returns
    (uk0, uk1, [accs...], rep_row_idx) for the distinct group keys of the
    batch — UNSORTED and possibly with rare duplicate keys (publication
    race); callers sort + consolidate the (small) result.

    expected_uniques sizes the table at ~2x that (L2-resident when the
    estimate is vocabulary-scale — the round-1 A/B lost because the 2n
    table was a 270 MB HBM random walk).  Returns None if the estimate
    was too small (probe-chain overflow): redo on the sort path with a
    bigger estimate."""
    lib = require_lib()
    n = k0.shape[0]
    dev = k0.device
    nacc = len(contribs)
    est = n if expected_uniques is None else min(max(expected_uniques, 512), n)
    cap = 1 << max(10, (2 * est - 1).bit_length())
    out_cap = min(n, cap)
    tk0 = torch.empty(cap, dtype=torch.int64, device=dev)
    tk1 = torch.zeros(cap, dtype=torch.int64, device=dev)
    rep = torch.zeros(cap, dtype=torch.int64, device=dev)
    taccs = [torch.zeros(cap, dtype=torch.int64, device=dev) for _ in range(nacc)]
    counter = torch.zeros(2, dtype=torch.int32, device=dev)  # [count, overflow]
    out_k0 = torch.empty(out_cap, dtype=torch.int64, device=dev)
    out_k1 = torch.empty(out_cap, dtype=torch.int64, device=dev)
    out_rep = torch.empty(out_cap, dtype=torch.int64, device=dev)
    out_accs = [
        torch.empty(out_cap, dtype=torch.int64, device=dev) for _ in range(nacc)
    ]
    carr = (ctypes.c_void_p * max(nacc, 1))(
        *[ctypes.c_void_p(c.contiguous().data_ptr()) for c in contribs]
    )
    tarr = (ctypes.c_void_p * max(nacc, 1))(
        *[ctypes.c_void_p(c.data_ptr()) for c in taccs]
    )
    oarr = (ctypes.c_void_p * max(nacc, 1))(
        *[ctypes.c_void_p(c.data_ptr()) for c in out_accs]
    )
    rc = lib.pw_hash_agg(
        ctypes.c_void_p(k0.contiguous().data_ptr()),
        ctypes.c_void_p(k1.contiguous().data_ptr()),
        carr,
        ctypes.c_int(nacc),
        ctypes.c_int64(n),
        ctypes.c_void_p(tk0.data_ptr()),
        ctypes.c_void_p(tk1.data_ptr()),
        tarr,
        ctypes.c_void_p(rep.data_ptr()),
        ctypes.c_int64(cap),
        ctypes.c_void_p(counter.data_ptr()),
        ctypes.c_void_p(out_k0.data_ptr()),
        ctypes.c_void_p(out_k1.data_ptr()),
        oarr,
        ctypes.c_void_p(out_rep.data_ptr()),
        ctypes.c_void_p(counter.data_ptr() + 4),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_hash_agg failed: hip error {rc}")
    # counter is the only host-visible size: one small D2H sync
    m, overflowed = (int(x) for x in counter.tolist())
    if overflowed:
        return None
    return (
        out_k0.narrow(0, 0, m),
        out_k1.narrow(0, 0, m),
        [a.narrow(0, 0, m) for a in out_accs],
        out_rep.narrow(0, 0, m),
    )


def seg_reduce_words_gpu(
    words: Sequence[torch.Tensor], contribs: Sequence[torch.Tensor]
) -> tuple[list, torch.Tensor, list]:
    """Fused segmented reduce over rows SORTED by `words` (≤8 int64 word
    columns): returns (unique word columns, first_row_idx, [per-seg int64
    sums...]) — the whole run-starts/compaction/segment-sum chain in two
    kernels + one tiny cumsum."""
    lib = require_lib()
    n = words[0].shape[0]
    dev = words[0].device
    nw = len(words)
    nacc = len(contribs)
    words = [w.contiguous() for w in words]
    nblocks = (n + 255) // 256
    block_counts = torch.empty(nblocks, dtype=torch.int32, device=dev)
    warr = (ctypes.c_void_p * nw)(
        *[ctypes.c_void_p(w.data_ptr()) for w in words]
    )
    rc = lib.pw_seg_reduce_count(
        warr,
        ctypes.c_int(nw),
        ctypes.c_int64(n),
        ctypes.c_void_p(block_counts.data_ptr()),
        ctypes.c_int64(nblocks),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_seg_reduce_count failed: hip error {rc}")
    csum = torch.cumsum(block_counts.to(torch.int64), 0)
    nseg = int(csum[-1].item())
    bases = torch.cat(
        [torch.zeros(1, dtype=torch.int64, device=dev), csum[:-1]]
    )
    out_words = [torch.empty(nseg, dtype=torch.int64, device=dev) for _ in range(nw)]
    out_first = torch.empty(nseg, dtype=torch.int64, device=dev)
    out_accs = [torch.zeros(nseg, dtype=torch.int64, device=dev) for _ in range(nacc)]
    carr = (ctypes.c_void_p * max(nacc, 1))(
        *[ctypes.c_void_p(c.contiguous().data_ptr()) for c in contribs]
    )
    oarr = (ctypes.c_void_p * max(nacc, 1))(
        *[ctypes.c_void_p(c.data_ptr()) for c in out_accs]
    )
    owarr = (ctypes.c_void_p * nw)(
        *[ctypes.c_void_p(w.data_ptr()) for w in out_words]
    )
    rc = lib.pw_seg_reduce_emit(
        warr,
        ctypes.c_int(nw),
        carr,
        ctypes.c_int(nacc),
        ctypes.c_int64(n),
        ctypes.c_void_p(bases.data_ptr()),
        owarr,
        ctypes.c_void_p(out_first.data_ptr()),
        oarr,
        ctypes.c_int64(nblocks),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_seg_reduce_emit failed: hip error {rc}")
    return out_words, out_first, out_accs


def seg_reduce_gpu(
    k0: torch.Tensor, k1: torch.Tensor, contribs: Sequence[torch.Tensor]
) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor, list]:
    """2-word convenience wrapper over seg_reduce_words_gpu."""
    out_words, out_first, out_accs = seg_reduce_words_gpu([k0, k1], contribs)
    return out_words[0], out_words[1], out_first, out_accs


def partition_gpu(dest: torch.Tensor, world: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Radix partition by destination rank: returns (perm, counts) with rows
    grouped by destination (exchange shuffle pack; pact.rs:56 analog)."""
    lib = require_lib()
    n = dest.shape[0]
    nblocks = max(1, min(2048, (n + 255) // 256))
    perm = torch.empty(n, dtype=torch.int64, device=dest.device)
    counts = torch.empty(world, dtype=torch.int64, device=dest.device)
    scratch = torch.empty(nblocks * world, dtype=torch.int64, device=dest.device)
    rc = lib.pw_partition(
        ctypes.c_void_p(dest.contiguous().data_ptr()),
        ctypes.c_int64(n),
        ctypes.c_int(world),
        ctypes.c_void_p(perm.data_ptr()),
        ctypes.c_void_p(counts.data_ptr()),
        ctypes.c_void_p(scratch.data_ptr()),
        ctypes.c_int64(nblocks),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_partition failed: hip error {rc}")
    return perm, counts


def gemm_bias_act_gpu(
    a: torch.Tensor,
    b_t: torch.Tensor,
    bias: torch.Tensor | None = None,
    act: str = "none",
) -> torch.Tensor:
    """Hand-written MFMA bf16 GEMM with fused bias(+GELU) epilogue.

    a (M,K) bf16, b_t = B^T (N,K) bf16 row-major (weights are static, so
    the transpose is one-time); bias (N,) float32; returns (M,N) bf16.
    mfma_f32_16x16x32_bf16 tiles, fp32 accumulate.
    """
    lib = require_lib()
    assert a.dtype == torch.bfloat16 and b_t.dtype == torch.bfloat16
    M, K = a.shape
    N, K2 = b_t.shape
    assert K == K2
    out = torch.empty((M, N), dtype=torch.bfloat16, device=a.device)
    if bias is not None:
        bias = bias.to(torch.float32).contiguous()
    act_code = {"none": 0, "gelu": 1}[act]
    rc = lib.pw_gemm_bf16(
        ctypes.c_void_p(a.contiguous().data_ptr()),
        ctypes.c_void_p(b_t.contiguous().data_ptr()),
        ctypes.c_void_p(bias.data_ptr() if bias is not None else 0),
        ctypes.c_void_p(out.data_ptr()),
        ctypes.c_int64(M),
        ctypes.c_int64(N),
        ctypes.c_int64(K),
        ctypes.c_int(act_code),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_gemm_bf16 failed: hip error {rc}")
    return out


def topk_gpu(scores: torch.Tensor, k: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Per-row top-k (higher=better) via the hand-written HIP kernel.

    scores (nq, m) float32 -> (vals (nq, k) f32, idx (nq, k) int64).
    """
    lib = require_lib()
    nq, m = scores.shape
    assert scores.dtype == torch.float32
    vals = torch.empty((nq, k), dtype=torch.float32, device=scores.device)
    idx = torch.empty((nq, k), dtype=torch.int64, device=scores.device)
    if nq == 0:
        return vals, idx
    rc = lib.pw_topk(
        ctypes.c_void_p(scores.contiguous().data_ptr()),
        ctypes.c_int64(nq),
        ctypes.c_int64(m),
        ctypes.c_int(k),
        ctypes.c_void_p(vals.data_ptr()),
        ctypes.c_void_p(idx.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_topk failed: hip error {rc}")
    return vals, idx


def merge_consolidate_gpu(
    a_words: Sequence[torch.Tensor],
    a_accs: Sequence[torch.Tensor],
    b_words: Sequence[torch.Tensor],
    b_accs: Sequence[torch.Tensor],
    compare_words: int | None = None,
) -> tuple[list[torch.Tensor], list[torch.Tensor], torch.Tensor]:
    """Fused LSM merge+consolidate of two unique lex-sorted row sets.

    Rows are nw (<=4) int64 words; comparison uses the first
    `compare_words` (default all — pass 2 when the tail words are
    key-determined).  acc slot 0 is the weight; merged rows with zero
    weight are dropped.  Returns (out_words, accs, rep) with rep
    indexing concat([A, B]) rows (A preferred on matches) for
    carried-column gathers.
    """
    lib = require_lib()
    m = a_words[0].shape[0]
    n = b_words[0].shape[0]
    nw = len(a_words)
    nwc = compare_words or nw
    nacc = len(a_accs)
    assert len(b_accs) == nacc and 1 <= nacc <= 8 and 1 <= nwc <= nw <= 4
    device = a_words[0].device
    total_diag = m + n
    nthreads = max(1, (total_diag + 7) // 8)
    counts = torch.empty(nthreads, dtype=torch.int32, device=device)
    wA = [t.contiguous() for t in a_words]
    wB = [t.contiguous() for t in b_words]
    aA = [t.contiguous() for t in a_accs]
    aB = [t.contiguous() for t in b_accs]
    rc = lib.pw_merge_consolidate_count(
        _ptr_arr(wA),
        _ptr_arr(aA),
        _ptr_arr(wB),
        _ptr_arr(aB),
        ctypes.c_int(nw),
        ctypes.c_int(nwc),
        ctypes.c_int(nacc),
        ctypes.c_int64(m),
        ctypes.c_int64(n),
        ctypes.c_void_p(counts.data_ptr()),
        ctypes.c_int64(nthreads),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_merge_consolidate_count failed: {rc}")
    csum = torch.cumsum(counts.to(torch.int64), 0)
    total = int(csum[-1].item())
    bases = torch.zeros(nthreads, dtype=torch.int64, device=device)
    bases[1:] = csum[:-1]
    out_words = [
        torch.empty(total, dtype=torch.int64, device=device)
        for _ in range(nw)
    ]
    out_accs = [
        torch.empty(total, dtype=torch.int64, device=device)
        for _ in range(nacc)
    ]
    rep = torch.empty(total, dtype=torch.int64, device=device)
    rc = lib.pw_merge_consolidate_emit(
        _ptr_arr(wA),
        _ptr_arr(aA),
        _ptr_arr(wB),
        _ptr_arr(aB),
        ctypes.c_int(nw),
        ctypes.c_int(nwc),
        ctypes.c_int(nacc),
        ctypes.c_int64(m),
        ctypes.c_int64(n),
        ctypes.c_void_p(bases.data_ptr()),
        ctypes.c_int64(nthreads),
        _ptr_arr(out_words),
        _ptr_arr(out_accs),
        ctypes.c_void_p(rep.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_merge_consolidate_emit failed: {rc}")
    return out_words, out_accs, rep


def radix_sort64_gpu(keys: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Hand-written LSD radix sort of int64 keys (signed order).

    Returns (sorted_keys, perm) with perm stable: equal keys keep their
    input order.  8x 8-bit digit passes; the (digit, block) scan runs as
    one torch cumsum between kernel launches.
    """
    lib = require_lib()
    n = keys.shape[0]
    device = keys.device
    perm = torch.arange(n, dtype=torch.int64, device=device)
    if n <= 1:
        return keys.clone(), perm
    nblocks = max(1, min(1024, (n + (64 * 64) - 1) // (64 * 64)))
    chunk = (n + nblocks * 64 - 1) // (nblocks * 64)
    ka = torch.empty(n, dtype=torch.int64, device=device)  # uint64 bits
    kb = torch.empty(n, dtype=torch.int64, device=device)
    pa = perm
    pb = torch.empty(n, dtype=torch.int64, device=device)
    rc = lib.pw_radix_flip(
        ctypes.c_void_p(keys.contiguous().data_ptr()),
        ctypes.c_void_p(ka.data_ptr()),
        ctypes.c_int64(n),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_radix_flip failed: {rc}")
    counts = torch.empty(256 * nblocks, dtype=torch.int32, device=device)
    for p in range(8):
        shift = p * 8
        rc = lib.pw_radix_count(
            ctypes.c_void_p(ka.data_ptr()),
            ctypes.c_int64(n),
            ctypes.c_int(shift),
            ctypes.c_int64(nblocks),
            ctypes.c_int64(chunk),
            ctypes.c_void_p(counts.data_ptr()),
            _stream_ptr(),
        )
        if rc != 0:
            raise RuntimeError(f"pw_radix_count failed: {rc}")
        csum = torch.cumsum(counts.to(torch.int64), 0)
        bases = torch.zeros_like(csum)
        bases[1:] = csum[:-1]
        rc = lib.pw_radix_scatter(
            ctypes.c_void_p(ka.data_ptr()),
            ctypes.c_void_p(pa.data_ptr()),
            ctypes.c_int64(n),
            ctypes.c_int(shift),
            ctypes.c_int64(nblocks),
            ctypes.c_int64(chunk),
            ctypes.c_void_p(bases.data_ptr()),
            ctypes.c_void_p(kb.data_ptr()),
            ctypes.c_void_p(pb.data_ptr()),
            _stream_ptr(),
        )
        if rc != 0:
            raise RuntimeError(f"pw_radix_scatter failed: {rc}")
        ka, kb = kb, ka
        pa, pb = pb, pa
    out = torch.empty(n, dtype=torch.int64, device=device)
    rc = lib.pw_radix_unflip(
        ctypes.c_void_p(ka.data_ptr()),
        ctypes.c_void_p(out.data_ptr()),
        ctypes.c_int64(n),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_radix_unflip failed: {rc}")
    return out, pa


def scan_positions_gpu(buf: torch.Tensor, target: int) -> torch.Tensor:
    """Ordered positions where buf == target (uint8 buffer) — the
    newline/separator scan of the ingest parse (replaces nonzero)."""
    lib = require_lib()
    n = buf.shape[0]
    device = buf.device
    nblocks = max(1, min(2048, (n + 4095) // 4096))
    counts = torch.empty(nblocks, dtype=torch.int64, device=device)
    rc = lib.pw_scan_positions(
        ctypes.c_void_p(buf.contiguous().data_ptr()),
        ctypes.c_int64(n),
        ctypes.c_int(int(target)),
        ctypes.c_void_p(counts.data_ptr()),
        ctypes.c_int64(nblocks),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_scan_positions failed: {rc}")
    csum = torch.cumsum(counts, 0)
    total = int(csum[-1].item())
    bases = torch.zeros(nblocks, dtype=torch.int64, device=device)
    bases[1:] = csum[:-1]
    out = torch.empty(total, dtype=torch.int64, device=device)
    rc = lib.pw_scan_emit(
        ctypes.c_void_p(buf.contiguous().data_ptr()),
        ctypes.c_int64(n),
        ctypes.c_int(int(target)),
        ctypes.c_void_p(bases.data_ptr()),
        ctypes.c_int64(nblocks),
        ctypes.c_void_p(out.data_ptr()),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_scan_emit failed: {rc}")
    return out


def sort_repair_gpu(
    k0_sorted: torch.Tensor,
    k1_sorted: torch.Tensor,
    perm: torch.Tensor,
    passes: int = 4,
) -> None:
    """In-place odd-even repair of word1 order inside equal-word0 runs
    (k_sort_repair) — the sync-free replacement for the 2-word sort fast
    path's collision check.  Mutates k1_sorted and perm."""
    lib = require_lib()
    n = k0_sorted.shape[0]
    if n < 2:
        return
    rc = lib.pw_sort_repair(
        ctypes.c_void_p(k0_sorted.data_ptr()),
        ctypes.c_void_p(k1_sorted.data_ptr()),
        ctypes.c_void_p(perm.data_ptr()),
        ctypes.c_int64(n),
        ctypes.c_int(int(passes)),
        _stream_ptr(),
    )
    if rc != 0:
        raise RuntimeError(f"pw_sort_repair failed: hip error {rc}")


def gather_cols_gpu(idx: torch.Tensor, cols: list) -> list:
    """Gather up to 8 8-byte columns through one shared int64 index in a
    single launch (k_gather_cols) — replaces per-column index_select on
    the arrange/merge/consolidate permutation paths.  Columns with a
    non-8-byte itemsize fall back to torch indexing."""
    lib = require_lib()
    m = idx.shape[0]
    outs: list = [None] * len(cols)
    fused_pos: list[int] = []
    for i, c in enumerate(cols):
        if c.dim() == 1 and c.element_size() == 8 and c.is_contiguous():
            fused_pos.append(i)
        else:
            outs[i] = c[idx] if c.dim() > 1 else c.index_select(0, idx)
    idx = idx.contiguous()
    for start in range(0, len(fused_pos), 8):
        group = fused_pos[start : start + 8]
        srcs = (ctypes.c_void_p * len(group))()
        dsts = (ctypes.c_void_p * len(group))()
        for k, i in enumerate(group):
            c = cols[i]
            out = torch.empty(m, dtype=c.dtype, device=c.device)
            outs[i] = out
            srcs[k] = c.data_ptr()
            dsts[k] = out.data_ptr()
        rc = lib.pw_gather_cols(
            ctypes.c_void_p(idx.data_ptr()),
            ctypes.c_int64(m),
            ctypes.c_int(len(group)),
            srcs,
            dsts,
            _stream_ptr(),
        )
        if rc != 0:
            raise RuntimeError(f"pw_gather_cols failed: {rc}")
    return outs


class DeviceHashTable:
    """Open-addressing device hash table over 128-bit keys -> int64 values
    (k_ht_build / k_ht_probe).  Build once per dictionary version; probe
    per delta batch.  Load factor <= 0.5 (nslots = next pow2 >= 2m)."""

    def __init__(self, klo: torch.Tensor, khi: torch.Tensor,
                 vals: torch.Tensor | None = None):
        lib = require_lib()
        m = klo.shape[0]
        device = klo.device
        nslots = 1 << max(4, (2 * m - 1).bit_length()) if m else 16
        self.nslots = nslots
        self.tab_lo = torch.empty(nslots, dtype=torch.int64, device=device)
        self.tab_hi = torch.empty(nslots, dtype=torch.int64, device=device)
        self.tab_val = torch.full((nslots,), -1, dtype=torch.int64, device=device)
        if m:
            rc = lib.pw_ht_build(
                ctypes.c_void_p(klo.contiguous().data_ptr()),
                ctypes.c_void_p(khi.contiguous().data_ptr()),
                ctypes.c_void_p(vals.contiguous().data_ptr()) if vals is not None else None,
                ctypes.c_int64(m),
                ctypes.c_void_p(self.tab_lo.data_ptr()),
                ctypes.c_void_p(self.tab_hi.data_ptr()),
                ctypes.c_void_p(self.tab_val.data_ptr()),
                ctypes.c_int64(nslots),
                _stream_ptr(),
            )
            if rc != 0:
                raise RuntimeError(f"pw_ht_build failed: hip error {rc}")

    def probe(self, qlo: torch.Tensor, qhi: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        """Returns (values, found); values are -1 where not found."""
        lib = require_lib()
        nq = qlo.shape[0]
        out = torch.empty(nq, dtype=torch.int64, device=qlo.device)
        found = torch.empty(nq, dtype=torch.bool, device=qlo.device)
        if nq == 0:
            return out, found
        rc = lib.pw_ht_probe(
            ctypes.c_void_p(qlo.contiguous().data_ptr()),
            ctypes.c_void_p(qhi.contiguous().data_ptr()),
            ctypes.c_int64(nq),
            ctypes.c_void_p(self.tab_lo.data_ptr()),
            ctypes.c_void_p(self.tab_hi.data_ptr()),
            ctypes.c_void_p(self.tab_val.data_ptr()),
            ctypes.c_int64(self.nslots),
            ctypes.c_void_p(out.data_ptr()),
            ctypes.c_void_p(found.data_ptr()),
            _stream_ptr(),
        )
        if rc != 0:
            raise RuntimeError(f"pw_ht_probe failed: hip error {rc}")
        return out, found


def gather_all(idx: torch.Tensor, tensors: list) -> list:
    """Gather a list of same-length tensors through one index — fused
    k_gather_cols on device, per-tensor index_select elsewhere."""
    if tensors and tensors[0].is_cuda and idx.shape[0] > 2048 and lib_available():
        return gather_cols_gpu(idx, list(tensors))
    return [
        t.index_select(0, idx) if t.dim() == 1 else t[idx] for t in tensors
    ]
