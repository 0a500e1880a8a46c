"""GPU-resident sorted arrangements — the engine's stateful data structure.

The MI355X-native replacement for differential-dataflow's Spine trace
(reference: spine_fueled.rs, merge_batcher.rs, ord.rs): instead of an LSM of
immutable batches with fueled merging, state lives as ONE consolidated
sorted columnar arrangement in HBM3E and every micro-batch merge is a full
sorted merge — at ~8 TB/s a full rewrite of even a 10^8-row arrangement is
sub-millisecond-scale, so the LSM's amortization trick buys nothing on this
hardware and costs random access.

Rows are ordered lexicographically by (key0, key1, vhash0, vhash1) where
vhash is the 128-bit hash of the row's values — consolidation (diff
summing, reference consolidation.rs) merges rows identical in key AND
value, exactly differential's (data, diff) semantics.
"""

from __future__ import annotations

from typing import Sequence

import os as _os

import torch

from pathway_amd.engine.batch import segmented_arange
from pathway_amd.engine.column import Column, concat_columns

#: opt-in for the hand-written radix sort kernel.  Measured on MI355X
#: (profiles/kernels_r02.md): pw_radix_sort64 reaches 2.05 Gkeys/s @16M,
#: rocPRIM's onesweep (torch.sort) 11 Gkeys/s — onesweep's decoupled-
#: lookback single-pass structure wins, so the library stays the default
#: and the pw kernel remains available and benchmarked.
_PW_PW_SORT = bool(_os.environ.get("PW_PW_SORT"))


def lex_sort_words(words: Sequence[torch.Tensor]) -> torch.Tensor:
    """Stable lexicographic argsort over parallel int64 word tensors.

    Fast path for 2-word (128-bit hash) keys: random hash keys almost never
    collide in the primary word, so sort by word0 alone and fall back to
    the full multi-pass only if adjacent duplicates exist (exact check)."""
    n = words[0].shape[0]
    perm = torch.arange(n, dtype=torch.int64, device=words[0].device)
    if n <= 1:
        return perm
    if len(words) == 2 and n > 2048:
        # INVARIANT: this fast path is sound only
        # because every caller sorts FULL row identities — equal (k0,k1)
        # rows are interchangeable for grouping/merging.  A caller
        # sorting by a strict prefix of a row's identity would break
        # silently; tests/test_fuzz_equivalence.py pins the property and
        # PW_DEBUG_SORT=1 verifies full lex order on every call.
        if words[0].is_cuda:
            from pathway_amd import ops

            if _PW_PW_SORT:
                k0_sorted, perm0 = ops.radix_sort64_gpu(words[0].contiguous())
            else:
                k0_sorted, perm0 = torch.sort(words[0])
            if ops.lib_available():
                # sync-free collision handling: odd-even repair of word1
                # order inside equal-word0 runs (k_sort_repair).  The
                # old host `.any()` check cost 0.27 ms/step in syncs —
                # profiles/kernels_r02.md.
                k1_sorted = words[1].index_select(0, perm0)
                ops.sort_repair_gpu(k0_sorted, k1_sorted, perm0)
                if _os.environ.get("PW_DEBUG_SORT"):
                    ok = bool(
                        (
                            (k0_sorted[1:] > k0_sorted[:-1])
                            | (
                                (k0_sorted[1:] == k0_sorted[:-1])
                                & (k1_sorted[1:] >= k1_sorted[:-1])
                            )
                        ).all()
                    )
                    assert ok, "sort repair produced non-lex order"
                return perm0
        else:
            k0_sorted, perm0 = torch.sort(words[0])
        k1_sorted = words[1].index_select(0, perm0)
        # only a k0 collision between rows with DIFFERENT k1 violates lex
        # order (equal (k0,k1) rows may appear in any relative order —
        # grouping and merging treat them identically)
        violation = bool(
            (
                (k0_sorted[1:] == k0_sorted[:-1])
                & (k1_sorted[1:] != k1_sorted[:-1])
            ).any()
        )
        if not violation:
            if _os.environ.get("PW_DEBUG_SORT"):
                ok = bool(
                    (
                        (k0_sorted[1:] > k0_sorted[:-1])
                        | (
                            (k0_sorted[1:] == k0_sorted[:-1])
                            & (k1_sorted[1:] >= k1_sorted[:-1])
                        )
                    ).all()
                )
                assert ok, "lex_sort_words fast path produced non-lex order"
            return perm0
    for w in reversed(words):
        keys = w.index_select(0, perm)
        p = torch.argsort(keys, stable=True)
        perm = perm.index_select(0, p)
    return perm


def rows_ne(words: Sequence[torch.Tensor], i_prev_mask: bool = True) -> torch.Tensor:
    """Mask marking rows that differ from their predecessor (run starts)."""
    n = words[0].shape[0]
    if words[0].is_cuda and n > 0 and len(words) <= 4:
        from pathway_amd import ops

        return ops.run_starts_gpu([w.contiguous() for w in words])
    starts = torch.ones(n, dtype=torch.bool, device=words[0].device)
    if n > 1:
        ne = torch.zeros(n - 1, dtype=torch.bool, device=words[0].device)
        for w in words:
            ne |= w[1:] != w[:-1]
        starts[1:] = ne
    return starts


def consolidate_sorted(
    words: list[torch.Tensor],
    weights: torch.Tensor,
    columns: dict[str, Column],
) -> tuple[list[torch.Tensor], torch.Tensor, dict[str, Column]]:
    """Sum weights over equal-row runs, drop zero-weight rows.

    Inputs must already be sorted by `words`.  Column values within a run are
    identical by construction (vhash is part of the sort key), so the first
    row of each run is kept.
    """
    n = weights.shape[0]
    if n == 0:
        return words, weights, columns
    import os as _os

    if (
        words[0].is_cuda
        and len(words) <= 8
        and not _os.environ.get("PW_NO_SEGRED")
    ):
        # fused path: run-starts + compaction + weight segment-sum in two
        # kernels (pw_seg_reduce)
        from pathway_amd import ops

        out_words, first_idx, (wsum,) = ops.seg_reduce_words_gpu(
            words, [weights]
        )
        keep = (wsum != 0).nonzero(as_tuple=True)[0]
        kept_first, kept_w, *kept_words = ops.gather_all(
            keep, [first_idx, wsum] + out_words
        )
        out_cols = {name: c.take(kept_first) for name, c in columns.items()}
        return kept_words, kept_w, out_cols
    starts = rows_ne(words)
    seg = torch.cumsum(starts.to(torch.int64), 0) - 1
    nseg = int(seg[-1]) + 1
    wsum = torch.zeros(nseg, dtype=torch.int64, device=weights.device)
    wsum.index_add_(0, seg, weights)
    first_idx = starts.nonzero(as_tuple=True)[0]
    keep = wsum != 0
    kept_first = first_idx.index_select(0, keep.nonzero(as_tuple=True)[0])
    kept_w = wsum.index_select(0, keep.nonzero(as_tuple=True)[0])
    out_words = [w.index_select(0, kept_first) for w in words]
    out_cols = {name: c.take(kept_first) for name, c in columns.items()}
    return out_words, kept_w, out_cols


def searchsorted_words(
    sorted_words: Sequence[torch.Tensor],
    query_words: Sequence[torch.Tensor],
    side: str = "left",
) -> torch.Tensor:
    """Vectorized multiword searchsorted (lexicographic) via binary search.

    GPU path: one HIP kernel (pw_searchsorted); the torch loop below is the
    CPU path and the kernel's numerics reference.
    """
    m = sorted_words[0].shape[0]
    nq = query_words[0].shape[0]
    device = query_words[0].device
    if query_words[0].is_cuda and len(sorted_words) <= 4 and m > 0 and nq > 0:
        from pathway_amd import ops

        return ops.searchsorted_gpu(sorted_words, query_words, side)
    lo = torch.zeros(nq, dtype=torch.int64, device=device)
    hi = torch.full((nq,), m, dtype=torch.int64, device=device)
    if m == 0 or nq == 0:
        return lo
    it = max(1, m.bit_length() + 1)
    for _ in range(it):
        active = lo < hi
        if not bool(active.any()):
            break
        mid = (lo + hi) >> 1
        mid_safe = mid.clamp(0, m - 1)
        # lexicographic compare sorted[mid] ? query
        lt = torch.zeros(nq, dtype=torch.bool, device=device)
        eq = torch.ones(nq, dtype=torch.bool, device=device)
        for sw, qw in zip(sorted_words, query_words):
            sv = sw.index_select(0, mid_safe)
            lt = lt | (eq & (sv < qw))
            eq = eq & (sv == qw)
        if side == "left":
            go_right = lt
        else:
            go_right = lt | eq
        lo = torch.where(active & go_right, mid + 1, lo)
        hi = torch.where(active & ~go_right, mid, hi)
    return lo


def merge_sorted_select(
    state_words: Sequence[torch.Tensor], delta_words: Sequence[torch.Tensor]
) -> torch.Tensor:
    """Source-selection permutation merging two SORTED row sets in O(m+n).

    Returns src_sel of length m+n indexing into concat([state, delta]):
    taking concat-columns with src_sel yields the merged sorted order
    (state rows precede equal delta rows).  Replaces the concat+re-sort of
    the whole arrangement — the LSM-merge analog, two binary-search kernels
    plus one gather instead of an O((m+n) log) sort.
    """
    m = state_words[0].shape[0]
    n = delta_words[0].shape[0]
    device = state_words[0].device
    if m == 0:
        return torch.arange(n, dtype=torch.int64, device=device)
    if n == 0:
        return torch.arange(m, dtype=torch.int64, device=device)
    # delta row i lands after all state rows ≤ it
    pos_d = searchsorted_words(state_words, delta_words, side="right")
    final_d = pos_d + torch.arange(n, dtype=torch.int64, device=device)
    # state row j lands after delta rows strictly before it
    pos_s = searchsorted_words(delta_words, state_words, side="left")
    final_s = pos_s + torch.arange(m, dtype=torch.int64, device=device)
    src_sel = torch.empty(m + n, dtype=torch.int64, device=device)
    src_sel[final_s] = torch.arange(m, dtype=torch.int64, device=device)
    src_sel[final_d] = m + torch.arange(n, dtype=torch.int64, device=device)
    return src_sel


class Arrangement:
    """Consolidated sorted multiset of weighted rows, keyed by a 128-bit key.

    key_words: [k0, k1] (m,) int64 each; vhash_words: [v0, v1]; weights (m,)
    int64 nonzero; columns: the row payload.
    """

    def __init__(self, device, column_protos: dict[str, Column]):
        self.device = device
        z = torch.zeros((0,), dtype=torch.int64, device=device)
        self.key_words = [z, z.clone()]
        self.vhash_words = [z.clone(), z.clone()]
        self.weights = z.clone()
        idx0 = torch.zeros((0,), dtype=torch.int64)
        self.columns: dict[str, Column] = {
            n: c.take(idx0.to(c._device())) for n, c in column_protos.items()
        }

    def __len__(self) -> int:
        return int(self.weights.shape[0])

    @property
    def words(self) -> list[torch.Tensor]:
        return self.key_words + self.vhash_words

    def key_range(self, query_keys: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        """[lo, hi) row ranges for (nq,2) query keys (by key prefix)."""
        q = [query_keys[:, 0].contiguous(), query_keys[:, 1].contiguous()]
        if query_keys.is_cuda and len(self) > 0 and query_keys.shape[0] > 0:
            from pathway_amd import ops

            return ops.key_range_gpu(self.key_words, q)
        lo = searchsorted_words(self.key_words, q, side="left")
        hi = searchsorted_words(self.key_words, q, side="right")
        return lo, hi

    def gather_ranges(
        self, lo: torch.Tensor, hi: torch.Tensor
    ) -> tuple[torch.Tensor, torch.Tensor]:
        """Expand [lo,hi) ranges → (row_indices, query_index per row)."""
        counts = hi - lo
        total = int(counts.sum())
        qidx = torch.repeat_interleave(
            torch.arange(counts.shape[0], dtype=torch.int64, device=counts.device),
            counts,
        )
        if total == 0:
            return torch.zeros((0,), dtype=torch.int64, device=counts.device), qidx
        offs = segmented_arange(counts)
        rows = torch.repeat_interleave(lo, counts) + offs
        return rows, qidx

    def merge(
        self,
        keys: torch.Tensor,
        vhash: tuple[torch.Tensor, torch.Tensor],
        weights: torch.Tensor,
        columns: dict[str, Column],
        key_determined_vhash: bool = False,
    ) -> None:
        """Merge a delta (unsorted ok) into the arrangement, consolidating.

        key_determined_vhash: caller guarantees vhash is a function of the
        key (e.g. group stores: values define the key) — sorting by the key
        words alone is then lexicographic for the full 4-word rows."""
        if keys.shape[0] == 0:
            return
        dwords = [keys[:, 0].contiguous(), keys[:, 1].contiguous(), vhash[0], vhash[1]]
        # 1. sort the (small) delta
        sort_words = dwords[:2] if key_determined_vhash else dwords
        dperm = lex_sort_words(sort_words)
        from pathway_amd import ops

        *dwords, dweights = ops.gather_all(dperm, dwords + [weights])
        dcols = {n: columns[n].take(dperm) for n in self.columns}
        # the delta may carry duplicate rows: consolidate it first so the
        # fused merge's unique-rows precondition holds
        dwords, dweights, dcols = consolidate_sorted(dwords, dweights, dcols)
        if dwords[0].shape[0] == 0:
            return
        if dwords[0].is_cuda:
            # 2a. fused merge-path merge+consolidate (pw HIP kernel):
            # one count+emit pair over 4-word rows; zero-weight rows drop
            # in-flight; carried columns gather once via rep
            from pathway_amd import ops

            out_words, out_accs, rep = ops.merge_consolidate_gpu(
                self.words,
                [self.weights],
                dwords,
                [dweights],
                compare_words=2 if key_determined_vhash else 4,
            )
            self.key_words = out_words[:2]
            self.vhash_words = out_words[2:]
            self.weights = out_accs[0]
            self.columns = {
                n: concat_columns([self.columns[n], dcols[n]]).take(rep)
                for n in self.columns
            }
            return
        # 2b. host path: O(m+n) sorted merge + consolidation scan
        mwords = self.words[:2] if key_determined_vhash else self.words
        qwords = dwords[:2] if key_determined_vhash else dwords
        src_sel = merge_sorted_select(mwords, qwords)
        all_words = [
            torch.cat([s, d]).index_select(0, src_sel)
            for s, d in zip(self.words, dwords)
        ]
        all_w = torch.cat([self.weights, dweights]).index_select(0, src_sel)
        all_cols = {
            n: concat_columns([self.columns[n], dcols[n]]).take(src_sel)
            for n in self.columns
        }
        out_words, out_w, out_cols = consolidate_sorted(all_words, all_w, all_cols)
        self.key_words = out_words[:2]
        self.vhash_words = out_words[2:]
        self.weights = out_w
        self.columns = out_cols

    def keys_tensor(self) -> torch.Tensor:
        return torch.stack(self.key_words, dim=1) if len(self) else torch.zeros(
            (0, 2), dtype=torch.int64, device=self.device
        )


class SpineArrangement:
    """Leveled arrangement — the GPU spine (reference spine_fueled.rs).

    State lives as ≤log₂(n) sorted runs with geometric compaction: a delta
    becomes a new run; adjacent runs merge (merge_sorted_select, O(sum))
    when within 2× of each other.  Probes binary-search every run and sum —
    amortizes maintenance to O(n log n) total instead of a full O(state)
    rewrite per micro-batch (which the flat Arrangement pays; that one is
    kept for small bounded states like reduce groups).
    Rows may appear in several runs with partial weights; probe results sum
    weights per (key, vhash) implicitly through downstream consolidation.
    """

    def __init__(self, device, column_protos: dict[str, Column]):
        self.device = device
        self.protos = column_protos
        self.levels: list[Arrangement] = []

    def __len__(self) -> int:
        return sum(len(l) for l in self.levels)

    def merge(self, keys, vhash, weights, columns, key_determined_vhash=False):
        if keys.shape[0] == 0:
            return
        run = Arrangement(self.device, columns)
        run.merge(keys, vhash, weights, columns, key_determined_vhash)
        self.levels.append(run)
        # geometric compaction
        while len(self.levels) >= 2 and (
            len(self.levels[-1]) * 2 >= len(self.levels[-2])
        ):
            b = self.levels.pop()
            a = self.levels.pop()
            if len(b) == 0:
                merged = a
            elif len(a) == 0:
                merged = b
            else:
                a.merge(
                    b.keys_tensor(),
                    (b.vhash_words[0], b.vhash_words[1]),
                    b.weights,
                    b.columns,
                )
                merged = a
            if len(merged):
                self.levels.append(merged)
        # keep levels sorted by size descending (compaction can disorder)
        self.levels.sort(key=len, reverse=True)

    def probe_rows(self, query_keys: torch.Tensor):
        """All matching rows for (nq,2) query keys across levels:
        returns (columns dict gathered, qidx, weights)."""
        device = self.device
        parts_cols: list[dict[str, Column]] = []
        parts_qidx: list[torch.Tensor] = []
        parts_w: list[torch.Tensor] = []
        for lvl in self.levels:
            if len(lvl) == 0:
                continue
            lo, hi = lvl.key_range(query_keys)
            rows, qidx = lvl.gather_ranges(lo, hi)
            if rows.shape[0] == 0:
                continue
            parts_cols.append({n: c.take(rows) for n, c in lvl.columns.items()})
            parts_qidx.append(qidx)
            parts_w.append(lvl.weights.index_select(0, rows))
        if not parts_qidx:
            z = torch.zeros((0,), dtype=torch.int64, device=device)
            idx0 = torch.zeros((0,), dtype=torch.int64)
            cols = {n: c.take(idx0.to(c._device())) for n, c in self.protos.items()}
            return cols, z, z.clone()
        if len(parts_qidx) == 1:
            return parts_cols[0], parts_qidx[0], parts_w[0]
        names = list(parts_cols[0].keys())
        cols = {n: concat_columns([p[n] for p in parts_cols]) for n in names}
        return cols, torch.cat(parts_qidx), torch.cat(parts_w)

    def count_for(self, query_keys: torch.Tensor) -> torch.Tensor:
        nq = query_keys.shape[0]
        out = torch.zeros(nq, dtype=torch.int64, device=self.device)
        for lvl in self.levels:
            if len(lvl) == 0:
                continue
            lo, hi = lvl.key_range(query_keys)
            rows, qidx = lvl.gather_ranges(lo, hi)
            if rows.shape[0]:
                out.index_add_(0, qidx, lvl.weights.index_select(0, rows))
        return out


class AdditiveState:
    """key -> additive accumulator columns (semigroup reduce fast path).

    Used for count/sum-style reducers (reference reduce.rs:163-560 semigroup
    states): merging a delta ADDS accumulators; rows with all-zero presence
    weight are dropped.  Sorted by key; single consolidated array.
    """

    def __init__(self, device, acc_names: list[str]):
        z = torch.zeros((0,), dtype=torch.int64, device=device)
        self.device = device
        self.key_words = [z, z.clone()]
        self.weights = z.clone()  # presence weight (count of underlying rows)
        self.accs: dict[str, torch.Tensor] = {}
        self.acc_names = acc_names

    def __len__(self) -> int:
        return int(self.weights.shape[0])

    def lookup(self, keys: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        """(positions, found_mask) of (nq,2) keys in the state."""
        q = [keys[:, 0].contiguous(), keys[:, 1].contiguous()]
        if keys.is_cuda and len(self) > 0 and keys.shape[0] > 0:
            from pathway_amd import ops

            return ops.lookup_gpu(self.key_words, q)
        pos = searchsorted_words(self.key_words, q, side="left")
        m = len(self)
        safe = pos.clamp(0, max(m - 1, 0))
        if m == 0:
            found = torch.zeros(pos.shape[0], dtype=torch.bool, device=pos.device)
        else:
            found = (
                (pos < m)
                & (self.key_words[0].index_select(0, safe) == q[0])
                & (self.key_words[1].index_select(0, safe) == q[1])
            )
        return pos, found
