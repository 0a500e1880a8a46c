"""DeltaBatch: a timestamped batch of keyed row updates.

The engine's unit of dataflow — the MI355X-native analog of a differential
batch (reference: Collection<S,(Key,Value)> updates at one timestamp,
dataflow.rs:281-440).  One DeltaBatch carries all updates for ONE timestamp:
  keys:  (n, 2) int64 device tensor — 128-bit row keys
  diffs: (n,)  int64 device tensor — multiplicities (+1 insert / -1 retract)
  columns: name -> Column (SoA device buffers)
"""

from __future__ import annotations

from typing import Any, Iterable

import torch

from pathway_amd.engine.column import Column, concat_columns


class DeltaBatch:
    __slots__ = ("keys", "columns", "diffs", "time", "consolidated")

    def __init__(
        self,
        keys: torch.Tensor,
        columns: dict[str, Column],
        diffs: torch.Tensor,
        time: int,
        consolidated: bool = False,
    ):
        assert keys.dim() == 2 and keys.shape[1] == 2, keys.shape
        self.keys = keys
        self.columns = columns
        self.diffs = diffs
        self.time = time
        #: producer guarantees rows are already consolidated (no equal-row
        #: cancellation possible) — sinks skip the consolidation sort
        self.consolidated = consolidated

    def __len__(self) -> int:
        return int(self.keys.shape[0])

    @property
    def device(self):
        return self.keys.device

    def is_empty(self) -> bool:
        return len(self) == 0

    def take(self, idx: torch.Tensor) -> "DeltaBatch":
        return DeltaBatch(
            self.keys.index_select(0, idx),
            {n: c.take(idx) for n, c in self.columns.items()},
            self.diffs.index_select(0, idx),
            self.time,
        )

    def filter(self, mask: torch.Tensor) -> "DeltaBatch":
        idx = mask.nonzero(as_tuple=True)[0]
        return self.take(idx)

    def with_columns(self, columns: dict[str, Column]) -> "DeltaBatch":
        return DeltaBatch(self.keys, columns, self.diffs, self.time)

    def with_keys(self, keys: torch.Tensor) -> "DeltaBatch":
        return DeltaBatch(keys, self.columns, self.diffs, self.time)

    def select_columns(self, names: Iterable[str]) -> "DeltaBatch":
        return DeltaBatch(
            self.keys, {n: self.columns[n] for n in names}, self.diffs, self.time
        )

    def to_device(self, device) -> "DeltaBatch":
        return DeltaBatch(
            self.keys.to(device),
            {n: c.to_device(device) for n, c in self.columns.items()},
            self.diffs.to(device),
            self.time,
        )

    @staticmethod
    def empty(column_protos: dict[str, Column], device="cpu", time: int = 0) -> "DeltaBatch":
        zero = torch.zeros((0,), dtype=torch.int64, device=device)
        return DeltaBatch(
            torch.zeros((0, 2), dtype=torch.int64, device=device),
            {n: c.take(zero.to(c._device())) for n, c in column_protos.items()},
            zero,
            time,
        )

    @staticmethod
    def concat(batches: list["DeltaBatch"]) -> "DeltaBatch":
        batches = [b for b in batches if b is not None]
        assert batches
        if len(batches) == 1:
            return batches[0]
        names = list(batches[0].columns.keys())
        return DeltaBatch(
            torch.cat([b.keys for b in batches]),
            {n: concat_columns([b.columns[n] for b in batches]) for n in names},
            torch.cat([b.diffs for b in batches]),
            batches[0].time,
        )

    def rows(self) -> list[tuple[Any, list[Any], int, int]]:
        """Host-side materialization: (key, values, time, diff) per row."""
        from pathway_amd.internals.api import BasePointer

        keys = self.keys.cpu().tolist()
        diffs = self.diffs.cpu().tolist()
        cols = {n: c.to_pylist() for n, c in self.columns.items()}
        names = list(self.columns.keys())
        out = []
        for i in range(len(self)):
            key = BasePointer.from_signed_pair(keys[i][0], keys[i][1])
            out.append((key, [cols[n][i] for n in names], self.time, diffs[i]))
        return out


def lex_sort_keys(keys: torch.Tensor) -> torch.Tensor:
    """Permutation sorting (n,2) keys lexicographically by (k0, k1).

    Two stable sorts: secondary word first, then primary — the standard
    LSD trick.  On GPU this is replaced by the HIP radix-sort kernel when
    available (ops/gpu_ops.py); this is the reference implementation.
    """
    n = keys.shape[0]
    if n <= 1:
        return torch.arange(n, dtype=torch.int64, device=keys.device)
    perm1 = torch.argsort(keys[:, 1], stable=True)
    k0 = keys[:, 0].index_select(0, perm1)
    perm0 = torch.argsort(k0, stable=True)
    return perm1.index_select(0, perm0)


def searchsorted_pairs(
    sorted_keys: torch.Tensor, query_keys: torch.Tensor, side: str = "left"
) -> torch.Tensor:
    """searchsorted over 128-bit keys sorted lexicographically.

    Primary bound from k0; ties on k0 are refined by a second searchsorted
    over k1 within the candidate run.  Implemented with two 1-D
    searchsorted calls (exact, fully vectorized).
    """
    sk0, sk1 = sorted_keys[:, 0].contiguous(), sorted_keys[:, 1].contiguous()
    qk0, qk1 = query_keys[:, 0].contiguous(), query_keys[:, 1].contiguous()
    lo = torch.searchsorted(sk0, qk0, side="left")
    hi = torch.searchsorted(sk0, qk0, side="right")
    # within [lo, hi) all k0 equal qk0; refine by k1.
    # Handle runs by per-query binary search on k1 via the "global" trick:
    # since (k0,k1) sorted lexicographically, within equal-k0 runs k1 is
    # sorted.  Use searchsorted on k1 restricted by offsetting the query into
    # the run: implement a small fixed-iteration binary search.
    span = hi - lo
    need = span > 0
    if not bool(need.any()):
        return lo
    result = lo.clone()
    # vectorized binary search over k1 in [lo, hi)
    lo_b = lo.clone()
    hi_b = hi.clone()
    max_iter = 64
    for _ in range(max_iter):
        active = lo_b < hi_b
        if not bool(active.any()):
            break
        mid = (lo_b + hi_b) >> 1
        mid_safe = mid.clamp(0, sk1.shape[0] - 1)
        mv = sk1.index_select(0, mid_safe)
        if side == "left":
            go_right = mv < qk1
        else:
            go_right = mv <= qk1
        lo_b = torch.where(active & go_right, mid + 1, lo_b)
        hi_b = torch.where(active & ~go_right, mid, hi_b)
    result = lo_b
    return result


def keys_equal_rows(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return (a[:, 0] == b[:, 0]) & (a[:, 1] == b[:, 1])


def segment_starts(sorted_keys: torch.Tensor) -> torch.Tensor:
    """Boolean mask marking the first row of each equal-key run."""
    n = sorted_keys.shape[0]
    if n == 0:
        return torch.zeros((0,), dtype=torch.bool, device=sorted_keys.device)
    prev_ne = torch.ones(n, dtype=torch.bool, device=sorted_keys.device)
    prev_ne[1:] = (sorted_keys[1:, 0] != sorted_keys[:-1, 0]) | (
        sorted_keys[1:, 1] != sorted_keys[:-1, 1]
    )
    return prev_ne


def segment_ids(sorted_keys: torch.Tensor) -> torch.Tensor:
    starts = segment_starts(sorted_keys)
    return torch.cumsum(starts.to(torch.int64), 0) - 1


def segmented_arange(counts: torch.Tensor) -> torch.Tensor:
    """[0..c0), [0..c1), ... concatenated — the join-expansion index helper."""
    total = int(counts.sum())
    if total == 0:
        return torch.zeros((0,), dtype=torch.int64, device=counts.device)
    ends = torch.cumsum(counts, 0)
    starts = ends - counts
    return torch.arange(total, dtype=torch.int64, device=counts.device) - torch.repeat_interleave(
        starts, counts
    )
