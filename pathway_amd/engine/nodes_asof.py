"""Tensor-native incremental asof join (reference stdlib/temporal/
asof_join.py semantics; engine analog of a differential join over a
time-ordered arrangement).

Both sides are kept as GPU-resident arrangements sorted by
(on-key-hash, time, row-vhash).  Per delta batch:

  * ΔL rows answer directly against the NEW right state (one multiword
    binary search — pw_searchsorted on gfx950).
  * ΔR rows invalidate exactly the left rows whose answer can change:
    for a right change at (k, t) that is the half-open time range
    [t, succ_old(k, t))  (backward)  /  (pred_old(k, t), t]  (forward),
    found with two more binary searches; affected rows re-answer against
    old and new state and emit (-old, +new) with unchanged pairs
    cancelled — the differential update is exact, never a recompute.

Weight algebra: for a left row with weight w and answer payload p the
output contribution is w·p; the node emits w_old·(new-old) + δw·new which
telescopes to the exact output delta (see step()).

The node handles Direction.BACKWARD / FORWARD, inner / left modes, int64
time columns (ints, datetimes, durations).  Other cases (NEAREST, float
times, right/outer) stay on the host RecomputeNode path
(stdlib/temporal/_asof_join.py).
"""

from __future__ import annotations

from typing import Any

import torch

from pathway_amd.engine import hashing
from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.column import Column, PointerColumn, concat_columns
from pathway_amd.engine.expression_eval import EvalContext, evaluate
from pathway_amd.engine.nodes import (
    Node,
    batch_vhash,
    consolidate_batch,
    _mask_missing,
)
from pathway_amd.engine.state import (
    consolidate_sorted,
    lex_sort_words,
    merge_sorted_select,
    searchsorted_words,
)
from pathway_amd.internals import dtype as dt

_TMAX = (1 << 62)  # +inf sentinel for time upper bounds


class _AsofSide:
    """Sorted arrangement: words = [k0, k1, t, v0, v1]; weights; columns
    (payload + '__rowkey__')."""

    def __init__(self, device):
        self.device = device
        z = torch.zeros(0, dtype=torch.int64, device=device)
        self.words = [z, z.clone(), z.clone(), z.clone(), z.clone()]
        self.weights = z.clone()
        self.cols: dict[str, Column] | None = None

    def __len__(self) -> int:
        return int(self.words[0].shape[0])

    def merge(self, dwords, dweights, dcols) -> None:
        if self.cols is None:
            idx0 = torch.zeros(0, dtype=torch.int64)
            self.cols = {n: c.take(idx0.to(c._device())) for n, c in dcols.items()}
        perm = lex_sort_words(dwords)
        dwords = [w.index_select(0, perm) for w in dwords]
        dweights = dweights.index_select(0, perm)
        dcols = {n: c.take(perm) for n, c in dcols.items()}
        sel = merge_sorted_select(self.words, dwords)
        all_words = [
            torch.cat([s, d]).index_select(0, sel)
            for s, d in zip(self.words, dwords)
        ]
        all_w = torch.cat([self.weights, dweights]).index_select(0, sel)
        all_cols = {
            n: concat_columns([self.cols[n], dcols[n]]).take(sel) for n in self.cols
        }
        self.words, self.weights, self.cols = consolidate_sorted(
            all_words, all_w, all_cols
        )


class AsofJoinNode(Node):
    """See module docstring.  lmap/rmap: output-name -> source column."""

    def __init__(
        self,
        left_node: Node,
        right_node: Node,
        left_on: list[Any],
        right_on: list[Any],
        left_time: Any,
        right_time: Any,
        lmap: dict[str, str],
        rmap: dict[str, str],
        mode: str,  # 'inner' | 'left'
        direction: str,  # 'backward' | 'forward'
        device,
        defaults: dict[str, Any] | None = None,
    ):
        super().__init__([left_node, right_node], device)
        self.left_on = left_on
        self.right_on = right_on
        self.left_time = left_time
        self.right_time = right_time
        self.lmap = lmap
        self.rmap = rmap
        self.mode = mode
        self.direction = direction
        self.defaults = defaults or {}
        self.L = _AsofSide(device)
        self.R = _AsofSide(device)
        self._xmeta_l: dict = {}
        self._xmeta_r: dict = {}

    def reset(self):
        self.L = _AsofSide(self.device)
        self.R = _AsofSide(self.device)
        self._xmeta_l = {}
        self._xmeta_r = {}

    def wants_frontier(self) -> bool:
        from pathway_amd.parallel import get_comm

        c = get_comm()
        return c is not None and c.world > 1

    # -- helpers ----------------------------------------------------------

    def _prep(self, b: DeltaBatch, on_exprs, time_expr, side: str):
        """(words5, weights, cols incl payload + __rowkey__) for a batch."""
        ctx = EvalContext(b.columns, b.keys, self.device)
        parts = []
        for e in on_exprs:
            c = evaluate(e, ctx)
            lo, hi = c.value_hash()
            parts.append((lo.to(self.device), hi.to(self.device)))
        if parts:
            k0, k1 = hashing.combine_value_hashes(parts)
        else:
            # no on-conditions: a single global asof group
            k0 = torch.zeros(len(b), dtype=torch.int64, device=self.device)
            k1 = k0.clone()
        tcol = evaluate(time_expr, ctx)
        t = tcol.tensor
        if t.dtype != torch.int64:
            raise TypeError("tensor asof path needs int64 time values")
        v0, v1 = batch_vhash(b)
        cols = dict(b.columns)
        cols["__rowkey__"] = PointerColumn(b.keys)
        return [k0, k1, t, v0, v1], b.diffs, cols

    def _ident(self, S: _AsofSide, idx, valid, proto_t):
        """(valid, matched time, matched vhash words) — the answer's
        identity triple; safe on an empty state."""
        z = torch.zeros_like(proto_t)
        if len(S) == 0:
            f = torch.zeros(
                proto_t.shape[0], dtype=torch.bool, device=self.device
            )
            return (f, z, z.clone(), z.clone())
        return (
            valid,
            torch.where(valid, S.words[2].index_select(0, idx), z),
            torch.where(valid, S.words[3].index_select(0, idx), z),
            torch.where(valid, S.words[4].index_select(0, idx), z),
        )

    def _probe(self, S: _AsofSide, qk0, qk1, qt):
        """(idx, valid) of the matched state row per query."""
        m = len(S)
        nq = qk0.shape[0]
        device = self.device
        if m == 0 or nq == 0:
            z = torch.zeros(nq, dtype=torch.int64, device=device)
            return z, torch.zeros(nq, dtype=torch.bool, device=device)
        sw = S.words[:3]
        if self.direction == "backward":
            pos = searchsorted_words(sw, [qk0, qk1, qt], side="right")
            idx = (pos - 1).clamp(min=0)
            valid = pos > 0
        else:
            pos = searchsorted_words(sw, [qk0, qk1, qt], side="left")
            idx = pos.clamp(max=m - 1)
            valid = pos < m
        valid = (
            valid
            & (S.words[0].index_select(0, idx) == qk0)
            & (S.words[1].index_select(0, idx) == qk1)
        )
        return idx, valid

    def _affected_ranges(self, R_old: _AsofSide, dk0, dk1, dt_):
        """Left-state index ranges whose answers a right change can touch."""
        device = self.device
        nd = dk0.shape[0]
        mL = len(self.L)
        if nd == 0 or mL == 0:
            z = torch.zeros(0, dtype=torch.int64, device=device)
            return z
        lw = self.L.words[:3]
        tmax = torch.full_like(dt_, _TMAX)
        mR = len(R_old)
        if self.direction == "backward":
            lo = searchsorted_words(lw, [dk0, dk1, dt_], side="left")
            if mR:
                posu = searchsorted_words(R_old.words[:3], [dk0, dk1, dt_], side="right")
                pu = posu.clamp(max=mR - 1)
                same = (
                    (posu < mR)
                    & (R_old.words[0].index_select(0, pu) == dk0)
                    & (R_old.words[1].index_select(0, pu) == dk1)
                )
                ub = torch.where(same, R_old.words[2].index_select(0, pu), tmax)
            else:
                ub = tmax
            hi = searchsorted_words(lw, [dk0, dk1, ub], side="left")
        else:
            if mR:
                posl = searchsorted_words(R_old.words[:3], [dk0, dk1, dt_], side="left")
                pl = (posl - 1).clamp(min=0)
                same = (
                    (posl > 0)
                    & (R_old.words[0].index_select(0, pl) == dk0)
                    & (R_old.words[1].index_select(0, pl) == dk1)
                )
                lb = torch.where(same, R_old.words[2].index_select(0, pl), -tmax)
            else:
                lb = -tmax
            lo = searchsorted_words(lw, [dk0, dk1, lb], side="right")
            hi = searchsorted_words(lw, [dk0, dk1, dt_], side="right")
        lens = (hi - lo).clamp(min=0)
        total = int(lens.sum())
        if total == 0:
            return torch.zeros(0, dtype=torch.int64, device=device)
        starts = torch.repeat_interleave(lo, lens)
        cum = torch.cumsum(lens, 0) - lens
        offs = torch.arange(total, dtype=torch.int64, device=device) - torch.repeat_interleave(cum, lens)
        return torch.unique(starts + offs)

    def _out_rows(self, rowkeys, lcols, qvalid, ridx, R: _AsofSide, weights, time):
        """Output DeltaBatch for matched/padded answers (None if empty)."""
        device = self.device
        if self.mode == "inner":
            keep = qvalid.nonzero(as_tuple=True)[0]
        else:
            keep = torch.arange(qvalid.shape[0], dtype=torch.int64, device=device)
        if keep.numel() == 0:
            return None
        rowkeys = rowkeys.index_select(0, keep)
        weights = weights.index_select(0, keep)
        qvalid = qvalid.index_select(0, keep)
        ridx = ridx.index_select(0, keep)
        out: dict[str, Column] = {}
        for out_name, src in self.lmap.items():
            out[out_name] = lcols[src].take(keep)
        for out_name, src in self.rmap.items():
            if R.cols is not None and src in R.cols and len(R) > 0:
                c = R.cols[src].take(ridx)
                out[out_name] = _mask_missing(c, qvalid, device)
                if out_name in self.defaults:
                    # fill pads with the provided default
                    vals = out[out_name].to_pylist()
                    dflt = self.defaults[out_name]
                    filled = [dflt if (v is None and not bool(qvalid[i])) else v
                              for i, v in enumerate(vals)]
                    from pathway_amd.engine.column import infer_and_build_column

                    out[out_name], _ = infer_and_build_column(filled, device)
            else:
                # right side never produced rows: all-None object column
                import numpy as np

                from pathway_amd.engine.column import ObjectColumn

                arr = np.empty(int(keep.numel()), dtype=object)
                out[out_name] = ObjectColumn(arr, dt.Optional(dt.ANY))
        return DeltaBatch(rowkeys, out, weights, time)

    # -- step -------------------------------------------------------------

    def step(self, time, inputs):
        bl = consolidate_batch(inputs[0])
        br = consolidate_batch(inputs[1])
        from pathway_amd.parallel import get_comm

        comm = get_comm()
        if comm is not None and comm.world > 1:
            from pathway_amd.engine.nodes_join import _exchange_side

            jl = jr = None
            if bl is not None:
                w5, _, _ = self._prep(bl, self.left_on, self.left_time, "l")
                jl = torch.stack(w5[:2], dim=1)
            if br is not None:
                w5, _, _ = self._prep(br, self.right_on, self.right_time, "r")
                jr = torch.stack(w5[:2], dim=1)
            bl, _ = _exchange_side(comm, bl, jl, time, self._xmeta_l)
            br, _ = _exchange_side(comm, br, jr, time, self._xmeta_r)
        if bl is None and br is None:
            return None
        device = self.device
        out_batches = []

        R_old = self.R

        dR = None
        if br is not None and len(br):
            dR = self._prep(br, self.right_on, self.right_time, "r")

        # 1. affected pre-existing left rows (computed against OLD states)
        aff = None
        if dR is not None and len(self.L):
            aff = self._affected_ranges(R_old, dR[0][0], dR[0][1], dR[0][2])
            if aff.numel() == 0:
                aff = None
        if aff is not None:
            a_k0 = self.L.words[0].index_select(0, aff)
            a_k1 = self.L.words[1].index_select(0, aff)
            a_t = self.L.words[2].index_select(0, aff)
            old_idx, old_valid = self._probe(R_old, a_k0, a_k1, a_t)
            old_ident = self._ident(R_old, old_idx, old_valid, a_t)
        # 2. merge ΔR into the right state (R becomes NEW)
        if dR is not None:
            # snapshot old arrays: merge builds new tensors, old refs stay
            snap = _AsofSide(device)
            snap.words = list(R_old.words)
            snap.weights = R_old.weights
            snap.cols = dict(R_old.cols) if R_old.cols is not None else None
            R_old = snap
            self.R.merge(*dR)

        # 3. re-answer affected rows against NEW right; emit changes
        if aff is not None:
            new_idx, new_valid = self._probe(self.R, a_k0, a_k1, a_t)
            new_ident = self._ident(self.R, new_idx, new_valid, a_t)
            unchanged = torch.ones_like(old_valid)
            for o, n in zip(old_ident, new_ident):
                unchanged = unchanged & (o == n)
            ch = (~unchanged).nonzero(as_tuple=True)[0]
            if ch.numel():
                affc = aff.index_select(0, ch)
                rowk = self.L.cols["__rowkey__"].pairs.index_select(0, affc)
                w_l = self.L.weights.index_select(0, affc)
                lc = {src: self.L.cols[src].take(affc) for src in self.lmap.values()}
                # old contribution retracted (against the OLD right state)
                ob = self._out_rows(
                    rowk,
                    lc,
                    old_ident[0].index_select(0, ch),
                    old_idx.index_select(0, ch),
                    R_old,
                    -w_l,
                    time,
                )
                if ob is not None:
                    out_batches.append(ob)
                nb = self._out_rows(
                    rowk,
                    lc,
                    new_ident[0].index_select(0, ch),
                    new_idx.index_select(0, ch),
                    self.R,
                    w_l,
                    time,
                )
                if nb is not None:
                    out_batches.append(nb)

        # 4. ΔL answers against the NEW right state; merge into left state
        if bl is not None and len(bl):
            dL = self._prep(bl, self.left_on, self.left_time, "l")
            lidx, lvalid = self._probe(self.R, dL[0][0], dL[0][1], dL[0][2])
            lb = self._out_rows(
                bl.keys,
                dict(bl.columns),
                lvalid,
                lidx,
                self.R,
                bl.diffs,
                time,
            )
            if lb is not None:
                out_batches.append(lb)
            self.L.merge(*dL)

        if not out_batches:
            return None
        return consolidate_batch(DeltaBatch.concat(out_batches))
