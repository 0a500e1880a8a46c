"""Tensor-native session-window assignment (reference windows.rs session
merge semantics).

Rows live in an arrangement sorted by (instance-hash, time, row-vhash).
A delta can only reshape the sessions of the instances it touches, so the
node re-derives session assignments for EXACTLY those instances — on
device, with segmented scans — against the old and the new state, and
emits the (-old, +new) assignment difference; other instances cost
nothing.  (Sessions merge and split non-locally within an instance, which
is why the invalidation unit is the instance, not a row neighborhood.)
"""

from __future__ import annotations

from typing import Any

import torch

from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.column import Column, PointerColumn, TensorColumn
from pathway_amd.engine.expression_eval import EvalContext, evaluate
from pathway_amd.engine.nodes import Node, batch_vhash, consolidate_batch
from pathway_amd.engine.nodes_asof import _AsofSide
from pathway_amd.engine.state import lex_sort_words, rows_ne, searchsorted_words
from pathway_amd.internals import dtype as dt


class SessionAssignNode(Node):
    """Output = input rows + _pw_window_start/_pw_window_end columns."""

    def __init__(
        self,
        input_node: Node,
        time_name: str,
        instance_name: str | None,
        max_gap: int,
        device,
    ):
        super().__init__([input_node], device)
        self.time_name = time_name
        self.instance_name = instance_name
        self.max_gap = max_gap
        self.S = _AsofSide(device)

    def reset(self):
        self.S = _AsofSide(self.device)
        self._xmeta = {}

    def wants_frontier(self) -> bool:
        from pathway_amd.parallel import get_comm

        c = get_comm()
        return c is not None and c.world > 1

    def _prep(self, b: DeltaBatch):
        ctx = EvalContext(b.columns, b.keys, self.device)
        if self.instance_name is not None:
            c = b.columns[self.instance_name]
            i0, i1 = c.value_hash()
            i0, i1 = i0.to(self.device), i1.to(self.device)
        else:
            i0 = torch.zeros(len(b), dtype=torch.int64, device=self.device)
            i1 = i0.clone()
        t = b.columns[self.time_name].tensor
        if t.dtype != torch.int64:
            raise TypeError("tensor session path needs int64 times")
        v0, v1 = batch_vhash(b)
        cols = dict(b.columns)
        cols["__rowkey__"] = PointerColumn(b.keys)
        return [i0, i1, t, v0, v1], b.diffs, cols

    def _instance_rows(self, S: _AsofSide, u0: torch.Tensor, u1: torch.Tensor):
        """Indices of all state rows belonging to the given instances."""
        device = self.device
        m = len(S)
        if m == 0 or u0.numel() == 0:
            return torch.zeros(0, dtype=torch.int64, device=device)
        lo = searchsorted_words(S.words[:2], [u0, u1], side="left")
        hi = searchsorted_words(S.words[:2], [u0, u1], side="right")
        lens = (hi - lo).clamp(min=0)
        total = int(lens.sum())
        if total == 0:
            return torch.zeros(0, dtype=torch.int64, device=device)
        starts = torch.repeat_interleave(lo, lens)
        cum = torch.cumsum(lens, 0) - lens
        offs = torch.arange(total, dtype=torch.int64, device=device) - torch.repeat_interleave(cum, lens)
        return starts + offs

    def _assign(self, S: _AsofSide, idx: torch.Tensor, time: int, sign: int):
        """Session-assignment DeltaBatch for the state rows at idx (which
        are contiguous per instance and time-sorted)."""
        device = self.device
        n = int(idx.numel())
        if n == 0:
            return None
        w0 = S.words[0].index_select(0, idx)
        w1 = S.words[1].index_select(0, idx)
        t = S.words[2].index_select(0, idx)
        # session boundaries: instance change OR gap > max_gap
        new_inst = torch.ones(n, dtype=torch.bool, device=device)
        if n > 1:
            new_inst[1:] = (w0[1:] != w0[:-1]) | (w1[1:] != w1[:-1])
        gap = torch.ones(n, dtype=torch.bool, device=device)
        if n > 1:
            gap[1:] = (t[1:] - t[:-1]) > self.max_gap
        bound = new_inst | gap
        sid = torch.cumsum(bound.to(torch.int64), 0) - 1
        first_idx = bound.nonzero(as_tuple=True)[0]
        nseg = int(first_idx.numel())
        last_idx = torch.cat(
            [first_idx[1:], torch.tensor([n], dtype=torch.int64, device=device)]
        ) - 1
        seg_start = t.index_select(0, first_idx)
        seg_end = t.index_select(0, last_idx)
        start_col = seg_start.index_select(0, sid)
        end_col = seg_end.index_select(0, sid)
        cols: dict[str, Column] = {
            name: c.take(idx)
            for name, c in S.cols.items()
            if name != "__rowkey__"
        }
        cols["_pw_window_start"] = TensorColumn(start_col, dt.INT)
        cols["_pw_window_end"] = TensorColumn(end_col, dt.INT)
        keys = S.cols["__rowkey__"].pairs.index_select(0, idx)
        diffs = S.weights.index_select(0, idx) * sign
        return DeltaBatch(keys, cols, diffs, time)

    def step(self, time, inputs):
        b = consolidate_batch(inputs[0])
        from pathway_amd.parallel import get_comm

        comm = get_comm()
        if comm is not None and comm.world > 1:
            # co-locate rows of the same instance: exchange by instance
            # hash (the reference centralizes time-column ops on one
            # worker; sharding by instance keeps the same correctness
            # with balanced load)
            from pathway_amd.engine.nodes_join import _exchange_side

            jk = None
            if b is not None and len(b):
                w5, _, _ = self._prep(b)
                jk = torch.stack(w5[:2], dim=1)
            if not hasattr(self, "_xmeta"):
                self._xmeta = {}
            b, _ = _exchange_side(comm, b, jk, time, self._xmeta)
        if b is None or len(b) == 0:
            return None
        device = self.device
        dwords, dweights, dcols = self._prep(b)

        # affected instances = distinct instance hashes of the delta
        perm = lex_sort_words(dwords[:2])
        si0 = dwords[0].index_select(0, perm)
        si1 = dwords[1].index_select(0, perm)
        starts = rows_ne([si0, si1])
        fidx = starts.nonzero(as_tuple=True)[0]
        u0 = si0.index_select(0, fidx)
        u1 = si1.index_select(0, fidx)

        S_old = self.S
        out = []
        old_idx = self._instance_rows(S_old, u0, u1)
        ob = self._assign(S_old, old_idx, time, -1) if old_idx.numel() else None
        if ob is not None:
            out.append(ob)

        snap = _AsofSide(device)
        snap.words = list(S_old.words)
        snap.weights = S_old.weights
        snap.cols = dict(S_old.cols) if S_old.cols is not None else None
        self.S.merge(dwords, dweights, dcols)

        new_idx = self._instance_rows(self.S, u0, u1)
        nb = self._assign(self.S, new_idx, time, 1) if new_idx.numel() else None
        if nb is not None:
            out.append(nb)
        if not out:
            return None
        res = DeltaBatch.concat(out)
        res.consolidated = False
        return consolidate_batch(res)
