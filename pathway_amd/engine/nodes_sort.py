"""Tensor-native differential prev/next pointer maintenance
(reference operators/prev_next.rs:775 add_prev_next_pointers — there built
on bidirectional trace cursors; here on a sorted GPU arrangement).

State: rows sorted by (instance-hash, key, rowkey) — the rowkey tiebreak
reproduces the reference's deterministic order for equal keys (unsigned
(hi, lo), the same order `repr(Pointer)` sorts in).  A delta touches only
the inserted/deleted rows and their immediate neighbors in the old and
new orders: those identities re-derive (prev, next) against both states
and emit the (-old, +new) difference.
"""

from __future__ import annotations

from typing import Any

import numpy as np
import torch

from pathway_amd.engine import hashing
from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.column import ObjectColumn, PointerColumn
from pathway_amd.engine.expression_eval import EvalContext, evaluate
from pathway_amd.engine.nodes import Node, consolidate_batch
from pathway_amd.engine.nodes_asof import _AsofSide
from pathway_amd.engine.state import searchsorted_words
from pathway_amd.internals import dtype as dt
from pathway_amd.internals.api import BasePointer

_SIGN = -0x8000000000000000  # xor flips the sign bit: unsigned order as signed


class SortPrevNextNode(Node):
    """Output per input row: prev / next row pointers in sort order."""

    def __init__(self, input_node: Node, key_expr: Any, instance_expr: Any, device):
        super().__init__([input_node], device)
        self.key_expr = key_expr
        self.instance_expr = instance_expr
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
        if self.instance_expr is not None:
            c = evaluate(self.instance_expr, ctx)
            i0, i1 = c.value_hash()
            i0, i1 = i0.to(self.device), i1.to(self.device)
        else:
            i0 = torch.zeros(len(b), dtype=torch.int64, device=self.device)
            i1 = i0.clone()
        kcol = evaluate(self.key_expr, ctx)
        k = kcol.tensor
        if k.dtype != torch.int64:
            raise TypeError("tensor sort path needs int64 keys")
        # rowkey tiebreak in unsigned (hi, lo) order
        thi = b.keys[:, 1] ^ _SIGN
        tlo = b.keys[:, 0] ^ _SIGN
        cols = {"__rowkey__": PointerColumn(b.keys)}
        return [i0, i1, k, thi, tlo], b.diffs, cols

    def _derive(self, S: _AsofSide, qwords):
        """(present, pos, prev_pair, next_pair, prev_ok, next_ok) per query
        identity against state S."""
        device = self.device
        nq = qwords[0].shape[0]
        m = len(S)
        if m == 0:
            z = torch.zeros(nq, dtype=torch.int64, device=device)
            f = torch.zeros(nq, dtype=torch.bool, device=device)
            zp = torch.zeros((nq, 2), dtype=torch.int64, device=device)
            return f, z, zp, zp, f, f
        pos = searchsorted_words(S.words, qwords, side="left")
        pc = pos.clamp(max=m - 1)
        present = pos < m
        for sw, qw in zip(S.words, qwords):
            present = present & (sw.index_select(0, pc) == qw)
        prev_i = (pos - 1).clamp(min=0)
        next_i = (pos + 1).clamp(max=m - 1)
        same_inst_prev = (
            (pos > 0)
            & (S.words[0].index_select(0, prev_i) == qwords[0])
            & (S.words[1].index_select(0, prev_i) == qwords[1])
        )
        same_inst_next = (
            (pos + 1 < m)
            & (S.words[0].index_select(0, next_i) == qwords[0])
            & (S.words[1].index_select(0, next_i) == qwords[1])
        )
        pairs = S.cols["__rowkey__"].pairs
        prev_pair = pairs.index_select(0, prev_i)
        next_pair = pairs.index_select(0, next_i)
        return present, pos, prev_pair, next_pair, same_inst_prev, same_inst_next

    def _gather_identities(self, S: _AsofSide, idx: torch.Tensor):
        idx = idx.clamp(min=0, max=max(len(S) - 1, 0))
        return [w.index_select(0, idx) for w in S.words], S.cols[
            "__rowkey__"
        ].pairs.index_select(0, idx)

    def step(self, time, inputs):
        b = consolidate_batch(inputs[0])
        from pathway_amd.parallel import get_comm

        comm = get_comm()
        if comm is not None and comm.world > 1:
            # co-locate per instance (global sort order needs all rows of
            # an instance on one rank — the prev_next.rs analog of the
            # reference's worker-local trace)
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

        S_old = self.S
        # neighborhood in the OLD state: insertion point and its flanks
        pos_old = (
            searchsorted_words(S_old.words, dwords, side="left")
            if len(S_old)
            else torch.zeros(len(b), dtype=torch.int64, device=device)
        )
        cand_old = torch.cat([pos_old - 1, pos_old, pos_old + 1])

        # snapshot old arrays, merge, then neighborhood in the NEW state
        snap = _AsofSide(device)
        snap.words = list(S_old.words)
        snap.weights = S_old.weights
        snap.cols = dict(S_old.cols) if S_old.cols is not None else None
        S_old = snap
        self.S.merge(dwords, dweights, dcols)
        S_new = self.S
        pos_new = searchsorted_words(S_new.words, dwords, side="left")
        cand_new = torch.cat([pos_new - 1, pos_new, pos_new + 1])

        # affected identities = rows at those positions in either state
        ids_w: list[torch.Tensor] = []
        ids_p: list[torch.Tensor] = []
        if len(S_old):
            w, p = self._gather_identities(S_old, cand_old)
            ids_w.append(torch.stack(w, dim=1))
            ids_p.append(p)
        if len(S_new):
            w, p = self._gather_identities(S_new, cand_new)
            ids_w.append(torch.stack(w, dim=1))
            ids_p.append(p)
        if not ids_w:
            return None
        allw = torch.cat(ids_w)
        allp = torch.cat(ids_p)
        packed = torch.cat([allw, allp], dim=1)
        uniq, uidx = torch.unique(packed, dim=0, return_inverse=False), None
        qwords = [uniq[:, j].contiguous() for j in range(5)]
        qpairs = uniq[:, 5:7]

        o_pres, _, o_pp, o_np, o_pok, o_nok = self._derive(S_old, qwords)
        n_pres, _, n_pp, n_np, n_pok, n_nok = self._derive(S_new, qwords)

        unchanged = (
            (o_pres == n_pres)
            & (o_pok == n_pok)
            & (o_nok == n_nok)
            & ((o_pp == n_pp).all(dim=1) | ~(o_pok & n_pok))
            & ((o_np == n_np).all(dim=1) | ~(o_nok & n_nok))
        )
        changed = (~unchanged).nonzero(as_tuple=True)[0]
        if changed.numel() == 0:
            return None

        def emit(pres, pp, np_, pok, nok, sign):
            keep = pres.index_select(0, changed).nonzero(as_tuple=True)[0]
            if keep.numel() == 0:
                return None
            sel = changed.index_select(0, keep)
            keys = qpairs.index_select(0, sel)
            ppl = pp.index_select(0, sel).cpu().tolist()
            npl = np_.index_select(0, sel).cpu().tolist()
            pokl = pok.index_select(0, sel).cpu().tolist()
            nokl = nok.index_select(0, sel).cpu().tolist()
            prev_vals = np.empty(len(ppl), dtype=object)
            next_vals = np.empty(len(npl), dtype=object)
            for i in range(len(ppl)):
                prev_vals[i] = (
                    BasePointer.from_signed_pair(*ppl[i]) if pokl[i] else None
                )
                next_vals[i] = (
                    BasePointer.from_signed_pair(*npl[i]) if nokl[i] else None
                )
            cols = {
                "prev": ObjectColumn(prev_vals, dt.Optional(dt.POINTER)),
                "next": ObjectColumn(next_vals, dt.Optional(dt.POINTER)),
            }
            diffs = torch.full(
                (int(keep.numel()),), sign, dtype=torch.int64, device=device
            )
            return DeltaBatch(keys, cols, diffs, time)

        out = []
        ob = emit(o_pres, o_pp, o_np, o_pok, o_nok, -1)
        if ob is not None:
            out.append(ob)
        nb = emit(n_pres, n_pp, n_np, n_pok, n_nok, 1)
        if nb is not None:
            out.append(nb)
        if not out:
            return None
        res = DeltaBatch.concat(out)
        res.consolidated = False
        return consolidate_batch(res)
