"""Join-family nodes: equi-join, semi-join (intersect/difference), keyed
merge (update_rows/update_cells), ix lookup, flatten.

Incremental equi-join (reference join.rs:218 join_core semantics):
  out_t = dL >< R_old  +  L_new >< dR   (dL><dR counted once)
Probing is a sorted-range lookup into the GPU arrangement + segmented
expansion (searchsorted + repeat_interleave), replaced by the HIP
merge-probe kernel on gfx950 for the hot path.
"""

from __future__ import annotations

from typing import Any, Sequence

import numpy as np
import torch

from pathway_amd.internals import dtype as dt
from pathway_amd.engine import hashing
from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.column import (
    Column,
    ObjectColumn,
    PointerColumn,
    StringColumn,
    TensorColumn,
)
from pathway_amd.engine.expression_eval import EvalContext, evaluate
from pathway_amd.engine.nodes import (
    Node,
    _salt,
    batch_vhash,
    consolidate_batch,
    null_column,
    unique_sorted_keys,
)
from pathway_amd.engine.state import Arrangement, SpineArrangement


def _join_keys(exprs: list[Any], b: DeltaBatch, device) -> torch.Tensor:
    ctx = EvalContext(b.columns, b.keys, device)
    cols = [evaluate(e, ctx) for e in exprs]
    parts = [
        (lo.to(device), hi.to(device)) for lo, hi in (c.value_hash() for c in cols)
    ]
    if not parts:
        z = torch.zeros(len(b), dtype=torch.int64, device=device)
        return torch.stack([z, z.clone()], dim=1)
    lo, hi = hashing.combine_value_hashes(parts)
    return torch.stack([lo, hi], dim=1)


def _exchange_side(comm, b: DeltaBatch | None, jk: torch.Tensor | None, time: int, meta_state: dict | None = None):
    """All-to-all one join side's delta rows by join-key shard.

    Every rank must participate in the collective even with an empty local
    batch (peers may still route rows here)."""
    from pathway_amd.parallel.exchange import exchange_bundle, shard_of

    if b is None:
        tensors, cols = exchange_bundle(comm, None, None, None, meta_state=meta_state)
    else:
        dest = shard_of(jk, comm.world)
        tensors, cols = exchange_bundle(
            comm,
            dest,
            {"jk": jk, "keys": b.keys, "diffs": b.diffs},
            dict(b.columns),
            meta_state=meta_state,
        )
    if tensors is None:
        return None, None
    nb = DeltaBatch(tensors["keys"], cols, tensors["diffs"], time)
    if len(nb) == 0:
        return None, None
    return nb, tensors["jk"]


class _SideStore:
    """Spine of one join side keyed by join key, payload = row
    (reference join.rs arrangements; GPU spine: state.SpineArrangement)."""

    def __init__(self, device):
        self.device = device
        self.spine: SpineArrangement | None = None

    @property
    def arr(self):  # back-compat for snapshots/tools
        return self.spine

    def ensure(self, cols: dict[str, Column]):
        if self.spine is None:
            self.spine = SpineArrangement(self.device, cols)

    def merge(self, jkeys: torch.Tensor, b: DeltaBatch):
        cols = dict(b.columns)
        cols["__id__"] = PointerColumn(b.keys)
        self.ensure(cols)
        parts = [
            (lo.to(self.device), hi.to(self.device))
            for lo, hi in (c.value_hash() for c in cols.values())
        ]
        v0, v1 = hashing.combine_value_hashes(parts)
        self.spine.merge(jkeys, (v0, v1), b.diffs, cols)

    def __len__(self) -> int:
        return len(self.spine) if self.spine is not None else 0

    def probe(self, jkeys: torch.Tensor):
        """(gathered columns incl __id__, qidx, weights) per matched row."""
        if self.spine is None or len(self.spine) == 0:
            z = torch.zeros((0,), dtype=torch.int64, device=self.device)
            return None, z, z.clone()
        return self.spine.probe_rows(jkeys)

    def count_for(self, keys2: torch.Tensor) -> torch.Tensor:
        nq = keys2.shape[0]
        if self.spine is None or len(self.spine) == 0:
            return torch.zeros(nq, dtype=torch.int64, device=self.device)
        return self.spine.count_for(keys2)

    rows_for = probe


class JoinNode(Node):
    """Equi-join.  mode: inner|left|right|outer.  key_mode: pair|left."""

    def __init__(
        self,
        left: Node,
        right: Node,
        left_on: list[Any],
        right_on: list[Any],
        left_out: dict[str, str],
        right_out: dict[str, str],
        mode: str,
        device,
        key_mode: str = "pair",
        left_id_name: str | None = None,
        right_id_name: str | None = None,
        probe_only_left: bool = False,
    ):
        super().__init__([left, right], device)
        self.left_on = left_on
        self.right_on = right_on
        self.left_out = left_out
        self.right_out = right_out
        self.mode = mode
        self.key_mode = key_mode
        self.left_id_name = left_id_name
        self.right_id_name = right_id_name
        #: as-of-now semantics: right deltas update state without emitting;
        #: only left (query) deltas produce output (answers frozen)
        self.probe_only_left = probe_only_left
        self.lstore = _SideStore(device)
        self.rstore = _SideStore(device)
        self.pair_salt = _salt("join_pair", self.node_id)
        self.lpad_salt = _salt("join_lpad", self.node_id)
        self.rpad_salt = _salt("join_rpad", self.node_id)
        self._left_proto: dict[str, Column] | None = None
        self._right_proto: dict[str, Column] | None = None

    def reset(self) -> None:
        self.lstore = _SideStore(self.device)
        self.rstore = _SideStore(self.device)
        self._left_proto = None
        self._right_proto = None
        self._xmeta_l = {}
        self._xmeta_r = {}

    def wants_frontier(self) -> bool:
        from pathway_amd.parallel import get_comm

        c = get_comm()
        return c is not None and c.world > 1

    def step(self, time, inputs):
        bl, br = inputs
        device = self.device
        out_parts: list[DeltaBatch] = []

        jl = _join_keys(self.left_on, bl, device) if bl is not None and len(bl) else None
        jr = _join_keys(self.right_on, br, device) if br is not None and len(br) else None

        # multi-worker: co-shard both sides by join key (RCCL all-to-all)
        from pathway_amd.parallel import get_comm

        comm = get_comm()
        if comm is not None and comm.world > 1:
            if not hasattr(self, "_xmeta_l"):
                self._xmeta_l = {}
                self._xmeta_r = {}
            bl, jl = _exchange_side(comm, bl, jl, time, self._xmeta_l)
            br, jr = _exchange_side(comm, br, jr, time, self._xmeta_r)

        if bl is not None and len(bl):
            self._left_proto = self._left_proto or dict(bl.columns)
        if br is not None and len(br):
            self._right_proto = self._right_proto or dict(br.columns)

        affected_list = []
        if jl is not None:
            affected_list.append(unique_sorted_keys(jl))
        if jr is not None:
            affected_list.append(unique_sorted_keys(jr))
        if not affected_list:
            return None
        affected = unique_sorted_keys(torch.cat(affected_list, dim=0))

        pad_left = self.mode in ("left", "outer")
        pad_right = self.mode in ("right", "outer")

        # old padded rows (pre-merge)
        old_lpad = self._padded_rows(affected, "left") if pad_left else None
        old_rpad = self._padded_rows(affected, "right") if pad_right else None

        if self.probe_only_left:
            # as-of-now: right state updates first (queries at time t see
            # the state AS OF t), then left deltas probe; no dR emissions.
            # Answers (incl. left-outer pads) are FROZEN at probe time —
            # later right changes must not retract them, so the generic
            # old/new pad mechanism below is skipped entirely.
            if jr is not None:
                self.rstore.merge(jr, br)
            if jl is not None:
                out_parts += self._emit_pairs(bl, jl, self.rstore, probe_is_right=False)
                if pad_left:
                    out_parts += self._asof_now_pads(bl, jl)
                self.lstore.merge(jl, bl)
            pad_left = pad_right = False
        else:
            # dR >< L_old
            if jr is not None:
                out_parts += self._emit_pairs(br, jr, self.lstore, probe_is_right=True)
            # merge left
            if jl is not None:
                self.lstore.merge(jl, bl)
            # dL >< R_old, then merge dR; the dL><dR term is emitted once
            # via _emit_delta_cross below
            if jl is not None:
                out_parts += self._emit_pairs(bl, jl, self.rstore, probe_is_right=False)
            if jr is not None:
                self.rstore.merge(jr, br)
            if jl is not None and jr is not None:
                out_parts += self._emit_delta_cross(bl, jl, br, jr)

        # new padded rows (post-merge)
        if pad_left:
            new_lpad = self._padded_rows(affected, "left")
            out_parts += _pad_delta(old_lpad, new_lpad, time, device)
        if pad_right:
            new_rpad = self._padded_rows(affected, "right")
            out_parts += _pad_delta(old_rpad, new_rpad, time, device)

        parts = [p for p in out_parts if p is not None and len(p)]
        if not parts:
            return None
        for p in parts:
            p.time = time
        return consolidate_batch(DeltaBatch.concat(parts))

    # -- pair emission --

    def _emit_pairs(self, b: DeltaBatch, jk: torch.Tensor, store: _SideStore, probe_is_right: bool):
        stored_cols, qidx, w = store.probe(jk)
        if qidx.shape[0] == 0:
            return []
        probe = b.take(qidx)
        diffs = probe.diffs * w
        stored_cols = dict(stored_cols)
        stored_ids = stored_cols.pop("__id__")
        if probe_is_right:
            lkeys, rkeys = stored_ids.pairs, probe.keys
            lcols, rcols = stored_cols, probe.columns
        else:
            lkeys, rkeys = probe.keys, stored_ids.pairs
            lcols, rcols = probe.columns, stored_cols
        return [self._make_pair_batch(lkeys, rkeys, lcols, rcols, diffs, b.time)]

    def _emit_delta_cross(self, bl, jl, br, jr):
        # match dL and dR directly: sort dR by key, range-probe with dL
        tmp = _SideStore(self.device)
        tmp.merge(jr, br)
        stored_cols, qidx, w = tmp.probe(jl)
        if qidx.shape[0] == 0:
            return []
        probe = bl.take(qidx)
        diffs = probe.diffs * w
        stored_cols = dict(stored_cols)
        stored_ids = stored_cols.pop("__id__")
        return [
            self._make_pair_batch(
                probe.keys, stored_ids.pairs, probe.columns, stored_cols, diffs, bl.time
            )
        ]

    def _make_pair_batch(self, lkeys, rkeys, lcols, rcols, diffs, time):
        device = self.device
        if self.key_mode == "left":
            okeys = lkeys
        elif self.key_mode == "right":
            okeys = rkeys
        else:
            lo, hi = hashing.derive_key_words(
                self.pair_salt,
                [
                    (lkeys[:, 0].contiguous(), lkeys[:, 1].contiguous()),
                    (rkeys[:, 0].contiguous(), rkeys[:, 1].contiguous()),
                ],
            )
            okeys = torch.stack([lo, hi], dim=1)
        out_cols: dict[str, Column] = {}
        for out_name, src in self.left_out.items():
            out_cols[out_name] = lcols[src]
        for out_name, src in self.right_out.items():
            out_cols[out_name] = rcols[src]
        if self.left_id_name:
            out_cols[self.left_id_name] = PointerColumn(lkeys)
        if self.right_id_name:
            out_cols[self.right_id_name] = PointerColumn(rkeys)
        return DeltaBatch(okeys, out_cols, diffs, time)

    def _asof_now_pads(self, bl: DeltaBatch, jl: torch.Tensor) -> list:
        """Left-outer pads for as-of-now probes: the rows of THIS left
        batch whose join key has no current right match, emitted with the
        batch's own diffs and never revisited (frozen answers)."""
        device = self.device
        ukeys = unique_sorted_keys(jl)
        ucnt = self.rstore.count_for(ukeys)
        from pathway_amd.engine.state import searchsorted_words

        pos = searchsorted_words(
            [ukeys[:, 0].contiguous(), ukeys[:, 1].contiguous()],
            [jl[:, 0].contiguous(), jl[:, 1].contiguous()],
            side="left",
        )
        row_cnt = ucnt.index_select(0, pos.clamp(0, max(ukeys.shape[0] - 1, 0)))
        un_idx = (row_cnt == 0).nonzero(as_tuple=True)[0]
        if un_idx.shape[0] == 0:
            return []
        probe = bl.take(un_idx)
        n = len(probe)
        if self.key_mode == "left":
            okeys = probe.keys
        else:
            lo, hi = hashing.derive_key_words(
                self.lpad_salt,
                [(probe.keys[:, 0].contiguous(), probe.keys[:, 1].contiguous())],
            )
            okeys = torch.stack([lo, hi], dim=1)
        out_cols: dict[str, Column] = {}
        for out_name, src in self.left_out.items():
            out_cols[out_name] = probe.columns[src]
        for out_name, src in self.right_out.items():
            proto = (self._right_proto or {}).get(src)
            if proto is None:
                out_cols[out_name] = ObjectColumn(np.empty(n, dtype=object), dt.ANY)
            else:
                out_cols[out_name] = null_column(proto, n, device)
        if self.left_id_name:
            out_cols[self.left_id_name] = PointerColumn(probe.keys)
        if self.right_id_name:
            out_cols[self.right_id_name] = ObjectColumn(
                np.empty(n, dtype=object), dt.Optional(dt.POINTER)
            )
        order = list(self.left_out.keys()) + list(self.right_out.keys())
        if self.left_id_name:
            order.append(self.left_id_name)
        if self.right_id_name:
            order.append(self.right_id_name)
        out_cols = {k: out_cols[k] for k in order}
        return [DeltaBatch(okeys, out_cols, probe.diffs, probe.time)]

    # -- padded rows for outer modes --

    def _padded_rows(self, affected: torch.Tensor, side: str) -> DeltaBatch | None:
        """Current unmatched rows of `side` among affected join keys."""
        device = self.device
        own = self.lstore if side == "left" else self.rstore
        other = self.rstore if side == "left" else self.lstore
        if len(own) == 0:
            return None
        other_cnt = other.count_for(affected)
        un_keys_mask = other_cnt == 0
        if not bool(un_keys_mask.any()):
            return None
        ukeys = affected.index_select(0, un_keys_mask.nonzero(as_tuple=True)[0])
        cols, qidx, w = own.rows_for(ukeys)
        if qidx.shape[0] == 0:
            return None
        cols = dict(cols)
        ids = cols.pop("__id__")
        n = qidx.shape[0]
        if (self.key_mode == "left" and side == "left") or (
            self.key_mode == "right" and side == "right"
        ):
            okeys = ids.pairs
        else:
            okeys_salt = self.lpad_salt if side == "left" else self.rpad_salt
            lo, hi = hashing.derive_key_words(
                okeys_salt,
                [(ids.pairs[:, 0].contiguous(), ids.pairs[:, 1].contiguous())],
            )
            okeys = torch.stack([lo, hi], dim=1)
        n = qidx.shape[0]
        out_cols: dict[str, Column] = {}
        own_map = self.left_out if side == "left" else self.right_out
        other_map = self.right_out if side == "left" else self.left_out
        other_proto = self._right_proto if side == "left" else self._left_proto
        for out_name, src in own_map.items():
            out_cols[out_name] = cols[src]
        for out_name, src in other_map.items():
            proto = (other_proto or {}).get(src)
            if proto is None:
                out_cols[out_name] = ObjectColumn(np.empty(n, dtype=object), dt.ANY)
            else:
                out_cols[out_name] = null_column(proto, n, device)
        if self.left_id_name:
            if side == "left":
                out_cols[self.left_id_name] = PointerColumn(ids.pairs)
            else:
                out_cols[self.left_id_name] = ObjectColumn(np.empty(n, dtype=object), dt.Optional(dt.POINTER))
        if self.right_id_name:
            if side == "right":
                out_cols[self.right_id_name] = PointerColumn(ids.pairs)
            else:
                out_cols[self.right_id_name] = ObjectColumn(np.empty(n, dtype=object), dt.Optional(dt.POINTER))
        # reorder columns to canonical order
        order = list(self.left_out.keys()) + list(self.right_out.keys())
        if self.left_id_name:
            order.append(self.left_id_name)
        if self.right_id_name:
            order.append(self.right_id_name)
        out_cols = {k: out_cols[k] for k in order}
        return DeltaBatch(okeys, out_cols, w, 0)


def _pad_delta(old: DeltaBatch | None, new: DeltaBatch | None, time, device):
    out = []
    if old is not None and len(old):
        out.append(DeltaBatch(old.keys, old.columns, -old.diffs, time))
    if new is not None and len(new):
        out.append(DeltaBatch(new.keys, new.columns, new.diffs, time))
    return out


class SemiJoinNode(Node):
    """intersect / difference / restrict: filter left rows by key presence
    in the right table (by row key)."""

    def __init__(self, left: Node, right: Node, mode: str, device):
        super().__init__([left, right], device)
        assert mode in ("intersect", "difference")
        self.mode = mode
        self.lstore = _SideStore(device)
        self.rcount: _SideStore = _SideStore(device)

    def reset(self) -> None:
        self.lstore = _SideStore(self.device)
        self.rcount = _SideStore(self.device)
        self._xmeta_l = {}
        self._xmeta_r = {}

    def wants_frontier(self) -> bool:
        from pathway_amd.parallel import get_comm

        c = get_comm()
        return c is not None and c.world > 1

    def step(self, time, inputs):
        bl, br = inputs
        from pathway_amd.parallel import get_comm

        comm = get_comm()
        if comm is not None and comm.world > 1:
            if not hasattr(self, "_xmeta_l"):
                self._xmeta_l = {}
                self._xmeta_r = {}
            bl, _ = _exchange_side(
                comm, bl, bl.keys if bl is not None else None, time, self._xmeta_l
            )
            br, _ = _exchange_side(
                comm, br, br.keys if br is not None else None, time, self._xmeta_r
            )
        device = self.device
        affected = []
        if bl is not None and len(bl):
            affected.append(unique_sorted_keys(bl.keys))
        if br is not None and len(br):
            affected.append(unique_sorted_keys(br.keys))
        if not affected:
            return None
        aff = unique_sorted_keys(torch.cat(affected, dim=0))
        old = self._visible(aff)
        if bl is not None and len(bl):
            self.lstore.merge(bl.keys, bl)
        if br is not None and len(br):
            empty_cols: dict[str, Column] = {}
            rb = DeltaBatch(br.keys, empty_cols, br.diffs, br.time)
            self.rcount.merge(br.keys, rb)
        new = self._visible(aff)
        parts = _pad_delta(old, new, time, device)
        parts = [p for p in parts if p is not None and len(p)]
        if not parts:
            return None
        return consolidate_batch(DeltaBatch.concat(parts))

    def _visible(self, aff: torch.Tensor) -> DeltaBatch | None:
        device = self.device
        if len(self.lstore) == 0:
            return None
        rc = self.rcount.count_for(aff)
        want = (rc > 0) if self.mode == "intersect" else (rc == 0)
        if not bool(want.any()):
            return None
        keys = aff.index_select(0, want.nonzero(as_tuple=True)[0])
        cols, qidx, w = self.lstore.rows_for(keys)
        if qidx.shape[0] == 0:
            return None
        cols = dict(cols)
        ids = cols.pop("__id__")
        return DeltaBatch(ids.pairs, cols, w, 0)


class KeyedMergeNode(Node):
    """update_rows / update_cells: right rows override left rows per key.

    update_rows: universe = left ∪ right, all columns overridden.
    update_cells: universe = left, only `override_cols` overridden.
    """

    def __init__(self, left: Node, right: Node, mode: str, override_cols: list[str], device):
        super().__init__([left, right], device)
        assert mode in ("rows", "cells")
        self.mode = mode
        self.override_cols = override_cols
        self.lstore = _SideStore(device)
        self.rstore = _SideStore(device)

    def reset(self) -> None:
        self.lstore = _SideStore(self.device)
        self.rstore = _SideStore(self.device)
        self._xmeta_l = {}
        self._xmeta_r = {}

    def wants_frontier(self) -> bool:
        from pathway_amd.parallel import get_comm

        c = get_comm()
        return c is not None and c.world > 1

    def step(self, time, inputs):
        bl, br = inputs
        from pathway_amd.parallel import get_comm

        comm = get_comm()
        if comm is not None and comm.world > 1:
            # the two tables of an update may shard the same row key to
            # different ranks at the source: co-locate by row key
            if not hasattr(self, "_xmeta_l"):
                self._xmeta_l = {}
                self._xmeta_r = {}
            bl, _ = _exchange_side(
                comm, bl, bl.keys if bl is not None else None, time, self._xmeta_l
            )
            br, _ = _exchange_side(
                comm, br, br.keys if br is not None else None, time, self._xmeta_r
            )
        device = self.device
        affected = []
        if bl is not None and len(bl):
            affected.append(unique_sorted_keys(bl.keys))
        if br is not None and len(br):
            affected.append(unique_sorted_keys(br.keys))
        if not affected:
            return None
        aff = unique_sorted_keys(torch.cat(affected, dim=0))
        old = self._visible(aff)
        if bl is not None and len(bl):
            self.lstore.merge(bl.keys, bl)
        if br is not None and len(br):
            self.rstore.merge(br.keys, br)
        new = self._visible(aff)
        parts = _pad_delta(old, new, time, device)
        parts = [p for p in parts if p is not None and len(p)]
        if not parts:
            return None
        return consolidate_batch(DeltaBatch.concat(parts))

    def _visible(self, aff: torch.Tensor) -> DeltaBatch | None:
        device = self.device
        lhas = len(self.lstore) > 0
        rhas = len(self.rstore) > 0
        if not lhas and not rhas:
            return None
        lcnt = self.lstore.count_for(aff) if lhas else torch.zeros(aff.shape[0], dtype=torch.int64, device=device)
        rcnt = self.rstore.count_for(aff) if rhas else torch.zeros(aff.shape[0], dtype=torch.int64, device=device)
        batches = []
        if self.mode == "rows":
            # right wins; keys only in left pass through
            rkeys_mask = rcnt > 0
            lonly_mask = (lcnt > 0) & (rcnt == 0)
            if bool(rkeys_mask.any()) and rhas:
                keys = aff.index_select(0, rkeys_mask.nonzero(as_tuple=True)[0])
                batches.append(self._gather_side(self.rstore, keys))
            if bool(lonly_mask.any()) and lhas:
                keys = aff.index_select(0, lonly_mask.nonzero(as_tuple=True)[0])
                batches.append(self._gather_side(self.lstore, keys))
        else:  # cells
            lmask = lcnt > 0
            if not bool(lmask.any()):
                return None
            keys = aff.index_select(0, lmask.nonzero(as_tuple=True)[0])
            lb = self._gather_side(self.lstore, keys)
            if lb is None:
                return None
            if rhas:
                # override cells where right row exists for same key
                rcols, qidx, w = self.rstore.rows_for(lb.keys)
                if qidx.shape[0]:
                    cols = dict(lb.columns)
                    for cname in self.override_cols:
                        cols[cname] = _scatter_override(
                            cols[cname], qidx, rcols[cname]
                        )
                    lb = lb.with_columns(cols)
            batches.append(lb)
        batches = [b for b in batches if b is not None and len(b)]
        if not batches:
            return None
        names = list(batches[0].columns.keys())
        batches = [b.select_columns(names) for b in batches]
        return DeltaBatch.concat(batches)

    def _gather_side(self, store: _SideStore, keys: torch.Tensor) -> DeltaBatch | None:
        cols, qidx, w = store.rows_for(keys)
        if qidx.shape[0] == 0:
            return None
        cols = dict(cols)
        ids = cols.pop("__id__")
        return DeltaBatch(ids.pairs, cols, w, 0)


def _scatter_override(base: Column, qidx: torch.Tensor, repl: Column) -> Column:
    """Replace base[qidx[j]] with repl[j]."""
    if isinstance(base, TensorColumn) and isinstance(repl, TensorColumn):
        t = base.tensor.clone()
        t[qidx] = repl.tensor.to(t.dtype)
        mask = base.mask.clone() if base.mask is not None else None
        if mask is not None:
            mask[qidx] = (
                repl.mask if repl.mask is not None else torch.ones_like(repl.tensor, dtype=torch.bool)
            )
        return TensorColumn(t, base.dtype, mask)
    if isinstance(base, StringColumn) and isinstance(repl, StringColumn) and base.pool is repl.pool:
        codes = base.codes.clone()
        codes[qidx] = repl.codes
        return StringColumn(codes, base.pool, base.dtype)
    vals = base.to_pylist()
    rv = repl.to_pylist()
    for j, q in enumerate(qidx.cpu().tolist()):
        vals[q] = rv[j]
    from pathway_amd.engine.column import obj_array
    return ObjectColumn(obj_array(vals), base.dtype)


class FlattenNode(Node):
    """flatten_table: explode a sequence column into rows."""

    def __init__(
        self, input_node: Node, flatten_name: str, device,
        origin_id: str | None = None,
    ):
        super().__init__([input_node], device)
        self.flatten_name = flatten_name
        self.origin_id = origin_id
        self.salt = _salt("flatten", self.node_id)

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0:
            return None
        device = self.device
        col = b.columns[self.flatten_name]
        vals = col.to_pylist()
        rep_idx = []
        flat_vals = []
        ordinals = []
        for i, v in enumerate(vals):
            if v is None:
                continue
            if isinstance(v, str):
                seq = list(v)
            else:
                seq = list(v)
            for j, x in enumerate(seq):
                rep_idx.append(i)
                ordinals.append(j)
                flat_vals.append(x)
        if not rep_idx:
            return None
        idx = torch.tensor(rep_idx, dtype=torch.int64, device=device)
        base = b.take(idx)
        from pathway_amd.engine.column import infer_and_build_column

        fcol, _ = infer_and_build_column(flat_vals, device)
        cols = dict(base.columns)
        cols[self.flatten_name] = fcol
        if self.origin_id:
            from pathway_amd.engine.column import PointerColumn

            cols[self.origin_id] = PointerColumn(base.keys.clone())
        ords = torch.tensor(ordinals, dtype=torch.int64, device=device)
        olo, ohi = hashing.value_hash_words(ords, 2)
        lo, hi = hashing.derive_key_words(
            self.salt,
            [
                (base.keys[:, 0].contiguous(), base.keys[:, 1].contiguous()),
                (olo, ohi),
            ],
        )
        keys = torch.stack([lo, hi], dim=1)
        return DeltaBatch(keys, cols, base.diffs, time)
