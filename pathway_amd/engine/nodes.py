"""Engine dataflow nodes — the MI355X-native operator set.

Each node mirrors one Graph-trait operator family of the reference
(src/engine/graph.rs:651-1055 — see SURVEY.md §2.3) but is implemented as a
GPU-columnar micro-batch transform over DeltaBatches:

  InputNode            connector_table / static table
  ExprMapNode          expression_table (select / with_columns)
  FilterNode           filter_table
  ReindexNode          reindex_table / with_id_from
  GroupReduceNode      group_by_table + reducers
  JoinNode             join_tables (inner/left/right/outer) + ix
  ConcatNode           concat_tables
  KeyedMergeNode       update_rows_table / update_cells_table
  SemiJoinNode         intersect_tables / subtract_table / restrict
  FlattenNode          flatten_table
  CaptureNode          capture for debug / tests
  OutputNode           output_table (sinks)

State lives in sorted GPU arrangements (engine/state.py).  All per-batch
work is torch ops (CPU & ROCm); hot paths are overridden by HIP kernels via
pathway_amd.ops when running on gfx950.
"""

from __future__ import annotations

from typing import Any, Sequence

import numpy as np
import torch

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.api import SHARD_MASK

MASK_SHARD = SHARD_MASK
from pathway_amd.engine import hashing
from pathway_amd.engine.batch import DeltaBatch, segment_starts
from pathway_amd.engine.column import (
    Column,
    ObjectColumn,
    PointerColumn,
    StringColumn,
    TensorColumn,
    column_from_pylist,
    concat_columns,
)
from pathway_amd.engine.expression_eval import EvalContext, evaluate
from pathway_amd.engine.reducers import REDUCERS, ReducerSpec
from pathway_amd.engine.state import (
    Arrangement,
    consolidate_sorted,
    lex_sort_words,
    searchsorted_words,
)

import os as _os

#: fused segmented-reduce pre-agg (pw_seg_reduce) — default ON on device;
#: PW_NO_SEGRED=1 falls back to the torch boundary chain (A/B)
_PW_NO_SEGRED = bool(_os.environ.get("PW_NO_SEGRED"))

#: opt-in HIP hash-aggregation pre-agg (PW_HASHAGG=1). Measured on MI355X:
#: the sort path wins at both 50k-distinct (2.95 vs 3.63 ms/step — atomic
#: contention on hot counters) and 50M-distinct (sort savings vanish when
#: distinct≈batch); rocprim onesweep radix sort is the better default.
_PW_HASHAGG = bool(_os.environ.get("PW_HASHAGG"))
_PW_NO_HASHAGG = bool(_os.environ.get("PW_NO_HASHAGG"))

_node_counter = [0]


def _salt(tag: str, node_id: int) -> int:
    from pathway_amd.internals.api import xxh64

    return xxh64(f"{tag}:{node_id}".encode(), 7)


class Node:
    n_outputs = 1

    def __init__(self, inputs: Sequence["Node"], device):
        self.inputs = list(inputs)
        self.device = device
        self.node_id = _node_counter[0]
        _node_counter[0] += 1
        self.name: str | None = None

    def step(self, time: int, inputs: list[DeltaBatch | None]) -> DeltaBatch | None:
        raise NotImplementedError

    def on_frontier(self, time: int) -> DeltaBatch | None:
        """Called when the frontier passes `time` (temporal behaviors)."""
        return None

    def wants_frontier(self) -> bool:
        return False

    def reset(self) -> None:
        """Clear mutable state before a fresh run (re-running a graph)."""


# ------------------------------------------------------------------ utils --

def batch_vhash(batch: DeltaBatch) -> tuple[torch.Tensor, torch.Tensor]:
    """128-bit row-value hash of a batch (the arrangement tiebreaker)."""
    n = len(batch)
    parts = []
    for name in batch.columns:
        lo, hi = batch.columns[name].value_hash()
        parts.append((lo.to(batch.device), hi.to(batch.device)))
    if not parts:
        z = torch.zeros(n, dtype=torch.int64, device=batch.device)
        return z, z.clone()
    return hashing.combine_value_hashes(parts)


def consolidate_batch(batch: DeltaBatch) -> DeltaBatch | None:
    """Sort by (key, vhash), sum diffs, drop zeros (consolidation.rs)."""
    if batch is None or len(batch) == 0:
        return None
    if batch.consolidated:
        return batch
    v0, v1 = batch_vhash(batch)
    words = [batch.keys[:, 0].contiguous(), batch.keys[:, 1].contiguous(), v0, v1]
    if len(batch) > 2048 and batch.keys.is_cuda:
        # consolidation only needs equal (key, vhash) rows adjacent — the
        # order itself is irrelevant for outputs.  Sorting by a fused
        # 128-bit row hash turns the 4-word multi-pass (4 stable
        # argsorts) into the single-sort fast path; equal rows share the
        # hash, so grouping is preserved (c0-collision guard inside
        # lex_sort_words).
        c0, c1 = hashing.hash128_words([w.contiguous() for w in words])
        perm = lex_sort_words([c0, c1])
    else:
        perm = lex_sort_words(words)
    from pathway_amd import ops

    *words, weights = ops.gather_all(perm, words + [batch.diffs])
    cols = {n: c.take(perm) for n, c in batch.columns.items()}
    out_words, out_w, out_cols = consolidate_sorted(words, weights, cols)
    if out_w.shape[0] == 0:
        return None
    keys = torch.stack(out_words[:2], dim=1)
    return DeltaBatch(keys, out_cols, out_w, batch.time)


def unique_sorted_keys(keys: torch.Tensor) -> torch.Tensor:
    words = [keys[:, 0].contiguous(), keys[:, 1].contiguous()]
    perm = lex_sort_words(words)
    sk = keys.index_select(0, perm)
    starts = segment_starts(sk)
    return sk.index_select(0, starts.nonzero(as_tuple=True)[0])


def null_column(proto: Column, n: int, device, dtype: dt.DType | None = None) -> Column:
    d = dtype or proto.dtype
    if isinstance(proto, TensorColumn):
        t = torch.zeros((n,), dtype=proto.tensor.dtype, device=device)
        mask = torch.zeros((n,), dtype=torch.bool, device=device)
        return TensorColumn(t, dt.Optional(dt.unoptionalize(d)), mask)
    if isinstance(proto, StringColumn):
        codes = torch.full((n,), -1, dtype=torch.int64, device=device)
        return StringColumn(codes, proto.pool, dt.Optional(dt.STR))
    arr = np.empty(n, dtype=object)
    return ObjectColumn(arr, dt.Optional(dt.unoptionalize(d)))


# ------------------------------------------------------------------ input --

class InputNode(Node):
    """Source of delta batches; wraps a Source with pull(time)."""

    def __init__(self, source: Any, device):
        super().__init__([], device)
        self.source = source

    def step(self, time: int, inputs: list[DeltaBatch | None]) -> DeltaBatch | None:
        return self.source.pull(time, self.device)

    def reset(self) -> None:
        r = getattr(self.source, "reset", None)
        if r is not None:
            r()


class ExprMapNode(Node):
    """select / with_columns: evaluate expressions, keep keys.

    extra_inputs: same-universe tables referenced in the expressions
    (reference: rowwise context over several universe-equal tables).
    With extras the node is STATEFUL: per-key arrangements of the main
    and extra tables, old-visible/merge/new-visible delta emission —
    aligning against the other table's per-step delta is wrong under
    retractions (the new main row would read the retracted old value)
    and silent when only the extra side changes.
    """

    def __init__(
        self,
        input_node: Node,
        exprs: dict[str, Any],
        device,
        extra_inputs: list[tuple[Node, str]] | None = None,
    ):
        extra_inputs = extra_inputs or []
        super().__init__([input_node] + [n for n, _ in extra_inputs], device)
        self.exprs = exprs
        self.extra_prefixes = [p for _, p in extra_inputs]
        if self.extra_prefixes:
            self.reset()

    def reset(self) -> None:
        if not self.extra_prefixes:
            return
        from pathway_amd.engine.nodes_join import _SideStore

        self.main_store = _SideStore(self.device)
        self.extra_stores = [_SideStore(self.device) for _ in self.extra_prefixes]

    def step(self, time, inputs):
        b = inputs[0]
        if not self.extra_prefixes:
            if b is None or len(b) == 0:
                return None
            ctx = EvalContext(b.columns, b.keys, self.device)
            out_cols = {name: evaluate(e, ctx) for name, e in self.exprs.items()}
            return DeltaBatch(b.keys, out_cols, b.diffs, time)
        ebs = list(inputs[1:])
        parts_in = [x for x in [b, *ebs] if x is not None and len(x)]
        if not parts_in:
            return None
        aff = unique_sorted_keys(
            torch.cat([x.keys for x in parts_in], dim=0)
        )
        old = self._eval_visible(aff)
        if b is not None and len(b):
            self.main_store.merge(b.keys, b)
        for st, eb in zip(self.extra_stores, ebs):
            if eb is not None and len(eb):
                st.merge(eb.keys, eb)
        new = self._eval_visible(aff)
        from pathway_amd.engine.nodes_join import _pad_delta

        parts = [p for p in _pad_delta(old, new, time, self.device) if p is not None and len(p)]
        if not parts:
            return None
        return consolidate_batch(DeltaBatch.concat(parts))

    def _eval_visible(self, aff: torch.Tensor):
        """Evaluate the expressions over the CURRENT state rows of the
        affected keys (None if no main rows)."""
        from pathway_amd.engine.nodes_join import _scatter_override

        cols, qidx, w = self.main_store.probe(aff)
        if cols is None or qidx.shape[0] == 0:
            return None
        cols = dict(cols)
        ids = cols.pop("__id__")
        keys = ids.pairs
        n = keys.shape[0]
        extra: dict[str, Column] = {}
        for st, prefix in zip(self.extra_stores, self.extra_prefixes):
            ecols, eqidx, _ew = st.probe(keys)
            if ecols is None:
                continue
            ecols = dict(ecols)
            ecols.pop("__id__", None)
            for cname, c in ecols.items():
                if eqidx.shape[0] == n and bool(
                    (eqidx == torch.arange(n, device=eqidx.device)).all()
                ):
                    extra[f"{prefix}{cname}"] = c
                else:
                    base = null_column(c, n, self.device)
                    extra[f"{prefix}{cname}"] = _scatter_override(
                        base, eqidx, c
                    )
        ctx = EvalContext(cols, keys, self.device, extra=extra)
        out_cols = {name: evaluate(e, ctx) for name, e in self.exprs.items()}
        return DeltaBatch(keys, out_cols, w, 0)


class FilterNode(Node):
    def __init__(self, input_node: Node, pred_expr: Any, device):
        super().__init__([input_node], device)
        self.pred = pred_expr

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0:
            return None
        ctx = EvalContext(b.columns, b.keys, self.device)
        col = evaluate(self.pred, ctx)
        if isinstance(col, TensorColumn):
            mask = col.tensor.to(torch.bool)
            if col.mask is not None:
                mask = mask & col.mask
        else:
            mask = torch.tensor(
                [bool(v) if v is not None else False for v in col.to_pylist()],
                dtype=torch.bool,
                device=b.device,
            )
        out = b.filter(mask)
        return out if len(out) else None


class ReindexNode(Node):
    """Assign new row keys from a pointer expression (reindex/with_id_from)."""

    def __init__(self, input_node: Node, key_expr: Any, device):
        super().__init__([input_node], device)
        self.key_expr = key_expr

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0:
            return None
        ctx = EvalContext(b.columns, b.keys, self.device)
        col = evaluate(self.key_expr, ctx)
        assert isinstance(col, PointerColumn), "reindex needs a pointer expression"
        return DeltaBatch(col.pairs, b.columns, b.diffs, time)


class DeriveKeyNode(Node):
    """Re-key with a salted hash of the existing key (concat_reindex etc.)."""

    def __init__(self, input_node: Node, tag: str, device):
        super().__init__([input_node], device)
        self.salt = _salt(tag, self.node_id)

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0:
            return None
        lo, hi = hashing.derive_key_words(
            self.salt, [(b.keys[:, 0].contiguous(), b.keys[:, 1].contiguous())]
        )
        return DeltaBatch(torch.stack([lo, hi], dim=1), b.columns, b.diffs, time)


class ConcatNode(Node):
    def __init__(self, input_nodes: Sequence[Node], device):
        super().__init__(input_nodes, device)

    def step(self, time, inputs):
        batches = [b for b in inputs if b is not None and len(b)]
        if not batches:
            return None
        names = list(batches[0].columns.keys())
        batches = [b.select_columns(names) for b in batches]
        return DeltaBatch.concat(batches)


# ----------------------------------------------------------------- reduce --

class GroupReduceNode(Node):
    """groupby().reduce() — incremental segmented reduce with retractions.

    Mirrors DataflowGraphInner::group_by_table (dataflow.rs:3761-3868):
    group key = hash of grouping values; per-reducer state merged per batch;
    on change emits (old row, -1) and (new row, +1) at batch time.
    """

    def __init__(
        self,
        input_node: Node,
        group_exprs: dict[str, Any],
        reducer_calls: dict[str, tuple[str, list[Any], dict]],
        device,
        sort_by: Any | None = None,
    ):
        super().__init__([input_node], device)
        self.group_exprs = group_exprs
        self.reducer_calls = reducer_calls
        self.sort_by = sort_by
        self.key_expr = None  # groupby(id=...): group key IS this pointer expr
        #: instance column name — shard bits of the group key come from the
        #: instance hash alone (ShardPolicy::LastKeyColumn, value.rs:96-118)
        self.instance_name: str | None = None
        self.seq = 0  # arrival sequence for earliest/latest

        # combined reduce state: sorted keys + additive acc tensors +
        # carried group-value columns (key-determined, first-write-wins)
        self.add_keys: list[torch.Tensor] | None = None
        self.add_accs: dict[str, torch.Tensor] = {}
        self.add_carried: dict[str, Column] = {}
        self.multiset_store: Arrangement | None = None
        self.multiset_colnames: list[str] = []
        self._spec_cache: dict[str, ReducerSpec] = {}

    def reset(self) -> None:
        self.seq = 0
        self._xmeta_add = {}
        self._xmeta_ms = {}
        self.add_keys = None
        self.add_accs = {}
        self.add_carried = {}
        self.multiset_store = None
        self.multiset_colnames = []

    # -- helpers --

    def _specs(self) -> dict[str, tuple[ReducerSpec, list[Any], dict]]:
        out = {}
        for out_name, (rname, args, kwargs) in self.reducer_calls.items():
            if rname == "stateful_many":
                combine = kwargs["_combine_many"]

                _n = max(len(args), 1)

                def host_agg(rows, _c=combine, _n=_n):
                    return _c(None, [(list(t)[:_n], w) for t, w in rows])

                spec = ReducerSpec(
                    name="stateful_many",
                    family="multiset",
                    n_args=len(args),
                    host_agg=host_agg,
                    out_dtype=lambda args_dt: dt.ANY,
                )
            elif rname == "udf_reducer":
                acc_cls = kwargs["_accumulator_cls"]

                _n = max(len(args), 1)

                def host_agg(rows, _cls=acc_cls, _n=_n):
                    acc = None
                    for t, w in rows:
                        row = list(t)[:_n]
                        for _ in range(w):
                            a = _cls.from_row(row)
                            if acc is None:
                                acc = a
                            else:
                                acc.update(a)
                    return acc.compute_result() if acc is not None else None

                spec = ReducerSpec(
                    name="udf_reducer",
                    family="multiset",
                    n_args=len(args),
                    host_agg=host_agg,
                    out_dtype=lambda args_dt: dt.ANY,
                )
            else:
                spec = REDUCERS.get(rname)
            if spec is None:
                raise NotImplementedError(f"reducer {rname}")
            out[out_name] = (spec, args, kwargs)
        return out

    def _additive_names(self):
        # count-style accumulators alias the presence weight __w__ (same
        # per-row contribution: diffs) — no separate state arrays for them
        names = []
        for out_name, (spec, args, kwargs) in self._specs().items():
            if spec.family == "additive":
                if spec.name == "avg":
                    names.append(f"{out_name}__sum")
                elif spec.name == "sum":
                    names.append(out_name)
        return names

    def wants_frontier(self) -> bool:
        # in multi-worker mode every rank must join the exchange collectives
        # each step, even with no local delta
        from pathway_amd.parallel import get_comm

        c = get_comm()
        return c is not None and c.world > 1

    def step(self, time, inputs):
        b = inputs[0]
        from pathway_amd.parallel import get_comm

        comm = get_comm()
        distributed = comm is not None and comm.world > 1
        if b is None or len(b) == 0:
            if not distributed:
                return None
            b = None
        device = self.device
        specs = self._specs()
        has_multiset = any(s.family == "multiset" for s, _, _ in specs.values())
        from pathway_amd.engine.state import rows_ne

        gkeys = rowkeys = diffs = None
        gcols: dict[str, Column] = {}
        arg_cols: dict[str, list[Column]] = {}
        ukeys_w = acc_deltas = gcols_first = None
        if b is not None:
            seq0 = self.seq
            self.seq += len(b)
            seq_col = TensorColumn(
                torch.arange(seq0, seq0 + len(b), dtype=torch.int64, device=device),
                dt.INT,
            )
            ctx = EvalContext(b.columns, b.keys, device, extra={"__seq__": seq_col})
            # 1. evaluate grouping columns and compute group keys
            gcols = {n: evaluate(e, ctx) for n, e in self.group_exprs.items()}
            if self.key_expr is not None:
                kc = evaluate(self.key_expr, ctx)
                assert isinstance(kc, PointerColumn), "groupby(id=...) needs a pointer"
                gkeys = kc.pairs
            else:
                parts = [
                    (lo.to(device), hi.to(device))
                    for lo, hi in (c.value_hash() for c in gcols.values())
                ]
                if parts:
                    glo, ghi = hashing.combine_value_hashes(parts)
                else:
                    glo = torch.zeros(len(b), dtype=torch.int64, device=device)
                    ghi = glo.clone()
                if self.instance_name and self.instance_name in gcols:
                    ilo, _ihi = gcols[self.instance_name].value_hash()
                    ilo = ilo.to(device)
                    glo = (glo & ~MASK_SHARD) | (ilo & MASK_SHARD)
                gkeys = torch.stack([glo, ghi], dim=1)

            # 2. evaluate reducer args
            for out_name, (spec, args, kwargs) in specs.items():
                arg_cols[out_name] = [evaluate(a, ctx) for a in args]

            rowkeys = b.keys
            diffs = b.diffs

            # 3. local pre-aggregation (the combiner — also what the
            # exchange ships for N>1)
            contribs = self._contributions(arg_cols, diffs, specs)
            hashagg = None
            uniq_est = getattr(self, "_uniq_est", None)
            # opt-in only: even with an LLC-resident estimate-sized table
            # (round-2 re-run) the atomic path loses the A/B to the radix
            # sort — 1.06B vs 1.62B ev/s on the wordcount headline (~80
            # duplicate rows per hot counter serialize, and the extra
            # mid-step size sync breaks the H2D/compute overlap).  Twice
            # measured, twice rejected: profiles/wordcount_r01.md and
            # profiles/kernels_r02.md.
            use_ha = _PW_HASHAGG
            if (
                torch.device(device).type == "cuda"
                and use_ha
                and all(c.dtype == torch.int64 for c in contribs.values())
            ):
                # sort-free path: one HIP hash-aggregation pass over the
                # batch (pw_hash_agg) — replaces the 4M-row radix sort +
                # segment-boundary chain with atomics into an open-
                # addressing table, then sort/consolidate only the (small)
                # distinct-key set
                from pathway_amd import ops

                names = list(contribs)
                ha = ops.hash_agg_gpu(
                    gkeys[:, 0].contiguous(),
                    gkeys[:, 1].contiguous(),
                    [contribs[nm] for nm in names],
                    expected_uniques=uniq_est,
                )
                if ha is None:
                    # estimate too small (probe overflow): redo on the
                    # sort path; quadruple the estimate for next step
                    self._uniq_est = (uniq_est or 1024) * 4
                    hashagg = None
                else:
                    uk0, uk1, accs, rep = ha
                    hashagg = True
            if hashagg:
                perm = lex_sort_words([uk0, uk1])
                uk0s = uk0.index_select(0, perm)
                uk1s = uk1.index_select(0, perm)
                starts = rows_ne([uk0s, uk1s])
                first_idx = starts.nonzero(as_tuple=True)[0]
                nseg = int(first_idx.numel())
                if nseg == uk0s.shape[0]:
                    # no duplicate slots (the common case)
                    ukeys_w = [uk0s, uk1s]
                    acc_deltas = {
                        nm: a.index_select(0, perm) for nm, a in zip(names, accs)
                    }
                    rep_first = rep.index_select(0, perm)
                else:
                    # rare publication race: merge duplicate-key slots
                    seg = torch.cumsum(starts.to(torch.int64), 0) - 1
                    ukeys_w = [
                        w.index_select(0, first_idx) for w in (uk0s, uk1s)
                    ]
                    acc_deltas = {}
                    for nm, a in zip(names, accs):
                        sa = a.index_select(0, perm)
                        out = torch.zeros(nseg, dtype=sa.dtype, device=device)
                        out.index_add_(0, seg, sa)
                        acc_deltas[nm] = out
                    rep_first = rep.index_select(0, perm).index_select(
                        0, first_idx
                    )
                gcols_first = {n: c.take(rep_first) for n, c in gcols.items()}
                hashagg = True
            if hashagg is None and (
                torch.device(device).type == "cuda"
                and not _PW_NO_SEGRED
                and all(c.dtype == torch.int64 for c in contribs.values())
            ):
                # sort + fused segmented reduce (pw_seg_reduce): the whole
                # run-starts/compaction/per-acc-sum chain in two kernels
                from pathway_amd import ops

                words = [gkeys[:, 0].contiguous(), gkeys[:, 1].contiguous()]
                perm = lex_sort_words(words)
                names = list(contribs)
                sw0, sw1, *sc_list = ops.gather_all(
                    perm, words + [contribs[nm] for nm in names]
                )
                uk0, uk1, first_sorted, accs = ops.seg_reduce_gpu(
                    sw0, sw1, sc_list
                )
                ukeys_w = [uk0, uk1]
                acc_deltas = dict(zip(names, accs))
                gfirst_rows = perm.index_select(0, first_sorted)
                gcols_first = {
                    n: c.take(gfirst_rows) for n, c in gcols.items()
                }
                hashagg = True
            if hashagg is None:
                # sort path: ONE lex sort of the batch's group keys, then
                # segmented sums per additive accumulator
                words = [gkeys[:, 0].contiguous(), gkeys[:, 1].contiguous()]
                perm = lex_sort_words(words)
                swords = [w.index_select(0, perm) for w in words]
                starts = rows_ne(swords)
                seg = torch.cumsum(starts.to(torch.int64), 0) - 1
                first_idx = starts.nonzero(as_tuple=True)[0]
                nseg = int(first_idx.numel())
                ukeys_w = [w.index_select(0, first_idx) for w in swords]
                # segmented sums over the sorted batch via prefix-sum
                # boundary differences — exact for int64, atomic-free
                # (index_add_ was the contention hot spot at high
                # duplicates-per-group)
                nrows = perm.shape[0]
                seg_ends = (
                    torch.cat(
                        [
                            first_idx[1:],
                            torch.tensor(
                                [nrows], dtype=torch.int64, device=device
                            ),
                        ]
                    )
                    - 1
                )
                acc_deltas = {}
                for name, c in contribs.items():
                    sc = c.index_select(0, perm)
                    if sc.dtype == torch.float64:
                        acc = torch.zeros(nseg, dtype=sc.dtype, device=device)
                        acc.index_add_(0, seg, sc)
                    else:
                        cs = torch.cumsum(sc, 0)
                        seg_tot = cs.index_select(0, seg_ends)
                        acc = seg_tot.clone()
                        acc[1:] -= seg_tot[:-1]
                    acc_deltas[name] = acc
                gfirst_rows = perm.index_select(0, first_idx)
                gcols_first = {n: c.take(gfirst_rows) for n, c in gcols.items()}

        if ukeys_w is not None and ukeys_w[0].is_cuda:
            # feed the hash-agg table-size estimate for the next step
            # (shape is host metadata — no device sync)
            self._uniq_est = max(int(ukeys_w[0].shape[0]), 512)

        # 3b. multi-worker: all-to-all-v of pre-aggregated partials by key
        # shard (RCCL over xGMI; pact.rs:56 analog with combiner)
        if distributed:
            from pathway_amd.parallel.exchange import exchange_bundle, shard_of

            if ukeys_w is not None:
                ukeys_t = torch.stack(ukeys_w, dim=1)
                dest = shard_of(ukeys_t, comm.world)
                tensors = {"k": ukeys_t}
                tensors.update({f"acc.{n}": t for n, t in acc_deltas.items()})
            else:
                dest = tensors = None
            if not hasattr(self, "_xmeta_add"):
                self._xmeta_add = {}
                self._xmeta_ms = {}
            tensors, gcols_first = exchange_bundle(
                comm, dest, tensors, gcols_first, meta_state=self._xmeta_add
            )
            ms_exchanged = None
            if has_multiset:
                # multiset values cannot be pre-combined: ship the raw rows
                if gkeys is not None:
                    destm = shard_of(gkeys, comm.world)
                    flat: dict[str, Column] = {}
                    for on_, cl_ in arg_cols.items():
                        if specs[on_][0].family == "multiset":
                            for i_, c_ in enumerate(cl_):
                                flat[f"a.{on_}.{i_}"] = c_
                    tensors_m = {"gkeys": gkeys, "rowkeys": rowkeys, "diffs": diffs}
                else:
                    destm = tensors_m = flat = None
                tensors_m, flat = exchange_bundle(
                    comm, destm, tensors_m, flat, meta_state=self._xmeta_ms
                )
                ms_exchanged = (tensors_m, flat)
            if tensors is None:
                return None  # no rank had data this step
            ukeys_t = tensors["k"]
            # re-consolidate: the same key may arrive from several ranks
            words2 = [ukeys_t[:, 0].contiguous(), ukeys_t[:, 1].contiguous()]
            perm2 = lex_sort_words(words2)
            sw2 = [w.index_select(0, perm2) for w in words2]
            starts2 = rows_ne(sw2)
            seg2 = torch.cumsum(starts2.to(torch.int64), 0) - 1
            first2 = starts2.nonzero(as_tuple=True)[0]
            nseg2 = int(first2.numel())
            acc_names = [k[4:] for k in tensors if k.startswith("acc.")]
            merged_accs = {}
            for name in acc_names:
                sc = tensors[f"acc.{name}"].index_select(0, perm2)
                acc = torch.zeros(nseg2, dtype=sc.dtype, device=device)
                acc.index_add_(0, seg2, sc)
                merged_accs[name] = acc
            acc_deltas = merged_accs
            ukeys_w = [w.index_select(0, first2) for w in sw2]
            gcols_first = {
                n: c.take(perm2.index_select(0, first2)) for n, c in gcols_first.items()
            }
            if has_multiset and ms_exchanged is not None and ms_exchanged[0] is not None:
                tensors_m, flat = ms_exchanged
                gkeys = tensors_m["gkeys"]
                rowkeys = tensors_m["rowkeys"]
                diffs = tensors_m["diffs"]
                for on_, cl_ in list(arg_cols.items()):
                    if specs[on_][0].family == "multiset":
                        ncols = sum(1 for k in flat if k.startswith(f"a.{on_}."))
                        arg_cols[on_] = [flat[f"a.{on_}.{i_}"] for i_ in range(ncols)]
        elif ukeys_w is None:
            return None

        # 4. affected keys + old output rows (pre-merge)
        changed = torch.stack(ukeys_w, dim=1)
        cw = ukeys_w
        old_presence, old_cols = self._current_rows(changed, cw, specs)

        # 5. merge states (deltas already consolidated per key)
        self._merge_state_pre(ukeys_w, acc_deltas, gcols_first)
        if has_multiset and gkeys is not None:
            self._merge_multiset(gkeys, arg_cols, rowkeys, diffs, specs)

        # 6. new output rows (post-merge)
        new_presence, new_cols = self._current_rows(changed, cw, specs)

        # 7. emit — sort-free consolidation: old/new rows are aligned on the
        # same (sorted, unique) changed keys, so cancellation is a direct
        # per-column equality compare instead of sort+vhash+segsum
        unchanged = old_presence & new_presence
        for name in old_cols:
            unchanged = unchanged & columns_equal_mask(
                old_cols[name], new_cols[name], device
            )
        out_batches = []
        idx = (old_presence & ~unchanged).nonzero(as_tuple=True)[0]
        if idx.numel():
            out_batches.append(
                DeltaBatch(
                    changed.index_select(0, idx),
                    {n: c.take(idx) for n, c in old_cols.items()},
                    torch.full((idx.shape[0],), -1, dtype=torch.int64, device=device),
                    time,
                )
            )
        idx = (new_presence & ~unchanged).nonzero(as_tuple=True)[0]
        if idx.numel():
            out_batches.append(
                DeltaBatch(
                    changed.index_select(0, idx),
                    {n: c.take(idx) for n, c in new_cols.items()},
                    torch.ones((idx.shape[0],), dtype=torch.int64, device=device),
                    time,
                )
            )
        if not out_batches:
            return None
        out = DeltaBatch.concat(out_batches)
        out.consolidated = True
        return out

    # -- state init --

    def _ensure_states(self, gcols: dict[str, Column], specs):
        pass  # state arrays initialize lazily in _merge_state_pre

    # -- additive state --

    def _contributions(self, arg_cols, diffs, specs) -> dict[str, torch.Tensor]:
        """Per-row additive accumulator contributions (incl. presence __w__)."""
        contribs: dict[str, torch.Tensor] = {"__w__": diffs}
        for out_name, (spec, args, kwargs) in specs.items():
            if spec.family != "additive":
                continue
            if spec.name == "count":
                pass  # aliases __w__
            elif spec.name == "sum":
                t = arg_cols[out_name][0]
                assert isinstance(t, TensorColumn), "sum needs a numeric column"
                v = t.tensor
                if v.dtype == torch.bool:
                    v = v.to(torch.int64)
                if t.mask is not None:
                    v = v * t.mask
                contribs[out_name] = v * diffs.to(v.dtype)
            elif spec.name == "avg":
                t = arg_cols[out_name][0]
                assert isinstance(t, TensorColumn)
                v = t.tensor.to(torch.float64)
                if t.mask is not None:
                    v = v * t.mask
                contribs[f"{out_name}__sum"] = v * diffs.to(torch.float64)
        return contribs

    def _merge_state_pre(self, ukeys_w, acc_deltas, gcols_first):
        """Merge consolidated (unique sorted keys, acc deltas, group values)
        into the combined state.  Group values are key-determined, so any
        row of an equal-key run is a valid carried representative."""
        device = self.device
        nseg = ukeys_w[0].shape[0]
        from pathway_amd.engine.state import rows_ne

        if self.add_keys is None:
            z = torch.zeros((0,), dtype=torch.int64, device=device)
            self.add_keys = [z, z.clone()]
            self.add_accs = {
                name: torch.zeros((0,), dtype=t.dtype, device=device)
                for name, t in acc_deltas.items()
            }
            idx0 = torch.zeros((0,), dtype=torch.int64)
            self.add_carried = {
                n: c.take(idx0.to(c._device())) for n, c in gcols_first.items()
            }
        all_words = [torch.cat([s, d]) for s, d in zip(self.add_keys, ukeys_w)]
        all_accs = {}
        for name in self.add_accs:
            d = acc_deltas.get(name)
            if d is None:
                d = torch.zeros(nseg, dtype=self.add_accs[name].dtype, device=device)
            st = self.add_accs[name]
            if st.dtype != d.dtype:
                st = st.to(d.dtype)
            all_accs[name] = torch.cat([st, d])
        all_carried = {
            n: concat_columns([self.add_carried[n], gcols_first[n]])
            for n in self.add_carried
        }
        if (
            torch.device(device).type == "cuda"
            and not _PW_NO_SEGRED
            and all(a.dtype == torch.int64 for a in all_accs.values())
            and 1 <= len(self.add_accs) <= 8
        ):
            # fused LSM merge+consolidate (one count+emit kernel pair):
            # unique-sorted state x unique-sorted delta, weight-0 rows
            # dropped, representative row indices for carried columns
            from pathway_amd import ops

            names2 = list(self.add_accs)
            # weight slot first (the kernel drops on acc[0] == 0)
            names2.sort(key=lambda nm: nm != "__w__")
            a_accs = [self.add_accs[nm].to(torch.int64) for nm in names2]
            b_accs = [
                acc_deltas.get(
                    nm, torch.zeros(nseg, dtype=torch.int64, device=device)
                )
                for nm in names2
            ]
            out_words, out_accs, rep = ops.merge_consolidate_gpu(
                self.add_keys, a_accs, ukeys_w, b_accs
            )
            self.add_keys = out_words
            self.add_accs = dict(zip(names2, out_accs))
            self.add_carried = {
                n: c.take(rep) for n, c in all_carried.items()
            }
            return
        # state and delta are both sorted: O(m+n) merge, no re-sort
        from pathway_amd.engine.state import merge_sorted_select

        perm2 = merge_sorted_select(self.add_keys, ukeys_w)
        all_words = [w.index_select(0, perm2) for w in all_words]
        if (
            torch.device(device).type == "cuda"
            and not _PW_NO_SEGRED
            and all(a.dtype == torch.int64 for a in all_accs.values())
        ):
            from pathway_amd import ops

            names2 = list(all_accs)
            sorted_accs = [
                all_accs[nm].index_select(0, perm2) for nm in names2
            ]
            out_words, first_idx, out_accs = ops.seg_reduce_words_gpu(
                all_words, sorted_accs
            )
            merged = dict(zip(names2, out_accs))
            keep = merged["__w__"] != 0
            kidx = keep.nonzero(as_tuple=True)[0]
            self.add_keys = [w.index_select(0, kidx) for w in out_words]
            self.add_accs = {
                name: acc.index_select(0, kidx) for name, acc in merged.items()
            }
            rep = perm2.index_select(0, first_idx).index_select(0, kidx)
            self.add_carried = {n: c.take(rep) for n, c in all_carried.items()}
            return
        starts2 = rows_ne(all_words)
        seg2 = torch.cumsum(starts2.to(torch.int64), 0) - 1
        first_idx = starts2.nonzero(as_tuple=True)[0]
        nseg2 = int(first_idx.numel())
        merged: dict[str, torch.Tensor] = {}
        for name, acc in all_accs.items():
            sa = acc.index_select(0, perm2)
            out = torch.zeros(nseg2, dtype=sa.dtype, device=device)
            out.index_add_(0, seg2, sa)
            merged[name] = out
        keep = merged["__w__"] != 0
        kidx = keep.nonzero(as_tuple=True)[0]
        self.add_keys = [
            w.index_select(0, first_idx).index_select(0, kidx) for w in all_words
        ]
        self.add_accs = {name: acc.index_select(0, kidx) for name, acc in merged.items()}
        rep = perm2.index_select(0, first_idx).index_select(0, kidx)
        self.add_carried = {n: c.take(rep) for n, c in all_carried.items()}

    def _merge_multiset(self, gkeys, arg_cols, rowkeys, diffs, specs):
        ms_cols: dict[str, Column] = {}
        for out_name, (spec, args, kwargs) in specs.items():
            if spec.family != "multiset":
                continue
            for i, c in enumerate(arg_cols[out_name]):
                ms_cols[f"{out_name}__{i}"] = c
        if not ms_cols:
            return
        # include row identity so distinct input rows stay distinct
        ms_cols["__rowkey__"] = PointerColumn(rowkeys)
        if self.multiset_store is None:
            self.multiset_store = Arrangement(self.device, ms_cols)
            self.multiset_colnames = list(ms_cols.keys())
        parts = [
            (lo.to(self.device), hi.to(self.device))
            for lo, hi in (c.value_hash() for c in ms_cols.values())
        ]
        v0, v1 = hashing.combine_value_hashes(parts)
        self.multiset_store.merge(gkeys, (v0, v1), diffs, ms_cols)

    # -- reading current rows for a set of keys --

    def _current_rows(self, changed: torch.Tensor, cw, specs):
        device = self.device
        nq = changed.shape[0]
        cols: dict[str, Column] = {}
        add_pos, add_found = self._additive_lookup(cw)
        w = self._gather_acc("__w__", add_pos, add_found)
        presence = w > 0
        # group columns carried in the combined state
        for name in self.group_exprs:
            c = self.add_carried.get(name)
            if c is None or len(c) == 0:
                cols[name] = column_from_pylist([None] * nq, dt.ANY, device)
            else:
                cols[name] = _mask_missing(c.take(add_pos), add_found, device)
        # additive reducer outputs
        w_cache: torch.Tensor | None = None

        def wacc():
            nonlocal w_cache
            if w_cache is None:
                w_cache = w
            return w_cache

        for out_name, (spec, args, kwargs) in specs.items():
            if spec.family == "additive":
                if spec.name == "avg":
                    sacc = self._gather_acc(f"{out_name}__sum", add_pos, add_found)
                    c = wacc()
                    vals = sacc.to(torch.float64) / c.clamp(min=1).to(torch.float64)
                    cols[out_name] = TensorColumn(vals, dt.FLOAT)
                elif spec.name == "count":
                    cols[out_name] = TensorColumn(wacc().to(torch.int64), dt.INT)
                else:
                    acc = self._gather_acc(out_name, add_pos, add_found)
                    odt = dt.FLOAT if acc.dtype == torch.float64 else dt.INT
                    cols[out_name] = TensorColumn(acc, odt)
            elif spec.family == "multiset":
                cols[out_name] = self._multiset_agg(out_name, spec, changed, nq, kwargs)
        return presence, cols


    def _gather_weights(self, store: Arrangement, pos, found):
        if len(store) == 0:
            return torch.zeros_like(pos)
        w = store.weights.index_select(0, pos.clamp(0, len(store) - 1))
        return w * found

    def _additive_lookup(self, cw):
        if self.add_keys is None or self.add_keys[0].shape[0] == 0:
            z = torch.zeros(cw[0].shape[0], dtype=torch.int64, device=self.device)
            return z, torch.zeros_like(z, dtype=torch.bool)
        if cw[0].is_cuda:
            from pathway_amd import ops

            return ops.lookup_gpu(self.add_keys, cw)
        pos = searchsorted_words(self.add_keys, cw, side="left")
        m = self.add_keys[0].shape[0]
        safe = pos.clamp(0, m - 1)
        found = (
            (pos < m)
            & (self.add_keys[0].index_select(0, safe) == cw[0])
            & (self.add_keys[1].index_select(0, safe) == cw[1])
        )
        return safe, found

    def _gather_acc(self, name, pos, found):
        acc = self.add_accs.get(name)
        if acc is None or acc.shape[0] == 0:
            return torch.zeros(pos.shape[0], dtype=torch.int64, device=self.device)
        v = acc.index_select(0, pos)
        return v * found

    def _multiset_agg(self, out_name, spec, changed, nq, rkw=None) -> Column:
        device = self.device
        store = self.multiset_store
        arg0 = f"{out_name}__0"
        if store is None or len(store) == 0:
            proto_dt = dt.ANY
            return column_from_pylist([None] * nq, proto_dt, device)
        lo, hi = store.key_range(changed)
        rows, qidx = store.gather_ranges(lo, hi)
        col0 = store.columns[arg0]
        # GPU fast path for min/max over numeric columns
        if (
            spec.segment_agg is not None
            and isinstance(col0, TensorColumn)
            and col0.mask is None
        ):
            vals = col0.tensor.index_select(0, rows)
            agg = spec.segment_agg(vals, qidx, nq)
            out = agg
            if col0.tensor.dtype == torch.int64:
                out = torch.where(
                    torch.isfinite(agg), agg, torch.zeros_like(agg)
                ).to(torch.int64)
                return TensorColumn(out, dt.INT)
            return TensorColumn(torch.where(torch.isfinite(agg), agg, torch.zeros_like(agg)), dt.FLOAT)
        # host path
        narg = spec.n_args if spec.n_args > 0 else 1
        arg_lists = []
        for i in range(narg):
            cname = f"{out_name}__{i}"
            if cname in store.columns:
                arg_lists.append(store.columns[cname].take(rows).to_pylist())
        weights = store.weights.index_select(0, rows).cpu().tolist()
        qidx_l = qidx.cpu().tolist()
        per_group: list[list] = [[] for _ in range(nq)]
        for j, q in enumerate(qidx_l):
            tup = tuple(al[j] for al in arg_lists)
            if len(tup) == 1:
                tup = (tup[0], None)
            per_group[q].append((tup, weights[j]))
        out_vals = []
        skip_nones = bool(rkw and rkw.get("skip_nones"))
        for rows_g in per_group:
            rows_g = [(t, w) for t, w in rows_g if w > 0]
            if skip_nones:
                rows_g = [(t, w) for t, w in rows_g if t[0] is not None]
            if not rows_g:
                out_vals.append(None)
                continue
            try:
                out_vals.append(spec.host_agg(rows_g))
            except ValueError:
                from pathway_amd.internals.api import ERROR

                out_vals.append(ERROR)
        col, _ = _build_from_values(out_vals, device)
        return col


def _build_from_values(vals, device):
    from pathway_amd.engine.column import infer_and_build_column

    return infer_and_build_column(vals, device)


def columns_equal_mask(a: Column, b: Column, device) -> torch.Tensor:
    """Per-row equality between two aligned columns (None == None)."""
    n = len(a)
    if isinstance(a, TensorColumn) and isinstance(b, TensorColumn):
        if a.tensor.dtype != b.tensor.dtype:
            eq = a.tensor.to(torch.float64) == b.tensor.to(torch.float64)
        else:
            eq = a.tensor == b.tensor
        ma = a.mask
        mb = b.mask
        if ma is None and mb is None:
            return eq
        ones = torch.ones(n, dtype=torch.bool, device=a.tensor.device)
        ma = ma if ma is not None else ones
        mb = mb if mb is not None else ones
        return (ma & mb & eq) | (~ma & ~mb)
    if isinstance(a, StringColumn) and isinstance(b, StringColumn) and a.pool is b.pool:
        return a.codes == b.codes
    if isinstance(a, PointerColumn) and isinstance(b, PointerColumn):
        return (a.pairs[:, 0] == b.pairs[:, 0]) & (a.pairs[:, 1] == b.pairs[:, 1])
    va, vb = a.to_pylist(), b.to_pylist()

    def _veq(x, y):
        if x is None or y is None:
            return x is None and y is None
        if isinstance(x, np.ndarray) or isinstance(y, np.ndarray):
            return (
                isinstance(x, np.ndarray)
                and isinstance(y, np.ndarray)
                and x.shape == y.shape
                and bool(np.array_equal(x, y))
            )
        r = x == y
        return bool(r) if isinstance(r, (bool, np.bool_)) else False

    return torch.tensor(
        [_veq(x, y) for x, y in zip(va, vb)],
        dtype=torch.bool,
        device=device,
    )


def _mask_missing(col: Column, found: torch.Tensor, device) -> Column:
    """Null out positions where found == False (no host sync on GPU)."""
    if not found.is_cuda and bool(found.all()):
        return col
    if isinstance(col, TensorColumn):
        mask = found & (col.mask if col.mask is not None else torch.ones_like(found))
        return TensorColumn(col.tensor, col.dtype, mask)
    if isinstance(col, StringColumn):
        codes = torch.where(found, col.codes, torch.full_like(col.codes, -1))
        return StringColumn(codes, col.pool, col.dtype)
    if isinstance(col, PointerColumn):
        return col  # pointer columns keep garbage at missing rows; filtered out
    vals = col.to_pylist()
    f = found.cpu().tolist()
    out = [v if ok else None for v, ok in zip(vals, f)]
    from pathway_amd.engine.column import obj_array
    return ObjectColumn(obj_array(out), col.dtype)


class ToStreamNode(Node):
    """table_to_stream (graph.rs): each update becomes an append-only event
    row keyed uniquely, with an is_upsert flag."""

    def __init__(self, input_node, device):
        super().__init__([input_node], device)
        self.salt = _salt("to_stream", self.node_id)
        self.seq = 0

    def reset(self):
        self.seq = 0

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0:
            return None
        n = len(b)
        seq = torch.arange(self.seq, self.seq + n, dtype=torch.int64, device=self.device)
        self.seq += n
        slo, shi = hashing.value_hash_words(seq, 2)
        lo, hi = hashing.derive_key_words(
            self.salt,
            [
                (b.keys[:, 0].contiguous(), b.keys[:, 1].contiguous()),
                (slo, shi),
            ],
        )
        cols = dict(b.columns)
        cols["is_upsert"] = TensorColumn(b.diffs > 0, dt.BOOL)
        cols["_pw_source_id"] = PointerColumn(b.keys)
        keys = torch.stack([lo, hi], dim=1)
        diffs = torch.ones(n, dtype=torch.int64, device=self.device)
        return DeltaBatch(keys, cols, diffs, time)


class StreamToTableNode(Node):
    """stream_to_table: interpret an event stream (with is_upsert) as
    upserts keyed by a source-id column."""

    def __init__(self, input_node, upsert_name: str, device):
        super().__init__([input_node], device)
        self.upsert_name = upsert_name
        self.state: dict | None = {}

    def reset(self):
        self.state = {}

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0:
            return None
        ups = b.columns[self.upsert_name]
        src = b.columns.get("_pw_source_id")
        if src is None:
            raise ValueError("stream_to_table needs a _pw_source_id column")
        # true upsert semantics (arrange_from_upsert, upsert.rs:346): a new
        # value for a key retracts the stored one; a deletion retracts the
        # STORED row even when the event carries different column values
        names = [
            n for n in b.columns if n not in (self.upsert_name, "_pw_source_id")
        ]
        if not hasattr(self, "state") or self.state is None:
            self.state = {}
        pairs = src.pairs.cpu().tolist()
        upl = [bool(v) for v in ups.to_pylist()]
        vals = {n: b.columns[n].to_pylist() for n in names}
        out_keys: list[tuple[int, int]] = []
        out_rows: list[tuple] = []
        out_diffs: list[int] = []
        for i, key in enumerate(map(tuple, pairs)):
            stored = self.state.get(key)
            if upl[i]:
                row = tuple(vals[n][i] for n in names)
                if stored is not None:
                    if stored == row:
                        continue
                    out_keys.append(key)
                    out_rows.append(stored)
                    out_diffs.append(-1)
                out_keys.append(key)
                out_rows.append(row)
                out_diffs.append(1)
                self.state[key] = row
            elif stored is not None:
                out_keys.append(key)
                out_rows.append(stored)
                out_diffs.append(-1)
                del self.state[key]
        if not out_keys:
            return None
        keys = torch.tensor(out_keys, dtype=torch.int64, device=self.device).reshape(
            len(out_keys), 2
        )
        diffs = torch.tensor(out_diffs, dtype=torch.int64, device=self.device)
        cols = {}
        for j, n in enumerate(names):
            proto = b.columns[n]
            cols[n] = column_from_pylist(
                [r[j] for r in out_rows], proto.dtype, self.device
            )
        return DeltaBatch(keys, cols, diffs, time)


class UnpackSnapshotsNode(Node):
    """unpack_snapshots (reference table.py:3056): whenever the input
    changes, append the FULL current table state as fresh rows keyed by
    (minibatch time, original row key).  The state lives as a sorted GPU
    arrangement; emission is one gather + one fused hash."""

    def __init__(self, input_node: Node, device):
        super().__init__([input_node], device)
        self.state: Arrangement | None = None

    def reset(self):
        self.state = None

    def step(self, time, inputs):
        b = consolidate_batch(inputs[0])
        if b is None or len(b) == 0:
            return None
        if self.state is None:
            self.state = Arrangement(self.device, dict(b.columns))
        v0, v1 = batch_vhash(b)
        self.state.merge(b.keys, (v0, v1), b.diffs, dict(b.columns))
        m = len(self.state)
        if m == 0:
            return None
        k0 = self.state.key_words[0]
        k1 = self.state.key_words[1]
        tconst = torch.full((m,), int(time), dtype=torch.int64, device=self.device)
        o0, o1 = hashing.hash128_words([k0, k1, tconst])
        keys = torch.stack([o0, o1], dim=1)
        cols = dict(self.state.columns)
        out = DeltaBatch(keys, cols, self.state.weights.clamp(min=1), time)
        out.consolidated = True
        return out


class FreezeAnswersNode(Node):
    """as-of-now answer freezing: the first emission per key wins; later
    positive updates for an answered key are dropped (retractions of the
    answered row pass through once)."""

    def __init__(self, input_node, device):
        super().__init__([input_node], device)
        self.answered: dict = {}

    def reset(self):
        self.answered = {}

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0:
            return None
        b = consolidate_batch(b)
        if b is None:
            return None
        keep = []
        keys = b.keys.cpu().tolist()
        diffs = b.diffs.cpu().tolist()
        for i in range(len(b)):
            k = tuple(keys[i])
            if diffs[i] > 0:
                if k not in self.answered:
                    self.answered[k] = True
                    keep.append(i)
            else:
                if self.answered.pop(k, None):
                    keep.append(i)
        if not keep:
            return None
        idx = torch.tensor(keep, dtype=torch.int64, device=self.device)
        return b.take(idx)
