"""Deduplicate node (reference stateful_reduce.rs:20 / stdlib.stateful).

Per instance keeps the last ACCEPTED value row; a new candidate row replaces
it iff acceptor(new_value, old_value) returns truthy (default: value
changed).  Host-side state — deduplicate is a control-plane op in practice.
"""

from __future__ import annotations

from typing import Any, Callable

import torch

from pathway_amd.internals.api import hash_values, BasePointer, Pointer
from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.column import column_from_pylist, infer_and_build_column
from pathway_amd.engine.expression_eval import EvalContext, evaluate
from pathway_amd.engine.nodes import Node, consolidate_batch


class DeduplicateNode(Node):
    def __init__(self, input_node, value_expr, instance_expr, acceptor, col_names, device):
        super().__init__([input_node], device)
        self.value_expr = value_expr
        self.instance_expr = instance_expr
        self.acceptor = acceptor
        self.col_names = col_names
        self.state: dict[Any, tuple[Any, Any, list]] = {}  # inst -> (key, value, row)

    def reset(self) -> None:
        self.state = {}
        self._xmeta = {}

    def wants_frontier(self) -> bool:
        from pathway_amd.parallel import get_comm

        c = get_comm()
        return c is not None and c.world > 1

    def step(self, time, inputs):
        b = inputs[0]
        from pathway_amd.parallel import get_comm

        comm = get_comm()
        if comm is not None and comm.world > 1:
            # sequential accept/reject per instance: co-locate instances
            from pathway_amd.engine.nodes_join import _exchange_side
            from pathway_amd.engine import hashing

            jk = None
            if b is not None and len(b):
                ctx0 = EvalContext(b.columns, b.keys, self.device)
                if self.instance_expr is not None:
                    c0 = evaluate(self.instance_expr, ctx0)
                    i0, i1 = c0.value_hash()
                else:
                    i0 = torch.zeros(len(b), dtype=torch.int64, device=self.device)
                    i1 = i0.clone()
                jk = torch.stack([i0.to(self.device), i1.to(self.device)], dim=1)
            if not hasattr(self, "_xmeta"):
                self._xmeta = {}
            b, _ = _exchange_side(comm, b, jk, time, self._xmeta)
        if b is None or len(b) == 0:
            return None
        ctx = EvalContext(b.columns, b.keys, self.device)
        vals = evaluate(self.value_expr, ctx).to_pylist()
        if self.instance_expr is not None:
            insts = evaluate(self.instance_expr, ctx).to_pylist()
        else:
            insts = [None] * len(b)
        names = list(b.columns.keys())
        cols = {n: c.to_pylist() for n, c in b.columns.items()}
        diffs = b.diffs.cpu().tolist()
        out_rows = []  # (key, row values, diff)
        for i in range(len(b)):
            if diffs[i] <= 0:
                continue  # deduplicate consumes insertions only
            inst = insts[i]
            new_val = vals[i]
            row = [cols[n][i] for n in names]
            prev = self.state.get(inst)
            if prev is None:
                accept = True
            elif self.acceptor is not None:
                try:
                    accept = bool(self.acceptor(new_val, prev[1]))
                except Exception:
                    accept = False
            else:
                accept = new_val != prev[1]
            if not accept:
                continue
            lo, hi = hash_values([inst, "__dedup__", self.node_id])
            key = Pointer(lo, hi)
            if prev is not None:
                out_rows.append((key, prev[2], -1))
            self.state[inst] = (key, new_val, row)
            out_rows.append((key, row, 1))
        if not out_rows:
            return None
        keys = torch.tensor(
            [list(k.as_signed_pair()) for k, _, _ in out_rows],
            dtype=torch.int64,
            device=self.device,
        ).reshape(len(out_rows), 2)
        dts = torch.tensor([d for _, _, d in out_rows], dtype=torch.int64, device=self.device)
        out_cols = {}
        for j, n in enumerate(names):
            vlist = [r[j] for _, r, _ in out_rows]
            col, _ = infer_and_build_column(vlist, self.device)
            out_cols[n] = col
        return consolidate_batch(DeltaBatch(keys, out_cols, dts, time))
