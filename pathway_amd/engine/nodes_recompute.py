"""RecomputeNode: incremental-output wrapper around a batch host function.

For operators whose incremental form is genuinely order-dependent (asof
join, global sort prev/next pointers), the node keeps each input's current
state, recomputes the full output on change, and emits the DIFF against the
previous output — output streams stay perfectly incremental even though the
inside recomputes (the reference instead maintains bidirectional cursors,
pathway/trace.rs:17; a future round can specialize the hot ones).
"""

from __future__ import annotations

from typing import Any, Callable

import torch

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.api import BasePointer, Pointer
from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.column import infer_and_build_column, column_from_pylist
from pathway_amd.engine.nodes import Node, consolidate_batch
from pathway_amd.engine.nodes_join import _SideStore


def _hashable(v):
    """Stable hashable identity for a row value (ndarrays by bytes)."""
    import numpy as np

    if isinstance(v, np.ndarray):
        return ("__nd__", v.shape, v.dtype.str, v.tobytes())
    if isinstance(v, (list, dict, set)):
        return repr(v)
    try:
        hash(v)
        return v
    except TypeError:
        return repr(v)


class RecomputeNode(Node):
    """fn(inputs: list[list[dict]], keys: list[list[Pointer]]) ->
    list[(Pointer, dict values)] — full output; node emits deltas."""

    def __init__(
        self,
        input_nodes,
        fn: Callable,
        out_columns: list[str],
        out_dtypes: dict[str, dt.DType] | None,
        device,
    ):
        super().__init__(input_nodes, device)
        self.fn = fn
        self.out_columns = out_columns
        self.out_dtypes = out_dtypes or {}
        self.stores = [_SideStore(device) for _ in input_nodes]
        self.prev_output: list[tuple[Any, tuple]] = []  # (key, values tuple)

    def reset(self):
        self.stores = [_SideStore(self.device) for _ in self.inputs]
        self.prev_output = []

    def step(self, time, inputs):
        changed = False
        for st, b in zip(self.stores, inputs):
            if b is not None and len(b):
                st.merge(b.keys, b)
                changed = True
        if not changed:
            return None
        # materialize inputs on host (summing weights across spine levels)
        in_rows = []
        in_keys = []
        for st in self.stores:
            rows = []
            keys = []
            if len(st):
                from collections import defaultdict

                acc: dict = defaultdict(int)
                payload: dict = {}
                for lvl in st.spine.levels:
                    if len(lvl) == 0:
                        continue
                    cols = {n: c.to_pylist() for n, c in lvl.columns.items()}
                    ids = cols.pop("__id__")
                    w = lvl.weights.cpu().tolist()
                    names = list(cols.keys())
                    for i in range(len(ids)):
                        row = tuple((n, _hashable(cols[n][i])) for n in names)
                        k = (repr(ids[i]), row)
                        acc[k] += w[i]
                        payload[k] = (ids[i], {n: cols[n][i] for n in names})
                for k, wsum in acc.items():
                    if wsum <= 0:
                        continue
                    key_obj, row = payload[k]
                    for _ in range(wsum):
                        rows.append(row)
                        keys.append(key_obj)
            in_rows.append(rows)
            in_keys.append(keys)
        new_output = self.fn(in_rows, in_keys)
        new_norm = [
            (key, tuple(vals[n] for n in self.out_columns)) for key, vals in new_output
        ]
        # diff vs previous (identities must be hashable: ndarray values
        # are keyed by their bytes)
        from collections import Counter

        def _ident(k, v):
            return (repr(k), tuple(_hashable(x) for x in v))

        old_c = Counter(_ident(k, v) for k, v in self.prev_output)
        new_c = Counter(_ident(k, v) for k, v in new_norm)
        key_by_repr = {repr(k): k for k, _ in self.prev_output}
        key_by_repr.update({repr(k): k for k, _ in new_norm})
        val_by_ident = {_ident(k, v): v for k, v in self.prev_output}
        val_by_ident.update({_ident(k, v): v for k, v in new_norm})
        out_keys = []
        out_vals = []
        out_diffs = []
        for item in set(old_c) | set(new_c):
            d = new_c.get(item, 0) - old_c.get(item, 0)
            if d != 0:
                krepr, _vid = item
                out_keys.append(key_by_repr[krepr])
                out_vals.append(val_by_ident[item])
                out_diffs.append(d)
        self.prev_output = new_norm
        if not out_keys:
            return None
        keys_t = torch.tensor(
            [list(k.as_signed_pair()) for k in out_keys],
            dtype=torch.int64,
            device=self.device,
        ).reshape(len(out_keys), 2)
        diffs_t = torch.tensor(out_diffs, dtype=torch.int64, device=self.device)
        cols = {}
        for j, n in enumerate(self.out_columns):
            vlist = [v[j] for v in out_vals]
            if n in self.out_dtypes:
                cols[n] = column_from_pylist(vlist, self.out_dtypes[n], self.device)
            else:
                cols[n], _ = infer_and_build_column(vlist, self.device)
        return consolidate_batch(DeltaBatch(keys_t, cols, diffs_t, time))
