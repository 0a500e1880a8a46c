"""External index node: use_external_index_as_of_now
(reference operators/external_index.rs:36-140 + brute_force_knn_integration.rs).

MI355X-native design: the index side is a GPU-resident (m, d) matrix kept
in HBM3E (broadcast to every worker in multi-GPU mode — queries stay
worker-local, like the reference's index-stream broadcast); each query
batch is one GEMM (queries × indexᵀ, hipBLASLt via torch on ROCm; bf16
MFMA path for large indexes) + top-k.  as-of-now semantics: answers are
frozen at query time; query retractions retract the stored answer.
"""

from __future__ import annotations

from typing import Any

import numpy as np
import torch

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.api import BasePointer
from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.column import ObjectColumn, TensorColumn, obj_array as _obj_array
from pathway_amd.engine.expression_eval import EvalContext, evaluate
from pathway_amd.engine.nodes import Node, consolidate_batch


def _as_matrix(col, device, dim: int | None = None) -> torch.Tensor:
    """Column of vectors (tuple / np.ndarray / tensor rows) → (n, d) f32."""
    if isinstance(col, TensorColumn) and col.tensor.dim() == 2:
        return col.tensor.to(device=device, dtype=torch.float32)
    vals = col.to_pylist()
    if not vals:
        return torch.zeros((0, dim or 0), dtype=torch.float32, device=device)
    arrs = [np.asarray(v, dtype=np.float32) for v in vals]
    return torch.from_numpy(np.stack(arrs)).to(device)


from pathway_amd.engine.ann import FlatIndexState, IvfFlatState, LshState

#: backwards-compatible name: the flat index with device tombstones
VectorIndexState = FlatIndexState

INDEX_KINDS = {
    "flat": FlatIndexState,
    "ivf": IvfFlatState,
    "lsh": LshState,
}


class ExternalIndexNode(Node):
    """inputs: [index_table, query_table].  Output universe = query rows;
    columns: _pw_index_reply_ids (tuple[Pointer]), _pw_index_reply_scores
    (tuple[float])."""

    def __init__(
        self,
        index_node: Node,
        query_node: Node,
        index_vec_col: str,
        query_vec_expr: Any,
        k: int,
        device,
        metric: str = "cos",
        filter_data_col: str | None = None,
        query_filter_expr: Any | None = None,
        query_k_expr: Any | None = None,
        index_kind: str = "flat",
        index_params: dict | None = None,
    ):
        super().__init__([index_node, query_node], device)
        self.index_vec_col = index_vec_col
        self.query_vec_expr = query_vec_expr
        self.k = k
        self.metric = metric
        self.filter_data_col = filter_data_col
        self.query_filter_expr = query_filter_expr
        self.query_k_expr = query_k_expr
        self.index_kind = index_kind
        self.index_params = index_params or {}
        self.state = INDEX_KINDS[index_kind](device, metric, **self.index_params)
        self.answers: dict[tuple[int, int], tuple] = {}  # query key -> values

    def reset(self):
        self.state = INDEX_KINDS[self.index_kind](
            self.device, self.metric, **self.index_params
        )
        self.answers = {}

    def step(self, time, inputs):
        bi, bq = inputs
        device = self.device
        # 1. apply index updates first (queries at time t see index at t)
        if bi is not None and len(bi):
            vec_col = bi.columns[self.index_vec_col]
            vecs = _as_matrix(vec_col, device)
            payloads = None
            if self.filter_data_col and self.filter_data_col in bi.columns:
                payloads = bi.columns[self.filter_data_col].to_pylist()
            self.state.update(bi.keys, vecs, bi.diffs, payloads)
        if bq is None or len(bq) == 0:
            return None
        # 2. answer queries
        ctx = EvalContext(bq.columns, bq.keys, device)
        qcol = evaluate(self.query_vec_expr, ctx)
        q = _as_matrix(qcol, device)
        ks = None
        if self.query_k_expr is not None:
            kcol = evaluate(self.query_k_expr, ctx)
            ks = [int(v) for v in kcol.to_pylist()]
        filt_fns = None
        if self.query_filter_expr is not None:
            fcol = evaluate(self.query_filter_expr, ctx)
            filt_fns = fcol.to_pylist()
        kmax = max(ks) if ks else self.k
        ids, scores, valid = self.state.search(q, kmax, None)
        ids_l = ids.cpu().tolist()
        scores_l = scores.cpu().tolist()
        diffs = bq.diffs.cpu().tolist()
        qkeys = bq.keys.cpu().tolist()
        out_rows = []
        payload = self.state.payload
        for i in range(len(bq)):
            key = tuple(qkeys[i])
            if diffs[i] > 0:
                klim = ks[i] if ks else self.k
                row_ids = []
                row_scores = []
                for j in range(len(ids_l[i])):
                    if len(row_ids) >= klim:
                        break
                    p = BasePointer.from_signed_pair(ids_l[i][j][0], ids_l[i][j][1])
                    if filt_fns is not None and filt_fns[i] is not None:
                        data = payload.get(tuple(ids_l[i][j]))
                        try:
                            if not _apply_filter(filt_fns[i], data):
                                continue
                        except Exception:
                            continue
                    row_ids.append(p)
                    row_scores.append(float(scores_l[i][j]))
                vals = (tuple(row_ids), tuple(row_scores))
                self.answers[key] = vals
                out_rows.append((qkeys[i], vals, diffs[i]))
            else:
                vals = self.answers.pop(key, ((), ()))
                out_rows.append((qkeys[i], vals, diffs[i]))
        if not out_rows:
            return None
        keys_t = torch.tensor(
            [r[0] for r in out_rows], dtype=torch.int64, device=device
        ).reshape(len(out_rows), 2)
        diffs_t = torch.tensor([r[2] for r in out_rows], dtype=torch.int64, device=device)
        ids_col = ObjectColumn(
            _obj_array([r[1][0] for r in out_rows]), dt.List(dt.POINTER)
        )
        sc_col = ObjectColumn(
            _obj_array([r[1][1] for r in out_rows]), dt.List(dt.FLOAT)
        )
        return DeltaBatch(
            keys_t,
            {"_pw_index_reply_ids": ids_col, "_pw_index_reply_scores": sc_col},
            diffs_t,
            time,
        )


def _apply_filter(filter_spec, data) -> bool:
    """JMESPath-style filter (reference external_integration/mod.rs:41-49).

    Accepts a callable, or a glob/equality dict {"field": value} against the
    payload dict."""
    if callable(filter_spec):
        return bool(filter_spec(data))
    if isinstance(filter_spec, dict) and isinstance(data, dict):
        import fnmatch

        for k, v in filter_spec.items():
            dv = data.get(k)
            if isinstance(v, str) and any(ch in v for ch in "*?["):
                if not (isinstance(dv, str) and fnmatch.fnmatch(dv, v)):
                    return False
            elif dv != v:
                return False
        return True
    if isinstance(filter_spec, str):
        from pathway_amd.stdlib.indexing.filters import eval_jmespath_filter

        return eval_jmespath_filter(filter_spec, data)
    return True
