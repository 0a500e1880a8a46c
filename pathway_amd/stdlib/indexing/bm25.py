"""BM25 full-text index (reference stdlib/indexing/bm25.py:41 TantivyBM25 +
src/external_integration/tantivy_integration.rs parity).

In-memory inverted index with Okapi BM25 scoring; same as-of-now semantics
and filter support as the KNN retrievers (served by IndexPort/
ExternalIndexNode machinery on the host — postings math is control-plane
sized next to the GEMM path; a GPU postings kernel is a later-round item).
"""

from __future__ import annotations

import math
import re
from collections import Counter, defaultdict
from dataclasses import dataclass
from typing import Any

import numpy as np
import torch

from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as ex
from pathway_amd.internals.api import BasePointer
from pathway_amd.internals.config import get_device


_TOKEN_RE = re.compile(r"[A-Za-z0-9_]+")


def _tokenize(text: str) -> list[str]:
    return [t.lower() for t in _TOKEN_RE.findall(text or "")]


class _BM25State:
    """Okapi BM25 over compiled CSR postings.

    Mutations (add/remove) update the source-of-truth doc maps and mark
    the compiled arrays dirty; the first search after a change rebuilds
    the CSR (vocab ids, per-term [doc_idx, tf] slabs, doc lengths) with
    numpy, and query scoring is fully vectorized (np.add.at scatter +
    argpartition top-k) — the tantivy-scale formulation of the round-1
    dict-loop."""

    def __init__(self, k1: float = 1.2, b: float = 0.75):
        self.k1 = k1
        self.b = b
        self.docs: dict[tuple[int, int], Counter] = {}
        self.doc_len: dict[tuple[int, int], int] = {}
        self.payload: dict[tuple[int, int], Any] = {}
        self._compiled = None  # (keys, doc_len_arr, vocab, offsets, pidx, ptf)

    def add(self, key, text: str, payload=None):
        toks = Counter(_tokenize(text))
        self.docs[key] = toks
        self.doc_len[key] = sum(toks.values())
        if payload is not None:
            self.payload[key] = payload
        self._compiled = None

    def remove(self, key):
        self.docs.pop(key, None)
        self.doc_len.pop(key, None)
        self.payload.pop(key, None)
        self._compiled = None

    def _compile(self):
        if self._compiled is not None:
            return self._compiled
        keys = list(self.docs.keys())
        n = len(keys)
        doc_len_arr = np.array(
            [self.doc_len[k] for k in keys], dtype=np.float64
        )
        vocab: dict[str, int] = {}
        term_ids_per_doc = []
        tfs_per_doc = []
        for k in keys:
            c = self.docs[k]
            tids = np.empty(len(c), dtype=np.int64)
            tfs = np.empty(len(c), dtype=np.int64)
            for j, (t, f) in enumerate(c.items()):
                tid = vocab.get(t)
                if tid is None:
                    tid = vocab[t] = len(vocab)
                tids[j] = tid
                tfs[j] = f
            term_ids_per_doc.append(tids)
            tfs_per_doc.append(tfs)
        if n:
            all_tids = np.concatenate(term_ids_per_doc)
            all_tfs = np.concatenate(tfs_per_doc)
            all_docs = np.repeat(
                np.arange(n, dtype=np.int64),
                [len(a) for a in term_ids_per_doc],
            )
            order = np.argsort(all_tids, kind="stable")
            s_tids = all_tids[order]
            pidx = all_docs[order]
            ptf = all_tfs[order].astype(np.float64)
            counts = np.bincount(s_tids, minlength=len(vocab))
            offsets = np.zeros(len(vocab) + 1, dtype=np.int64)
            np.cumsum(counts, out=offsets[1:])
        else:
            pidx = np.zeros(0, dtype=np.int64)
            ptf = np.zeros(0, dtype=np.float64)
            offsets = np.zeros(1, dtype=np.int64)
        self._compiled = (keys, doc_len_arr, vocab, offsets, pidx, ptf)
        return self._compiled

    def search(self, query: str, k: int, filter_spec=None):
        from pathway_amd.engine.nodes_index import _apply_filter

        n = len(self.docs)
        if n == 0:
            return []
        keys, doc_len_arr, vocab, offsets, pidx, ptf = self._compile()
        avgdl = float(doc_len_arr.mean()) if n else 1.0
        scores = np.zeros(n, dtype=np.float64)
        norm = self.k1 * (1 - self.b + self.b * doc_len_arr / max(avgdl, 1e-9))
        for t in set(_tokenize(query)):
            tid = vocab.get(t)
            if tid is None:
                continue
            s, e = offsets[tid], offsets[tid + 1]
            df = e - s
            idf = math.log(1 + (n - df + 0.5) / (df + 0.5))
            d = pidx[s:e]
            f = ptf[s:e]
            contrib = idf * (f * (self.k1 + 1)) / (f + norm[d])
            np.add.at(scores, d, contrib)
        nz = np.nonzero(scores > 0)[0]
        if nz.size == 0:
            return []
        # rank: top-k among scored docs (argpartition then sort)
        take = min(max(k * 4, k), nz.size) if filter_spec is not None else min(k, nz.size)
        part = nz[np.argpartition(-scores[nz], take - 1)[:take]]
        ranked = part[np.argsort(-scores[part], kind="stable")]
        out = []
        for di in ranked:
            key = keys[di]
            if filter_spec is not None:
                try:
                    if not _apply_filter(filter_spec, self.payload.get(key)):
                        continue
                except Exception:
                    continue
            out.append((key, float(scores[di])))
            if len(out) >= k:
                return out
        if filter_spec is not None and len(out) < k:
            # filtered short: fall back to the full ranked list
            full = nz[np.argsort(-scores[nz], kind="stable")]
            for di in full[take:]:
                key = keys[di]
                try:
                    if not _apply_filter(filter_spec, self.payload.get(key)):
                        continue
                except Exception:
                    continue
                out.append((key, float(scores[di])))
                if len(out) >= k:
                    break
        return out


class TantivyBM25:
    """InnerIndex retriever over the text column (reference bm25.py:41)."""

    def __init__(
        self,
        data_column: ex.ColumnReference,
        metadata_column: ex.ColumnReference | None = None,
        *,
        ram_budget: int = 50_000_000,
        in_memory_index: bool = True,
        k1: float = 1.2,
        b: float = 0.75,
    ):
        self.data_column = data_column
        self.metadata_column = metadata_column
        self.k1 = k1
        self.b = b
        self.embedder = None  # API parity with KNN retrievers

    def query_as_of_now(self, query_column, number_of_matches: int = 3, metadata_filter=None):
        from pathway_amd.engine.nodes_index import ExternalIndexNode
        from pathway_amd.internals.table import Table

        data_table = self.data_column.table
        query_table = query_column.table
        index_src = data_table
        text_name = self.data_column.name
        filter_col = None
        if self.metadata_column is not None:
            index_src = index_src.with_columns(_pw_meta=self.metadata_column)
            filter_col = "_pw_meta"
        node = _BM25IndexNode(
            index_src._node,
            query_table._node,
            text_name,
            query_column,
            number_of_matches,
            get_device(),
            filter_data_col=filter_col,
            query_filter_expr=metadata_filter,
            k1=self.k1,
            b=self.b,
        )
        dtypes = {
            "_pw_index_reply_ids": dt.List(dt.POINTER),
            "_pw_index_reply_scores": dt.List(dt.FLOAT),
        }
        return Table(node, dtypes, query_table._universe)

    query = query_as_of_now


@dataclass
class TantivyBM25Factory:
    ram_budget: int = 50_000_000
    in_memory_index: bool = True

    def build_index(self, data_column, metadata_column=None, **kwargs) -> TantivyBM25:
        return TantivyBM25(data_column, metadata_column)


from pathway_amd.engine.nodes import Node as _Node


class _BM25IndexNode(_Node):
    """Engine node sharing ExternalIndexNode's protocol, BM25 scoring."""

    def __init__(
        self,
        index_node,
        query_node,
        text_col: str,
        query_expr,
        k: int,
        device,
        filter_data_col=None,
        query_filter_expr=None,
        k1: float = 1.2,
        b: float = 0.75,
    ):
        super().__init__([index_node, query_node], device)
        self.text_col = text_col
        self.query_expr = query_expr
        self.k = int(k) if not isinstance(k, ex.ColumnExpression) else 16
        self.k_expr = k if isinstance(k, ex.ColumnExpression) else None
        self.filter_data_col = filter_data_col
        self.query_filter_expr = query_filter_expr
        self.state = _BM25State(k1, b)
        self.answers: dict = {}

    def reset(self):
        self.state = _BM25State(self.state.k1, self.state.b)
        self.answers = {}

    def step(self, time, inputs):
        from pathway_amd.engine.batch import DeltaBatch
        from pathway_amd.engine.column import obj_array, ObjectColumn
        from pathway_amd.engine.expression_eval import EvalContext, evaluate

        bi, bq = inputs
        if bi is not None and len(bi):
            texts = bi.columns[self.text_col].to_pylist()
            payloads = (
                bi.columns[self.filter_data_col].to_pylist()
                if self.filter_data_col and self.filter_data_col in bi.columns
                else [None] * len(bi)
            )
            diffs = bi.diffs.cpu().tolist()
            keys = bi.keys.cpu().tolist()
            for i in range(len(bi)):
                key = tuple(keys[i])
                if diffs[i] > 0:
                    self.state.add(key, texts[i], payloads[i])
                else:
                    self.state.remove(key)
        if bq is None or len(bq) == 0:
            return None
        ctx = EvalContext(bq.columns, bq.keys, self.device)
        queries = evaluate(self.query_expr, ctx).to_pylist()
        ks = None
        if self.k_expr is not None:
            ks = [int(v) for v in evaluate(self.k_expr, ctx).to_pylist()]
        filts = None
        if self.query_filter_expr is not None:
            filts = evaluate(self.query_filter_expr, ctx).to_pylist()
        qkeys = bq.keys.cpu().tolist()
        diffs = bq.diffs.cpu().tolist()
        out_rows = []
        for i in range(len(bq)):
            key = tuple(qkeys[i])
            if diffs[i] > 0:
                klim = ks[i] if ks else self.k
                hits = self.state.search(
                    queries[i] or "", klim, filts[i] if filts else None
                )
                ids = tuple(
                    BasePointer.from_signed_pair(h[0][0], h[0][1]) for h in hits
                )
                scores = tuple(float(h[1]) for h in hits)
                vals = (ids, scores)
                self.answers[key] = vals
                out_rows.append((qkeys[i], vals, diffs[i]))
            else:
                vals = self.answers.pop(key, ((), ()))
                out_rows.append((qkeys[i], vals, diffs[i]))
        keys_t = torch.tensor(
            [r[0] for r in out_rows], dtype=torch.int64, device=self.device
        ).reshape(len(out_rows), 2)
        diffs_t = torch.tensor(
            [r[2] for r in out_rows], dtype=torch.int64, device=self.device
        )
        return DeltaBatch(
            keys_t,
            {
                "_pw_index_reply_ids": ObjectColumn(
                    obj_array([r[1][0] for r in out_rows]), dt.List(dt.POINTER)
                ),
                "_pw_index_reply_scores": ObjectColumn(
                    obj_array([r[1][1] for r in out_rows]), dt.List(dt.FLOAT)
                ),
            },
            diffs_t,
            time,
        )
