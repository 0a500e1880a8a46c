"""Rerankers (reference xpacks/llm/rerankers.py:17-330)."""
from __future__ import annotations

from typing import Any

import numpy as np

import pathway_amd.internals.common as common
from pathway_amd.internals import dtype as dt
from pathway_amd.internals.common import UDF
from pathway_amd.internals.expression import ApplyExpression


def rerank_topk_filter(docs: Any, scores: Any, k: int = 5):
    """Filter (docs, scores) tuples to the top-k by score."""
    def fil(docs_v, scores_v):
        pairs = sorted(zip(docs_v, scores_v), key=lambda p: -p[1])[:k]
        return tuple(d for d, _ in pairs), tuple(s for _, s in pairs)

    return common.apply_with_type(fil, dt.ANY_TUPLE, docs, scores)


class EncoderReranker(UDF):
    """Embedding-similarity reranker (reference :220): score = cos of the
    MI355X-native encoder embeddings."""

    def __init__(self, embedder=None, **kwargs):
        super().__init__()
        if embedder is None:
            from pathway_amd.xpacks.llm.embedders import SentenceTransformerEmbedder

            embedder = SentenceTransformerEmbedder()
        self.embedder = embedder

    def __wrapped__(self, doc: str, query: str, **kwargs) -> float:
        vecs = self.embedder._embed_many([doc or "", query or ""])
        a, b = np.asarray(vecs[0]), np.asarray(vecs[1])
        return float(np.dot(a, b) / ((np.linalg.norm(a) * np.linalg.norm(b)) or 1.0))

    def __call__(self, doc, query, **kwargs):
        expr = ApplyExpression(self.__wrapped__, dt.FLOAT, doc, query, **kwargs)
        embed_many = self.embedder._embed_many

        def batch_fun(docs, queries, **kw):
            n = len(docs)
            vecs = embed_many([d or "" for d in docs] + [q or "" for q in queries])
            out = []
            for i in range(n):
                a, b = np.asarray(vecs[i]), np.asarray(vecs[n + i])
                out.append(float(np.dot(a, b)))
            return out

        expr._batch_fun = batch_fun
        return expr


CrossEncoderReranker = EncoderReranker  # cross-encoder forward lands later


class LLMReranker(UDF):
    def __init__(self, llm, **kwargs):
        super().__init__()
        self.llm = llm

    def __wrapped__(self, doc: str, query: str, **kwargs) -> float:
        resp = self.llm.__wrapped__(
            f"Rate 1-5 relevance of doc to query.\ndoc: {doc}\nquery: {query}"
        )
        import re

        m = re.search(r"[1-5]", str(resp))
        return float(m.group(0)) if m else 1.0


class FlashRankReranker(UDF):
    def __init__(self, model: str = "ms-marco-TinyBERT-L-2-v2", **kwargs):
        super().__init__()

    def __wrapped__(self, doc: str, query: str, **kwargs) -> float:
        raise RuntimeError("flashrank unavailable offline; use EncoderReranker")
