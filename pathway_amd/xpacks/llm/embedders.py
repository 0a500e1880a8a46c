"""Embedders (reference xpacks/llm/embedders.py:77-870 API surface).

SentenceTransformerEmbedder runs the MI355X-native encoder forward
(_encoder.py) — bf16 on gfx950, batch 1024 like the reference's default.
Provider-backed embedders (OpenAI/LiteLLM/Gemini/...) keep the reference
API but require network credentials at call time.
"""

from __future__ import annotations

from typing import Any

import numpy as np

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.common import UDF
from pathway_amd.internals.expression import ApplyExpression, ColumnExpression, wrap_expr


class BaseEmbedder(UDF):
    def get_embedding_dimension(self, **kwargs) -> int:
        probe = self._embed_many(["."])[0]
        return int(len(probe))

    def _embed_many(self, texts: list[str], **kwargs) -> list[np.ndarray]:
        raise NotImplementedError

    def __wrapped__(self, text: str, **kwargs) -> np.ndarray:
        return self._embed_many([text])[0]

    def __call__(self, *args: Any, **kwargs: Any) -> ColumnExpression:
        expr = ApplyExpression(
            self.__wrapped__, dt.Array(1, dt.FLOAT), *args, **kwargs
        )
        embed_many = self._embed_many

        def batch_fun(texts, **kw):
            return embed_many([t if t is not None else "" for t in texts])

        expr._batch_fun = batch_fun
        return expr


class SentenceTransformerEmbedder(BaseEmbedder):
    """MI355X-native encoder embedder (reference embedders.py:454: wraps the
    sentence_transformers lib; here the forward is our own bf16 module)."""

    def __init__(
        self,
        model: str = "pathway-native/bge-small-like",
        call_kwargs: dict = {},
        device: str | None = None,
        batch_size: int = 1024,
        **init_kwargs: Any,
    ):
        super().__init__()
        from pathway_amd.xpacks.llm._encoder import get_encoder

        self.model = model
        self.batch_size = batch_size
        self._encoder = get_encoder(device=device)

    def _embed_many(self, texts: list[str], **kwargs) -> list[np.ndarray]:
        return self._encoder.encode(texts, batch_size=self.batch_size)


NativeEncoderEmbedder = SentenceTransformerEmbedder


class _NetworkEmbedder(BaseEmbedder):
    provider = "generic"

    def __init__(self, model: str | None = None, *, capacity: int | None = None,
                 retry_strategy: Any = None, cache_strategy: Any = None, **kwargs: Any):
        super().__init__(cache_strategy=cache_strategy)
        self.model = model
        self.kwargs = kwargs

    def _embed_many(self, texts: list[str], **kwargs) -> list[np.ndarray]:
        raise RuntimeError(
            f"{type(self).__name__} needs network access to the {self.provider} "
            "API, which is unavailable in this environment; use "
            "SentenceTransformerEmbedder (local MI355X-native encoder) instead"
        )


class OpenAIEmbedder(_NetworkEmbedder):
    """OpenAI /embeddings protocol (reference embedders.py:77) — a real
    client for any OpenAI-compatible endpoint (pass base_url for vLLM /
    TEI in openai mode); verified against the fake HTTP service."""

    provider = "OpenAI"

    def __init__(self, model: str | None = "text-embedding-3-small", *,
                 api_key: str | None = None, base_url: str | None = None,
                 **kwargs):
        super().__init__(model, **kwargs)
        self.api_key = api_key
        self.base_url = (base_url or "https://api.openai.com/v1").rstrip("/")

    def _embed_many(self, texts: list[str], **kwargs) -> list[np.ndarray]:
        from pathway_amd.io import _rest

        headers = {}
        if self.api_key:
            headers["Authorization"] = f"Bearer {self.api_key}"
        out = _rest.request(
            "POST", f"{self.base_url}/embeddings",
            body={"model": self.model, "input": texts, **kwargs},
            headers=headers,
        )
        data = sorted(out["data"], key=lambda d: d.get("index", 0))
        return [np.asarray(d["embedding"], dtype=np.float32) for d in data]


class LiteLLMEmbedder(OpenAIEmbedder):
    """LiteLLM proxy speaks the OpenAI embeddings protocol."""

    provider = "LiteLLM"


class GeminiEmbedder(_NetworkEmbedder):
    provider = "Gemini"


class BedrockEmbedder(_NetworkEmbedder):
    provider = "AWS Bedrock"


class MarengoEmbedder(_NetworkEmbedder):
    provider = "TwelveLabs Marengo"


def contextful(*args, **kwargs):
    raise NotImplementedError("reference-deprecated API")
