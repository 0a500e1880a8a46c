"""Text splitters (reference xpacks/llm/splitters.py:21-250)."""
from __future__ import annotations

from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.common import UDF


class BaseSplitter(UDF):
    def chunk(self, text: str, metadata: dict | None = None) -> list[tuple[str, dict]]:
        raise NotImplementedError

    def __wrapped__(self, text: str, metadata: Any = None, **kwargs):
        meta = metadata.value if hasattr(metadata, "value") else (metadata or {})
        return tuple(
            (c, _json(m)) for c, m in self.chunk(text or "", dict(meta))
        )


def _json(v):
    from pathway_amd.internals.json import Json

    return Json(v)


class NullSplitter(BaseSplitter):
    def chunk(self, text, metadata=None):
        return [(text, metadata or {})]


class TokenCountSplitter(BaseSplitter):
    """Split into chunks of min_tokens..max_tokens whitespace tokens
    (the reference counts tiktoken tokens; offline we count words)."""

    def __init__(self, min_tokens: int = 50, max_tokens: int = 500, encoding_name: str = "cl100k_base", **kwargs):
        super().__init__()
        self.min_tokens = min_tokens
        self.max_tokens = max_tokens

    def chunk(self, text, metadata=None):
        words = text.split()
        out = []
        i = 0
        while i < len(words):
            j = min(i + self.max_tokens, len(words))
            out.append((" ".join(words[i:j]), dict(metadata or {})))
            i = j
        return out or [("", dict(metadata or {}))]


class RecursiveSplitter(BaseSplitter):
    """Recursive character splitter (langchain-style, reference :150)."""

    def __init__(
        self,
        chunk_size: int = 500,
        chunk_overlap: int = 0,
        separators: list[str] | None = None,
        encoding_name: str = "cl100k_base",
        model_name: str | None = None,
        **kwargs,
    ):
        super().__init__()
        self.chunk_size = chunk_size
        self.chunk_overlap = chunk_overlap
        self.separators = separators or ["\n\n", "\n", ". ", " ", ""]

    def _split(self, text: str, seps: list[str]) -> list[str]:
        if len(text) <= self.chunk_size:
            return [text] if text else []
        if not seps:
            return [
                text[i : i + self.chunk_size]
                for i in range(0, len(text), max(self.chunk_size - self.chunk_overlap, 1))
            ]
        sep, rest = seps[0], seps[1:]
        if sep == "":
            return self._split(text, rest) if rest else self._split(text, [])
        parts = text.split(sep)
        out: list[str] = []
        cur = ""
        for p in parts:
            cand = (cur + sep + p) if cur else p
            if len(cand) <= self.chunk_size:
                cur = cand
            else:
                if cur:
                    out.append(cur)
                if len(p) > self.chunk_size:
                    out.extend(self._split(p, rest))
                    cur = ""
                else:
                    cur = p
        if cur:
            out.append(cur)
        return out

    def chunk(self, text, metadata=None):
        return [(c, dict(metadata or {})) for c in self._split(text, self.separators)] or [
            ("", dict(metadata or {}))
        ]
