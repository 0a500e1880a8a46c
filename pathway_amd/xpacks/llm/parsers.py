"""Document parsers (reference xpacks/llm/parsers.py:56-1200 surface)."""
from __future__ import annotations

from typing import Any

from pathway_amd.internals.common import UDF
from pathway_amd.internals.json import Json


class Utf8Parser(UDF):
    """bytes -> [(text, metadata)] (reference Utf8Parser)."""

    async def parse(self, contents: bytes) -> list[tuple[str, dict]]:
        return [(contents.decode("utf-8", errors="replace"), {})]

    def __wrapped__(self, contents: bytes, **kwargs):
        text = contents.decode("utf-8", errors="replace") if isinstance(contents, (bytes, bytearray)) else str(contents)
        return ((text, Json({})),)


ParseUtf8 = Utf8Parser


class _HeavyParser(UDF):
    dependency = "an external parsing service/library"

    def __wrapped__(self, contents: bytes, **kwargs):
        raise RuntimeError(
            f"{type(self).__name__} requires {self.dependency}, unavailable "
            "in this offline environment; Utf8Parser works locally"
        )


class UnstructuredParser(_HeavyParser):
    dependency = "the unstructured library"

    def __init__(self, mode: str = "single", post_processors=None, **kwargs):
        super().__init__()


class DoclingParser(_HeavyParser):
    dependency = "docling"


class PypdfParser(_HeavyParser):
    dependency = "pypdf"


class PaddleParser(_HeavyParser):
    dependency = "paddleocr"


class ImageParser(_HeavyParser):
    dependency = "a vision LLM"


class SlideParser(_HeavyParser):
    dependency = "a vision LLM"


def default_vision_llm():
    """Default vision-capable chat for image/slide parsing (reference
    parsers.py default_vision_llm) — network-backed models are unavailable
    offline; the echo chat stands in so pipelines still construct."""
    from pathway_amd.xpacks.llm.llms import EchoChat

    return EchoChat()


#: legacy alias (reference parsers.py ParseUnstructured deprecation)
ParseUnstructured = UnstructuredParser

#: legacy alias kept for API parity
PaddleOCRParser = PaddleParser


class AudioParser(UDF):
    """Audio transcription parser (reference parsers.py AudioParser) —
    requires a speech-to-text backend; unavailable in the offline image."""

    def __init__(self, *args: Any, **kwargs: Any):
        super().__init__()

    def __wrapped__(self, contents: bytes, **kwargs: Any):
        raise NotImplementedError(
            "AudioParser needs a speech-to-text backend (no network in this environment)"
        )


class TwelveLabsVideoParser(UDF):
    """Video parser backed by the TwelveLabs API (reference parsers.py) —
    network service, unavailable offline."""

    def __init__(self, *args: Any, **kwargs: Any):
        super().__init__()

    def __wrapped__(self, contents: bytes, **kwargs: Any):
        raise NotImplementedError(
            "TwelveLabsVideoParser needs the TwelveLabs API (no network in this environment)"
        )
