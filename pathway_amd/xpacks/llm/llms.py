"""LLM chat wrappers (reference xpacks/llm/llms.py:43-1059 surface)."""
from __future__ import annotations

from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.common import UDF


def prompt_chat_single_qa(question: str):
    from pathway_amd.internals.expression import MakeTupleExpression
    import json

    return question


class BaseChat(UDF):
    pass


class _NetworkChat(BaseChat):
    provider = "generic"

    def __init__(self, model: str | None = None, *, capacity=None, retry_strategy=None,
                 cache_strategy=None, temperature=None, max_tokens=None, **kwargs):
        super().__init__(cache_strategy=cache_strategy)
        self.model = model
        self.kwargs = kwargs

    def __wrapped__(self, messages: Any, **kwargs) -> str:
        raise RuntimeError(
            f"{type(self).__name__} needs network access to the {self.provider} API, "
            "unavailable in this environment; use EchoChat / your own local model"
        )


class OpenAIChat(_NetworkChat):
    provider = "OpenAI"


class LiteLLMChat(_NetworkChat):
    provider = "LiteLLM"


class CohereChat(_NetworkChat):
    provider = "Cohere"


class BedrockChat(_NetworkChat):
    provider = "AWS Bedrock"


class HFPipelineChat(BaseChat):
    """transformers-pipeline chat (reference :600) — transformers is
    importable offline but has no weights; loading a local path works."""

    def __init__(self, model: str | None = None, call_kwargs: dict = {}, device: str = "gpu", **kwargs):
        super().__init__()
        self.model = model
        self.call_kwargs = call_kwargs
        self._pipeline = None

    def _load(self):
        if self._pipeline is None:
            import transformers

            self._pipeline = transformers.pipeline(
                "text-generation", model=self.model
            )
        return self._pipeline

    def __wrapped__(self, messages: Any, **kwargs) -> str:
        pipe = self._load()
        out = pipe(messages, **{**self.call_kwargs, **kwargs})
        return out[0]["generated_text"] if isinstance(out, list) else str(out)

    def crop_to_max_length(self, input_string: str, max_prompt_length: int = 500) -> str:
        words = input_string.split()
        return " ".join(words[-max_prompt_length:])


class EchoChat(BaseChat):
    """Offline deterministic chat for tests/benchmarks: echoes the last
    user message with a fixed prefix."""

    def __init__(self, prefix: str = "ECHO: ", **kwargs):
        super().__init__()
        self.prefix = prefix

    def __wrapped__(self, messages: Any, **kwargs) -> str:
        if isinstance(messages, str):
            return self.prefix + messages
        try:
            last = messages[-1]
            content = last.get("content") if isinstance(last, dict) else str(last)
        except Exception:
            content = str(messages)
        return self.prefix + str(content)
