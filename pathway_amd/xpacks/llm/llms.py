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
        self.temperature = temperature
        self.max_tokens = max_tokens
        self.kwargs = kwargs

    def __wrapped__(self, messages: Any, **kwargs) -> str:
        raise RuntimeError(
            f"{type(self).__name__} needs network access to the {self.provider} API, "
            "unavailable in this environment; use EchoChat / your own local model"
        )


class OpenAIChat(_NetworkChat):
    """OpenAI chat-completions protocol (reference llms.py:43 OpenAIChat).

    A real client for any OpenAI-compatible endpoint (OpenAI, vLLM,
    llama.cpp server, TGI in openai mode): pass ``base_url`` to point it
    at the server.  The request/response shapes are the standard
    ``/chat/completions`` JSON — exercised against the fake HTTP service
    in tests (no network in this environment, so the public api.openai.com
    default is only reachable in deployments with egress)."""

    provider = "OpenAI"

    def __init__(self, model: str | None = None, *, api_key: str | None = None,
                 base_url: str | None = None, **kwargs):
        super().__init__(model, **kwargs)
        self.api_key = api_key
        self.base_url = (base_url or "https://api.openai.com/v1").rstrip("/")

    def __wrapped__(self, messages: Any, **kwargs) -> str:
        from pathway_amd.io import _rest

        if isinstance(messages, str):
            messages = [{"role": "user", "content": messages}]
        elif isinstance(messages, tuple):
            messages = [
                dict(m) if isinstance(m, dict) else {"role": "user", "content": str(m)}
                for m in messages
            ]
        body: dict[str, Any] = {"model": self.model, "messages": messages}
        if self.temperature is not None:
            body["temperature"] = self.temperature
        if self.max_tokens is not None:
            body["max_tokens"] = self.max_tokens
        body.update(kwargs)
        headers = {}
        if self.api_key:
            headers["Authorization"] = f"Bearer {self.api_key}"
        out = _rest.request(
            "POST", f"{self.base_url}/chat/completions", body=body,
            headers=headers,
        )
        return out["choices"][0]["message"]["content"]


class LiteLLMChat(OpenAIChat):
    """LiteLLM proxy speaks the OpenAI protocol; same client with the
    proxy's base_url (reference llms.py LiteLLMChat)."""

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
