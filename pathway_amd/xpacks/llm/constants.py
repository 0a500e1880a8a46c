"""Shared llm xpack constants (reference xpacks/llm/constants.py)."""

#: default capacity for async LLM/embedder calls
DEFAULT_ASYNC_CAPACITY = 10
