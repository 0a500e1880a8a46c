"""LLM xpack (reference xpacks/llm) — lands with the index/RAG phase."""
