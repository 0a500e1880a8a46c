"""RAG prompt templates (reference xpacks/llm/prompts.py)."""
from __future__ import annotations


def prompt_qa(query: str, docs, information_not_found_response: str = "No information found.") -> str:
    context = "\n\n".join(
        d if isinstance(d, str) else str(d) for d in (docs or [])
    )
    return (
        "Answer the question based only on the context.\n"
        f"Context:\n{context}\n\nQuestion: {query}\n"
        f"If the context lacks the answer, reply: {information_not_found_response}"
    )


def prompt_short_qa(query: str, docs, **kwargs) -> str:
    return prompt_qa(query, docs, **kwargs)


def prompt_citing_qa(query: str, docs, **kwargs) -> str:
    return prompt_qa(query, docs, **kwargs)


def prompt_summarize(text_list) -> str:
    joined = "\n".join(text_list or [])
    return f"Summarize the following texts:\n{joined}"


def prompt_query_rewrite(query: str, **kwargs) -> str:
    return f"Rewrite the search query to be more specific: {query}"
