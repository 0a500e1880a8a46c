"""LLM xpack (reference xpacks/llm): embedders, llms, parsers, splitters,
rerankers, DocumentStore, VectorStore, RAG question answering, servers."""
from pathway_amd.xpacks.llm import (
    embedders,
    llms,
    parsers,
    prompts,
    question_answering,
    rerankers,
    splitters,
)
from pathway_amd.xpacks.llm.document_store import DocumentStore
from pathway_amd.xpacks.llm.vector_store import VectorStoreClient, VectorStoreServer

__all__ = [
    "embedders",
    "llms",
    "parsers",
    "prompts",
    "question_answering",
    "rerankers",
    "splitters",
    "DocumentStore",
    "VectorStoreServer",
    "VectorStoreClient",
]

from pathway_amd.xpacks.llm import constants, mcp_server, servers, utils  # noqa: E402
