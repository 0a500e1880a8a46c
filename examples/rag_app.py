"""Live RAG server: documents folder -> DocumentStore -> REST endpoints.

POST /v1/retrieve {"query": "...", "k": 3}
POST /v1/pw_ai_answer {"prompt": "..."}
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pathway_amd as pw
from pathway_amd.xpacks.llm.document_store import DocumentStore
from pathway_amd.xpacks.llm.embedders import SentenceTransformerEmbedder
from pathway_amd.xpacks.llm.llms import EchoChat
from pathway_amd.xpacks.llm.question_answering import BaseRAGQuestionAnswerer
from pathway_amd.xpacks.llm.splitters import TokenCountSplitter
from pathway_amd.stdlib.indexing.nearest_neighbors import BruteForceKnnFactory

docs = pw.io.fs.read("./documents", format="binary", with_metadata=True, mode="static")
store = DocumentStore(
    docs,
    retriever_factory=BruteForceKnnFactory(embedder=SentenceTransformerEmbedder()),
    splitter=TokenCountSplitter(max_tokens=300),
)
qa = BaseRAGQuestionAnswerer(EchoChat(), store, search_topk=4)
qa.run_server(host="0.0.0.0", port=8000)
