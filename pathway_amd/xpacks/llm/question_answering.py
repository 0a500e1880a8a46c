"""RAG question answering (reference xpacks/llm/question_answering.py:184-460)."""
from __future__ import annotations

from typing import Any

import pathway_amd.internals.common as common
import pathway_amd.reducers as reducers
from pathway_amd.internals import dtype as dt
from pathway_amd.internals import thisclass
from pathway_amd.internals.json import Json
from pathway_amd.internals.schema import Schema

this = thisclass.this


def answer_with_geometric_rag_strategy(
    questions, documents, llm, n_starting_documents: int = 2, factor: int = 2,
    max_iterations: int = 4, strict_prompt: bool = False,
):
    """Adaptive-k strategy (reference :184): try n docs, double until the
    LLM answers (synchronous engine: evaluated per batch)."""
    from pathway_amd.xpacks.llm.prompts import prompt_qa

    def answer(question, docs):
        n = n_starting_documents
        for _ in range(max_iterations):
            subset = list(docs or [])[:n]
            prompt = prompt_qa(question, subset)
            resp = llm.__wrapped__(prompt)
            if resp and "No information" not in str(resp):
                return str(resp)
            n *= factor
        return "No information found."

    return questions.select(
        result=common.apply_with_type(answer, dt.STR, this.prompt
                                      if "prompt" in questions._dtypes else this.query,
                                      this.docs)
    )


answer_with_geometric_rag_strategy_from_index = answer_with_geometric_rag_strategy


class BaseRAGQuestionAnswerer:
    """DocumentStore + LLM answerer (reference :250)."""

    class AnswerQuerySchema(Schema):
        prompt: str
        filters: str | None
        model: str | None
        return_context_docs: bool | None

    class SummarizeQuerySchema(Schema):
        text_list: Any

    def __init__(self, llm, indexer, *, default_llm_name: str | None = None,
                 prompt_template: Any = None, search_topk: int = 6, **kwargs):
        self.llm = llm
        self.indexer = indexer  # DocumentStore or VectorStoreServer
        self.search_topk = search_topk
        from pathway_amd.xpacks.llm.prompts import prompt_qa

        self.prompt_template = prompt_template or prompt_qa

    def _store(self):
        return getattr(self.indexer, "document_store", self.indexer)

    def answer_query(self, pw_ai_queries):
        store = self._store()
        queries = pw_ai_queries.select(
            query=this.prompt,
            k=self.search_topk,
            metadata_filter=this.filters
            if "filters" in pw_ai_queries._dtypes
            else None,
            filepath_globpattern=None,
        )
        docs = store.retrieve_query(queries)
        llm = self.llm
        template = self.prompt_template

        def make_answer(prompt, result):
            rv = result.value if hasattr(result, "value") else result
            texts = [r.get("text") for r in (rv or [])]
            return str(llm.__wrapped__(template(prompt, texts)))

        answered = pw_ai_queries.select(
            result=common.apply_with_type(
                make_answer, dt.STR, this.prompt,
                _align(docs, pw_ai_queries),
            )
        )
        return answered

    answer = answer_query
    pw_ai_answer = answer_query

    def summarize_query(self, summarize_queries):
        from pathway_amd.xpacks.llm.prompts import prompt_summarize

        llm = self.llm

        def do(text_list):
            return str(llm.__wrapped__(prompt_summarize(list(text_list or []))))

        return summarize_queries.select(
            result=common.apply_with_type(do, dt.STR, this.text_list)
        )

    pw_summarize = summarize_query

    def build_server(self, host: str, port: int, **kwargs):
        from pathway_amd.xpacks.llm.servers import QARestServer

        self._server = QARestServer(host, port, self)
        return self._server

    def run_server(self, host: str = "127.0.0.1", port: int = 8000, threaded: bool = False, **kwargs):
        self.build_server(host, port)
        return self._server.run(threaded=threaded, **kwargs)


def _align(docs_table, query_table):
    from pathway_amd.internals import expression as ex

    return ex.ColumnReference(docs_table.with_universe_of(query_table), "result")


class AdaptiveRAGQuestionAnswerer(BaseRAGQuestionAnswerer):
    pass


class DeckRetriever(BaseRAGQuestionAnswerer):
    def __init__(self, *args, **kwargs):
        raise NotImplementedError("slide decks need vision parsers (offline)")
