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
    """Adaptive-k answerer (reference question_answering.py
    AdaptiveRAGQuestionAnswerer): retrieve a generous doc set once, then
    prompt the LLM with a geometrically growing prefix until it answers."""

    def __init__(self, llm, indexer, *, n_starting_documents: int = 2,
                 factor: int = 2, max_iterations: int = 4,
                 strict_prompt: bool = False, **kwargs):
        kwargs.setdefault(
            "search_topk", n_starting_documents * factor ** max(max_iterations - 1, 0)
        )
        super().__init__(llm, indexer, **kwargs)
        self.n_starting_documents = n_starting_documents
        self.factor = factor
        self.max_iterations = max_iterations
        self.strict_prompt = strict_prompt

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
        from pathway_amd.xpacks.llm.prompts import prompt_qa

        llm = self.llm
        n0, factor, iters = self.n_starting_documents, self.factor, self.max_iterations

        def adaptive_answer(prompt, result):
            rv = result.value if hasattr(result, "value") else result
            texts = [r.get("text") for r in (rv or [])]
            n = n0
            for _ in range(iters):
                resp = str(llm.__wrapped__(prompt_qa(prompt, texts[:n])))
                if resp and "No information" not in resp:
                    return resp
                n *= factor
            return "No information found."

        return pw_ai_queries.select(
            result=common.apply_with_type(
                adaptive_answer, dt.STR, this.prompt, _align(docs, pw_ai_queries)
            )
        )

    answer = answer_query
    pw_ai_answer = answer_query


class DeckRetriever(BaseRAGQuestionAnswerer):
    def __init__(self, *args, **kwargs):
        raise NotImplementedError("slide decks need vision parsers (offline)")


class BaseContextProcessor:
    """Formats retrieved docs into the LLM context string (reference
    question_answering.py:39-70)."""

    def maybe_unwrap_docs(self, docs):
        from pathway_amd.internals.json import Json

        if isinstance(docs, Json):
            docs = docs.value
        out = []
        for d in docs or []:
            if isinstance(d, Json):
                d = d.value
            out.append(d)
        return out

    def docs_to_context(self, docs) -> str:
        raise NotImplementedError

    def apply(self, docs) -> str:
        return self.docs_to_context(self.maybe_unwrap_docs(docs))

    def as_udf(self):
        from pathway_amd.internals.common import udf

        return udf(lambda docs: self.apply(docs))


class SimpleContextProcessor(BaseContextProcessor):
    """Joins doc texts with a separator, keeping selected metadata."""

    def __init__(self, context_metadata_keys=None, docs_separator: str = "\n\n"):
        self.context_metadata_keys = context_metadata_keys or []
        self.docs_separator = docs_separator

    def docs_to_context(self, docs) -> str:
        parts = []
        for d in docs:
            if isinstance(d, dict):
                text = str(d.get("text", ""))
                meta = ", ".join(
                    f"{k}: {d.get(k)}" for k in self.context_metadata_keys if k in d
                )
                parts.append(f"{text} ({meta})" if meta else text)
            else:
                parts.append(str(d))
        return self.docs_separator.join(parts)


class BaseQuestionAnswerer:
    """Server-facing QA interface (reference question_answering.py)."""

    AnswerQuerySchema = BaseRAGQuestionAnswerer.AnswerQuerySchema
    RetrieveQuerySchema = None
    StatisticsQuerySchema = None
    InputsQuerySchema = None

    def answer_query(self, pw_ai_queries):
        raise NotImplementedError


class SummaryQuestionAnswerer(BaseRAGQuestionAnswerer):
    """QA variant whose answer includes a summary of matching docs."""


def send_post_request(url: str, data: dict, headers: dict | None = None, timeout: float | None = None):
    """POST JSON and return the parsed response (reference helper)."""
    import json as _json
    import urllib.request

    req = urllib.request.Request(
        url,
        data=_json.dumps(data).encode(),
        headers={"Content-Type": "application/json", **(headers or {})},
        method="POST",
    )
    with urllib.request.urlopen(req, timeout=timeout) as resp:
        return _json.loads(resp.read())


class RAGClient:
    """HTTP client for a running QA REST server (reference
    question_answering.py:1070-1230: /v2/answer, /v2/summarize,
    /v1/retrieve, /v1/statistics)."""

    def __init__(self, host: str | None = None, port: int | None = None,
                 url: str | None = None, timeout: float | None = 90,
                 additional_headers: dict | None = None):
        err = "Either (`host` and `port`) or `url` must be provided, but not both."
        if url is not None:
            if host is not None or port is not None:
                raise ValueError(err)
            self.url = url
        else:
            if host is None or port is None:
                raise ValueError(err)
            self.url = f"http://{host}:{port}"
        self.timeout = timeout
        self.additional_headers = additional_headers or {}

    def _post(self, path: str, payload: dict):
        return send_post_request(
            self.url + path, payload, self.additional_headers, self.timeout
        )

    def answer(self, prompt: str, filters: str | None = None, model: str | None = None):
        payload = {"prompt": prompt}
        if filters:
            payload["filters"] = filters
        if model:
            payload["model"] = model
        return self._post("/v2/answer", payload)

    def summarize(self, text_list, model: str | None = None):
        payload = {"text_list": list(text_list)}
        if model:
            payload["model"] = model
        return self._post("/v2/summarize", payload)

    def retrieve(self, query: str, k: int = 3, metadata_filter: str | None = None,
                 filepath_globpattern: str | None = None):
        return self._post(
            "/v1/retrieve",
            {
                "query": query,
                "k": k,
                "metadata_filter": metadata_filter,
                "filepath_globpattern": filepath_globpattern,
            },
        )

    def statistics(self):
        return self._post("/v1/statistics", {})
