"""DataIndex (reference stdlib/indexing/data_index.py:206-560).

Composes an inner retriever (BruteForceKnn / USearchKnn / TantivyBM25) with
the data table: query results come back as per-query tuples of the data
table's columns (collapse_rows=True) or as one row per match.
"""

from __future__ import annotations

from typing import Any

import pathway_amd.reducers as reducers
from pathway_amd.internals import expression as ex
from pathway_amd.internals import thisclass

this = thisclass.this


class DataIndex:
    def __init__(self, data_table, inner_index, embedder: Any = None):
        self.data_table = data_table
        self.inner = inner_index
        if embedder is not None and getattr(self.inner, "embedder", None) is None:
            self.inner.embedder = embedder

    def _reply(self, query_column, number_of_matches, metadata_filter):
        return self.inner.query_as_of_now(
            query_column,
            number_of_matches=number_of_matches,
            metadata_filter=metadata_filter,
        )

    def query_as_of_now(
        self,
        query_column: ex.ColumnReference,
        *,
        number_of_matches: Any = 3,
        collapse_rows: bool = True,
        metadata_filter: Any = None,
        with_distances: bool = True,
    ):
        reply = self._reply(query_column, number_of_matches, metadata_filter)
        data = self.data_table
        # one row per (query, match): flatten ids, keep rank order
        flat = reply.select(
            _pw_qid=this.id,
            _pw_match=this._pw_index_reply_ids,
            _pw_scores=this._pw_index_reply_scores,
        ).flatten(this._pw_match)
        # align score with its match via positional get over the tuple
        import pathway_amd.internals.common as common

        # fetch data rows by pointer
        docs = data.ix(flat._pw_match, optional=True)
        # docs has flat's universe; combine
        combined = flat.with_columns(
            **{n: ex.ColumnReference(docs, n) for n in data._dtypes}
        )
        if not collapse_rows:
            return combined
        gb = combined.groupby(this._pw_qid)
        red_kwargs = {
            n: reducers.tuple(ex.ColumnReference(combined, n)) for n in data._dtypes
        }
        if with_distances:
            red_kwargs["_pw_index_reply_score"] = reducers.tuple(
                ex.ColumnReference(combined, "_pw_scores")
            )
        collapsed = gb.reduce(_pw_qid=this._pw_qid, **red_kwargs)
        keyed = collapsed.with_id_from_expr(
            ex.ColumnReference(collapsed, "_pw_qid")
        ).without("_pw_qid")
        return keyed.with_universe_of(reply)

    def query(self, query_column, *, number_of_matches: Any = 3, collapse_rows: bool = True, metadata_filter: Any = None, with_distances: bool = True):
        return self.query_as_of_now(
            query_column,
            number_of_matches=number_of_matches,
            collapse_rows=collapse_rows,
            metadata_filter=metadata_filter,
            with_distances=with_distances,
        )
