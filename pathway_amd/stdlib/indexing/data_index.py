"""DataIndex (reference stdlib/indexing/data_index.py:278) — plumbing only
for round 1; query paths land with the index phase."""
from __future__ import annotations

from typing import Any


class DataIndex:
    def __init__(self, data_table, inner_index, embedder=None):
        self.data_table = data_table
        self.inner = inner_index
        self.embedder = embedder

    def query(self, query_column, *, number_of_matches: int = 3, collapse_rows: bool = True, **kwargs):
        raise NotImplementedError("index query lands with the index phase")

    def query_as_of_now(self, query_column, *, number_of_matches: int = 3, **kwargs):
        return self.inner.query(
            self.data_table, query_column, number_of_matches
        )
