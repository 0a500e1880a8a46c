"""Class-based AsyncTransformer (reference stdlib/utils/async_transformer.py
+ src/engine/dataflow/async_transformer.rs:297).

The reference feeds rows to a Python asyncio transformer on a separate
runtime and re-ingests completed results as a new input stream; rows
whose invoke() has not completed are `Pending`, failures land in
`.failed`.  This build routes invoke() through the engine's async UDF
executor (concurrent per batch, capacity/timeout/retry options — the
same machinery as @pw.udf(executor=async_executor)) and splits the
output into the reference's result views:

  .successful — rows whose invoke() completed without error
  .failed     — rows whose invoke() raised (values are Error)
  .finished   — all completed rows
  .output_table / .result — alias of .successful
"""

from __future__ import annotations

import asyncio
from typing import Any

from pathway_amd import udfs
from pathway_amd.internals import expression as ex
from pathway_amd.internals.schema import SchemaMetaclass


class AsyncTransformer:
    output_schema: SchemaMetaclass

    def __init__(self, input_table=None, instance=None, autocommit_duration_ms=None,
                 **kwargs):
        self._input_table = input_table
        self._instance = instance
        self._kwargs = kwargs
        self._executor_options: dict[str, Any] = {}
        self._cache_strategy = None
        self._result_cache: dict[str, Any] = {}
        self.open()

    async def invoke(self, *args, **kwargs) -> dict:
        raise NotImplementedError

    def open(self) -> None:
        """Called once before the first invoke (reference hook)."""

    def close(self) -> None:
        """Called when the transformer is dropped (reference hook)."""

    # -- options (reference with_options: capacity, timeout, retries) --

    def with_options(self, capacity: int | None = None,
                     timeout: float | None = None,
                     retry_strategy: Any = None,
                     cache_strategy: Any = None,
                     **kwargs: Any) -> "AsyncTransformer":
        if capacity is not None:
            self._executor_options["capacity"] = capacity
        if timeout is not None:
            self._executor_options["timeout"] = timeout
        if retry_strategy is not None:
            self._executor_options["retry_strategy"] = retry_strategy
        if cache_strategy is not None:
            self._cache_strategy = cache_strategy
        return self

    def with_instance(self, instance) -> "AsyncTransformer":
        self._instance = instance
        return self

    # -- result views --

    def _computed(self):
        cached = self._result_cache.get("computed")
        if cached is not None:
            return cached
        table = self._input_table
        out_names = self.output_schema.column_names()
        transformer = self

        async def fun_all(**row):
            res = await transformer.invoke(**row)
            return tuple(res[n] for n in out_names)

        wrapped = udfs.udf(
            fun_all,
            executor=udfs.async_executor(**self._executor_options),
            cache_strategy=self._cache_strategy,
        )
        cols = {n: ex.ColumnReference(table, n) for n in table._dtypes}
        tup = table.select(_pw_res=wrapped(**cols))
        out = tup.select(
            **{
                n: ex.DeclareTypeExpression(
                    tup._pw_res[i], self.output_schema.__columns__[n].dtype
                )
                for i, n in enumerate(out_names)
            }
        )
        self._result_cache["computed"] = out
        return out

    @property
    def finished(self):
        """All rows whose invoke() completed (ok or error)."""
        return self._computed()

    def _error_mask(self, out):
        """Boolean expression: True where invoke() raised.  An apply on
        an Error row yields Error; fill_error turns that into True."""
        from pathway_amd.internals.expression import FillErrorExpression

        first = self.output_schema.column_names()[0]
        probe = ex.ApplyExpression(
            lambda *a: False, None, ex.ColumnReference(out, first)
        )
        return FillErrorExpression(probe, True)

    @property
    def successful(self):
        """Rows whose invoke() completed without raising."""
        out = self._computed()
        return out.remove_errors()

    @property
    def failed(self):
        """Rows whose invoke() raised (reference .failed view)."""
        out = self._computed()
        return out.filter(self._error_mask(out))

    @property
    def result(self):
        return self.successful

    @property
    def output_table(self):
        return self.successful
