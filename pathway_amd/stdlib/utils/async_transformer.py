"""Class-based AsyncTransformer (reference stdlib/utils/async_transformer.py).

Synchronous engine: invoke() coroutines are gathered per batch (the
reference feeds an asyncio loop on a separate thread,
src/engine/dataflow/async_transformer.rs:297).
"""
from __future__ import annotations

import asyncio

from pathway_amd.internals import expression as ex
from pathway_amd.internals.schema import SchemaMetaclass


class AsyncTransformer:
    output_schema: SchemaMetaclass

    def __init__(self, input_table=None, instance=None, **kwargs):
        self._input_table = input_table
        self._kwargs = kwargs

    async def invoke(self, *args, **kwargs) -> dict:
        raise NotImplementedError

    def open(self) -> None:
        pass

    def close(self) -> None:
        pass

    @property
    def successful(self):
        return self.result

    @property
    def result(self):
        table = self._input_table
        out_names = self.output_schema.column_names()
        transformer = self

        def make_fun(name):
            def fun(**row):
                async def run():
                    return await transformer.invoke(**row)

                res = asyncio.get_event_loop_policy().new_event_loop().run_until_complete(run())
                return res[name]

            return fun

        # evaluate invoke once per row for all outputs via tuple apply
        def fun_all(**row):
            loop = asyncio.new_event_loop()
            try:
                res = loop.run_until_complete(transformer.invoke(**row))
            finally:
                loop.close()
            return tuple(res[n] for n in out_names)

        cols = {n: ex.ColumnReference(table, n) for n in table._dtypes}
        tup = table.select(
            _pw_res=ex.ApplyExpression(fun_all, None, **cols)
        )
        out = tup.select(
            **{
                n: ex.DeclareTypeExpression(
                    tup._pw_res[i], self.output_schema.__columns__[n].dtype
                )
                for i, n in enumerate(out_names)
            }
        )
        return out

    @property
    def output_table(self):
        return self.result

    def with_options(self, **kwargs) -> "AsyncTransformer":
        return self

    def with_instance(self, instance) -> "AsyncTransformer":
        return self
