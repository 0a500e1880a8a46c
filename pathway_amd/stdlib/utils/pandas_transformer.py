"""@pw.pandas_transformer (reference stdlib/utils/pandas_transformer.py)."""
from __future__ import annotations

from typing import Any, Callable


def pandas_transformer(output_schema, output_universe: Any = None):
    def decorator(fun: Callable):
        def wrapper(*tables):
            from pathway_amd.debug import table_from_pandas, table_to_pandas

            dfs = [table_to_pandas(t) for t in tables]
            out = fun(*dfs)
            return table_from_pandas(out, schema=output_schema)

        return wrapper

    return decorator
