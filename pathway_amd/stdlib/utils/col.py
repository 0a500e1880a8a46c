"""Column helpers (reference stdlib/utils/col.py)."""
from __future__ import annotations

from typing import Any


def flatten_column(column, origin_id: str = "origin_id"):
    table = column.table
    return table.flatten(column)


def unpack_col(column, *unpacked_columns: Any, schema=None):
    """Unpack a tuple column into named columns."""
    table = column.table
    names = []
    for c in unpacked_columns:
        names.append(c if isinstance(c, str) else c.name)
    if schema is not None:
        names = schema.column_names()
    kwargs = {n: column[i] for i, n in enumerate(names)}
    return table.select(**kwargs)


def multiapply_all_rows(*cols, fun, result_col_names):
    raise NotImplementedError


def apply_all_rows(*cols, fun, result_col_name):
    raise NotImplementedError


def groupby_reduce_majority(column_group, column_val):
    tab = column_group.table.groupby(column_group, column_val).reduce(
        column_group, column_val
    )
    raise NotImplementedError
