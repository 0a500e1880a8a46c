"""pw.reducers — user-facing reducer factory functions
(reference python/pathway/reducers.py surface)."""

from __future__ import annotations

from typing import Any

from pathway_amd.internals.expression import ReducerExpression


def count(*args: Any) -> ReducerExpression:
    return ReducerExpression("count", *args[:0])


def sum(expr: Any) -> ReducerExpression:  # noqa: A001
    return ReducerExpression("sum", expr)


def avg(expr: Any) -> ReducerExpression:
    return ReducerExpression("avg", expr)


def min(expr: Any) -> ReducerExpression:  # noqa: A001
    return ReducerExpression("min", expr)


def max(expr: Any) -> ReducerExpression:  # noqa: A001
    return ReducerExpression("max", expr)


def argmin(expr: Any, id_expr: Any = None) -> ReducerExpression:
    if id_expr is not None:
        return ReducerExpression("argmin", expr, id_expr)
    return ReducerExpression("argmin", expr)


def argmax(expr: Any, id_expr: Any = None) -> ReducerExpression:
    if id_expr is not None:
        return ReducerExpression("argmax", expr, id_expr)
    return ReducerExpression("argmax", expr)


def unique(expr: Any) -> ReducerExpression:
    return ReducerExpression("unique", expr)


def any(expr: Any) -> ReducerExpression:  # noqa: A001
    return ReducerExpression("any", expr)


def sorted_tuple(expr: Any, *, skip_nones: bool = False) -> ReducerExpression:
    return ReducerExpression("sorted_tuple", expr, skip_nones=skip_nones)


def tuple(expr: Any, *, skip_nones: bool = False) -> ReducerExpression:  # noqa: A001
    return ReducerExpression("tuple", expr, skip_nones=skip_nones)


def ndarray(expr: Any, *, skip_nones: bool = False) -> ReducerExpression:
    return ReducerExpression("ndarray", expr, skip_nones=skip_nones)


def count_distinct(expr: Any) -> ReducerExpression:
    return ReducerExpression("count_distinct", expr)


def count_distinct_approximate(expr: Any) -> ReducerExpression:
    return ReducerExpression("count_distinct_approximate", expr)


def earliest(expr: Any) -> ReducerExpression:
    return ReducerExpression("earliest", expr)


def latest(expr: Any) -> ReducerExpression:
    return ReducerExpression("latest", expr)


def stateful_many(combine_many):
    """Custom python reducer over grouped rows (reference custom_reducers)."""

    def reducer(*exprs: Any) -> ReducerExpression:
        e = ReducerExpression("stateful_many", *exprs)
        e._combine_many = combine_many
        return e

    return reducer


def stateful_single(combine_single):
    def combine_many(state, rows):
        for row, cnt in rows:
            for _ in range(cnt):
                state = combine_single(state, *row)
        return state

    return stateful_many(combine_many)


def udf_reducer(reducer_cls):
    """Reducer from a BaseCustomAccumulator subclass."""

    def reducer(*exprs: Any) -> ReducerExpression:
        e = ReducerExpression("udf_reducer", *exprs)
        e._accumulator_cls = reducer_cls
        return e

    return reducer
