"""Custom accumulator reducers (reference internals/custom_reducers.py:36-430)."""

from __future__ import annotations

from typing import Any

from pathway_amd.internals.expression import ReducerExpression


class BaseCustomAccumulator:
    """Subclass with from_row / update / compute_result (+ optional retract)."""

    @classmethod
    def from_row(cls, row):
        raise NotImplementedError

    def update(self, other) -> None:
        raise NotImplementedError

    def retract(self, other) -> None:
        raise NotImplementedError

    def neutral(self):
        raise NotImplementedError

    def compute_result(self) -> Any:
        raise NotImplementedError


def udf_reducer(accumulator: type[BaseCustomAccumulator]):
    def reducer(*exprs: Any) -> ReducerExpression:
        e = ReducerExpression("udf_reducer", *exprs)
        e._accumulator_cls = accumulator
        return e

    return reducer


def stateful_many(combine_many):
    from pathway_amd.reducers import stateful_many as _sm

    return _sm(combine_many)


def stateful_single(combine_single):
    from pathway_amd.reducers import stateful_single as _ss

    return _ss(combine_single)


def mark_stub(fun):
    return fun
