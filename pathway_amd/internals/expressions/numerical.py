"""expr.num namespace (reference internals/expressions/numerical.py)."""

from __future__ import annotations

from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.expression import ColumnExpression, MethodCallExpression


class NumericalNamespace:
    def __init__(self, expr: ColumnExpression):
        self._expr = expr

    def abs(self):
        return MethodCallExpression("num.abs", self._expr, return_type=dt.ANY)

    def round(self, decimals: Any = 0):
        return MethodCallExpression("num.round", self._expr, decimals, return_type=dt.ANY)

    def fill_na(self, default_value: Any):
        return MethodCallExpression(
            "num.fill_na", self._expr, default_value, return_type=dt.ANY
        )
