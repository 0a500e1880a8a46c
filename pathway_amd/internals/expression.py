"""Column expression AST (reference internals/expression.py:88-1240 behavior).

Expressions are built by operator overloading on ``ColumnReference``/
``ColumnExpression`` and evaluated columnar-batch-at-a-time by the engine
(pathway_amd/engine/expression_eval.py): device-representable subtrees run
as torch ops on the GPU; host subtrees (str/json/python UDFs) run vectorized
on the host, batched at the boundary like the reference's GIL-wrapped
BatchWrapper (graph.rs:463).
"""

from __future__ import annotations

import operator
from typing import TYPE_CHECKING, Any, Callable, Iterable

from pathway_amd.internals import dtype as dt

if TYPE_CHECKING:
    from pathway_amd.internals.table import Table


class ColumnExpression:
    _dtype: dt.DType | None = None

    # -- arithmetic --
    def __add__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.add, "+")

    def __radd__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.add, "+")

    def __sub__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.sub, "-")

    def __rsub__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.sub, "-")

    def __mul__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.mul, "*")

    def __rmul__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.mul, "*")

    def __truediv__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.truediv, "/")

    def __rtruediv__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.truediv, "/")

    def __floordiv__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.floordiv, "//")

    def __rfloordiv__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.floordiv, "//")

    def __mod__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.mod, "%")

    def __rmod__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.mod, "%")

    def __pow__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.pow, "**")

    def __rpow__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.pow, "**")

    def __matmul__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.matmul, "@")

    def __rmatmul__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.matmul, "@")

    def __lshift__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.lshift, "<<")

    def __rshift__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.rshift, ">>")

    # -- comparisons --
    def __eq__(self, other):  # type: ignore[override]
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.eq, "==")

    def __ne__(self, other):  # type: ignore[override]
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.ne, "!=")

    def __lt__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.lt, "<")

    def __le__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.le, "<=")

    def __gt__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.gt, ">")

    def __ge__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.ge, ">=")

    # -- boolean --
    def __and__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.and_, "&")

    def __rand__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.and_, "&")

    def __or__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.or_, "|")

    def __ror__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.or_, "|")

    def __xor__(self, other):
        return ColumnBinaryOpExpression(self, wrap_expr(other), operator.xor, "^")

    def __rxor__(self, other):
        return ColumnBinaryOpExpression(wrap_expr(other), self, operator.xor, "^")

    def __invert__(self):
        return ColumnUnaryOpExpression(self, operator.not_, "~")

    def __neg__(self):
        return ColumnUnaryOpExpression(self, operator.neg, "-")

    def __pos__(self):
        return self

    def __abs__(self):
        return ColumnUnaryOpExpression(self, operator.abs, "abs")

    def __hash__(self) -> int:
        return id(self)

    def __bool__(self):
        raise RuntimeError(
            "ColumnExpression is not a boolean; use &, | and ~ instead of and/or/not"
        )

    # -- item access --
    def __getitem__(self, item):
        return GetExpression(self, wrap_expr(item), check_if_exists=False)

    def get(self, index, default=None):
        return GetExpression(
            self, wrap_expr(index), wrap_expr(default), check_if_exists=True
        )

    # -- misc methods (reference expression.py public surface) --
    def is_none(self):
        return IsNoneExpression(self)

    def is_not_none(self):
        return IsNotNoneExpression(self)

    def to_string(self):
        return MethodCallExpression("to_string", self)

    def as_int(self, unwrap: bool = False, default=None):
        return ConvertExpression(self, dt.INT, unwrap=unwrap, default=wrap_expr(default))

    def as_float(self, unwrap: bool = False, default=None):
        return ConvertExpression(self, dt.FLOAT, unwrap=unwrap, default=wrap_expr(default))

    def as_str(self, unwrap: bool = False, default=None):
        return ConvertExpression(self, dt.STR, unwrap=unwrap, default=wrap_expr(default))

    def as_bool(self, unwrap: bool = False, default=None):
        return ConvertExpression(self, dt.BOOL, unwrap=unwrap, default=wrap_expr(default))

    # -- namespaces --
    @property
    def dt(self):
        from pathway_amd.internals.expressions.date_time import DateTimeNamespace

        return DateTimeNamespace(self)

    @property
    def str(self):
        from pathway_amd.internals.expressions.string import StringNamespace

        return StringNamespace(self)

    @property
    def num(self):
        from pathway_amd.internals.expressions.numerical import NumericalNamespace

        return NumericalNamespace(self)

    @property
    def bin(self):
        from pathway_amd.internals.expressions.string import BinaryNamespace

        return BinaryNamespace(self)

    # -- introspection --
    @property
    def _deps(self) -> tuple["ColumnExpression", ...]:
        return ()

    def _col_refs(self) -> list["ColumnReference"]:
        out: list[ColumnReference] = []

        def rec(e: ColumnExpression):
            if isinstance(e, ColumnReference):
                out.append(e)
            for d in e._deps:
                rec(d)

        rec(self)
        return out


class ColumnConstExpression(ColumnExpression):
    def __init__(self, value: Any):
        self._value = value

    def __repr__(self):
        return f"const({self._value!r})"


class ColumnReference(ColumnExpression):
    """Reference to ``table.column`` (or ``pw.this.column`` before resolution)."""

    def __init__(self, table: Any, name: str):
        self._table = table
        self._name = name

    @property
    def table(self):
        return self._table

    @property
    def name(self) -> str:
        return self._name

    def __repr__(self):
        return f"<{self._name}>"

    def to_column_expression(self) -> ColumnExpression:
        return self


class ColumnBinaryOpExpression(ColumnExpression):
    def __init__(self, left: ColumnExpression, right: ColumnExpression, op: Callable, symbol: str):
        self._left = left
        self._right = right
        self._operator = op
        self._symbol = symbol

    @property
    def _deps(self):
        return (self._left, self._right)

    def __repr__(self):
        return f"({self._left!r} {self._symbol} {self._right!r})"


class ColumnUnaryOpExpression(ColumnExpression):
    def __init__(self, expr: ColumnExpression, op: Callable, symbol: str):
        self._expr = expr
        self._operator = op
        self._symbol = symbol

    @property
    def _deps(self):
        return (self._expr,)

    def __repr__(self):
        return f"({self._symbol}{self._expr!r})"


class ReducerExpression(ColumnExpression):
    def __init__(self, reducer: Any, *args: Any, **kwargs: Any):
        self._reducer = reducer
        self._args = tuple(wrap_expr(a) for a in args)
        self._kwargs = kwargs

    @property
    def _deps(self):
        return self._args

    def __repr__(self):
        return f"{self._reducer}({', '.join(map(repr, self._args))})"


class ApplyExpression(ColumnExpression):
    def __init__(
        self,
        fun: Callable,
        return_type: Any,
        *args: Any,
        _check_for_disallowed_types: bool = True,
        propagate_none: bool = False,
        deterministic: bool = True,
        max_batch_size: int | None = None,
        **kwargs: Any,
    ):
        self._fun = fun
        self._return_type = dt.wrap(return_type) if return_type is not None else dt.ANY
        self._args = tuple(wrap_expr(a) for a in args)
        self._kwargs = {k: wrap_expr(v) for k, v in kwargs.items()}
        self._propagate_none = propagate_none
        self._deterministic = deterministic
        self._max_batch_size = max_batch_size

    @property
    def _deps(self):
        return self._args + tuple(self._kwargs.values())

    def __repr__(self):
        return f"apply({getattr(self._fun, '__name__', '?')})"


class AsyncApplyExpression(ApplyExpression):
    pass


class FullyAsyncApplyExpression(ApplyExpression):
    autocommit_duration_ms: int | None = 1500


class CastExpression(ColumnExpression):
    def __init__(self, expr: ColumnExpression, target: Any):
        self._expr = expr
        self._target = dt.wrap(target)

    @property
    def _deps(self):
        return (self._expr,)


class ConvertExpression(ColumnExpression):
    """Json / Any → typed conversion (as_int etc.)."""

    def __init__(self, expr, target: dt.DType, unwrap: bool = False, default=None):
        self._expr = expr
        self._target = target
        self._unwrap = unwrap
        self._default = default

    @property
    def _deps(self):
        return (self._expr,) if self._default is None else (self._expr, self._default)


class DeclareTypeExpression(ColumnExpression):
    def __init__(self, expr: ColumnExpression, target: Any):
        self._expr = expr
        self._target = dt.wrap(target)

    @property
    def _deps(self):
        return (self._expr,)


class CoalesceExpression(ColumnExpression):
    def __init__(self, *args: Any):
        self._args = tuple(wrap_expr(a) for a in args)

    @property
    def _deps(self):
        return self._args


class RequireExpression(ColumnExpression):
    def __init__(self, value: Any, *args: Any):
        self._value = wrap_expr(value)
        self._args = tuple(wrap_expr(a) for a in args)

    @property
    def _deps(self):
        return (self._value,) + self._args


class IfElseExpression(ColumnExpression):
    def __init__(self, if_: Any, then: Any, else_: Any):
        self._if = wrap_expr(if_)
        self._then = wrap_expr(then)
        self._else = wrap_expr(else_)

    @property
    def _deps(self):
        return (self._if, self._then, self._else)


class IsNoneExpression(ColumnExpression):
    def __init__(self, expr: ColumnExpression):
        self._expr = expr

    @property
    def _deps(self):
        return (self._expr,)


class IsNotNoneExpression(ColumnExpression):
    def __init__(self, expr: ColumnExpression):
        self._expr = expr

    @property
    def _deps(self):
        return (self._expr,)


class PointerExpression(ColumnExpression):
    def __init__(self, table: Any, *args: Any, optional: bool = False, instance=None):
        self._table = table
        self._args = tuple(wrap_expr(a) for a in args)
        self._optional = optional
        self._instance = wrap_expr(instance) if instance is not None else None

    @property
    def _deps(self):
        extra = (self._instance,) if self._instance is not None else ()
        return self._args + extra


class MakeTupleExpression(ColumnExpression):
    def __init__(self, *args: Any):
        self._args = tuple(wrap_expr(a) for a in args)

    @property
    def _deps(self):
        return self._args


class GetExpression(ColumnExpression):
    def __init__(
        self,
        obj: ColumnExpression,
        index: ColumnExpression,
        default: ColumnExpression | None = None,
        check_if_exists: bool = True,
    ):
        self._object = obj
        self._index = index
        self._default = default if default is not None else ColumnConstExpression(None)
        self._check_if_exists = check_if_exists

    @property
    def _deps(self):
        return (self._object, self._index, self._default)


class MethodCallExpression(ColumnExpression):
    """Namespace method call (.dt.year(), .str.upper(), ...)."""

    def __init__(self, name: str, *args: Any, return_type: dt.DType | None = None):
        self._method = name
        self._args = tuple(wrap_expr(a) for a in args)
        self._return_type = return_type

    @property
    def _deps(self):
        return self._args

    def __repr__(self):
        return f".{self._method}({', '.join(map(repr, self._args[1:]))})"


class UnwrapExpression(ColumnExpression):
    def __init__(self, expr: Any):
        self._expr = wrap_expr(expr)

    @property
    def _deps(self):
        return (self._expr,)


class FillErrorExpression(ColumnExpression):
    def __init__(self, expr: Any, replacement: Any):
        self._expr = wrap_expr(expr)
        self._replacement = wrap_expr(replacement)

    @property
    def _deps(self):
        return (self._expr, self._replacement)


def wrap_expr(value: Any) -> ColumnExpression:
    if isinstance(value, ColumnExpression):
        return value
    return ColumnConstExpression(value)


def smart_name(expr: ColumnExpression) -> str | None:
    if isinstance(expr, ColumnReference):
        return expr.name
    return None


def get_expression_info(expr: ColumnExpression) -> str:
    return repr(expr)
