"""Lightweight expression dtype inference (reference type_interpreter)."""

from __future__ import annotations

from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as ex


def infer_dtype(e: ex.ColumnExpression, dtypes: dict[str, dt.DType]) -> dt.DType:
    if isinstance(e, ex.ColumnConstExpression):
        return dt.dtype_of_value(e._value)
    if isinstance(e, ex.ColumnReference):
        if e.name == "id":
            return dt.POINTER
        return dtypes.get(e.name, dt.ANY)
    if isinstance(e, ex.ColumnBinaryOpExpression):
        a = infer_dtype(e._left, dtypes)
        b = infer_dtype(e._right, dtypes)
        sym = e._symbol
        if sym in ("==", "!=", "<", "<=", ">", ">="):
            return dt.BOOL
        if sym == "/":
            return dt.FLOAT
        if sym in ("&", "|", "^") and dt.unoptionalize(a) == dt.BOOL:
            return a
        ua, ub = dt.unoptionalize(a), dt.unoptionalize(b)
        out: dt.DType
        if ua == dt.FLOAT or ub == dt.FLOAT:
            out = dt.FLOAT
        elif ua == dt.STR and sym in ("+", "*"):
            out = dt.STR
        elif ua in (dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC) and sym == "-" and ub in (
            dt.DATE_TIME_NAIVE,
            dt.DATE_TIME_UTC,
        ):
            out = dt.DURATION
        else:
            out = ua if ua != dt.ANY else ub
        if a.is_optional() or b.is_optional():
            return dt.Optional(out)
        return out
    if isinstance(e, ex.ColumnUnaryOpExpression):
        if e._symbol == "~":
            inner = infer_dtype(e._expr, dtypes)
            return inner
        return infer_dtype(e._expr, dtypes)
    if isinstance(e, (ex.CastExpression, ex.DeclareTypeExpression)):
        return e._target
    if isinstance(e, ex.ConvertExpression):
        return e._target if e._unwrap else dt.Optional(e._target)
    if isinstance(e, ex.CoalesceExpression):
        args = [infer_dtype(a, dtypes) for a in e._args]
        out = args[0]
        for a in args[1:]:
            out = dt.types_lca(out, a)
        non_opt = any(not a.is_optional() and a != dt.NONE for a in args)
        if non_opt:
            out = dt.unoptionalize(out)
        return out
    if isinstance(e, ex.RequireExpression):
        inner = infer_dtype(e._value, dtypes)
        return dt.Optional(dt.unoptionalize(inner))
    if isinstance(e, ex.IfElseExpression):
        return dt.types_lca(
            infer_dtype(e._then, dtypes), infer_dtype(e._else, dtypes)
        )
    if isinstance(e, (ex.IsNoneExpression, ex.IsNotNoneExpression)):
        return dt.BOOL
    if isinstance(e, ex.PointerExpression):
        return dt.Optional(dt.POINTER) if e._optional else dt.POINTER
    if isinstance(e, ex.MakeTupleExpression):
        return dt.Tuple(*[infer_dtype(a, dtypes) for a in e._args])
    if isinstance(e, ex.GetExpression):
        obj = infer_dtype(e._object, dtypes)
        if dt.unoptionalize(obj) == dt.JSON:
            return dt.JSON
        return dt.ANY
    if isinstance(e, ex.MethodCallExpression):
        if e._return_type is not None:
            return e._return_type
        return dt.ANY
    if isinstance(e, ex.UnwrapExpression):
        return dt.unoptionalize(infer_dtype(e._expr, dtypes))
    if isinstance(e, ex.FillErrorExpression):
        return infer_dtype(e._expr, dtypes)
    if isinstance(e, ex.ApplyExpression):
        return e._return_type
    if isinstance(e, ex.ReducerExpression):
        from pathway_amd.engine.reducers import REDUCERS

        spec = REDUCERS.get(e._reducer)
        if spec is not None and spec.out_dtype is not None:
            args_dt = [infer_dtype(a, dtypes) for a in e._args] or [dt.ANY]
            try:
                return spec.out_dtype(args_dt)
            except Exception:
                return dt.ANY
        return dt.ANY
    return dt.ANY
