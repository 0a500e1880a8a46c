"""Columnar expression evaluation over DeltaBatches.

The engine-side analog of the reference's expression interpreter
(src/engine/expression.rs:112-338, applied batch-wise in expression_table,
dataflow.rs:1504): device-representable subtrees evaluate as torch ops on
GPU tensors; host subtrees (strings, tuples, Json, python UDFs) evaluate
vectorized on the host, batched at the boundary.
"""

from __future__ import annotations

import math
import operator
from typing import Any, Callable

import numpy as np
import torch

from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as expr
from pathway_amd.internals.api import ERROR, BasePointer, Pointer, hash_values
from pathway_amd.engine import hashing
from pathway_amd.engine.column import (
    Column,
    ObjectColumn,
    PointerColumn,
    StringColumn,
    TensorColumn,
    column_from_pylist,
    infer_and_build_column,
)


class EvalContext:
    """Evaluation context: named input columns + row keys of one batch."""

    def __init__(
        self,
        columns: dict[str, Column],
        keys: torch.Tensor,
        device,
        extra: dict[str, Column] | None = None,
    ):
        self.columns = columns
        self.keys = keys
        self.device = device
        self.extra = extra or {}

    @property
    def n(self) -> int:
        return int(self.keys.shape[0])

    def resolve(self, name: str) -> Column:
        if name == "id":
            return PointerColumn(self.keys)
        if name in self.columns:
            return self.columns[name]
        if name in self.extra:
            return self.extra[name]
        raise KeyError(f"unknown column {name!r}; have {list(self.columns)}")


_FLOAT_OPS = {"+", "-", "*", "/", "//", "%", "**"}
_CMP_OPS = {"==", "!=", "<", "<=", ">", ">="}
_BOOL_OPS = {"&", "|", "^"}


def _as_tensor_op(col: Column) -> torch.Tensor | None:
    if isinstance(col, TensorColumn) and col.mask is None:
        return col.tensor
    return None


def _host_values(col: Column) -> list[Any]:
    return col.to_pylist()


def evaluate(e: expr.ColumnExpression, ctx: EvalContext) -> Column:
    if isinstance(e, expr.ColumnConstExpression):
        return _const_column(e._value, ctx)
    if isinstance(e, expr.ColumnReference):
        return ctx.resolve(e.name)
    if isinstance(e, expr.ColumnBinaryOpExpression):
        return _eval_binary(e, ctx)
    if isinstance(e, expr.ColumnUnaryOpExpression):
        return _eval_unary(e, ctx)
    if isinstance(e, expr.CastExpression):
        return _eval_cast(evaluate(e._expr, ctx), e._target, ctx)
    if isinstance(e, expr.DeclareTypeExpression):
        col = evaluate(e._expr, ctx)
        col = _maybe_specialize(col, e._target, ctx)
        return col
    if isinstance(e, expr.ConvertExpression):
        return _eval_convert(e, ctx)
    if isinstance(e, expr.CoalesceExpression):
        return _eval_coalesce(e, ctx)
    if isinstance(e, expr.RequireExpression):
        return _eval_require(e, ctx)
    if isinstance(e, expr.IfElseExpression):
        return _eval_if_else(e, ctx)
    if isinstance(e, expr.IsNoneExpression):
        return _eval_is_none(evaluate(e._expr, ctx), ctx, negate=False)
    if isinstance(e, expr.IsNotNoneExpression):
        return _eval_is_none(evaluate(e._expr, ctx), ctx, negate=True)
    if isinstance(e, expr.PointerExpression):
        return _eval_pointer(e, ctx)
    if isinstance(e, expr.MakeTupleExpression):
        cols = [evaluate(a, ctx) for a in e._args]
        vals = [_host_values(c) for c in cols]
        return ObjectColumn(
            __import__("pathway_amd.engine.column", fromlist=["obj_array"]).obj_array(
                [tuple(row) for row in zip(*vals)] if vals else [()] * ctx.n
            ),
            dt.ANY_TUPLE,
        )
    if isinstance(e, expr.GetExpression):
        return _eval_get(e, ctx)
    if isinstance(e, expr.MethodCallExpression):
        return _eval_method(e, ctx)
    if isinstance(e, expr.UnwrapExpression):
        return _eval_unwrap(evaluate(e._expr, ctx))
    if isinstance(e, expr.FillErrorExpression):
        return _eval_fill_error(e, ctx)
    if isinstance(e, (expr.AsyncApplyExpression, expr.FullyAsyncApplyExpression)):
        return _eval_apply(e, ctx, is_async=True)
    if isinstance(e, expr.ApplyExpression):
        return _eval_apply(e, ctx, is_async=False)
    if isinstance(e, expr.ReducerExpression):
        raise TypeError("reducer expression outside of reduce()")
    raise NotImplementedError(f"cannot evaluate {type(e).__name__}")


def _const_column(value: Any, ctx: EvalContext) -> Column:
    n = ctx.n
    vdt = dt.dtype_of_value(value)
    if vdt in (dt.INT, dt.FLOAT, dt.BOOL):
        td = {dt.INT: torch.int64, dt.FLOAT: torch.float64, dt.BOOL: torch.bool}[vdt]
        return TensorColumn(
            torch.full((n,), value, dtype=td, device=ctx.device), vdt
        )
    if vdt == dt.STR:
        return StringColumn.from_strings([value] * n, device=ctx.device)
    return column_from_pylist([value] * n, vdt, device=ctx.device)


def _combine_masks(a: Column, b: Column) -> torch.Tensor | None:
    ma = a.mask if isinstance(a, TensorColumn) else None
    mb = b.mask if isinstance(b, TensorColumn) else None
    if ma is None:
        return mb
    if mb is None:
        return ma
    return ma & mb


def _eval_binary(e: expr.ColumnBinaryOpExpression, ctx: EvalContext) -> Column:
    a = evaluate(e._left, ctx)
    b = evaluate(e._right, ctx)
    sym = e._symbol

    # pointer/string equality on device representations
    if sym in ("==", "!=") and isinstance(a, PointerColumn) and isinstance(b, PointerColumn):
        eq = (a.pairs[:, 0] == b.pairs[:, 0]) & (a.pairs[:, 1] == b.pairs[:, 1])
        return TensorColumn(eq if sym == "==" else ~eq, dt.BOOL)
    if (
        sym in ("==", "!=")
        and isinstance(a, StringColumn)
        and isinstance(b, StringColumn)
        and a.pool is b.pool
    ):
        eq = a.codes == b.codes
        return TensorColumn(eq if sym == "==" else ~eq, dt.BOOL)

    ta = a.tensor if isinstance(a, TensorColumn) else None
    tb = b.tensor if isinstance(b, TensorColumn) else None
    if ta is not None and tb is not None:
        mask = _combine_masks(a, b)
        out = _tensor_binop(ta, tb, sym)
        if out is not None:
            out_dt = _binop_dtype(a.dtype, b.dtype, sym)
            if mask is not None and out_dt != dt.BOOL:
                return TensorColumn(out, dt.Optional(out_dt), mask)
            if mask is not None and sym in _CMP_OPS:
                # comparisons with None propagate None in reference semantics
                return TensorColumn(out, dt.Optional(dt.BOOL), mask)
            return TensorColumn(out, out_dt)

    # host fallback
    av = _host_values(a)
    bv = _host_values(b)
    f = e._operator
    out_vals = []
    for x, y in zip(av, bv):
        if x is ERROR or y is ERROR:
            out_vals.append(ERROR)
            continue
        if x is None or y is None:
            if sym == "==":
                out_vals.append(x is None and y is None)
            elif sym == "!=":
                out_vals.append(not (x is None and y is None))
            else:
                out_vals.append(None)
            continue
        try:
            out_vals.append(f(x, y))
        except Exception:
            out_vals.append(ERROR)
    col, _ = infer_and_build_column(out_vals, device=ctx.device)
    return col


def _tensor_binop(ta: torch.Tensor, tb: torch.Tensor, sym: str) -> torch.Tensor | None:
    try:
        if sym == "+":
            return ta + tb
        if sym == "-":
            return ta - tb
        if sym == "*":
            return ta * tb
        if sym == "/":
            return ta.to(torch.float64) / tb.to(torch.float64)
        if sym == "//":
            # integer zero-divisors: CUDA does not trap like CPU does —
            # fall back to the host path so those rows become ERROR values
            if (
                ta.dtype == torch.int64
                and tb.dtype == torch.int64
                and bool((tb == 0).any())
            ):
                return None
            return torch.div(ta, tb, rounding_mode="floor")
        if sym == "%":
            if (
                ta.dtype == torch.int64
                and tb.dtype == torch.int64
                and bool((tb == 0).any())
            ):
                return None
            return ta - torch.div(ta, tb, rounding_mode="floor") * tb
        if sym == "**":
            if ta.dtype == torch.int64 and tb.dtype == torch.int64:
                return torch.pow(ta.to(torch.float64), tb.to(torch.float64)).to(torch.int64)
            return torch.pow(ta.to(torch.float64), tb.to(torch.float64))
        if sym == "==":
            return ta == tb
        if sym == "!=":
            return ta != tb
        if sym == "<":
            return ta < tb
        if sym == "<=":
            return ta <= tb
        if sym == ">":
            return ta > tb
        if sym == ">=":
            return ta >= tb
        if sym == "&":
            return ta & tb
        if sym == "|":
            return ta | tb
        if sym == "^":
            return ta ^ tb
        if sym == "<<":
            return ta << tb
        if sym == ">>":
            return ta >> tb
        if sym == "@":
            return ta @ tb
    except RuntimeError:
        return None
    return None


def _binop_dtype(a: dt.DType, b: dt.DType, sym: str) -> dt.DType:
    if sym in _CMP_OPS:
        return dt.BOOL
    if sym == "/":
        return dt.FLOAT
    a, b = dt.unoptionalize(a), dt.unoptionalize(b)
    if sym in _BOOL_OPS and a == dt.BOOL and b == dt.BOOL:
        return dt.BOOL
    if a == dt.FLOAT or b == dt.FLOAT:
        return dt.FLOAT
    if a == dt.DATE_TIME_NAIVE or a == dt.DATE_TIME_UTC:
        if sym == "-" and b in (dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC):
            return dt.DURATION
        return a
    if a == dt.DURATION and b in (dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC):
        return b
    return a if a != dt.ANY else b


def _eval_unary(e: expr.ColumnUnaryOpExpression, ctx: EvalContext) -> Column:
    a = evaluate(e._expr, ctx)
    if isinstance(a, TensorColumn):
        if e._symbol == "~":
            out = ~a.tensor if a.tensor.dtype == torch.bool else ~a.tensor
            return TensorColumn(out, a.dtype, a.mask)
        if e._symbol == "-":
            return TensorColumn(-a.tensor, a.dtype, a.mask)
        if e._symbol == "abs":
            return TensorColumn(a.tensor.abs(), a.dtype, a.mask)
    vals = [
        (ERROR if v is ERROR else None if v is None else e._operator(v))
        for v in _host_values(a)
    ]
    col, _ = infer_and_build_column(vals, device=ctx.device)
    return col


def _eval_cast(col: Column, target: dt.DType, ctx: EvalContext) -> Column:
    base = dt.unoptionalize(target)
    if isinstance(col, TensorColumn):
        if base == dt.FLOAT:
            return TensorColumn(col.tensor.to(torch.float64), target, col.mask)
        if base == dt.INT:
            return TensorColumn(col.tensor.to(torch.int64), target, col.mask)
        if base == dt.BOOL:
            return TensorColumn(col.tensor.to(torch.bool), target, col.mask)
        if base == dt.STR:
            vals = [None if v is None else _to_str(v) for v in col.to_pylist()]
            return StringColumn.from_strings(vals, device=ctx.device)
    vals = []
    conv: Callable[[Any], Any]
    if base == dt.INT:
        conv = int
    elif base == dt.FLOAT:
        conv = float
    elif base == dt.BOOL:
        conv = bool
    elif base == dt.STR:
        conv = _to_str
    else:
        conv = lambda v: v
    for v in _host_values(col):
        if v is None or v is ERROR:
            vals.append(v)
        else:
            try:
                vals.append(conv(v))
            except Exception:
                vals.append(ERROR)
    return column_from_pylist(vals, target, device=ctx.device)


def _to_str(v: Any) -> str:
    if isinstance(v, bool):
        return "True" if v else "False"
    if isinstance(v, float) and v.is_integer() and not math.isinf(v):
        return f"{v:.1f}"
    return str(v)


def _eval_convert(e: expr.ConvertExpression, ctx: EvalContext) -> Column:
    from pathway_amd.internals.json import Json

    col = evaluate(e._expr, ctx)
    default = (
        evaluate(e._default, ctx)
        if e._default is not None
        else None
    )
    dvals = _host_values(default) if default is not None else None
    target = e._target
    out = []
    for i, v in enumerate(_host_values(col)):
        if isinstance(v, Json):
            v = v.value
        if v is None:
            out.append(dvals[i] if dvals is not None else None)
            continue
        if v is ERROR:
            out.append(ERROR)
            continue
        ok = (
            (target == dt.INT and isinstance(v, int) and not isinstance(v, bool))
            or (target == dt.FLOAT and isinstance(v, (int, float)) and not isinstance(v, bool))
            or (target == dt.STR and isinstance(v, str))
            or (target == dt.BOOL and isinstance(v, bool))
        )
        if ok:
            out.append(float(v) if target == dt.FLOAT else v)
        elif e._unwrap:
            out.append(ERROR)
        else:
            out.append(dvals[i] if dvals is not None else None)
    tdt = target if e._unwrap else dt.Optional(target)
    return column_from_pylist(out, tdt, device=ctx.device)


def _eval_coalesce(e: expr.CoalesceExpression, ctx: EvalContext) -> Column:
    cols = [evaluate(a, ctx) for a in e._args]
    # fast path: tensor columns
    if all(isinstance(c, TensorColumn) for c in cols):
        out_t = cols[-1].tensor
        out_m = cols[-1].mask
        for c in reversed(cols[:-1]):
            if c.mask is None:
                out_t, out_m = c.tensor, None
            else:
                if out_t.dtype != c.tensor.dtype:
                    common = torch.promote_types(out_t.dtype, c.tensor.dtype)
                    out_t = out_t.to(common)
                    ct = c.tensor.to(common)
                else:
                    ct = c.tensor
                out_t = torch.where(c.mask, ct, out_t)
                out_m = c.mask | out_m if out_m is not None else None
        dtype = dt.types_lca(cols[0].dtype, cols[-1].dtype)
        if out_m is not None and bool(out_m.all()):
            out_m = None
        if out_m is None:
            dtype = dt.unoptionalize(dtype)
        return TensorColumn(out_t, dtype, out_m)
    valss = [_host_values(c) for c in cols]
    out = []
    for row in zip(*valss):
        val = None
        for v in row:
            if v is not None:
                val = v
                break
        out.append(val)
    col, _ = infer_and_build_column(out, device=ctx.device)
    return col


def _eval_require(e: expr.RequireExpression, ctx: EvalContext) -> Column:
    val = evaluate(e._value, ctx)
    args = [evaluate(a, ctx) for a in e._args]
    none_mask = None
    for a in args:
        m = _none_mask(a)
        none_mask = m if none_mask is None else (none_mask | m)
    if none_mask is None or not bool(none_mask.any()):
        return val
    if isinstance(val, TensorColumn):
        mask = (~none_mask) & (
            val.mask if val.mask is not None else torch.ones_like(none_mask)
        )
        return TensorColumn(val.tensor, dt.Optional(dt.unoptionalize(val.dtype)), mask)
    vals = _host_values(val)
    nm = none_mask.cpu().tolist()
    out = [None if bad else v for v, bad in zip(vals, nm)]
    return column_from_pylist(out, dt.Optional(dt.unoptionalize(val.dtype)), ctx.device)


def _none_mask(col: Column) -> torch.Tensor:
    """True where the value IS None."""
    if isinstance(col, TensorColumn):
        if col.mask is None:
            return torch.zeros(len(col), dtype=torch.bool, device=col.tensor.device)
        return ~col.mask
    if isinstance(col, StringColumn):
        return col.codes < 0
    vals = col.to_pylist()
    return torch.tensor([v is None for v in vals], dtype=torch.bool)


def _eval_if_else(e: expr.IfElseExpression, ctx: EvalContext) -> Column:
    cond = evaluate(e._if, ctx)
    then = evaluate(e._then, ctx)
    els = evaluate(e._else, ctx)
    if (
        isinstance(cond, TensorColumn)
        and isinstance(then, TensorColumn)
        and isinstance(els, TensorColumn)
        and then.tensor.dtype == els.tensor.dtype
    ):
        c = cond.tensor.to(torch.bool)
        out = torch.where(c, then.tensor, els.tensor)
        mask = None
        mt = then.mask if then.mask is not None else None
        me = els.mask if els.mask is not None else None
        if mt is not None or me is not None:
            ones = torch.ones_like(c)
            mask = torch.where(c, mt if mt is not None else ones, me if me is not None else ones)
        return TensorColumn(out, dt.types_lca(then.dtype, els.dtype), mask)
    cv = _host_values(cond)
    tv = _host_values(then)
    ev = _host_values(els)
    out_vals = [ERROR if c is ERROR else (t if c else f) for c, t, f in zip(cv, tv, ev)]
    col, _ = infer_and_build_column(out_vals, device=ctx.device)
    return col


def _eval_is_none(col: Column, ctx: EvalContext, negate: bool) -> Column:
    m = _none_mask(col).to(ctx.device if not isinstance(col, ObjectColumn) else "cpu")
    return TensorColumn(~m if negate else m, dt.BOOL)


def _eval_pointer(e: expr.PointerExpression, ctx: EvalContext) -> Column:
    cols = [evaluate(a, ctx) for a in e._args]
    if e._instance is not None:
        cols.append(evaluate(e._instance, ctx))
    parts = [c.value_hash() for c in cols]
    parts = [(lo.to(ctx.device), hi.to(ctx.device)) for lo, hi in parts]
    lo, hi = hashing.combine_value_hashes(parts)
    return PointerColumn(torch.stack([lo, hi], dim=1))


def _eval_get(e: expr.GetExpression, ctx: EvalContext) -> Column:
    from pathway_amd.internals.json import Json

    obj = evaluate(e._object, ctx)
    idx = evaluate(e._index, ctx)
    dflt = evaluate(e._default, ctx)
    ov, iv, dv = _host_values(obj), _host_values(idx), _host_values(dflt)
    out = []
    for o, i, d in zip(ov, iv, dv):
        if o is ERROR or i is ERROR:
            out.append(ERROR)
            continue
        try:
            if isinstance(o, Json):
                val = o.value
                got = val[i]
                out.append(Json(got))
            else:
                out.append(o[i])
        except (KeyError, IndexError, TypeError):
            if e._check_if_exists:
                out.append(d)
            else:
                out.append(ERROR)
    col, _ = infer_and_build_column(out, device=ctx.device)
    return col


def _eval_unwrap(col: Column) -> Column:
    if isinstance(col, TensorColumn):
        if col.mask is not None and not bool(col.mask.all()):
            raise ValueError("cannot unwrap, None value present")
        return TensorColumn(col.tensor, dt.unoptionalize(col.dtype))
    vals = col.to_pylist()
    if any(v is None for v in vals):
        raise ValueError("cannot unwrap, None value present")
    return col


def _eval_fill_error(e: expr.FillErrorExpression, ctx: EvalContext) -> Column:
    try:
        col = evaluate(e._expr, ctx)
    except Exception:
        return evaluate(e._replacement, ctx)
    vals = _host_values(col)
    if not any(v is ERROR for v in vals):
        return col
    rep = _host_values(evaluate(e._replacement, ctx))
    out = [r if v is ERROR else v for v, r in zip(vals, rep)]
    c, _ = infer_and_build_column(out, device=ctx.device)
    return c


def _eval_apply(e: expr.ApplyExpression, ctx: EvalContext, is_async: bool) -> Column:
    arg_cols = [evaluate(a, ctx) for a in e._args]
    kw_cols = {k: evaluate(v, ctx) for k, v in e._kwargs.items()}
    arg_vals = [_host_values(c) for c in arg_cols]
    kw_vals = {k: _host_values(c) for k, c in kw_cols.items()}
    n = ctx.n
    fun = e._fun
    batch_fun = getattr(e, "_batch_fun", None)
    if batch_fun is not None:
        # batched UDF (embedders etc.): one call per batch, GIL-amortized —
        # reference BatchWrapper::WithGil + max_expression_batch_size
        out = batch_fun(*arg_vals, **kw_vals)
        rt = e._return_type
        if rt == dt.ANY:
            col, _ = infer_and_build_column(out, device=ctx.device)
            return col
        return column_from_pylist(out, rt, device=ctx.device)
    if is_async:
        import asyncio
        import inspect

        async def run_all():
            coros = []
            for i in range(n):
                args = [a[i] for a in arg_vals]
                kwargs = {k: v[i] for k, v in kw_vals.items()}
                if any(a is ERROR for a in args) or any(
                    v is ERROR for v in kwargs.values()
                ):
                    # Value::Error propagates without invoking the UDF
                    async def _err():
                        return ERROR

                    coros.append(_err())
                elif e._propagate_none and any(a is None for a in args):
                    async def _none():
                        return None

                    coros.append(_none())
                else:
                    r = fun(*args, **kwargs)
                    if inspect.isawaitable(r):
                        coros.append(r)
                    else:
                        async def _wrap(rv=r):
                            return rv

                        coros.append(_wrap())
            return await asyncio.gather(*coros, return_exceptions=True)

        results = asyncio.get_event_loop_policy().new_event_loop().run_until_complete(
            run_all()
        )
        out = [ERROR if isinstance(r, Exception) else r for r in results]
    else:
        out = []
        for i in range(n):
            args = [a[i] for a in arg_vals]
            kwargs = {k: v[i] for k, v in kw_vals.items()}
            if any(a is ERROR for a in args) or any(
                v is ERROR for v in kwargs.values()
            ):
                out.append(ERROR)  # Value::Error propagates, UDF not called
                continue
            if e._propagate_none and (
                any(a is None for a in args) or any(v is None for v in kwargs.values())
            ):
                out.append(None)
                continue
            try:
                out.append(fun(*args, **kwargs))
            except Exception as exc:  # noqa: BLE001
                from pathway_amd.internals.errors import record_error

                record_error(f"{type(exc).__name__}: {exc}", trace=getattr(fun, "__name__", "?"))
                out.append(ERROR)
    rt = e._return_type
    if rt == dt.ANY:
        col, _ = infer_and_build_column(out, device=ctx.device)
        return col
    return column_from_pylist(out, rt, device=ctx.device)


def _maybe_specialize(col: Column, target: dt.DType, ctx: EvalContext) -> Column:
    """declare_type: re-type an ANY/object column into a device column."""
    base = dt.unoptionalize(target)
    if isinstance(col, ObjectColumn) and base in (dt.INT, dt.FLOAT, dt.BOOL, dt.STR):
        return column_from_pylist(col.to_pylist(), target, device=ctx.device)
    col2 = col
    try:
        col2.dtype = target  # type: ignore[misc]
    except Exception:
        pass
    return col2


# ---------------------------------------------------------------- methods --

def _eval_method(e: expr.MethodCallExpression, ctx: EvalContext) -> Column:
    name = e._method
    cols = [evaluate(a, ctx) for a in e._args]
    impl = _METHODS.get(name)
    if impl is None:
        raise NotImplementedError(f"method {name!r} not implemented")
    return impl(cols, ctx)


def _host_method(fun: Callable[..., Any], result_dtype: dt.DType | None = None):
    def impl(cols: list[Column], ctx: EvalContext) -> Column:
        valss = [_host_values(c) for c in cols]
        out = []
        for row in zip(*valss):
            if any(v is ERROR for v in row):
                out.append(ERROR)
            elif any(v is None for v in row):
                out.append(None)
            else:
                try:
                    out.append(fun(*row))
                except Exception:
                    out.append(ERROR)
        if result_dtype is not None:
            return column_from_pylist(out, result_dtype, device=ctx.device)
        col, _ = infer_and_build_column(out, device=ctx.device)
        return col

    return impl


def _tensor_method(tf: Callable[[torch.Tensor], torch.Tensor], result_dtype: dt.DType):
    def impl(cols: list[Column], ctx: EvalContext) -> Column:
        c = cols[0]
        if isinstance(c, TensorColumn):
            return TensorColumn(tf(c.tensor), result_dtype, c.mask)
        return _host_method(lambda v: tf(torch.tensor([v])).item(), result_dtype)(
            cols, ctx
        )

    return impl


_METHODS: dict[str, Callable[[list[Column], EvalContext], Column]] = {
    "to_string": lambda cols, ctx: _eval_cast(cols[0], dt.STR, ctx),
    # --- str namespace ---
    "str.lower": _host_method(lambda s: s.lower(), dt.STR),
    "str.upper": _host_method(lambda s: s.upper(), dt.STR),
    "str.reversed": _host_method(lambda s: s[::-1], dt.STR),
    "str.len": _host_method(len, dt.INT),
    "str.strip": _host_method(lambda s, *a: s.strip(*a), dt.STR),
    "str.lstrip": _host_method(lambda s, *a: s.lstrip(*a), dt.STR),
    "str.rstrip": _host_method(lambda s, *a: s.rstrip(*a), dt.STR),
    "str.startswith": _host_method(lambda s, p: s.startswith(p), dt.BOOL),
    "str.endswith": _host_method(lambda s, p: s.endswith(p), dt.BOOL),
    "str.count": _host_method(lambda s, p: s.count(p), dt.INT),
    "str.find": _host_method(lambda s, p, *a: s.find(p, *a), dt.INT),
    "str.rfind": _host_method(lambda s, p, *a: s.rfind(p, *a), dt.INT),
    "str.replace": _host_method(lambda s, a, b, *r: s.replace(a, b, *r), dt.STR),
    "str.split": _host_method(lambda s, *a: tuple(s.split(*a)) if a else tuple(s.split()), dt.ANY_TUPLE),
    "str.slice": _host_method(lambda s, a, b: s[a:b], dt.STR),
    "str.title": _host_method(lambda s: s.title(), dt.STR),
    "str.swapcase": _host_method(lambda s: s.swapcase(), dt.STR),
    "str.ljust": _host_method(lambda s, *a: s.ljust(*a), dt.STR),
    "str.rjust": _host_method(lambda s, *a: s.rjust(*a), dt.STR),
    "str.removeprefix": _host_method(lambda s, p: s.removeprefix(p), dt.STR),
    "str.removesuffix": _host_method(lambda s, p: s.removesuffix(p), dt.STR),
    "str.parse_int": _host_method(int, dt.Optional(dt.INT)),
    "str.parse_float": _host_method(float, dt.Optional(dt.FLOAT)),
    "str.parse_bool": _host_method(
        lambda s: True if s.lower() in ("true", "yes", "on", "1") else (False if s.lower() in ("false", "no", "off", "0") else None),
        dt.Optional(dt.BOOL),
    ),
    # --- num namespace ---
    "num.abs": _tensor_method(torch.abs, dt.FLOAT),
    "num.round": lambda cols, ctx: _num_round(cols, ctx),
    "num.fill_na": lambda cols, ctx: _num_fill_na(cols, ctx),
    # --- dt namespace (int64 ns tensors) ---
    "dt.nanosecond": _tensor_method(lambda t: t % 1000, dt.INT),
    "dt.timestamp_ns": _tensor_method(lambda t: t, dt.INT),
}


def _num_round(cols: list[Column], ctx: EvalContext) -> Column:
    c = cols[0]
    nd = 0
    if len(cols) > 1:
        ndv = cols[1].to_pylist()
        nd = ndv[0] if ndv else 0
    if isinstance(c, TensorColumn) and c.tensor.dtype == torch.float64:
        scale = 10.0 ** nd
        return TensorColumn(torch.round(c.tensor * scale) / scale, c.dtype, c.mask)
    return _host_method(lambda v: round(v, nd))(cols[:1], ctx)


def _num_fill_na(cols: list[Column], ctx: EvalContext) -> Column:
    c, repl = cols[0], cols[1]
    if isinstance(c, TensorColumn):
        t = c.tensor
        rv = repl.to_pylist()[0] if len(repl) else 0
        if t.dtype == torch.float64:
            t = torch.nan_to_num(t, nan=float(rv))
        if c.mask is not None:
            fill = torch.full_like(t, rv)
            t = torch.where(c.mask, t, fill)
        return TensorColumn(t, dt.unoptionalize(c.dtype))
    return _host_method(lambda v, r: r if v is None else v)(cols, ctx)


def register_method(name: str, impl: Callable[[list[Column], EvalContext], Column]) -> None:
    _METHODS[name] = impl
