"""Top-level expression helpers: apply, cast, coalesce, if_else, udf …
(reference internals/common.py / internals/udfs)."""

from __future__ import annotations

import functools
from typing import Any, Callable

from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as ex


def apply(fun: Callable, *args: Any, **kwargs: Any) -> ex.ApplyExpression:
    import typing

    hints = typing.get_type_hints(fun) if callable(fun) else {}
    ret = hints.get("return", None)
    return ex.ApplyExpression(fun, ret, *args, **kwargs)


def apply_with_type(fun: Callable, ret_type: Any, *args: Any, **kwargs: Any) -> ex.ApplyExpression:
    return ex.ApplyExpression(fun, ret_type, *args, **kwargs)


def apply_async(fun: Callable, *args: Any, **kwargs: Any) -> ex.AsyncApplyExpression:
    import typing

    hints = typing.get_type_hints(fun) if callable(fun) else {}
    ret = hints.get("return", None)
    return ex.AsyncApplyExpression(fun, ret, *args, **kwargs)


def apply_full_async(fun: Callable, *args: Any, **kwargs: Any) -> ex.FullyAsyncApplyExpression:
    import typing

    hints = typing.get_type_hints(fun) if callable(fun) else {}
    ret = hints.get("return", None)
    return ex.FullyAsyncApplyExpression(fun, ret, *args, **kwargs)


def declare_type(target_type: Any, col: Any) -> ex.DeclareTypeExpression:
    return ex.DeclareTypeExpression(ex.wrap_expr(col), target_type)


def cast(target_type: Any, col: Any) -> ex.CastExpression:
    return ex.CastExpression(ex.wrap_expr(col), target_type)


def coalesce(*args: Any) -> ex.CoalesceExpression:
    return ex.CoalesceExpression(*args)


def require(val: Any, *deps: Any) -> ex.RequireExpression:
    return ex.RequireExpression(val, *deps)


def if_else(if_clause: Any, then_clause: Any, else_clause: Any) -> ex.IfElseExpression:
    return ex.IfElseExpression(if_clause, then_clause, else_clause)


def make_tuple(*args: Any) -> ex.MakeTupleExpression:
    return ex.MakeTupleExpression(*args)


def unwrap(col: Any) -> ex.UnwrapExpression:
    return ex.UnwrapExpression(col)


def fill_error(col: Any, replacement: Any) -> ex.FillErrorExpression:
    return ex.FillErrorExpression(col, replacement)


class UDF:
    """Base class for user-defined functions (reference internals/udfs).

    Subclass and define __wrapped__, or use the @udf decorator.
    """

    def __init__(
        self,
        *,
        return_type: Any = None,
        deterministic: bool = False,
        propagate_none: bool = False,
        executor: Any = None,
        cache_strategy: Any = None,
        max_batch_size: int | None = None,
    ):
        self.return_type = return_type
        self.deterministic = deterministic
        self.propagate_none = propagate_none
        self.executor = executor
        self.cache_strategy = cache_strategy
        self.max_batch_size = max_batch_size

    def __wrapped__(self, *args: Any, **kwargs: Any) -> Any:
        raise NotImplementedError

    def _resolve_return_type(self) -> Any:
        if self.return_type is not None:
            return self.return_type
        import typing

        try:
            hints = typing.get_type_hints(self.__wrapped__)
            return hints.get("return", None)
        except Exception:
            return None

    def __call__(self, *args: Any, **kwargs: Any) -> ex.ColumnExpression:
        import asyncio
        import inspect

        fun = self.__wrapped__
        ret = self._resolve_return_type()
        is_async = inspect.iscoroutinefunction(fun)
        fun2 = fun
        if self.cache_strategy is not None:
            fun2 = self.cache_strategy.wrap(fun)
        from pathway_amd.udfs import AsyncExecutor, FullyAsyncExecutor

        if isinstance(self.executor, AsyncExecutor):
            wrapped = _apply_executor_options(fun2, self.executor)
            cls = (
                ex.FullyAsyncApplyExpression
                if isinstance(self.executor, FullyAsyncExecutor)
                else ex.AsyncApplyExpression
            )
            return cls(wrapped, ret, *args, propagate_none=self.propagate_none, **kwargs)
        if is_async:
            return ex.AsyncApplyExpression(
                fun2, ret, *args, propagate_none=self.propagate_none, **kwargs
            )
        return ex.ApplyExpression(
            fun2,
            ret,
            *args,
            propagate_none=self.propagate_none,
            deterministic=self.deterministic,
            max_batch_size=self.max_batch_size,
            **kwargs,
        )


def _apply_executor_options(fun: Callable, executor) -> Callable:
    """Wrap `fun` per the AsyncExecutor knobs (reference udfs/executors.py:
    capacity semaphore, per-attempt timeout, retry strategy)."""
    import asyncio
    import inspect

    sems: dict[int, Any] = {}

    async def awrap(*a: Any, **k: Any):
        async def call():
            r = fun(*a, **k)
            return await r if inspect.isawaitable(r) else r

        async def timed():
            if executor.timeout is not None:
                return await asyncio.wait_for(call(), executor.timeout)
            return await call()

        async def retried():
            if executor.retry_strategy is not None:
                return await executor.retry_strategy.invoke(timed)
            return await timed()

        if executor.capacity:
            loop = asyncio.get_running_loop()
            sem = sems.get(id(loop))
            if sem is None:
                sem = sems[id(loop)] = asyncio.Semaphore(executor.capacity)
            async with sem:
                return await retried()
        return await retried()

    functools.update_wrapper(awrap, fun, updated=())
    return awrap


def udf(
    fun: Callable | None = None,
    /,
    *,
    return_type: Any = None,
    deterministic: bool = False,
    propagate_none: bool = False,
    executor: Any = None,
    cache_strategy: Any = None,
    max_batch_size: int | None = None,
):
    """@pw.udf decorator."""

    def wrapper(f: Callable) -> UDF:
        u = UDF(
            return_type=return_type,
            deterministic=deterministic,
            propagate_none=propagate_none,
            executor=executor,
            cache_strategy=cache_strategy,
            max_batch_size=max_batch_size,
        )
        u.__wrapped__ = f  # type: ignore[method-assign]
        functools.update_wrapper(u, f, updated=())
        return u

    if fun is None:
        return wrapper
    return wrapper(fun)


def table_transformer(
    func: Callable | None = None,
    /,
    *,
    allow_superset: Any = True,
    ignore_primary_keys: Any = True,
    locals: Any = None,
):
    """Decorator marking a function as a Table -> Table transformer."""

    def wrapper(f: Callable) -> Callable:
        return f

    if func is None:
        return wrapper
    return wrapper(func)
