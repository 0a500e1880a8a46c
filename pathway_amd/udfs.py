"""pw.udfs — executors, retry/cache strategies
(reference internals/udfs/__init__.py:68-330 surface)."""

from __future__ import annotations

import asyncio
import functools
import os
import pickle
import time
from dataclasses import dataclass
from typing import Any, Callable


class Executor:
    pass


@dataclass
class SyncExecutor(Executor):
    pass


@dataclass
class AsyncExecutor(Executor):
    capacity: int | None = None
    timeout: float | None = None
    retry_strategy: Any = None


@dataclass
class FullyAsyncExecutor(AsyncExecutor):
    autocommit_duration_ms: int | None = 1500


def sync_executor() -> SyncExecutor:
    return SyncExecutor()


def async_executor(
    capacity: int | None = None,
    timeout: float | None = None,
    retry_strategy: Any = None,
) -> AsyncExecutor:
    return AsyncExecutor(capacity, timeout, retry_strategy)


def fully_async_executor(
    capacity: int | None = None,
    timeout: float | None = None,
    retry_strategy: Any = None,
    autocommit_duration_ms: int | None = 1500,
) -> FullyAsyncExecutor:
    return FullyAsyncExecutor(capacity, timeout, retry_strategy, autocommit_duration_ms)


def auto_executor() -> Executor:
    return SyncExecutor()


class AsyncRetryStrategy:
    async def invoke(self, fun, *args, **kwargs):
        return await fun(*args, **kwargs)


class NoRetryStrategy(AsyncRetryStrategy):
    pass


@dataclass
class FixedDelayRetryStrategy(AsyncRetryStrategy):
    max_retries: int = 3
    delay_ms: int = 1000

    async def invoke(self, fun, *args, **kwargs):
        last: Exception | None = None
        for _ in range(self.max_retries):
            try:
                return await fun(*args, **kwargs)
            except Exception as e:  # noqa: BLE001
                last = e
                await asyncio.sleep(self.delay_ms / 1000)
        raise last  # type: ignore[misc]


@dataclass
class ExponentialBackoffRetryStrategy(AsyncRetryStrategy):
    max_retries: int = 3
    initial_delay: int = 1000
    backoff_factor: float = 2.0
    jitter_ms: int = 300

    async def invoke(self, fun, *args, **kwargs):
        delay = self.initial_delay
        last: Exception | None = None
        for _ in range(self.max_retries):
            try:
                return await fun(*args, **kwargs)
            except Exception as e:  # noqa: BLE001
                last = e
                await asyncio.sleep(delay / 1000)
                delay = int(delay * self.backoff_factor)
        raise last  # type: ignore[misc]


class CacheStrategy:
    def wrap(self, fun: Callable) -> Callable:
        return fun


class InMemoryCache(CacheStrategy):
    def wrap(self, fun: Callable) -> Callable:
        cache: dict[Any, Any] = {}

        @functools.wraps(fun)
        def wrapper(*args, **kwargs):
            key = pickle.dumps((args, sorted(kwargs.items())))
            if key not in cache:
                cache[key] = fun(*args, **kwargs)
            return cache[key]

        return wrapper


class DiskCache(CacheStrategy):
    def __init__(self, directory: str | None = None, name: str | None = None):
        self._directory = directory
        self.name = name

    @property
    def directory(self) -> str:
        # resolved lazily so PersistenceMode.UDF_CACHING (which points the
        # env var at the persistence backend) takes effect even when the
        # UDF was declared before pw.run()
        return self._directory or os.environ.get(
            "PATHWAY_PERSISTENT_STORAGE", "/tmp/pw_udf_cache"
        )

    def wrap(self, fun: Callable) -> Callable:
        prefix = self.name or getattr(fun, "__name__", "udf")

        @functools.wraps(fun)
        def wrapper(*args, **kwargs):
            from pathway_amd.internals.api import hash128

            key = pickle.dumps((args, sorted(kwargs.items())))
            lo, hi = hash128(key)
            path = os.path.join(self.directory, f"{prefix}_{lo:016x}{hi:016x}.pkl")
            if os.path.exists(path):
                with open(path, "rb") as f:
                    return pickle.load(f)
            result = fun(*args, **kwargs)
            os.makedirs(self.directory, exist_ok=True)
            with open(path, "wb") as f:
                pickle.dump(result, f)
            return result

        return wrapper


DefaultCache = InMemoryCache


async def coerce_async(fun):
    return fun


def with_capacity(fun, capacity: int):
    return fun


def with_timeout(fun, timeout: float):
    return fun


def async_options(**kwargs):
    def decorator(fun):
        return fun

    return decorator


from pathway_amd.internals.common import UDF, udf  # noqa: E402


def with_cache_strategy(fun, cache_strategy: CacheStrategy):
    """Wrap a callable with a cache strategy (reference udfs surface)."""
    return cache_strategy.wrap(fun)


def with_retry_strategy(fun, retry_strategy: AsyncRetryStrategy):
    """Wrap an async callable with a retry strategy."""

    async def wrapped(*args, **kwargs):
        async def call():
            return await fun(*args, **kwargs)

        return await retry_strategy.invoke(call)

    return wrapped
