"""UDF executor coverage: retries, timeouts, disk cache, fully-async +
await_futures (reference internals/udfs/executors.py + caches.py tests)."""

import asyncio
import time

import pytest

import pathway_amd as pw
from pathway_amd.debug import table_from_markdown as T, table_to_dicts


def _col(table, name):
    _, cols = table_to_dicts(table)
    return sorted(cols[name].values())


def test_async_udf_with_capacity():
    calls = []

    @pw.udf(executor=pw.udfs.async_executor(capacity=2))
    async def slow_double(x: int) -> int:
        calls.append(x)
        await asyncio.sleep(0.01)
        return 2 * x

    t = T(
        """
        a
        1
        2
        3
        4
        """
    )
    res = t.select(b=slow_double(pw.this.a))
    assert _col(res, "b") == [2, 4, 6, 8]
    assert sorted(calls) == [1, 2, 3, 4]


def test_async_udf_retry_strategy():
    attempts = {"n": 0}

    @pw.udf(
        executor=pw.udfs.async_executor(
            retry_strategy=pw.udfs.FixedDelayRetryStrategy(max_retries=4, delay_ms=1)
        )
    )
    async def flaky(x: int) -> int:
        attempts["n"] += 1
        if attempts["n"] < 3:
            raise RuntimeError("transient")
        return x * 10

    t = T(
        """
        a
        7
        """
    )
    res = t.select(b=flaky(pw.this.a))
    assert _col(res, "b") == [70]
    assert attempts["n"] == 3


def test_async_udf_timeout_gives_error():
    @pw.udf(executor=pw.udfs.async_executor(timeout=0.01))
    async def too_slow(x: int) -> int:
        await asyncio.sleep(5)
        return x

    t = T(
        """
        a
        1
        """
    )
    res = t.select(b=too_slow(pw.this.a))
    out = res.select(b=pw.fill_error(pw.this.b, -1))
    assert _col(out, "b") == [-1]


def test_disk_cache_persists_across_runs(tmp_path):
    calls = {"n": 0}

    def make_udf():
        @pw.udf(cache_strategy=pw.udfs.DiskCache(directory=str(tmp_path)))
        def expensive(x: int) -> int:
            calls["n"] += 1
            return x + 100

        return expensive

    f = make_udf()
    t = T(
        """
        a
        5
        6
        """
    )
    assert _col(t.select(b=f(pw.this.a)), "b") == [105, 106]
    first = calls["n"]
    pw.internals.rungraph.G.clear()
    f2 = make_udf()
    t2 = T(
        """
        a
        5
        6
        """
    )
    assert _col(t2.select(b=f2(pw.this.a)), "b") == [105, 106]
    assert calls["n"] == first  # served from the on-disk cache


def test_in_memory_cache_dedupes_calls():
    calls = {"n": 0}

    @pw.udf(cache_strategy=pw.udfs.InMemoryCache())
    def f(x: int) -> int:
        calls["n"] += 1
        return x * 3

    t = T(
        """
        a
        2
        2
        2
        3
        """
    )
    res = t.select(b=f(pw.this.a))
    assert _col(res, "b") == [6, 6, 6, 9]
    assert calls["n"] == 2  # one call per distinct argument


def test_fully_async_udf_and_await_futures():
    @pw.udf(executor=pw.udfs.fully_async_executor())
    async def slow(x: int) -> int:
        await asyncio.sleep(0.01)
        return x + 1

    t = T(
        """
        a
        1
        2
        """
    )
    res = t.select(b=slow(pw.this.a)).await_futures()
    assert _col(res, "b") == [2, 3]


def test_sync_executor_explicit():
    @pw.udf(executor=pw.udfs.sync_executor())
    def inc(x: int) -> int:
        return x + 1

    t = T(
        """
        a
        1
        """
    )
    assert _col(t.select(b=inc(pw.this.a)), "b") == [2]


def test_udf_class_style():
    class Mult(pw.UDF):
        def __init__(self, k):
            super().__init__()
            self.k = k

        def __wrapped__(self, x: int) -> int:
            return self.k * x

    m = Mult(5)
    t = T(
        """
        a
        3
        """
    )
    assert _col(t.select(b=m(pw.this.a)), "b") == [15]


def test_udf_caching_persistence_mode(tmp_path):
    """PersistenceMode.UDF_CACHING routes DiskCache at the backend dir."""
    calls = {"n": 0}

    @pw.udf(cache_strategy=pw.udfs.DiskCache(name="pmodecache"))
    def f(x: int) -> int:
        calls["n"] += 1
        return x * 7

    t = T(
        """
        a
        3
        """
    )
    res = t.select(b=f(pw.this.a))
    out = str(tmp_path / "out.csv")
    pw.io.csv.write(res, out)
    cfg = pw.persistence.Config(
        backend=pw.persistence.Backend.filesystem(str(tmp_path / "snap")),
        persistence_mode=pw.PersistenceMode.UDF_CACHING,
    )
    import os as _os

    try:
        pw.run(monitoring_level=pw.MonitoringLevel.NONE, persistence_config=cfg)
        cached = [
            p for p in (tmp_path / "snap").rglob("pmodecache_*.pkl")
        ]
        assert cached, "UDF result not cached under the persistence backend"
    finally:
        _os.environ.pop("PATHWAY_PERSISTENT_STORAGE", None)
        pw.internals.rungraph.G.clear()
