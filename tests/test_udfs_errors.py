"""UDF system, custom reducers, error propagation (reference test_udfs.py +
test_error_messages patterns)."""

import pytest

import pathway_amd as pw
from pathway_amd.debug import (
    assert_table_equality,
    assert_table_equality_wo_index,
    table_from_markdown as T,
)


def test_udf_cache_strategy():
    calls = [0]

    @pw.udf(cache_strategy=pw.udfs.InMemoryCache(), deterministic=True)
    def slow_double(x: int) -> int:
        calls[0] += 1
        return 2 * x

    t = T(
        """
        a
        1
        1
        2
        """
    )
    res = t.select(b=slow_double(pw.this.a))
    keys, cols = pw.debug.table_to_dicts(res)
    assert sorted(cols["b"].values()) == [2, 2, 4]
    assert calls[0] == 2  # cached across identical inputs


def test_async_udf():
    @pw.udf
    async def aincr(x: int) -> int:
        import asyncio

        await asyncio.sleep(0.001)
        return x + 1

    t = T(
        """
        a
        1
        2
        """
    )
    res = t.select(b=aincr(pw.this.a))
    assert_table_equality(
        res,
        T(
            """
            b
            2
            3
            """
        ),
    )


def test_stateful_many_reducer():
    def combine(state, rows):
        total = 0
        for row, cnt in rows:
            total += row[0] * cnt
        return total * 10

    ssum = pw.reducers.stateful_many(combine)
    t = T(
        """
        g | v
        a | 1
        a | 2
        b | 5
        """
    )
    res = t.groupby(pw.this.g).reduce(pw.this.g, s=ssum(pw.this.v))
    keys, cols = pw.debug.table_to_dicts(res)
    got = {cols["g"][k]: cols["s"][k] for k in keys}
    assert got == {"a": 30, "b": 50}


def test_custom_accumulator_reducer():
    class StdDevAcc(pw.BaseCustomAccumulator):
        def __init__(self, cnt, s, s2):
            self.cnt, self.s, self.s2 = cnt, s, s2

        @classmethod
        def from_row(cls, row):
            (v,) = row
            return cls(1, v, v * v)

        def update(self, other):
            self.cnt += other.cnt
            self.s += other.s
            self.s2 += other.s2

        def compute_result(self) -> float:
            mean = self.s / self.cnt
            return self.s2 / self.cnt - mean * mean

    stddev = pw.internals.custom_reducers.udf_reducer(StdDevAcc)
    import pathway_amd.internals.custom_reducers  # noqa: F401

    from pathway_amd.internals.custom_reducers import udf_reducer

    stddev = udf_reducer(StdDevAcc)
    t = T(
        """
        v
        1
        3
        """
    )
    res = t.groupby().reduce(var=stddev(pw.this.v))
    keys, cols = pw.debug.table_to_dicts(res)
    assert list(cols["var"].values()) == [1.0]


def test_error_value_and_fill_error():
    t = T(
        """
        a | b
        6 | 2
        4 | 0
        """
    )
    res = t.select(q=pw.fill_error(pw.this.a // pw.this.b, -1))
    assert_table_equality_wo_index(
        res,
        T(
            """
            q
            3
            -1
            """
        ),
    )


def test_udf_error_recorded():
    from pathway_amd.internals import errors

    errors._global_error_rows.clear()

    @pw.udf
    def boom(x: int) -> int:
        raise ValueError("kaboom")

    t = T(
        """
        a
        1
        """
    )
    res = t.select(b=pw.fill_error(boom(pw.this.a), 0))
    keys, cols = pw.debug.table_to_dicts(res)
    assert list(cols["b"].values()) == [0]
    assert any("kaboom" in m for m, _ in errors._global_error_rows)


def test_unwrap_and_require():
    t = T(
        """
        a | b
        1 | 5
        2 |
        """
    )
    res = t.select(c=pw.require(pw.this.a * 10, pw.this.b))
    keys, cols = pw.debug.table_to_dicts(res)
    assert sorted(str(v) for v in cols["c"].values()) == ["10", "None"]


def test_expression_namespaces():
    t = T(
        """
        s     | x
        Hello | 2.7
        """
    )
    res = t.select(
        up=pw.this.s.str.upper(),
        swap=pw.this.s.str.swap_case(),
        l=pw.this.s.str.len(),
        r=pw.this.x.num.round(1),
        a=(-pw.this.x).num.abs(),
    )
    keys, cols = pw.debug.table_to_dicts(res)
    k = keys[0]
    assert cols["up"][k] == "HELLO"
    assert cols["swap"][k] == "hELLO"
    assert cols["l"][k] == 5
    assert abs(cols["r"][k] - 2.7) < 1e-9
    assert abs(cols["a"][k] - 2.7) < 1e-9


def test_datetime_namespace():
    import pandas as pd

    t = pw.debug.table_from_pandas(
        pd.DataFrame({"ts": [pd.Timestamp("2024-03-05 10:30:00")]})
    )
    res = t.select(
        y=pw.this.ts.dt.year(),
        m=pw.this.ts.dt.month(),
        d=pw.this.ts.dt.day(),
        hh=pw.this.ts.dt.hour(),
    )
    keys, cols = pw.debug.table_to_dicts(res)
    k = keys[0]
    assert (cols["y"][k], cols["m"][k], cols["d"][k], cols["hh"][k]) == (2024, 3, 5, 10)


def test_json_column():
    from pathway_amd.internals.json import Json

    t = pw.debug.table_from_rows(
        pw.schema_from_types(j=dict),
        [(Json({"a": {"b": 7}, "l": [1, 2, 3]}),)],
    )
    res = t.select(
        b=pw.this.j["a"]["b"].as_int(),
        l1=pw.this.j["l"][1].as_int(),
    )
    keys, cols = pw.debug.table_to_dicts(res)
    k = keys[0]
    assert cols["b"][k] == 7
    assert cols["l1"][k] == 2


def test_terminate_on_error_raises_at_sink():
    from pathway_amd.internals.config import pathway_config

    @pw.udf
    def boom(x: int) -> int:
        raise ValueError("nope")

    t = T(
        """
        a
        1
        """
    )
    res = t.select(b=pw.declare_type(pw.Type.ANY, boom(pw.this.a)))
    assert pathway_config.terminate_on_error
    with pytest.raises(RuntimeError, match="Error value"):
        pw.debug.table_to_dicts(res)


def test_global_error_log_records_udf_failures():
    from pathway_amd.internals.errors import _global_error_rows

    before = len(_global_error_rows)

    @pw.udf
    def boom(x: int) -> int:
        if x == 2:
            raise ValueError("bad row 2")
        return x

    t = T(
        """
        a
        1
        2
        """
    )
    res = t.select(b=pw.fill_error(boom(pw.this.a), -1))
    _, cols = pw.debug.table_to_dicts(res)
    assert sorted(cols["b"].values()) == [-1, 1]
    log = pw.global_error_log()
    _, lcols = pw.debug.table_to_dicts(log)
    msgs = list(lcols["message"].values())
    assert any("bad row 2" in m for m in msgs)
    assert len(msgs) > before


def test_iterate_iteration_limit():
    def step(t):
        return t.select(v=pw.if_else(pw.this.v < 100, pw.this.v * 2, pw.this.v))

    t = T(
        """
        v
        1
        """
    )
    limited = pw.iterate(step, iteration_limit=3, t=t)
    _, cols = pw.debug.table_to_dicts(limited)
    assert list(cols["v"].values()) == [8]  # 3 doublings only
