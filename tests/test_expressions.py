"""Expression-method coverage: str/dt/num namespaces, Json access, tuples,
casts (modeled on the reference's test_expressions / test_datetimes suites).
"""

import math
from typing import Optional

import pytest

import pathway_amd as pw
from pathway_amd.debug import (
    assert_table_equality,
    assert_table_equality_wo_index,
    table_from_markdown as T,
    table_from_rows,
    table_to_dicts,
)
from pathway_amd.internals.schema import schema_from_types


def _one_col(table, name):
    _, cols = table_to_dicts(table)
    return sorted(cols[name].values())


def test_str_methods_basic():
    t = T(
        """
        s
        __Hello__
        world
        """
    )
    res = t.select(
        up=pw.this.s.str.upper(),
        low=pw.this.s.str.lower(),
        stripped=pw.this.s.str.strip("_"),
        n=pw.this.s.str.len(),
    )
    _, cols = table_to_dicts(res)
    rows = sorted(
        zip(cols["up"].values(), cols["low"].values(), cols["stripped"].values(), cols["n"].values())
    )
    assert rows == [
        ("WORLD", "world", "world", 5),
        ("__HELLO__", "__hello__", "Hello", 9),
    ]


def test_str_find_replace_count():
    t = table_from_rows(schema_from_types(s=str), [("abcabca",)])
    res = t.select(
        f=pw.this.s.str.find("bc"),
        rf=pw.this.s.str.rfind("bc"),
        c=pw.this.s.str.count("a"),
        rep=pw.this.s.str.replace("a", "X"),
        rev=pw.this.s.str.reversed(),
    )
    _, cols = table_to_dicts(res)
    (f,), (rf,), (c,), (rep,), (rev,) = (
        cols["f"].values(), cols["rf"].values(), cols["c"].values(),
        cols["rep"].values(), cols["rev"].values(),
    )
    assert (f, rf, c, rep, rev) == (1, 4, 3, "XbcXbcX", "acbacba")


def test_str_affixes_justify():
    t = table_from_rows(schema_from_types(s=str), [("prefix_core_suffix",)])
    res = t.select(
        a=pw.this.s.str.removeprefix("prefix_"),
        b=pw.this.s.str.removesuffix("_suffix"),
        sw=pw.this.s.str.startswith("pre"),
        ew=pw.this.s.str.endswith("fix"),
        lj=pw.this.s.str.slice(0, 3).str.ljust(5, "."),
        rj=pw.this.s.str.slice(0, 3).str.rjust(5, "."),
    )
    _, cols = table_to_dicts(res)
    get = lambda n: next(iter(cols[n].values()))
    assert get("a") == "core_suffix"
    assert get("b") == "prefix_core"
    assert get("sw") is True and get("ew") is True
    assert get("lj") == "pre.." and get("rj") == "..pre"


def test_str_parse_and_split():
    t = table_from_rows(schema_from_types(s=str), [("10,2.5,true",)])
    parts = t.select(p=pw.this.s.str.split(","))
    res = parts.select(
        i=pw.this.p.get(0).str.parse_int(),
        f=pw.this.p.get(1).str.parse_float(),
        b=pw.this.p.get(2).str.parse_bool(),
    )
    _, cols = table_to_dicts(res)
    get = lambda n: next(iter(cols[n].values()))
    assert get("i") == 10 and get("f") == 2.5 and get("b") is True


def test_str_title_swapcase_bytes():
    t = table_from_rows(schema_from_types(s=str), [("hello World",)])
    res = t.select(
        ti=pw.this.s.str.title(),
        sc=pw.this.s.str.swap_case(),
        by=pw.this.s.str.to_bytes(),
    )
    _, cols = table_to_dicts(res)
    get = lambda n: next(iter(cols[n].values()))
    assert get("ti") == "Hello World"
    assert get("sc") == "HELLO wORLD"
    assert get("by") == b"hello World"


def test_datetime_parse_fields_format():
    t = table_from_rows(schema_from_types(s=str), [("2023-03-25 12:34:56",)])
    d = t.select(ts=pw.this.s.dt.strptime("%Y-%m-%d %H:%M:%S"))
    res = d.select(
        y=pw.this.ts.dt.year(),
        mo=pw.this.ts.dt.month(),
        da=pw.this.ts.dt.day(),
        h=pw.this.ts.dt.hour(),
        mi=pw.this.ts.dt.minute(),
        se=pw.this.ts.dt.second(),
        wd=pw.this.ts.dt.weekday(),
        s2=pw.this.ts.dt.strftime("%Y/%m/%d"),
    )
    _, cols = table_to_dicts(res)
    get = lambda n: next(iter(cols[n].values()))
    assert (get("y"), get("mo"), get("da")) == (2023, 3, 25)
    assert (get("h"), get("mi"), get("se")) == (12, 34, 56)
    assert get("wd") == 5  # Saturday
    assert get("s2") == "2023/03/25"


def test_datetime_round_floor_timestamp():
    t = table_from_rows(schema_from_types(s=str), [("2023-03-25 12:34:56",)])
    d = t.select(ts=pw.this.s.dt.strptime("%Y-%m-%d %H:%M:%S"))
    res = d.select(
        fl=pw.this.ts.dt.floor("1h").dt.strftime("%H:%M"),
        ro=pw.this.ts.dt.round("1h").dt.strftime("%H:%M"),
        unix=pw.this.ts.dt.timestamp(unit="s"),
    )
    _, cols = table_to_dicts(res)
    get = lambda n: next(iter(cols[n].values()))
    assert get("fl") == "12:00"
    assert get("ro") == "13:00"
    assert get("unix") == 1679747696.0


def test_duration_components():
    t = T(
        """
        a | b
        1 | 2
        """
    )
    d = t.select(
        t1=pw.cast(str, "2023-01-01 00:00:00").dt.strptime("%Y-%m-%d %H:%M:%S"),
        t2=pw.cast(str, "2023-01-02 03:00:30").dt.strptime("%Y-%m-%d %H:%M:%S"),
    )
    res = d.select(
        days=(pw.this.t2 - pw.this.t1).dt.days(),
        hours=(pw.this.t2 - pw.this.t1).dt.hours(),
        secs=(pw.this.t2 - pw.this.t1).dt.seconds(),
    )
    _, cols = table_to_dicts(res)
    get = lambda n: next(iter(cols[n].values()))
    assert get("days") == 1
    assert get("hours") == 27
    assert get("secs") == 27 * 3600 + 30


def test_from_timestamp():
    t = table_from_rows(schema_from_types(x=int), [(1679747696,)])
    res = t.select(
        s=pw.this.x.dt.from_timestamp(unit="s").dt.strftime("%Y-%m-%d %H:%M:%S")
    )
    assert _one_col(res, "s") == ["2023-03-25 12:34:56"]


def test_num_namespace():
    t = table_from_rows(
        schema_from_types(x=float, y=Optional[float]),
        [(-2.567, None), (1.234, 5.0)],
    )
    res = t.select(
        a=pw.this.x.num.abs(),
        r=pw.this.x.num.round(2),
        f=pw.this.y.num.fill_na(0.0),
    )
    _, cols = table_to_dicts(res)
    rows = sorted(zip(cols["a"].values(), cols["r"].values(), cols["f"].values()))
    assert rows == [(1.234, 1.23, 5.0), (2.567, -2.57, 0.0)]


def test_json_access():
    import json as _json

    t = table_from_rows(schema_from_types(s=str), [('{"a": 1, "b": {"c": [10, 20]}, "s": "x"}',)])
    j = t.select(j=pw.apply_with_type(lambda s: _json.loads(s), pw.Json, pw.this.s))
    res = j.select(
        a=pw.this.j["a"].as_int(),
        c1=pw.this.j["b"]["c"][1].as_int(),
        s=pw.this.j["s"].as_str(),
        missing=pw.this.j.get("zzz"),
    )
    _, cols = table_to_dicts(res)
    get = lambda n: next(iter(cols[n].values()))
    assert get("a") == 1 and get("c1") == 20 and get("s") == "x"
    assert get("missing") is None


def test_make_tuple_and_get():
    t = T(
        """
        a | b
        1 | 2
        3 | 4
        """
    )
    res = t.select(tup=pw.make_tuple(pw.this.a, pw.this.b * 10))
    res2 = res.select(x=pw.this.tup.get(0), y=pw.this.tup.get(1), z=pw.this.tup.get(5, -1))
    expected = T(
        """
        x | y  | z
        1 | 20 | -1
        3 | 40 | -1
        """
    )
    assert_table_equality(res2, expected)


def test_cast_declare_unwrap_require():
    t = table_from_rows(
        schema_from_types(x=Optional[int], s=str), [(5, "7"), (None, "8")]
    )
    res = t.select(
        c=pw.cast(float, pw.this.s.str.parse_int()),
        u=pw.unwrap(pw.coalesce(pw.this.x, 0)),
        r=pw.require(pw.this.x + 1, pw.this.x),
    )
    _, cols = table_to_dicts(res)
    rows = sorted(zip(cols["c"].values(), cols["u"].values(), cols["r"].values()),
                  key=lambda r: r[0])
    assert rows == [(7.0, 5, 6), (8.0, 0, None)]


def test_if_else_chain_and_bools():
    t = T(
        """
        a
        -2
        0
        3
        """
    )
    res = t.select(
        sign=pw.if_else(pw.this.a > 0, 1, pw.if_else(pw.this.a < 0, -1, 0)),
        both=(pw.this.a >= 0) & (pw.this.a < 3),
        either=(pw.this.a < 0) | (pw.this.a >= 3),
        inv=~(pw.this.a == 0),
    )
    _, cols = table_to_dicts(res)
    rows = sorted(zip(cols["sign"].values(), cols["both"].values(),
                      cols["either"].values(), cols["inv"].values()))
    assert rows == [(-1, False, True, True), (0, True, False, False), (1, False, True, True)]


def test_to_string_and_is_none():
    t = table_from_rows(schema_from_types(x=Optional[int]), [(1,), (None,)])
    res = t.select(
        s=pw.this.x.to_string(),
        none=pw.this.x.is_none(),
        some=pw.this.x.is_not_none(),
    )
    _, cols = table_to_dicts(res)
    rows = sorted(
        zip(cols["s"].values(), cols["none"].values(), cols["some"].values()),
        key=lambda r: str(r[0]),
    )
    assert rows == [("1", False, True), (None, True, False)]


def test_int_float_arithmetic_matrix():
    t = T(
        """
        a | b
        7 | 2
        """
    )
    res = t.select(
        q=pw.this.a // pw.this.b,
        r=pw.this.a % pw.this.b,
        p=pw.this.a ** pw.this.b,
        tdiv=pw.this.a / pw.this.b,
        neg=-pw.this.a,
        xor=pw.this.a ^ pw.this.b,
    )
    _, cols = table_to_dicts(res)
    get = lambda n: next(iter(cols[n].values()))
    assert (get("q"), get("r"), get("p")) == (3, 1, 49)
    assert get("tdiv") == 3.5 and get("neg") == -7 and get("xor") == 5


def test_dt_timezone_aware_arithmetic_and_conversions():
    """Reference date_time.py:855-1600: wall-clock add/subtract in a time
    zone (DST-aware), to_duration, weeks, utc_from_timestamp, swapcase."""
    import pandas as pd

    from pathway_amd.internals.rungraph import G

    G.clear()
    t = T(
        """
        s   | n
        AbC | 3
        """
    )
    r = t.select(
        sw=t.s.str.swapcase(),
        dur=t.n.dt.to_duration("h"),
        wk=pw.this.n.dt.to_duration("W").dt.weeks(),
        utc=t.n.dt.utc_from_timestamp("s"),
    )
    _k, c = table_to_dicts(r)
    assert list(c["sw"].values())[0] == "aBc"
    assert pd.Timedelta(list(c["dur"].values())[0]) == pd.Timedelta(hours=3)
    assert list(c["wk"].values())[0] == 3
    assert pd.Timestamp(list(c["utc"].values())[0]) == pd.Timestamp(
        3, unit="s", tz="UTC"
    )
    G.clear()
    t2 = T(
        """
        x
        1
        """
    )
    r2 = t2.select(
        a=pw.apply_with_type(
            lambda _: pd.Timestamp("2024-03-10 01:30"),
            pw.DateTimeNaive,
            pw.this.x,
        )
    ).select(
        plus=pw.this.a.dt.add_duration_in_timezone(
            pd.Timedelta(hours=1), "America/New_York"
        ),
        minus=pw.this.a.dt.subtract_duration_in_timezone(
            pd.Timedelta(hours=1), "America/New_York"
        ),
        diff=pw.this.a.dt.subtract_date_time_in_timezone(
            pd.Timestamp("2024-03-10 03:30"), "America/New_York"
        ),
    )
    _k2, c2 = table_to_dicts(r2)
    # 2024-03-10 02:00 does not exist in New York (DST spring forward):
    # one wall-clock hour past 01:30 lands at 03:30
    assert pd.Timestamp(list(c2["plus"].values())[0]) == pd.Timestamp(
        "2024-03-10 03:30"
    )
    assert pd.Timestamp(list(c2["minus"].values())[0]) == pd.Timestamp(
        "2024-03-10 00:30"
    )
    # 01:30 -> 03:30 spans ONE absolute hour across the gap
    assert pd.Timedelta(list(c2["diff"].values())[0]) == pd.Timedelta(hours=-1)
