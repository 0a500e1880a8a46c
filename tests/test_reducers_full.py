"""Every reducer, including retraction behavior under update streams
(modeled on the reference's test_reducers.py coverage)."""

import numpy as np
import pytest

import pathway_amd as pw
from pathway_amd.debug import (
    assert_table_equality,
    assert_table_equality_wo_index,
    table_from_markdown as T,
    table_to_dicts,
)


def _rows(table, *names):
    _, cols = table_to_dicts(table)
    ids = list(cols[names[0]].keys())
    return sorted(tuple(cols[n][i] for n in names) for i in ids)


GRID = """
g | v | w
a | 3 | 1.0
a | 1 | 2.0
a | 2 | 4.0
b | 5 | 8.0
b | 4 | 16.0
"""


def test_min_max_sum_avg_count():
    t = T(GRID)
    r = t.groupby(pw.this.g).reduce(
        pw.this.g,
        mn=pw.reducers.min(pw.this.v),
        mx=pw.reducers.max(pw.this.v),
        sm=pw.reducers.sum(pw.this.v),
        av=pw.reducers.avg(pw.this.w),
        ct=pw.reducers.count(),
    )
    assert _rows(r, "g", "mn", "mx", "sm", "av", "ct") == [
        ("a", 1, 3, 6, 7.0 / 3, 3),
        ("b", 4, 5, 9, 12.0, 2),
    ]


def test_argmin_argmax_point_back_to_rows():
    t = T(GRID)
    r = t.groupby(pw.this.g).reduce(
        pw.this.g,
        lo=pw.reducers.argmin(pw.this.v),
        hi=pw.reducers.argmax(pw.this.v),
    )
    # argmin/argmax return row pointers; dereference through ix
    r2 = r.select(
        pw.this.g,
        lo_v=t.ix(r.lo).v,
        hi_v=t.ix(r.hi).v,
    )
    assert _rows(r2, "g", "lo_v", "hi_v") == [("a", 1, 3), ("b", 4, 5)]


def test_any_unique_count_distinct():
    t = T(
        """
        g | v | u
        a | 1 | 9
        a | 1 | 9
        a | 2 | 9
        b | 7 | 8
        """
    )
    r = t.groupby(pw.this.g).reduce(
        pw.this.g,
        anyv=pw.reducers.any(pw.this.v),
        uniq=pw.reducers.unique(pw.this.u),
        nd=pw.reducers.count_distinct(pw.this.v),
        nda=pw.reducers.count_distinct_approximate(pw.this.v),
    )
    rows = _rows(r, "g", "anyv", "uniq", "nd", "nda")
    assert [x[:2][0] for x in rows] == ["a", "b"]
    (ga, gb) = rows
    assert ga[1] in (1, 2) and ga[2] == 9 and ga[3] == 2 and ga[4] == 2
    assert gb[1] == 7 and gb[2] == 8 and gb[3] == 1 and gb[4] == 1


def test_tuple_sorted_tuple_ndarray():
    t = T(GRID)
    r = t.groupby(pw.this.g).reduce(
        pw.this.g,
        st=pw.reducers.sorted_tuple(pw.this.v),
        nd=pw.reducers.ndarray(pw.this.w),
    )
    _, cols = table_to_dicts(r)
    byg = {cols["g"][i]: (cols["st"][i], cols["nd"][i]) for i in cols["g"]}
    assert byg["a"][0] == (1, 2, 3)
    assert byg["b"][0] == (4, 5)
    assert sorted(byg["a"][1].tolist()) == [1.0, 2.0, 4.0]
    assert isinstance(byg["a"][1], np.ndarray)


def test_tuple_reducer_preserves_multiset():
    t = T(
        """
        g | v
        a | 5
        a | 5
        a | 6
        """
    )
    r = t.groupby(pw.this.g).reduce(
        pw.this.g, tup=pw.reducers.tuple(pw.this.v)
    )
    _, cols = table_to_dicts(r)
    (tup,) = cols["tup"].values()
    assert sorted(tup) == [5, 5, 6]


def test_earliest_latest_over_stream():
    t = T(
        """
        g | v | __time__
        a | 1 |    2
        a | 2 |    4
        a | 3 |    6
        """
    )
    r = t.groupby(pw.this.g).reduce(
        pw.this.g,
        first=pw.reducers.earliest(pw.this.v),
        last=pw.reducers.latest(pw.this.v),
    )
    assert _rows(r, "g", "first", "last") == [("a", 1, 3)]


def test_min_retraction_stream():
    # deleting the current minimum must promote the next value
    t = T(
        """
        g | v | __time__ | __diff__
        a | 1 |    2     |    1
        a | 2 |    2     |    1
        a | 1 |    4     |   -1
        """,
        id_from=["g", "v"],
    )
    r = t.groupby(pw.this.g).reduce(pw.this.g, mn=pw.reducers.min(pw.this.v))
    assert _rows(r, "g", "mn") == [("a", 2)]


def test_stateful_single():
    @pw.reducers.stateful_single
    def max_len(state, val):
        return max(state or 0, len(val))

    t = T(
        """
        g | w
        a | xx
        a | yyyy
        b | z
        """
    )
    r = t.groupby(pw.this.g).reduce(pw.this.g, ml=max_len(pw.this.w))
    assert _rows(r, "g", "ml") == [("a", 4), ("b", 1)]


def test_stateful_many():
    @pw.reducers.stateful_many
    def counter(state, rows):
        s = state or 0
        for row, cnt in rows:
            s += cnt * row[0]
        return s

    t = T(
        """
        g | v
        a | 2
        a | 3
        b | 10
        """
    )
    r = t.groupby(pw.this.g).reduce(pw.this.g, s=counter(pw.this.v))
    assert _rows(r, "g", "s") == [("a", 5), ("b", 10)]


def test_udf_reducer_accumulator():
    from pathway_amd.internals.custom_reducers import BaseCustomAccumulator

    class SumSquares(BaseCustomAccumulator):
        def __init__(self, s):
            self.s = s

        @classmethod
        def from_row(cls, row):
            return cls(row[0] ** 2)

        def update(self, other):
            self.s += other.s

        def retract(self, other):
            self.s -= other.s

        def compute_result(self):
            return self.s

    ssq = pw.reducers.udf_reducer(SumSquares)
    t = T(
        """
        g | v | __time__ | __diff__
        a | 3 |    2     |    1
        a | 4 |    2     |    1
        a | 3 |    4     |   -1
        """,
        id_from=["g", "v"],
    )
    r = t.groupby(pw.this.g).reduce(pw.this.g, s=ssq(pw.this.v))
    assert _rows(r, "g", "s") == [("a", 16)]


def test_reduce_without_groupby_global():
    t = T(
        """
        v
        1
        2
        3
        """
    )
    r = t.reduce(total=pw.reducers.sum(pw.this.v), n=pw.reducers.count())
    assert _rows(r, "total", "n") == [(6, 3)]


def test_groupby_multiple_keys():
    t = T(
        """
        a | b | v
        1 | x | 10
        1 | y | 20
        1 | x | 30
        2 | x | 40
        """
    )
    r = t.groupby(pw.this.a, pw.this.b).reduce(
        pw.this.a, pw.this.b, s=pw.reducers.sum(pw.this.v)
    )
    assert _rows(r, "a", "b", "s") == [(1, "x", 40), (1, "y", 20), (2, "x", 40)]


def test_sorted_tuple_skip_nones():
    from typing import Optional

    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.schema import schema_from_types

    t = table_from_rows(
        schema_from_types(g=str, v=Optional[int]),
        [("a", 2), ("a", None), ("a", 1)],
    )
    r = t.groupby(pw.this.g).reduce(
        pw.this.g, st=pw.reducers.sorted_tuple(pw.this.v, skip_nones=True)
    )
    _, cols = table_to_dicts(r)
    (st,) = cols["st"].values()
    assert st == (1, 2)


def test_groupby_sort_by_orders_reducers():
    t = T(
        """
        g | v | o
        a | 10 | 3
        a | 20 | 1
        a | 30 | 2
        """
    )
    r = t.groupby(pw.this.g, sort_by=pw.this.o).reduce(
        pw.this.g,
        first=pw.reducers.earliest(pw.this.v),
        last=pw.reducers.latest(pw.this.v),
        tup=pw.reducers.tuple(pw.this.v),
    )
    _, cols = table_to_dicts(r)
    (first,) = cols["first"].values()
    (last,) = cols["last"].values()
    (tup,) = cols["tup"].values()
    # ordered by o: 20 (o=1), 30 (o=2), 10 (o=3)
    assert first == 20 and last == 10
    assert tup == (20, 30, 10)
