"""Additional pw.sql coverage (reference sql tests: expressions, grouping,
joins, unions)."""

import pytest

import pathway_amd as pw
from pathway_amd.debug import table_from_markdown as T, table_to_dicts


def _rows(table, *names):
    _, cols = table_to_dicts(table)
    ids = list(cols[names[0]].keys())
    return sorted(tuple(cols[n][i] for n in names) for i in ids)


def test_sql_where_and_or_not():
    t = T(
        """
        a | b
        1 | 10
        2 | 20
        3 | 30
        4 | 40
        """
    )
    r = pw.sql("SELECT a FROM tab WHERE (a > 1 AND b < 40) OR a = 4", tab=t)
    assert _rows(r, "a") == [(2,), (3,), (4,)]


def test_sql_arithmetic_parens():
    t = T(
        """
        x
        2
        5
        """
    )
    r = pw.sql("SELECT x, (x + 1) * 3 AS y FROM tab", tab=t)
    assert _rows(r, "x", "y") == [(2, 9), (5, 18)]


def test_sql_group_by_having():
    t = T(
        """
        g | v
        a | 1
        a | 2
        b | 5
        b | 7
        c | 1
        """
    )
    r = pw.sql(
        "SELECT g, SUM(v) AS s, COUNT(*) AS n FROM tab GROUP BY g HAVING SUM(v) > 2",
        tab=t,
    )
    assert _rows(r, "g", "s", "n") == [("a", 3, 2), ("b", 12, 2)]


def test_sql_union_all():
    t1 = T(
        """
        a
        1
        """
    )
    t2 = T(
        """
        a
        2
        """
    )
    r = pw.sql("SELECT a FROM t1 UNION ALL SELECT a FROM t2", t1=t1, t2=t2)
    assert _rows(r, "a") == [(1,), (2,)]


def test_sql_join_on():
    l = T(
        """
        k | v
        1 | x
        2 | y
        """
    )
    rt = T(
        """
        k | w
        1 | 10
        3 | 30
        """
    )
    r = pw.sql("SELECT l.v AS v, r.w AS w FROM l JOIN r ON l.k = r.k", l=l, r=rt)
    assert _rows(r, "v", "w") == [("x", 10)]
