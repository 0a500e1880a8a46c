"""Core Table API tests (modeled on the reference test_common.py patterns)."""

import pytest

import pathway_amd as pw
from pathway_amd.debug import (
    assert_table_equality,
    assert_table_equality_wo_index,
    table_from_markdown as T,
)


def test_select_arithmetic():
    t = T(
        """
        a | b
        1 | 2
        3 | 4
        """
    )
    res = t.select(pw.this.a, s=pw.this.a + pw.this.b, d=pw.this.b - pw.this.a)
    expected = T(
        """
        a | s | d
        1 | 3 | 1
        3 | 7 | 1
        """
    )
    assert_table_equality(res, expected)


def test_select_strings():
    t = T(
        """
        name
        Alice
        Bob
        """
    )
    res = t.select(upper=pw.this.name.str.upper(), l=pw.this.name.str.len())
    expected = T(
        """
        upper | l
        ALICE | 5
        BOB   | 3
        """
    )
    assert_table_equality(res, expected)


def test_filter():
    t = T(
        """
        a
        1
        2
        3
        4
        """
    )
    res = t.filter(pw.this.a > 2)
    expected = T(
        """
        a | __pos__
        3 | 3
        4 | 4
        """
    ).select(pw.this.a)
    assert_table_equality_wo_index(res, expected)


def test_groupby_count_sum():
    t = T(
        """
        word  | v
        apple | 1
        pear  | 2
        apple | 3
        """
    )
    res = t.groupby(pw.this.word).reduce(
        pw.this.word, cnt=pw.reducers.count(), total=pw.reducers.sum(pw.this.v)
    )
    expected = T(
        """
        word  | cnt | total
        apple | 2   | 4
        pear  | 1   | 2
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_groupby_min_max_avg():
    t = T(
        """
        g | v
        a | 5
        a | 1
        b | 7
        """
    )
    res = t.groupby(pw.this.g).reduce(
        pw.this.g,
        mn=pw.reducers.min(pw.this.v),
        mx=pw.reducers.max(pw.this.v),
        av=pw.reducers.avg(pw.this.v),
    )
    expected = T(
        """
        g | mn | mx | av
        a | 1  | 5  | 3.0
        b | 7  | 7  | 7.0
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_groupby_retraction_stream():
    t = T(
        """
        word  | __time__ | __diff__
        apple | 0        | 1
        apple | 2        | 1
        apple | 4        | -1
        """
    )
    res = t.groupby(pw.this.word).reduce(pw.this.word, cnt=pw.reducers.count())
    expected = T(
        """
        word  | cnt
        apple | 1
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_join_inner():
    t1 = T(
        """
        a | k
        1 | x
        2 | y
        3 | z
        """
    )
    t2 = T(
        """
        b | k
        10 | x
        20 | y
        30 | w
        """
    )
    res = t1.join(t2, t1.k == t2.k).select(t1.a, t2.b, pw.this.k)
    expected = T(
        """
        a | b  | k
        1 | 10 | x
        2 | 20 | y
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_join_left():
    t1 = T(
        """
        a | k
        1 | x
        3 | z
        """
    )
    t2 = T(
        """
        b  | k
        10 | x
        """
    )
    res = t1.join_left(t2, t1.k == t2.k).select(t1.a, t2.b)
    expected = T(
        """
        a | b
        1 | 10
        3 |
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_concat_and_update_rows():
    t1 = T(
        """
        a
        1
        2
        """
    )
    t2 = T(
        """
        a
        3
        4
        """
    )
    res = t1.concat_reindex(t2)
    assert_table_equality_wo_index(
        res,
        T(
            """
            a
            1
            2
            3
            4
            """
        ),
    )


def test_ix():
    t = T(
        """
        a | ptr_target
        1 | 10
        2 | 20
        """
    )
    target = T(
        """
        v
        100
        200
        """
    )
    # build pointers to target rows via with_id_from
    keyed = target.with_id_from(pw.this.v)
    q = t.select(p=t.pointer_from(pw.this.ptr_target * 10))
    res = keyed.ix(q.p)
    expected = T(
        """
        v
        100
        200
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_if_else_coalesce():
    t = T(
        """
        a | b
        1 |
        2 | 5
        """
    )
    res = t.select(
        c=pw.if_else(pw.this.a > 1, pw.this.a * 10, pw.this.a),
        d=pw.coalesce(pw.this.b, 0),
    )
    expected = T(
        """
        c  | d
        1  | 0
        20 | 5
        """
    )
    assert_table_equality(res, expected)


def test_apply_udf():
    t = T(
        """
        a
        1
        2
        """
    )

    @pw.udf
    def double(x: int) -> int:
        return 2 * x

    res = t.select(b=double(pw.this.a))
    expected = T(
        """
        b
        2
        4
        """
    )
    assert_table_equality(res, expected)


def test_flatten():
    t = T(
        """
        s
        ab
        c
        """
    )
    res = t.flatten(pw.this.s)
    expected = T(
        """
        s
        a
        b
        c
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_difference_intersect():
    t1 = T(
        """
        a
        1
        2
        3
        """
    )
    t2 = t1.filter(pw.this.a > 1)
    diff = t1.difference(t2)
    assert_table_equality_wo_index(
        diff,
        T(
            """
            a
            1
            """
        ),
    )
    inter = t1.intersect(t2)
    assert_table_equality_wo_index(
        inter,
        T(
            """
            a
            2
            3
            """
        ),
    )


def test_update_cells():
    t1 = T(
        """
        a | b
        1 | 2
        3 | 4
        """
    )
    t2 = t1.filter(pw.this.a == 1).select(b=pw.this.b * 100)
    res = t1.update_cells(t2)
    expected = T(
        """
        a | b
        1 | 200
        3 | 4
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_groupby_argmax_tuple():
    t = T(
        """
        g | v
        a | 5
        a | 1
        b | 7
        """
    )
    res = t.groupby(pw.this.g).reduce(
        pw.this.g,
        st=pw.reducers.sorted_tuple(pw.this.v),
    )
    keys, cols = pw.debug.table_to_dicts(res)
    vals = {cols["g"][k]: cols["st"][k] for k in keys}
    assert vals == {"a": (1, 5), "b": (7,)}


def test_sql_basic():
    t = T(
        """
        a | b
        1 | 2
        3 | 4
        """
    )
    res = pw.sql("SELECT a, a + b AS s FROM tab WHERE a > 1", tab=t)
    expected = T(
        """
        a | s
        3 | 7
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_sql_groupby():
    t = T(
        """
        g | v
        a | 1
        a | 2
        b | 3
        """
    )
    res = pw.sql("SELECT g, SUM(v) AS s FROM tab GROUP BY g", tab=t)
    expected = T(
        """
        g | s
        a | 3
        b | 3
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_join_result_filter_and_reduce():
    t1 = T(
        """
        a | k
        1 | x
        2 | y
        5 | x
        """
    )
    t2 = T(
        """
        b  | k
        10 | x
        20 | y
        """
    )
    j = t1.join(t2, t1.k == t2.k)
    res = j.filter(pw.this.a > 1).select(pw.this.a, pw.this.b)
    assert_table_equality_wo_index(
        res,
        T(
            """
            a | b
            2 | 20
            5 | 10
            """
        ),
    )
    red = t1.join(t2, t1.k == t2.k).groupby(pw.this.k).reduce(
        pw.this.k, s=pw.reducers.sum(pw.left.a), sb=pw.reducers.sum(pw.right.b)
    )
    assert_table_equality_wo_index(
        red,
        T(
            """
            k | s | sb
            x | 6 | 20
            y | 2 | 20
            """
        ),
    )


def test_groupby_instance_colocation():
    t = T(
        """
        g | inst | v
        a | 1    | 1
        a | 1    | 2
        b | 1    | 3
        a | 2    | 4
        """
    )
    res = t.groupby(pw.this.g, instance=pw.this.inst).reduce(
        pw.this.g, s=pw.reducers.sum(pw.this.v)
    )
    keys, cols = pw.debug.table_to_dicts(res)
    got = sorted((cols["g"][k], cols["s"][k]) for k in keys)
    assert got == [("a", 3), ("a", 4), ("b", 3)]
    # rows with the same instance share the shard bits of their group key
    from pathway_amd.internals.api import SHARD_MASK

    shards = {}
    gk, ck = pw.debug.table_to_dicts(
        t.groupby(pw.this.g, instance=pw.this.inst).reduce(
            pw.this.g, i=pw.reducers.any(pw.this.inst)
        )
    )
    for k in gk:
        shards.setdefault(ck["i"][k], set()).add(k.lo & SHARD_MASK)
    for inst, sh in shards.items():
        assert len(sh) == 1, f"instance {inst} split across shards {sh}"


def test_row_transformer():
    @pw.transformer
    class fib_tr:
        class series(pw.ClassArg):
            n = pw.input_attribute()

            @pw.output_attribute
            def fib(self) -> int:
                if self.n <= 1:
                    return self.n
                p1 = self.pointer_from(self.n - 1)
                p2 = self.pointer_from(self.n - 2)
                return (
                    self.transformer.series[p1].fib
                    + self.transformer.series[p2].fib
                )

    t = T(
        """
        n
        0
        1
        2
        3
        4
        5
        """
    ).with_id_from(pw.this.n)
    res = fib_tr(series=t).series
    keys, cols = pw.debug.table_to_dicts(res)
    assert sorted(cols["fib"].values()) == [0, 1, 1, 2, 3, 5]


def test_sql_join():
    t1 = T(
        """
        a | k
        1 | x
        2 | y
        """
    )
    t2 = T(
        """
        b  | k
        10 | x
        20 | y
        """
    )
    res = pw.sql("SELECT a, b, a + b AS s FROM l JOIN r ON l.k = r.k", l=t1, r=t2)
    assert_table_equality_wo_index(
        res,
        T(
            """
            a | b  | s
            1 | 10 | 11
            2 | 20 | 22
            """
        ),
    )


def test_select_with_ix_updates_under_retraction():
    """Regression: a select combining own columns with an ix()-derived
    column must re-evaluate from STATE when either side changes —
    aligning against the other table's per-step delta read the retracted
    old value (and missed extra-side-only changes)."""
    from pathway_amd import reducers
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.rungraph import G
    from pathway_amd.internals.schema import column_definition, schema_builder
    from pathway_amd.internals import thisclass

    this = thisclass.this
    G.clear()
    schema = schema_builder(
        {
            "uid": column_definition(primary_key=True, dtype=int),
            "g": column_definition(dtype=str),
            "v": column_definition(dtype=int),
        }
    )
    t = table_from_rows(
        schema,
        [(1, "a", 5, 0, 1), (2, "a", 9, 1, 1), (2, "a", 9, 2, -1)],
        is_stream=True,
    )
    g = t.groupby(this.g).reduce(
        g=this.g, best=reducers.argmax(this.v), w=reducers.max(this.v)
    )
    g2 = g.select(this.g, this.w, vv=t.ix(g.best).v)
    keys, cols = pw.debug.table_to_dicts(g2)
    # after insert(5) -> insert(9) -> retract(9): back to the uid=1 row
    rows = [(cols["g"][k], cols["w"][k], cols["vv"][k]) for k in keys]
    assert rows == [("a", 5, 5)]
