"""API-surface behaviors mirroring reference test_common.py patterns."""

import pytest

import pathway_amd as pw
from pathway_amd.debug import (
    assert_table_equality,
    assert_table_equality_wo_index,
    table_from_markdown as T,
)


def test_rename_without_chain():
    t = T(
        """
        a | b | c
        1 | 2 | 3
        """
    )
    r = t.rename_columns(x="a").without("b")
    assert set(r.column_names()) == {"x", "c"}
    r2 = t.rename_by_dict({"a": "p", "b": "q"})
    assert set(r2.column_names()) == {"p", "q", "c"}
    r3 = t.with_prefix("u_")
    assert set(r3.column_names()) == {"u_a", "u_b", "u_c"}
    r4 = t.with_suffix("_v")
    assert set(r4.column_names()) == {"a_v", "b_v", "c_v"}


def test_having_and_restrict():
    t = T(
        """
        v
        10
        20
        30
        """
    ).with_id_from(pw.this.v)
    q = T(
        """
        p
        10
        30
        """
    )
    res = t.having(q.p)
    assert_table_equality_wo_index(
        res,
        T(
            """
            v
            10
            30
            """
        ),
    )


def test_ix_ref():
    t = T(
        """
        name  | score
        alice | 5
        bob   | 7
        """
    ).with_id_from(pw.this.name)
    q = T(
        """
        who
        bob
        alice
        """
    )
    res = t.ix_ref(q.who, context=q)
    keys, cols = pw.debug.table_to_dicts(res)
    got = sorted(cols["score"].values())
    assert got == [5, 7]


def test_with_universe_of_and_select_across():
    t1 = T(
        """
        a
        1
        2
        """
    )
    t2 = t1.select(b=pw.this.a * 10)
    combined = t1.select(pw.this.a, b=t2.b)
    assert_table_equality(
        combined,
        T(
            """
            a | b
            1 | 10
            2 | 20
            """
        ),
    )


def test_update_types_and_schema():
    t = T(
        """
        a
        1
        """
    )
    r = t.update_types(a=float)
    assert "FLOAT" in repr(r.schema)
    s = pw.schema_from_types(x=int, y=str)
    assert s.column_names() == ["x", "y"]
    union = s | pw.schema_from_types(z=float)
    assert union.column_names() == ["x", "y", "z"]


def test_schema_class_and_defaults():
    class S(pw.Schema):
        a: int = pw.column_definition(primary_key=True)
        b: str = pw.column_definition(default_value="d")

    assert S.primary_key_columns() == ["a"]
    assert S.default_values() == {"b": "d"}
    t = pw.debug.table_from_rows(S, [(1, "x"), (2, "y")])
    keys, cols = pw.debug.table_to_dicts(t)
    assert sorted(cols["b"].values()) == ["x", "y"]


def test_empty_table_ops():
    t = pw.Table.empty(a=int)
    res = t.select(b=pw.this.a + 1)
    keys, cols = pw.debug.table_to_dicts(res)
    assert keys == []
    g = t.groupby(pw.this.a).reduce(pw.this.a, c=pw.reducers.count())
    keys, _ = pw.debug.table_to_dicts(g)
    assert keys == []


def test_concat_reindex_three():
    ts = [
        T(
            f"""
            a
            {i}
            """
        )
        for i in range(3)
    ]
    res = ts[0].concat_reindex(ts[1], ts[2])
    keys, cols = pw.debug.table_to_dicts(res)
    assert sorted(cols["a"].values()) == [0, 1, 2]


def test_geometric_rag_strategy():
    from pathway_amd.xpacks.llm.llms import BaseChat
    from pathway_amd.xpacks.llm.question_answering import (
        answer_with_geometric_rag_strategy,
    )

    class StubChat(BaseChat):
        def __init__(self):
            super().__init__()
            self.calls = 0

        def __wrapped__(self, prompt, **kw):
            self.calls += 1
            # fails with 1 doc, answers once both docs are in the prompt
            return "42" if "ctx two" in prompt else "No information"

    chat = StubChat()
    qs = T(
        """
        query  | docs
        what?  |
        """
    ).select(pw.this.query, docs=pw.make_tuple("ctx one", "ctx two"))
    res = answer_with_geometric_rag_strategy(qs, None, chat, 1, 2, 3)
    keys, cols = pw.debug.table_to_dicts(res)
    out = list(cols["result"].values())[0]
    assert out == "42"
    assert chat.calls == 2  # adaptive doubling: 1 doc failed, 2 docs answered
