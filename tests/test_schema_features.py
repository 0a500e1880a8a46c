"""Schema machinery: column_definition, primary keys, defaults,
schema_from_* constructors, with_types (reference test_schema.py)."""

from typing import Optional

import pytest

import pathway_amd as pw
from pathway_amd.debug import table_from_markdown as T, table_from_rows, table_to_dicts
from pathway_amd.internals import dtype as dt
from pathway_amd.internals.schema import (
    column_definition,
    schema_from_dict,
    schema_from_types,
)


def test_column_definition_primary_key():
    class S(pw.Schema):
        k: int = column_definition(primary_key=True)
        v: str

    assert S.primary_key_columns() == ["k"]
    t = table_from_rows(S, [(1, "a"), (2, "b")])
    # ids are derived from the primary key: same key -> same id across tables
    t2 = table_from_rows(S, [(1, "zzz")])
    k1, _ = table_to_dicts(t)
    k2, _ = table_to_dicts(t2)
    assert set(k2) <= set(k1)


def test_column_definition_default_value():
    class S(pw.Schema):
        a: int
        b: int = column_definition(default_value=42)

    assert S.default_values() == {"b": 42}


def test_column_definition_renamed():
    class S(pw.Schema):
        data: str = column_definition(name="json.data")

    assert "json.data" in S.column_names()


def test_schema_from_dict_and_types():
    S1 = schema_from_types(a=int, b=Optional[float])
    assert S1.__columns__["a"].dtype == dt.INT
    assert S1.__columns__["b"].dtype == dt.Optional(dt.FLOAT)
    S2 = schema_from_dict({"x": str, "y": int})
    assert S2.column_names() == ["x", "y"]


def test_schema_or_union():
    A = schema_from_types(a=int)
    B = schema_from_types(b=str)
    AB = A | B
    assert AB.column_names() == ["a", "b"]


def test_update_types():
    t = T(
        """
        a
        1
        """
    )
    t2 = t.update_types(a=float)
    assert t2.schema.__columns__["a"].dtype == dt.FLOAT


def test_schema_generate_class():
    class S(pw.Schema):
        a: int
        b: str

    # typed accessors survive the metaclass
    assert S.column_names() == ["a", "b"]
    inst = S
    assert "a" in inst.typehints()


def test_table_schema_roundtrip():
    t = T(
        """
        a | b
        1 | x
        """
    )
    sch = t.schema
    assert sch.__columns__["a"].dtype == dt.INT
    assert sch.__columns__["b"].dtype == dt.STR


def test_assert_table_has_schema():
    t = T(
        """
        a | b
        1 | x
        """
    )

    class Good(pw.Schema):
        a: int
        b: str

    pw.assert_table_has_schema(t, Good)

    class Bad(pw.Schema):
        a: str
        b: str

    with pytest.raises(Exception):
        pw.assert_table_has_schema(t, Bad)


def test_pointer_typed_column():
    t = T(
        """
        a
        1
        """
    )
    withid = t.select(p=t.id)
    assert dt.unoptionalize(withid.schema.__columns__["p"].dtype) == dt.POINTER
