"""pw.Json value-type semantics (reference test_json.py patterns)."""

import pytest

import pathway_amd as pw
from pathway_amd.debug import table_from_rows, table_to_dicts
from pathway_amd.internals.json import Json
from pathway_amd.internals.schema import schema_from_types


def _one(table, name):
    _, cols = table_to_dicts(table)
    (v,) = cols[name].values()
    return v


def test_json_scalars_and_coercion():
    t = table_from_rows(
        schema_from_types(j=pw.Json),
        [(Json({"i": 7, "f": 2.5, "s": "txt", "b": True, "n": None}),)],
    )
    res = t.select(
        i=pw.this.j["i"].as_int(),
        f=pw.this.j["f"].as_float(),
        fi=pw.this.j["i"].as_float(),  # int json -> float coercion
        s=pw.this.j["s"].as_str(),
        b=pw.this.j["b"].as_bool(),
        n=pw.this.j["n"],
    )
    assert _one(res, "i") == 7
    assert _one(res, "f") == 2.5
    assert _one(res, "fi") == 7.0
    assert _one(res, "s") == "txt"
    assert _one(res, "b") is True
    assert _one(res, "n") is None or isinstance(_one(res, "n"), Json)


def test_json_nested_chain_and_missing():
    t = table_from_rows(
        schema_from_types(j=pw.Json),
        [(Json({"a": {"b": [10, {"c": 20}]}}),)],
    )
    res = t.select(
        x=pw.this.j["a"]["b"][0].as_int(),
        y=pw.this.j["a"]["b"][1]["c"].as_int(),
        # [] on a missing key is an ERROR value (reference semantics);
        # .get() is the None-returning accessor
        missing=pw.fill_error(pw.this.j["zz"].as_int(), -1),
        got=pw.this.j.get("zz"),
    )
    assert _one(res, "x") == 10
    assert _one(res, "y") == 20
    assert _one(res, "missing") == -1
    assert _one(res, "got") is None


def test_json_as_list_and_flatten():
    t = table_from_rows(
        schema_from_types(j=pw.Json),
        [(Json({"items": [3, 1, 2]}),)],
    )
    lst = t.select(l=pw.apply_with_type(lambda j: list(j["items"].as_list()), list, pw.this.j))
    flat = lst.flatten(pw.this.l)
    _, cols = table_to_dicts(flat)
    vals = sorted(v.value if isinstance(v, Json) else v for v in cols["l"].values())
    assert vals == [1, 2, 3]


def test_json_in_group_key():
    t = table_from_rows(
        schema_from_types(j=pw.Json, v=int),
        [
            (Json({"k": 1}), 10),
            (Json({"k": 1}), 5),
            (Json({"k": 2}), 7),
        ],
    )
    r = t.groupby(pw.this.j).reduce(pw.this.j, s=pw.reducers.sum(pw.this.v))
    _, cols = table_to_dicts(r)
    got = sorted(
        (v.value["k"] if isinstance(v, Json) else v["k"], s)
        for v, s in zip(cols["j"].values(), cols["s"].values())
    )
    assert got == [(1, 15), (2, 7)]


def test_json_equality_and_filter():
    t = table_from_rows(
        schema_from_types(j=pw.Json),
        [(Json({"t": "a"}),), (Json({"t": "b"}),)],
    )
    res = t.filter(pw.this.j["t"].as_str() == "a")
    _, cols = table_to_dicts(res)
    assert len(cols["j"]) == 1
