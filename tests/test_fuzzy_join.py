"""Fuzzy join (reference _fuzzy_join.py semantics)."""

import pathway_amd as pw
from pathway_amd.debug import table_from_markdown as T, table_to_dicts
from pathway_amd.internals.rungraph import G
from pathway_amd.stdlib.ml.smart_table_ops import (
    FuzzyJoinFeatureGeneration,
    FuzzyJoinNormalization,
    fuzzy_match_tables,
    fuzzy_self_match,
    smart_fuzzy_match,
    _edges_for,
)


def _pairs(res, lt, rt, lcol, rcol):
    _k, cols = table_to_dicts(res)
    _lk, lcols = table_to_dicts(lt)
    _rk, rcols = table_to_dicts(rt)
    lmap = {repr(k): v for k, v in lcols[lcol].items()}
    rmap = {repr(k): v for k, v in rcols[rcol].items()}
    out = {}
    for k in cols["left"]:
        lv = lmap[repr(cols["left"][k])]
        rv = rmap[repr(cols["right"][k])]
        out[(lv, rv)] = cols["weight"][k]
    return out


def test_smart_fuzzy_match_basic():
    G.clear()
    lt = T(
        """
        name
        john doe
        jane smith
        alice cooper
        """
    )
    rt = T(
        """
        name
        doe john
        smith jane
        bob marley
        """
    )
    res = smart_fuzzy_match(lt.name, rt.name)
    got = _pairs(res, lt, rt, "name", "name")
    assert set(got) == {("john doe", "doe john"), ("jane smith", "smith jane")}
    assert all(w > 0 for w in got.values())


def test_mutual_best_is_one_to_one():
    G.clear()
    lt = T(
        """
        name
        red apple
        green apple
        """
    )
    rt = T(
        """
        name
        apple pie
        """
    )
    res = smart_fuzzy_match(lt.name, rt.name)
    got = _pairs(res, lt, rt, "name", "name")
    # only ONE left row may claim the single right row
    assert len(got) == 1
    assert list(got)[0][1] == "apple pie"


def test_normalization_downweights_common_tokens():
    G.clear()
    # "inc" appears everywhere; the rare surname dominates under LOGWEIGHT
    lt = T(
        """
        name
        acme inc
        zorblax inc
        """
    )
    rt = T(
        """
        name
        zorblax inc
        inc acme
        """
    )
    res = smart_fuzzy_match(lt.name, rt.name)
    got = _pairs(res, lt, rt, "name", "name")
    assert got[("zorblax inc", "zorblax inc")] >= got[("acme inc", "inc acme")]


def test_letters_feature_generation():
    G.clear()
    lt = T(
        """
        name
        abc
        """
    )
    rt = T(
        """
        name
        cab
        """
    )
    res = smart_fuzzy_match(
        lt.name, rt.name,
        feature_generation=FuzzyJoinFeatureGeneration.LETTERS,
    )
    got = _pairs(res, lt, rt, "name", "name")
    assert ("abc", "cab") in got


def test_fuzzy_self_match_dedups_pairs():
    G.clear()
    t = T(
        """
        name
        widget mark one
        widget mark two
        unrelated thing
        """
    )
    res = smart_fuzzy_match(t.name, t.name)
    _k, cols = table_to_dicts(res)
    # symmetric: each pair appears once (left < right), and the two
    # widget rows match each other
    assert len(cols["left"]) == 1
    for k in cols["left"]:
        assert repr(cols["left"][k]) != repr(cols["right"][k])


def test_fuzzy_match_tables_and_projections():
    G.clear()
    lt = T(
        """
        first | last
        john  | doe
        jane  | smith
        """
    )
    rt = T(
        """
        a    | b
        doe  | john
        smith| jane
        """
    )
    res = fuzzy_match_tables(lt, rt)
    _k, cols = table_to_dicts(res)
    assert len(cols["left"]) == 2
    G.clear()
    lt2 = T(
        """
        first | last
        john  | doe
        """
    )
    rt2 = T(
        """
        a    | b
        john | doe
        """
    )
    res2 = fuzzy_match_tables(
        lt2, rt2,
        left_projection={"first": "f", "last": "l"},
        right_projection={"a": "f", "b": "l"},
    )
    _k2, cols2 = table_to_dicts(res2)
    # both buckets match the same pair: weights summed into one row
    assert len(cols2["left"]) == 1


def test_by_hand_match_overrides():
    G.clear()
    lt = T(
        """
        name
        alpha beta
        gamma delta
        """
    )
    rt = T(
        """
        name
        alpha beta
        gamma delta
        """
    )
    # force the CROSSED match by hand: alpha->gamma row
    lkeys, lcols = table_to_dicts(lt)
    rkeys, rcols = table_to_dicts(rt)
    G.clear()
    lt = T(
        """
        name
        alpha beta
        gamma delta
        """
    )
    rt = T(
        """
        name
        alpha beta
        gamma delta
        """
    )
    edges_l = _edges_for(lt, lt.name, FuzzyJoinFeatureGeneration.AUTO)
    # hand-match: pair the "alpha beta" rows explicitly with weight 99
    hand = lt.join(rt, lt.name == rt.name).select(
        left=lt.id, right=rt.id, weight=99.0
    ).filter(pw.this.weight > 0)
    res = smart_fuzzy_match(lt.name, rt.name, by_hand_match=hand)
    _k, cols = table_to_dicts(res)
    weights = sorted(cols["weight"].values())
    assert weights.count(99.0) == 2  # both identical pairs forced by hand


def test_fuzzy_match_updates_incrementally():
    """A retraction + new insert re-pairs the match (the node diffs its
    output across engine times)."""
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.schema import schema_builder, column_definition

    G.clear()
    schema = schema_builder(
        {
            "uid": column_definition(primary_key=True, dtype=int),
            "name": column_definition(dtype=str),
        }
    )
    # time 0: left matches "acme corp"; time 1: that right row is
    # retracted and a better candidate arrives
    rt_rows = [
        (1, "acme corp", 0, 1),
        (1, "acme corp", 1, -1),
        (2, "acme corp holdings", 1, 1),
    ]
    lt = table_from_rows(
        schema, [(10, "acme corp", 0, 1)], is_stream=True
    )
    rt = table_from_rows(schema, rt_rows, is_stream=True)
    res = smart_fuzzy_match(lt.name, rt.name)
    _k, cols = table_to_dicts(res)
    # final state: exactly one pair, against the surviving right row
    assert len(cols["left"]) == 1
    _lk, lcols = table_to_dicts(lt)
    _rk, rcols = table_to_dicts(rt)
    rmap = {repr(k): v for k, v in rcols["name"].items()}
    for k in cols["right"]:
        assert rmap[repr(cols["right"][k])] == "acme corp holdings"
