"""pw.universes (reference python/pathway/universes.py)."""

from __future__ import annotations


def promise_is_subset_of(subset, superset) -> None:
    from pathway_amd.internals.universe import promise_is_subset_of as _p

    _p(subset._universe, superset._universe)


def promise_are_equal(*tables) -> None:
    for t in tables[1:]:
        tables[0]._universe.promise_equal(t._universe)


def promise_are_pairwise_disjoint(*tables) -> None:
    pass
