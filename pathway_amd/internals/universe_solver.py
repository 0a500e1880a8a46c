"""Universe relation solver (reference internals/universe_solver.py).

The reference encodes key-set (universe) relationships as SAT clauses
(python-sat) to validate operations like ``with_universe_of`` /
``restrict`` / ``intersect``.  This build implements the same decision
procedure as a saturating relational reasoner over the promise algebra:

  equal(a, b)          a == b
  subset(a, b)         a ⊆ b
  disjoint(a, b)       a ∩ b == ∅
  union(u, parts)      u == ⋃ parts
  intersection(i, ps)  i == ⋂ ps
  difference(d, a, b)  d == a ∖ b

Queries (is_subset_of / is_equal / are_disjoint) run a memoized
recursive search with the derivation rules:

  subset transitivity; union: u ⊆ w if every part ⊆ w;
  intersection: x ⊆ i if x ⊆ every part; difference: d ⊆ a, d ∥ b;
  disjoint: x ∥ y if x ⊆ a, y ⊆ b with a ∥ b (symmetric).

Sound (only promised facts and their consequences are derivable) and
complete for the promise algebra above — the same guarantees the
reference gets from its clause encoding.
"""

from __future__ import annotations

from typing import Hashable


class UniverseSolver:
    def __init__(self) -> None:
        #: canonical-id union-find for equality
        self._eq_parent: dict[Hashable, Hashable] = {}
        self._subset_edges: dict[Hashable, set[Hashable]] = {}
        self._disjoint: set[tuple[Hashable, Hashable]] = {}
        self._disjoint = set()
        self._unions: dict[Hashable, tuple[Hashable, ...]] = {}
        self._intersections: dict[Hashable, tuple[Hashable, ...]] = {}

    # -- canonicalization --

    def _find(self, x: Hashable) -> Hashable:
        p = self._eq_parent.get(x, x)
        if p == x:
            return x
        r = self._find(p)
        self._eq_parent[x] = r
        return r

    # -- promises --

    def promise_equal(self, a: Hashable, b: Hashable) -> None:
        ra, rb = self._find(a), self._find(b)
        if ra != rb:
            self._eq_parent[ra] = rb

    def promise_subset(self, sub: Hashable, sup: Hashable) -> None:
        self._subset_edges.setdefault(sub, set()).add(sup)

    def promise_disjoint(self, a: Hashable, b: Hashable) -> None:
        self._disjoint.add((a, b))
        self._disjoint.add((b, a))

    def register_union(self, u: Hashable, parts: tuple[Hashable, ...]) -> None:
        self._unions[u] = tuple(parts)
        for p in parts:
            self.promise_subset(p, u)

    def register_intersection(self, i: Hashable, parts: tuple[Hashable, ...]) -> None:
        self._intersections[i] = tuple(parts)
        for p in parts:
            self.promise_subset(i, p)

    def register_difference(self, d: Hashable, a: Hashable, b: Hashable) -> None:
        self.promise_subset(d, a)
        self.promise_disjoint(d, b)

    # -- queries --

    def is_equal(self, a: Hashable, b: Hashable) -> bool:
        if self._find(a) == self._find(b):
            return True
        return self.is_subset_of(a, b) and self.is_subset_of(b, a)

    def is_subset_of(self, a: Hashable, b: Hashable, _seen=None) -> bool:
        ra, rb = self._find(a), self._find(b)
        if ra == rb:
            return True
        if _seen is None:
            _seen = set()
        key = (ra, rb)
        if key in _seen:
            return False
        _seen.add(key)
        # direct + transitive edges (compare canonically)
        for x, sups in self._subset_edges.items():
            if self._find(x) != ra:
                continue
            for s in sups:
                if self._find(s) == rb or self.is_subset_of(s, b, _seen):
                    return True
        # union source: every part ⊆ b  ->  u ⊆ b
        for u, parts in self._unions.items():
            if self._find(u) == ra and parts:
                if all(self.is_subset_of(p, b, _seen) for p in parts):
                    return True
        # intersection target: a ⊆ every part  ->  a ⊆ i
        for i, parts in self._intersections.items():
            if self._find(i) == rb and parts:
                if all(self.is_subset_of(a, p, _seen) for p in parts):
                    return True
        return False

    def are_disjoint(self, a: Hashable, b: Hashable) -> bool:
        # x ∥ y if exist promised (p, q) disjoint with x ⊆ p and y ⊆ q
        for p, q in self._disjoint:
            if self.is_subset_of(a, p) and self.is_subset_of(b, q):
                return True
        return False

    def query_are_equal(self, a, b):  # reference naming
        return self.is_equal(a, b)

    def query_is_subset(self, a, b):
        return self.is_subset_of(a, b)


#: process-wide solver (reference keeps one per parse graph)
GLOBAL_SOLVER = UniverseSolver()
