"""Universe (key-set) tracking.

The reference validates universe relationships with a SAT solver
(internals/universe_solver.py); here a light union-find over 'equal'
relations plus parent links for 'subset' is enough for the same API checks.
"""

from __future__ import annotations

import itertools

_ids = itertools.count()


class Universe:
    def __init__(self, parent: "Universe | None" = None):
        self.id = next(_ids)
        self._parent = parent
        self._equal_root: "Universe" = self

    def subuniverse(self) -> "Universe":
        return Universe(parent=self)

    def root(self) -> "Universe":
        u = self
        while u._equal_root is not u:
            u = u._equal_root
        self._equal_root = u
        return u

    def is_equal(self, other: "Universe") -> bool:
        return self.root() is other.root()

    def promise_equal(self, other: "Universe") -> None:
        self.root()._equal_root = other.root()

    def is_subset_of(self, other: "Universe") -> bool:
        if self.is_equal(other):
            return True
        u: Universe | None = self
        while u is not None:
            if u.is_equal(other):
                return True
            u = u._parent
        return False

    def __repr__(self) -> str:
        return f"<universe {self.id}>"


def promise_are_pairwise_disjoint(*universes: Universe) -> None:
    pass  # advisory in this implementation


def promise_is_subset_of(sub: Universe, sup: Universe) -> None:
    sub._parent = sup
