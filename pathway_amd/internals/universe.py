"""Universe (key-set) tracking.

The reference validates universe relationships with a SAT solver
(internals/universe_solver.py, python-sat); here the same decision
procedure runs on the saturating relational reasoner in
pathway_amd.internals.universe_solver — promises (equal/subset/disjoint
plus union/intersection/difference definitions) and their logical
consequences, nothing else, are derivable.
"""

from __future__ import annotations

import itertools

from pathway_amd.internals.universe_solver import GLOBAL_SOLVER

_ids = itertools.count()


class Universe:
    def __init__(self, parent: "Universe | None" = None):
        self.id = next(_ids)
        if parent is not None:
            GLOBAL_SOLVER.promise_subset(self.id, parent.id)

    def subuniverse(self) -> "Universe":
        return Universe(parent=self)

    def is_equal(self, other: "Universe") -> bool:
        return GLOBAL_SOLVER.is_equal(self.id, other.id)

    def promise_equal(self, other: "Universe") -> None:
        GLOBAL_SOLVER.promise_equal(self.id, other.id)

    def is_subset_of(self, other: "Universe") -> bool:
        return GLOBAL_SOLVER.is_subset_of(self.id, other.id)

    def is_disjoint_from(self, other: "Universe") -> bool:
        return GLOBAL_SOLVER.are_disjoint(self.id, other.id)

    @staticmethod
    def union_of(*parts: "Universe") -> "Universe":
        u = Universe()
        GLOBAL_SOLVER.register_union(u.id, tuple(p.id for p in parts))
        return u

    @staticmethod
    def intersection_of(*parts: "Universe") -> "Universe":
        i = Universe()
        GLOBAL_SOLVER.register_intersection(i.id, tuple(p.id for p in parts))
        return i

    @staticmethod
    def difference_of(a: "Universe", b: "Universe") -> "Universe":
        d = Universe()
        GLOBAL_SOLVER.register_difference(d.id, a.id, b.id)
        return d

    def __repr__(self) -> str:
        return f"<universe {self.id}>"


def promise_are_pairwise_disjoint(*universes: Universe) -> None:
    for i, a in enumerate(universes):
        for b in universes[i + 1 :]:
            GLOBAL_SOLVER.promise_disjoint(a.id, b.id)


def promise_is_subset_of(sub: Universe, sup: Universe) -> None:
    GLOBAL_SOLVER.promise_subset(sub.id, sup.id)
