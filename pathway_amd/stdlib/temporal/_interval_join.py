"""interval_join (reference stdlib/temporal/interval_join.py, 1619 LoC).

Inner mode is implemented compositionally: bucket the right side by
floor(t/W) with W = upper-lower, expand each left row to its ≤2 candidate
buckets, equi-join on bucket (+extra conditions), then filter the exact
bounds — all stages run on the engine's GPU join/filter path.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

from pathway_amd.internals import expression as ex
from pathway_amd.internals import thisclass
from pathway_amd.internals.table import Table


@dataclass
class Interval:
    lower_bound: Any
    upper_bound: Any


def interval(lower_bound: Any, upper_bound: Any) -> Interval:
    return Interval(lower_bound, upper_bound)


class IntervalJoinResult:
    def __init__(self, left, right, left_time, right_time, itv, on, mode, how=None):
        self._left = left
        self._right = right
        self._left_time = left_time
        self._right_time = right_time
        self._interval = itv
        self._on = list(on)
        self._mode = mode

    def select(self, *args: Any, **kwargs: Any) -> Table:
        if self._mode != "inner":
            raise NotImplementedError(
                f"interval_join mode {self._mode!r} lands with the temporal phase"
            )
        left, right = self._left, self._right
        lo, hi = self._interval.lower_bound, self._interval.upper_bound
        import pandas as pd

        def as_num(v):
            if isinstance(v, pd.Timedelta):
                return int(v.value)
            return v

        lon, hin = as_num(lo), as_num(hi)
        W = hin - lon
        lt = thisclass.substitute_this(
            ex.wrap_expr(self._left_time), {thisclass.this: left, thisclass.left: left}
        )
        rt = thisclass.substitute_this(
            ex.wrap_expr(self._right_time), {thisclass.this: right, thisclass.right: right}
        )
        if W == 0:
            lx = left.with_columns(_pw_b=lt + lon)
            rx = right.with_columns(_pw_b=rt)
        else:
            l0 = left.with_columns(_pw_b=(lt + lon) // W)
            l1 = left.with_columns(_pw_b=(lt + hin) // W).filter(
                (thisclass.this._pw_b) != ((self._rebind(lt, "left") + lon) // W)
            )
            lx = l0.concat_reindex(l1)
            rx = right.with_columns(_pw_b=rt // W)
        conds = [lx._pw_b == rx._pw_b]
        for c in self._on:
            c = thisclass.substitute_this(
                ex.wrap_expr(c), {thisclass.left: lx, thisclass.right: rx}
            )
            conds.append(self._remap_cond(c, lx, rx))
        j = lx.join(rx, *conds)
        # precise bound filter on the joined pairs
        ltj = self._rewrap(lt, thisclass.left)
        rtj = self._rewrap(rt, thisclass.right)
        out_kwargs = dict(kwargs)
        out_kwargs["_pw_lt"] = ltj
        out_kwargs["_pw_rt"] = rtj
        sel = j.select(*args, **out_kwargs)
        if W == 0:
            flt = sel.filter(thisclass.this._pw_rt == thisclass.this._pw_lt + lon)
        else:
            flt = sel.filter(
                (thisclass.this._pw_rt >= thisclass.this._pw_lt + lon)
                & (thisclass.this._pw_rt <= thisclass.this._pw_lt + hin)
            )
        return flt.without("_pw_lt", "_pw_rt")

    def _rebind(self, e, side):
        return e

    def _rewrap(self, e, marker):
        """Rebind a time expr of the original table to the join marker side."""
        if isinstance(e, ex.ColumnReference):
            return ex.ColumnReference(marker, e.name)
        return e

    def _remap_cond(self, c, lx, rx):
        def rec(e):
            if isinstance(e, ex.ColumnReference):
                if e.table is self._left:
                    return ex.ColumnReference(lx, e.name)
                if e.table is self._right:
                    return ex.ColumnReference(rx, e.name)
                return e
            new = object.__new__(type(e))
            new.__dict__.update(e.__dict__)
            for attr, val in list(e.__dict__.items()):
                if isinstance(val, ex.ColumnExpression):
                    new.__dict__[attr] = rec(val)
            return new

        return rec(c)


def interval_join(
    self: Table,
    other: Table,
    self_time: Any,
    other_time: Any,
    interval: Interval,
    *on: Any,
    behavior=None,
    how: Any = None,
) -> IntervalJoinResult:
    return IntervalJoinResult(self, other, self_time, other_time, interval, on, "inner")


def interval_join_inner(self, other, self_time, other_time, interval, *on, **kw):
    return IntervalJoinResult(self, other, self_time, other_time, interval, on, "inner")


def interval_join_left(self, other, self_time, other_time, interval, *on, **kw):
    return IntervalJoinResult(self, other, self_time, other_time, interval, on, "left")


def interval_join_right(self, other, self_time, other_time, interval, *on, **kw):
    return IntervalJoinResult(self, other, self_time, other_time, interval, on, "right")


def interval_join_outer(self, other, self_time, other_time, interval, *on, **kw):
    return IntervalJoinResult(self, other, self_time, other_time, interval, on, "outer")
