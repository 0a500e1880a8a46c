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


def _behave(table, time_expr, behavior, side):
    """Apply a CommonBehavior cutoff to one join input."""
    cutoff = getattr(behavior, "cutoff", None)
    if cutoff is None:
        return table
    from pathway_amd.engine.nodes_temporal import ForgetNode, FreezeNode
    from pathway_amd.internals.config import get_device

    te = thisclass.substitute_this(
        ex.wrap_expr(time_expr),
        {thisclass.this: table, thisclass.left: table, thisclass.right: table},
    )
    import pandas as pd

    c = cutoff
    if isinstance(c, pd.Timedelta):
        c = int(c.value)
    cls = FreezeNode if getattr(behavior, "keep_results", True) else ForgetNode
    node = cls(table._node, te + c, te, get_device())
    return Table(node, table._dtypes, table._universe)


class IntervalJoinResult:
    def __init__(self, left, right, left_time, right_time, itv, on, mode,
                 how=None, behavior=None):
        if behavior is not None:
            # cutoff behavior: freeze (keep_results) or forget each input
            # past watermark - cutoff before joining (reference
            # interval_join behavior semantics)
            left = _behave(left, left_time, behavior, "left")
            right = _behave(right, right_time, behavior, "right")
        self._left = left
        self._right = right
        self._left_time = left_time
        self._right_time = right_time
        self._interval = itv
        self._on = list(on)
        self._mode = mode

    def select(self, *args: Any, **kwargs: Any) -> Table:
        if self._mode != "inner":
            return self._select_outer(*args, **kwargs)
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
            lx = left.with_columns(_pw_b=lt + lon, _pw_olid=left.id)
            rx = right.with_columns(_pw_b=rt, _pw_orid=right.id)
        else:
            l0 = left.with_columns(_pw_b=(lt + lon) // W, _pw_olid=left.id)
            l1 = left.with_columns(
                _pw_b=(lt + hin) // W, _pw_olid=left.id
            ).filter(
                (thisclass.this._pw_b) != ((self._rebind(lt, "left") + lon) // W)
            )
            lx = l0.concat_reindex(l1)
            rx = right.with_columns(_pw_b=rt // W, _pw_orid=right.id)
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

    def _select_outer(self, *args: Any, **kwargs: Any) -> Table:
        """left/right/outer interval joins: inner pairs + None-padded
        unmatched rows (plain column references only in select)."""
        import pathway_amd.internals.common as common
        import pathway_amd.reducers as reducers

        left, right = self._left, self._right
        inner = IntervalJoinResult(
            left, right, self._left_time, self._right_time, self._interval,
            self._on, "inner",
        )
        # resolve requested columns to (side, name, out_name)
        wanted: list[tuple[str, str, str]] = []
        for a in args:
            a2 = thisclass.substitute_this(
                ex.wrap_expr(a), {thisclass.left: left, thisclass.right: right}
            )
            if not isinstance(a2, ex.ColumnReference):
                raise NotImplementedError("outer interval select needs plain columns")
            side = "l" if (a2.table is left or getattr(a2.table, "_node", None) is left._node) else "r"
            wanted.append((side, a2.name, a2.name))
        for name, e in kwargs.items():
            e2 = thisclass.substitute_this(
                ex.wrap_expr(e), {thisclass.left: left, thisclass.right: right}
            )
            if not isinstance(e2, ex.ColumnReference):
                raise NotImplementedError("outer interval select needs plain columns")
            side = "l" if (e2.table is left or getattr(e2.table, "_node", None) is left._node) else "r"
            wanted.append((side, e2.name, name))
        sel_kwargs = {out: (thisclass.left[src] if side == "l" else thisclass.right[src])
                      for side, src, out in wanted}
        sel_kwargs["_pw_lid"] = thisclass.left._pw_olid
        sel_kwargs["_pw_rid"] = thisclass.right._pw_orid
        inner_t = inner.select(**sel_kwargs)
        parts = [inner_t.without("_pw_lid", "_pw_rid")]
        if self._mode in ("left", "outer"):
            matched = inner_t.groupby(inner_t._pw_lid).reduce(
                _pw_k=thisclass.this._pw_lid
            ).with_id_from_expr(ex.ColumnReference(None, "_pw_k"))
            unmatched = left.difference(matched)
            pad = {
                out: (ex.ColumnReference(unmatched, src) if side == "l" else None)
                for side, src, out in wanted
            }
            parts.append(unmatched.select(**pad))
        if self._mode in ("right", "outer"):
            matched = inner_t.groupby(inner_t._pw_rid).reduce(
                _pw_k=thisclass.this._pw_rid
            ).with_id_from_expr(ex.ColumnReference(None, "_pw_k"))
            unmatched = right.difference(matched)
            pad = {
                out: (ex.ColumnReference(unmatched, src) if side == "r" else None)
                for side, src, out in wanted
            }
            parts.append(unmatched.select(**pad))
        out = parts[0].concat_reindex(*parts[1:]) if len(parts) > 1 else parts[0]
        return out

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
    return IntervalJoinResult(
        self, other, self_time, other_time, interval, on, "inner",
        behavior=behavior,
    )


def interval_join_inner(self, other, self_time, other_time, interval, *on, behavior=None, **kw):
    return IntervalJoinResult(self, other, self_time, other_time, interval, on, "inner", behavior=behavior)


def interval_join_left(self, other, self_time, other_time, interval, *on, behavior=None, **kw):
    return IntervalJoinResult(self, other, self_time, other_time, interval, on, "left", behavior=behavior)


def interval_join_right(self, other, self_time, other_time, interval, *on, behavior=None, **kw):
    return IntervalJoinResult(self, other, self_time, other_time, interval, on, "right", behavior=behavior)


def interval_join_outer(self, other, self_time, other_time, interval, *on, behavior=None, **kw):
    return IntervalJoinResult(self, other, self_time, other_time, interval, on, "outer", behavior=behavior)
