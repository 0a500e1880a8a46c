"""window_join (reference stdlib/temporal/window_join.py, 1217 LoC).

Both sides are window-assigned (same Window), then equi-joined on
(window_start, window_end) + extra conditions — runs on the GPU join path.
"""

from __future__ import annotations

from typing import Any

from pathway_amd.internals import expression as ex
from pathway_amd.internals import thisclass
from pathway_amd.internals.table import Table
from pathway_amd.stdlib.temporal._window import Window, WindowedTable


class WindowJoinResult:
    def __init__(self, left, right, left_time, right_time, window, on, mode):
        self._mode = mode
        lw = WindowedTable(left, left_time, window)
        rw = WindowedTable(right, right_time, window)
        self._lx, _ = lw._prepare()
        self._rx, _ = rw._prepare()
        self._left, self._right = left, right
        self._on = list(on)

    def select(self, *args: Any, **kwargs: Any) -> Table:
        lx, rx = self._lx, self._rx
        conds = [
            lx._pw_window_start == rx._pw_window_start,
            lx._pw_window_end == rx._pw_window_end,
        ]
        for c in self._on:
            c = thisclass.substitute_this(
                ex.wrap_expr(c), {thisclass.left: lx, thisclass.right: rx}
            )
            conds.append(_remap(c, self._left, self._right, lx, rx))
        join_fn = {
            "inner": lx.join_inner,
            "left": lx.join_left,
            "right": lx.join_right,
            "outer": lx.join_outer,
        }[self._mode]
        j = join_fn(rx, *conds)
        args = [
            _remap(
                thisclass.substitute_this(
                    ex.wrap_expr(a), {thisclass.left: lx, thisclass.right: rx}
                ),
                self._left,
                self._right,
                lx,
                rx,
            )
            for a in args
        ]
        kwargs = {
            k: _remap(
                thisclass.substitute_this(
                    ex.wrap_expr(v), {thisclass.left: lx, thisclass.right: rx}
                ),
                self._left,
                self._right,
                lx,
                rx,
            )
            for k, v in kwargs.items()
        }
        return j.select(*args, **kwargs)


def _remap(e, left, right, lx, rx):
    if isinstance(e, ex.ColumnReference):
        if e.table is left or getattr(e.table, "_node", None) is left._node:
            return ex.ColumnReference(lx, e.name)
        if e.table is right or getattr(e.table, "_node", None) is right._node:
            return ex.ColumnReference(rx, e.name)
        return e
    if not isinstance(e, ex.ColumnExpression):
        return e
    new = object.__new__(type(e))
    new.__dict__.update(e.__dict__)
    for attr, val in list(e.__dict__.items()):
        if isinstance(val, ex.ColumnExpression):
            new.__dict__[attr] = _remap(val, left, right, lx, rx)
        elif isinstance(val, tuple) and any(isinstance(v, ex.ColumnExpression) for v in val):
            new.__dict__[attr] = tuple(
                _remap(v, left, right, lx, rx) if isinstance(v, ex.ColumnExpression) else v
                for v in val
            )
    return new


def window_join(self, other, self_time, other_time, window: Window, *on, how: Any = "inner", **kw):
    mode = how.value if hasattr(how, "value") else (how or "inner")
    return WindowJoinResult(self, other, self_time, other_time, window, on, mode)


def window_join_inner(self, other, self_time, other_time, window, *on, **kw):
    return WindowJoinResult(self, other, self_time, other_time, window, on, "inner")


def window_join_left(self, other, self_time, other_time, window, *on, **kw):
    return WindowJoinResult(self, other, self_time, other_time, window, on, "left")


def window_join_right(self, other, self_time, other_time, window, *on, **kw):
    return WindowJoinResult(self, other, self_time, other_time, window, on, "right")


def window_join_outer(self, other, self_time, other_time, window, *on, **kw):
    return WindowJoinResult(self, other, self_time, other_time, window, on, "outer")
