"""asof_join (reference stdlib/temporal/asof_join.py, 1109 LoC).

Built on RecomputeNode: both sides sort-merged on the host per change,
output diffs stay incremental.  asof_now_join matches the current state of
the right side (same semantics in this synchronous engine).
"""

from __future__ import annotations

import bisect
from enum import Enum
from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as ex
from pathway_amd.internals import thisclass
from pathway_amd.internals.api import hash_values, Pointer
from pathway_amd.internals.config import get_device
from pathway_amd.internals.table import Table
from pathway_amd.internals.universe import Universe


class Direction(Enum):
    BACKWARD = "backward"
    FORWARD = "forward"
    NEAREST = "nearest"


class AsofJoinResult:
    def __init__(self, left, right, left_time, right_time, on, mode, direction=Direction.BACKWARD, defaults=None):
        self._left = left
        self._right = right
        self._left_time = left_time
        self._right_time = right_time
        self._on = list(on)
        self._mode = mode
        self._direction = direction
        self._defaults = defaults or {}

    def select(self, *args: Any, **kwargs: Any) -> Table:
        from pathway_amd.engine.nodes_recompute import RecomputeNode

        left, right = self._left, self._right
        lt = thisclass.substitute_this(
            ex.wrap_expr(self._left_time), {thisclass.this: left, thisclass.left: left}
        )
        rt = thisclass.substitute_this(
            ex.wrap_expr(self._right_time), {thisclass.this: right, thisclass.right: right}
        )
        if not isinstance(lt, ex.ColumnReference) or not isinstance(rt, ex.ColumnReference):
            raise NotImplementedError("asof_join times must be plain columns")
        lt_name, rt_name = lt.name, rt.name
        on_pairs = []
        for c in self._on:
            c = thisclass.substitute_this(
                ex.wrap_expr(c), {thisclass.left: left, thisclass.right: right}
            )
            if not (isinstance(c, ex.ColumnBinaryOpExpression) and c._symbol == "=="):
                raise ValueError("asof_join on-conditions must be equalities")
            on_pairs.append((c._left.name, c._right.name))

        out_exprs: dict[str, Any] = {}
        for a in args:
            a2 = thisclass.substitute_this(
                ex.wrap_expr(a), {thisclass.left: left, thisclass.right: right}
            )
            if not isinstance(a2, ex.ColumnReference):
                raise ValueError("positional args must be column refs")
            side = "l" if (a2.table is left or getattr(a2.table, "_node", None) is left._node) else "r"
            out_exprs[a2.name] = (side, a2.name)
        for name, e in kwargs.items():
            e2 = thisclass.substitute_this(
                ex.wrap_expr(e), {thisclass.left: left, thisclass.right: right}
            )
            if not isinstance(e2, ex.ColumnReference):
                raise NotImplementedError("asof select supports plain column refs")
            side = "l" if (e2.table is left or getattr(e2.table, "_node", None) is left._node) else "r"
            out_exprs[name] = (side, e2.name)

        mode = self._mode
        direction = self._direction
        defaults = {
            (k.name if isinstance(k, ex.ColumnReference) else k): v
            for k, v in self._defaults.items()
        }
        out_columns = list(out_exprs.keys())

        # --- tensor-native incremental path (engine/nodes_asof.py) ---
        import os

        int_like = (dt.INT, dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC, dt.DURATION)
        lt_dt = dt.unoptionalize(left._dtypes.get(lt_name, dt.ANY))
        rt_dt = dt.unoptionalize(right._dtypes.get(rt_name, dt.ANY))
        if (
            not os.environ.get("PW_ASOF_HOST")
            and direction in (Direction.BACKWARD, Direction.FORWARD)
            and mode in ("inner", "left")
            and lt_dt in int_like
            and rt_dt in int_like
        ):
            from pathway_amd.engine.nodes_asof import AsofJoinNode

            lmap = {n: src for n, (side, src) in out_exprs.items() if side == "l"}
            rmap = {n: src for n, (side, src) in out_exprs.items() if side == "r"}
            node = AsofJoinNode(
                left._node,
                right._node,
                [ex.ColumnReference(left, ln) for (ln, _) in on_pairs],
                [ex.ColumnReference(right, rn) for (_, rn) in on_pairs],
                ex.ColumnReference(left, lt_name),
                ex.ColumnReference(right, rt_name),
                lmap,
                rmap,
                mode,
                direction.value,
                get_device(),
                defaults=defaults,
            )
            out_dtypes = {}
            for name, (side, src) in out_exprs.items():
                srcd = (left if side == "l" else right)._dtypes.get(src, dt.ANY)
                opt = side == "r" and mode == "left"
                out_dtypes[name] = dt.Optional(srcd) if opt else srcd
            return Table(node, out_dtypes, Universe())

        def fn(in_rows, in_keys):
            lrows, rrows = in_rows
            lkeys, rkeys = in_keys
            # group right by on-columns, sorted by time
            rgroups: dict[tuple, list] = {}
            for row, key in zip(rrows, rkeys):
                g = tuple(row[rn] for (_, rn) in on_pairs)
                rgroups.setdefault(g, []).append((row[rt_name], row, key))
            for g in rgroups:
                rgroups[g].sort(key=lambda x: (x[0],))
            lgroups: dict[tuple, list] = {}
            for row, key in zip(lrows, lkeys):
                g = tuple(row[ln] for (ln, _) in on_pairs)
                lgroups.setdefault(g, []).append((row[lt_name], row, key))
            out = []
            matched_right = set()

            def emit(lrow, lkey, rrow, rkey):
                okey_vals = [repr(lkey) if lkey is not None else None,
                             repr(rkey) if rkey is not None else None]
                lo, hi = hash_values(okey_vals)
                vals = {}
                for name, (side, src) in out_exprs.items():
                    if side == "l":
                        vals[name] = lrow[src] if lrow is not None else defaults.get(name)
                    else:
                        vals[name] = rrow[src] if rrow is not None else defaults.get(name)
                out.append((Pointer(lo, hi), vals))

            for g, lrs in lgroups.items():
                rl = rgroups.get(g, [])
                rtimes = [x[0] for x in rl]
                for ltv, lrow, lkey in lrs:
                    idx = None
                    if direction == Direction.BACKWARD:
                        i = bisect.bisect_right(rtimes, ltv) - 1
                        idx = i if i >= 0 else None
                    elif direction == Direction.FORWARD:
                        i = bisect.bisect_left(rtimes, ltv)
                        idx = i if i < len(rtimes) else None
                    else:  # NEAREST
                        if rtimes:
                            i = bisect.bisect_right(rtimes, ltv) - 1
                            j = i + 1
                            cand = []
                            if i >= 0:
                                cand.append((abs(ltv - rtimes[i]), i))
                            if j < len(rtimes):
                                cand.append((abs(rtimes[j] - ltv), j))
                            idx = min(cand)[1] if cand else None
                    if idx is not None:
                        rtv, rrow, rkey = rl[idx]
                        matched_right.add(repr(rkey))
                        emit(lrow, lkey, rrow, rkey)
                    elif mode in ("left", "outer"):
                        emit(lrow, lkey, None, None)
            if mode in ("right", "outer"):
                for g, rl in rgroups.items():
                    for rtv, rrow, rkey in rl:
                        if repr(rkey) not in matched_right:
                            emit(None, None, rrow, rkey)
            return out

        out_dtypes = {}
        for name, (side, src) in out_exprs.items():
            srcd = (left if side == "l" else right)._dtypes.get(src, dt.ANY)
            opt = (side == "r" and mode in ("left", "outer")) or (
                side == "l" and mode in ("right", "outer")
            )
            out_dtypes[name] = dt.Optional(srcd) if opt else srcd

        node = RecomputeNode(
            [left._node, right._node], fn, out_columns, out_dtypes, get_device()
        )
        return Table(node, out_dtypes, Universe())


def asof_join(self, other, self_time, other_time, *on, how=None, defaults=None, direction=Direction.BACKWARD, **kw):
    mode = how.value if hasattr(how, "value") else (how or "inner")
    return AsofJoinResult(self, other, self_time, other_time, on, mode, direction, defaults)


def asof_join_left(self, other, self_time, other_time, *on, defaults=None, direction=Direction.BACKWARD, **kw):
    return AsofJoinResult(self, other, self_time, other_time, on, "left", direction, defaults)


def asof_join_right(self, other, self_time, other_time, *on, defaults=None, direction=Direction.BACKWARD, **kw):
    return AsofJoinResult(self, other, self_time, other_time, on, "right", direction, defaults)


def asof_join_outer(self, other, self_time, other_time, *on, defaults=None, direction=Direction.BACKWARD, **kw):
    return AsofJoinResult(self, other, self_time, other_time, on, "outer", direction, defaults)


class AsofNowJoinResult:
    """asof_now: each left row is answered with the right side AS OF its
    arrival; later right-side changes do not retro-update the answer
    (reference asof_now_join / use-as-of-now semantics)."""

    def __init__(self, left, right, on, mode):
        from pathway_amd.internals.joins import JoinMode, JoinResult

        self._inner = JoinResult(
            left,
            right,
            list(on),
            JoinMode(mode),
            assign_id=left.id,
            probe_only_left=True,
        )
        self._left = left

    def select(self, *args, **kwargs):
        return self._inner.select(*args, **kwargs)


def asof_now_join(self, other, *on, how=None, **kw):
    mode = how.value if hasattr(how, "value") else (how or "inner")
    return AsofNowJoinResult(self, other, on, mode)


def asof_now_join_inner(self, other, *on, **kw):
    return asof_now_join(self, other, *on, how="inner")


def asof_now_join_left(self, other, *on, **kw):
    return asof_now_join(self, other, *on, how="left")
