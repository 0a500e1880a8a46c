"""Windows + windowby (reference stdlib/temporal/_window.py:39-880;
engine window assignment: src/engine/dataflow/windows.rs:166).

Window assignment is expression-level (tensor ops on device for numeric
time columns; host apply for datetimes), followed by the standard
GroupReduce — so windowed aggregation runs on the same GPU segmented-reduce
path as plain groupby.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as ex
from pathway_amd.internals import thisclass
from pathway_amd.internals.table import Table


class Window:
    pass


@dataclass
class TumblingWindow(Window):
    duration: Any
    origin: Any = None
    shift: Any = None


@dataclass
class SlidingWindow(Window):
    hop: Any
    duration: Any = None
    ratio: int | None = None
    origin: Any = None


@dataclass
class SessionWindow(Window):
    predicate: Any = None
    max_gap: Any = None


@dataclass
class IntervalsOverWindow(Window):
    at: Any = None
    lower_bound: Any = None
    upper_bound: Any = None
    is_outer: bool = True


def tumbling(duration: Any = None, origin: Any = None, shift: Any = None) -> TumblingWindow:
    if duration is None and shift is not None:
        duration = shift
    return TumblingWindow(duration, origin, shift)


def sliding(hop: Any, duration: Any = None, ratio: int | None = None, origin: Any = None) -> SlidingWindow:
    return SlidingWindow(hop, duration, ratio, origin)


def session(predicate: Any = None, max_gap: Any = None) -> SessionWindow:
    return SessionWindow(predicate, max_gap)


def intervals_over(
    *, at: Any = None, lower_bound: Any = None, upper_bound: Any = None, is_outer: bool = True
) -> IntervalsOverWindow:
    return IntervalsOverWindow(at, lower_bound, upper_bound, is_outer)


def _is_numeric(v: Any) -> bool:
    return isinstance(v, (int, float)) and not isinstance(v, bool)


def _as_number(v: Any):
    import pandas as pd

    if isinstance(v, pd.Timedelta):
        return int(v.value)
    if isinstance(v, pd.Timestamp):
        return int(v.value)
    return v


class WindowedTable:
    """Result of windowby; reduce() completes the windowed aggregation."""

    def __init__(self, table: Table, time_expr, window: Window, behavior=None, instance=None):
        self._table = table
        self._time_expr = table._resolve(time_expr)
        self._window = window
        self._behavior = behavior
        self._instance = table._resolve(instance) if instance is not None else None

    def _prepare(self) -> tuple[Table, list[str]]:
        """Assign windows: return (table with _pw_window_{start,end}, gb names)."""
        t = self._table
        w = self._window
        texpr = self._time_expr
        if isinstance(w, TumblingWindow):
            d = w.duration
            origin = w.origin if w.origin is not None else 0
            if _is_numeric(d) and _is_numeric(origin):
                start = ((texpr - origin) // d) * d + origin
            else:
                dn = _as_number(d)
                on = _as_number(origin) if w.origin is not None else None

                def f(tv):
                    tvn = _as_number(tv)
                    o = on if on is not None else 0
                    s = ((tvn - o) // dn) * dn + o
                    return _back_like(s, tv)

                start = ex.ApplyExpression(f, None, texpr)
            end = _add_like(start, d)
            cols: dict[str, Any] = {
                "_pw_window_start": start,
                "_pw_window_end": end,
            }
            extra = self._with_instance(cols)
            out = t.with_columns(**cols, **extra)
            return out, ["_pw_window_start", "_pw_window_end"] + list(extra.keys())
        if isinstance(w, SlidingWindow):
            hop = w.hop
            duration = w.duration if w.duration is not None else (
                w.ratio * hop if w.ratio is not None else hop
            )
            origin = w.origin if w.origin is not None else 0
            nwin = (
                int(math.ceil(_as_number(duration) / _as_number(hop)))
                if _as_number(duration) is not None
                else 1
            )
            parts = []
            for k in range(nwin):
                if _is_numeric(hop) and _is_numeric(origin):
                    smax = ((texpr - origin) // hop) * hop + origin
                    start = smax - k * hop
                else:
                    hn, dn2 = _as_number(hop), _as_number(duration)
                    on = _as_number(origin)

                    def f(tv, k=k, hn=hn, on=on):
                        tvn = _as_number(tv)
                        s = ((tvn - on) // hn) * hn + on - k * hn
                        return _back_like(s, tv)

                    start = ex.ApplyExpression(f, None, texpr)
                end = _add_like(start, duration)
                cols = {"_pw_window_start": start, "_pw_window_end": end}
                extra = self._with_instance(cols)
                part = t.with_columns(**cols, **extra)
                # keep only windows that actually contain the row
                part = part.filter(
                    (ex.ColumnReference(part, "_pw_window_end") > self._retime(part))
                    & (ex.ColumnReference(part, "_pw_window_start") <= self._retime(part))
                )
                parts.append(part)
            out = parts[0].concat_reindex(*parts[1:]) if len(parts) > 1 else parts[0]
            return out, ["_pw_window_start", "_pw_window_end"] + list(
                self._with_instance({}).keys()
            )
        if isinstance(w, SessionWindow):
            return self._prepare_session(w)
        if isinstance(w, IntervalsOverWindow):
            return self._prepare_intervals_over(w)
        raise TypeError(f"unknown window {w!r}")

    def _prepare_session(self, w: "SessionWindow"):
        """Session windows: merge rows whose gap ≤ max_gap (or predicate).

        Order-dependent merge → RecomputeNode host path (reference
        windows.rs session merge); downstream groupby stays incremental.
        """
        from pathway_amd.engine.nodes_recompute import RecomputeNode
        from pathway_amd.internals import dtype as dt
        from pathway_amd.internals.config import get_device
        from pathway_amd.internals.universe import Universe

        t = self._table
        texpr = self._time_expr
        if not isinstance(texpr, ex.ColumnExpression):
            raise TypeError("session windowby needs a time expression")
        # materialize the needed columns (time + instance + all original)
        cols = {n: ex.ColumnReference(t, n) for n in t._dtypes}
        cols["_pw_t"] = texpr
        if self._instance is not None:
            cols["_pw_instance"] = self._instance
        src = t.select(**cols)
        max_gap = _as_number(w.max_gap) if w.max_gap is not None else None
        predicate = w.predicate

        import os

        tdt = dt.unoptionalize(src._dtypes.get("_pw_t", dt.ANY))
        if (
            not os.environ.get("PW_SESSION_HOST")
            and predicate is None
            and isinstance(max_gap, int)
            and tdt in (dt.INT, dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC, dt.DURATION)
        ):
            # tensor path: affected-instance recompute on device
            from pathway_amd.engine.nodes_session import SessionAssignNode

            node = SessionAssignNode(
                src._node,
                "_pw_t",
                "_pw_instance" if self._instance is not None else None,
                max_gap,
                get_device(),
            )
            out_dtypes = dict(src._dtypes)
            out_dtypes["_pw_window_start"] = tdt
            out_dtypes["_pw_window_end"] = tdt
            gb = ["_pw_window_start", "_pw_window_end"]
            if self._instance is not None:
                gb.append("_pw_instance")
            return Table(node, out_dtypes, Universe()), gb
        out_columns = list(src._dtypes.keys()) + [
            "_pw_window_start",
            "_pw_window_end",
        ]

        def fn(in_rows, in_keys):
            rows, keys = in_rows[0], in_keys[0]
            groups: dict = {}
            for row, key in zip(rows, keys):
                g = row.get("_pw_instance")
                groups.setdefault(g, []).append((row["_pw_t"], row, key))
            out = []
            for g, rl in groups.items():
                rl.sort(key=lambda x: x[0])
                sess: list = []
                for tv, row, key in rl:
                    if sess:
                        prev_t = sess[-1][0]
                        merge = (
                            predicate(prev_t, tv)
                            if predicate is not None
                            else (max_gap is not None and tv - prev_t <= max_gap)
                        )
                    else:
                        merge = False
                    if sess and not merge:
                        out.extend(_emit_session(sess))
                        sess = []
                    sess.append((tv, row, key))
                if sess:
                    out.extend(_emit_session(sess))
            return out

        def _emit_session(sess):
            start = sess[0][0]
            end = sess[-1][0]
            for tv, row, key in sess:
                vals = dict(row)
                vals["_pw_window_start"] = start
                vals["_pw_window_end"] = end
                yield (key, vals)

        out_dtypes = dict(src._dtypes)
        out_dtypes["_pw_window_start"] = src._dtypes.get("_pw_t", dt.ANY)
        out_dtypes["_pw_window_end"] = src._dtypes.get("_pw_t", dt.ANY)
        node = RecomputeNode([src._node], fn, out_columns, out_dtypes, get_device())
        out = Table(node, out_dtypes, Universe())
        gb = ["_pw_window_start", "_pw_window_end"]
        if self._instance is not None:
            gb.append("_pw_instance")
        return out, gb

    def _prepare_intervals_over(self, w: "IntervalsOverWindow"):
        """intervals_over: windows [at+lb, at+ub] centered on the `at`
        column's values — an interval join + groupby on the at-point."""
        from pathway_amd.stdlib.temporal._interval_join import (
            IntervalJoinResult,
            interval,
        )
        from pathway_amd.internals import thisclass

        at_ref = w.at
        if not isinstance(at_ref, ex.ColumnReference):
            raise TypeError("intervals_over(at=...) must be a column reference")
        at_table = at_ref.table
        t = self._table
        lb = _as_number(w.lower_bound)
        ub = _as_number(w.upper_bound)
        ij = IntervalJoinResult(
            at_table, t, at_ref, self._time_expr, interval(lb, ub), [], "inner"
        )
        sel = {n: ex.ColumnReference(thisclass.right, n) for n in t._dtypes}
        sel["_pw_window_location"] = ex.ColumnReference(thisclass.left, at_ref.name)
        joined = ij.select(**sel)
        joined = joined.with_columns(
            _pw_window_start=ex.ColumnReference(joined, "_pw_window_location") + lb,
            _pw_window_end=ex.ColumnReference(joined, "_pw_window_location") + ub,
        )
        return joined, ["_pw_window_location", "_pw_window_start", "_pw_window_end"]

    def _retime(self, part: Table):
        """time expression rebound to the expanded table."""
        e = self._time_expr
        if isinstance(e, ex.ColumnReference):
            return ex.ColumnReference(part, e.name)
        return e

    def _with_instance(self, cols: dict) -> dict:
        if self._instance is None:
            return {}
        return {"_pw_instance": self._instance}

    def reduce(self, *args: Any, **kwargs: Any) -> Table:
        prepared, gb_names = self._prepare()
        prepared = self._apply_behavior(prepared)
        gb_refs = [ex.ColumnReference(prepared, n) for n in gb_names]
        grouped = prepared.groupby(*gb_refs)
        return grouped.reduce(*args, **kwargs)

    def _apply_behavior(self, prepared: Table) -> Table:
        """Wire delay/cutoff behaviors as buffer/forget/freeze nodes
        (reference temporal_behavior.py:10-100 + time_column.rs)."""
        from pathway_amd.stdlib.temporal.temporal_behavior import (
            CommonBehavior,
            ExactlyOnceBehavior,
        )
        from pathway_amd.engine.nodes_temporal import (
            BufferNode,
            ForgetNode,
            FreezeNode,
        )
        from pathway_amd.internals.config import get_device

        beh = self._behavior
        if beh is None:
            return prepared
        time_expr = self._retime(prepared)
        start_ref = ex.ColumnReference(prepared, "_pw_window_start")
        end_ref = ex.ColumnReference(prepared, "_pw_window_end")
        if isinstance(beh, ExactlyOnceBehavior):
            shift = _as_number(beh.shift) if beh.shift is not None else 0
            beh = CommonBehavior(delay=None, cutoff=shift, keep_results=True)
            buf = BufferNode(
                prepared._node, end_ref + shift, time_expr, get_device()
            )
            prepared = Table(buf, prepared._dtypes, prepared._universe)
            node = FreezeNode(
                prepared._node,
                ex.ColumnReference(prepared, "_pw_window_end") + shift,
                self._retime(prepared),
                get_device(),
            )
            node.wm_source = buf  # the buffer sees the unfiltered stream
            return Table(node, prepared._dtypes, prepared._universe)
        if isinstance(beh, CommonBehavior):
            buf = None
            if beh.delay is not None:
                buf = BufferNode(
                    prepared._node,
                    start_ref + _as_number(beh.delay),
                    time_expr,
                    get_device(),
                )
                prepared = Table(buf, prepared._dtypes, prepared._universe)
            if beh.cutoff is not None:
                thr = (
                    ex.ColumnReference(prepared, "_pw_window_end")
                    + _as_number(beh.cutoff)
                )
                cls = FreezeNode if beh.keep_results else ForgetNode
                node = cls(
                    prepared._node, thr, self._retime(prepared), get_device()
                )
                if buf is not None:
                    node.wm_source = buf
                prepared = Table(node, prepared._dtypes, prepared._universe)
        return prepared


def _add_like(start_expr, d):
    return start_expr + d


def _back_like(num, proto):
    import pandas as pd

    if isinstance(proto, pd.Timestamp):
        return type(proto)(pd.Timestamp(num, unit="ns"))
    return num


def windowby(
    self: Table,
    time_expr: Any,
    *,
    window: Window,
    behavior=None,
    instance: Any = None,
    **kwargs: Any,
) -> WindowedTable:
    return WindowedTable(self, time_expr, window, behavior, instance)
