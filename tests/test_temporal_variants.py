"""Temporal-join variant coverage: asof directions/modes, interval join
right/outer, window join modes (modeled on the reference's
test_asof_join.py / test_interval_join.py / test_window_join.py)."""

import pytest

import pathway_amd as pw
from pathway_amd.debug import table_from_markdown as T, table_to_dicts
from pathway_amd.stdlib.temporal import Direction


def _rows(table, *names):
    _, cols = table_to_dicts(table)
    ids = list(cols[names[0]].keys())
    return sorted(
        (tuple(cols[n][i] for n in names) for i in ids),
        key=lambda r: tuple((x is None, x) for x in r),
    )


LEFT = """
t  | v
1  | a
5  | b
9  | c
"""

RIGHT = """
s  | w
2  | X
6  | Y
"""


def test_asof_backward():
    l, r = T(LEFT), T(RIGHT)
    res = l.asof_join(
        r, l.t, r.s, how=pw.JoinMode.INNER, direction=Direction.BACKWARD
    ).select(pw.left.v, pw.right.w)
    # backward: right row with largest s <= t
    assert _rows(res, "v", "w") == [("b", "X"), ("c", "Y")]


def test_asof_forward():
    l, r = T(LEFT), T(RIGHT)
    res = l.asof_join(
        r, l.t, r.s, how=pw.JoinMode.INNER, direction=Direction.FORWARD
    ).select(pw.left.v, pw.right.w)
    # forward: right row with smallest s >= t
    assert _rows(res, "v", "w") == [("a", "X"), ("b", "Y")]


def test_asof_nearest():
    l, r = T(LEFT), T(RIGHT)
    res = l.asof_join(
        r, l.t, r.s, how=pw.JoinMode.INNER, direction=Direction.NEAREST
    ).select(pw.left.v, pw.right.w)
    # nearest: 1→2(X), 5→6(Y), 9→6(Y)
    assert _rows(res, "v", "w") == [("a", "X"), ("b", "Y"), ("c", "Y")]


def test_asof_left_with_defaults():
    l, r = T(LEFT), T(RIGHT)
    res = l.asof_join_left(
        r, l.t, r.s, defaults={r.w: "none"}, direction=Direction.BACKWARD
    ).select(pw.left.v, pw.right.w)
    assert _rows(res, "v", "w") == [("a", "none"), ("b", "X"), ("c", "Y")]


def test_interval_join_right_and_outer():
    l = T(
        """
        t | a
        0 | p
        10 | q
        """
    )
    r = T(
        """
        s | b
        1 | u
        50 | v
        """
    )
    inner_right = l.interval_join_right(
        r, l.t, r.s, pw.temporal.interval(-2, 2)
    ).select(pw.left.a, pw.right.b)
    assert _rows(inner_right, "a", "b") == [("p", "u"), (None, "v")]

    outer = l.interval_join_outer(
        r, l.t, r.s, pw.temporal.interval(-2, 2)
    ).select(pw.left.a, pw.right.b)
    assert _rows(outer, "a", "b") == [("p", "u"), ("q", None), (None, "v")]


def test_interval_join_with_exact_on_condition():
    l = T(
        """
        t | k | a
        0 | 1 | p
        0 | 2 | q
        """
    )
    r = T(
        """
        s | k | b
        1 | 1 | u
        1 | 3 | v
        """
    )
    res = l.interval_join(
        r, l.t, r.s, pw.temporal.interval(-2, 2), l.k == r.k
    ).select(pw.left.a, pw.right.b)
    assert _rows(res, "a", "b") == [("p", "u")]


def test_window_join_left():
    l = T(
        """
        t | a
        1 | p
        7 | q
        """
    )
    r = T(
        """
        t | b
        2 | u
        """
    )
    res = l.window_join_left(
        r, l.t, r.t, pw.temporal.tumbling(duration=5)
    ).select(pw.left.a, pw.right.b)
    assert _rows(res, "a", "b") == [("p", "u"), ("q", None)]


def test_window_join_sliding_multiplicity():
    l = T(
        """
        t | a
        3 | p
        """
    )
    r = T(
        """
        t | b
        4 | u
        """
    )
    # hop 2, duration 4: windows [0,4), [2,6) — t=3 in both; t=4 in [2,6),[4,8)
    res = l.window_join_inner(
        r, l.t, r.t, pw.temporal.sliding(hop=2, duration=4)
    ).select(pw.left.a, pw.right.b)
    assert _rows(res, "a", "b") == [("p", "u")]


def test_windowby_sliding_counts():
    t = T(
        """
        t | v
        0 | 1
        1 | 1
        3 | 1
        """
    )
    res = t.windowby(t.t, window=pw.temporal.sliding(hop=2, duration=4)).reduce(
        start=pw.this._pw_window_start,
        n=pw.reducers.sum(pw.this.v),
    )
    assert _rows(res, "start", "n") == [(-2, 2), (0, 3), (2, 1)]


def test_session_window_max_gap_merges():
    t = T(
        """
        t | v
        1 | 1
        2 | 1
        9 | 1
        """
    )
    res = t.windowby(
        t.t, window=pw.temporal.session(max_gap=3)
    ).reduce(n=pw.reducers.sum(pw.this.v))
    assert sorted(x[0] for x in _rows(res, "n")) == [1, 2]


def test_asof_now_join_left_pads():
    t = T(
        """
        a | __time__
        1 |    2
        2 |    4
        """
    )
    r = T(
        """
        a | b | __time__
        1 | x |    2
        """
    )
    res = t.asof_now_join_left(r, t.a == r.a).select(t.a, r.b)
    assert _rows(res, "a", "b") == [(1, "x"), (2, None)]


def test_windowby_with_instance():
    t = T(
        """
        t | u | v
        1 | a | 1
        2 | a | 1
        1 | b | 5
        """
    )
    res = t.windowby(
        t.t, window=pw.temporal.tumbling(duration=5), instance=t.u
    ).reduce(u=pw.this._pw_instance, s=pw.reducers.sum(pw.this.v))
    assert _rows(res, "u", "s") == [("a", 2), ("b", 5)]


def test_interval_join_with_behavior_cutoff():
    # late left rows past the cutoff are dropped from the join output
    l = T(
        """
        t | a | __time__
        1 | p |    2
        10 | q |    4
        1 | r |    8
        """
    )
    r = T(
        """
        s | b | __time__
        1 | u |    2
        10 | w |    4
        """
    )
    res = l.interval_join(
        r,
        l.t,
        r.s,
        pw.temporal.interval(-1, 1),
        behavior=pw.temporal.common_behavior(cutoff=2),
    ).select(pw.left.a, pw.right.b)
    got = _rows(res, "a", "b")
    # p joined u (on time), q joined w; the r-row (t=1) arrived when the
    # watermark (10) had passed 1+cutoff → dropped
    assert ("p", "u") in got and ("q", "w") in got
    assert ("r", "u") not in got


def test_window_factory_params():
    # origin shifts the tumbling grid
    t = T(
        """
        t | v
        1 | 1
        4 | 1
        """
    )
    r = t.windowby(
        t.t, window=pw.temporal.tumbling(duration=5, origin=1)
    ).reduce(start=pw.this._pw_window_start, n=pw.reducers.count())
    assert _rows(r, "start", "n") == [(1, 2)]


def test_sliding_ratio():
    t = T(
        """
        t | v
        0 | 1
        3 | 1
        """
    )
    r = t.windowby(
        t.t, window=pw.temporal.sliding(hop=2, ratio=2)
    ).reduce(start=pw.this._pw_window_start, n=pw.reducers.count())
    assert _rows(r, "start", "n") == [(-2, 1), (0, 2), (2, 1)]


def test_session_predicate():
    t = T(
        """
        t | v
        1 | 1
        2 | 1
        10 | 1
        """
    )
    r = t.windowby(
        t.t, window=pw.temporal.session(predicate=lambda a, b: abs(a - b) <= 3)
    ).reduce(n=pw.reducers.count())
    assert sorted(x[0] for x in _rows(r, "n")) == [1, 2]


def test_asof_now_left_pads_frozen_before_first_right_row():
    """as-of-now answers (incl. left-outer pads) freeze at probe time:
    left rows arriving before ANY right row keep their null-padded
    answer even after the right side later fills (r2 regression —
    reference use-as-of-now semantics)."""
    import datetime

    import pathway_amd as pw
    import pathway_amd.stdlib.temporal.time_utils as tu
    from pathway_amd.internals.rungraph import G

    tu.utc_now.cache_clear()
    G.clear()
    t = pw.debug.table_from_markdown("a\n1\n2\n")
    right = tu.utc_now(
        refresh_rate=datetime.timedelta(milliseconds=100),
        initial_delay=datetime.timedelta(milliseconds=300),
        max_ticks=2,
    ).reduce(timestamp_utc=pw.reducers.latest(pw.this.timestamp_utc))
    res = t.asof_now_join_left(right).select(
        pw.left.a, ts=pw.right.timestamp_utc
    )
    _, cols = pw.debug.table_to_dicts(res)
    assert sorted(cols["a"].values()) == [1, 2]
    assert all(v is None for v in cols["ts"].values())
    tu.utc_now.cache_clear()
