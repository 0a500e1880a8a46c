"""Streaming semantics, windows, iterate, temporal joins."""

import pytest

import pathway_amd as pw
from pathway_amd.debug import (
    assert_table_equality_wo_index,
    table_from_markdown as T,
)


def test_update_stream_times():
    t = T(
        """
        a | __time__ | __diff__
        1 | 0        | 1
        2 | 2        | 1
        1 | 4        | -1
        """
    )
    cap = t._capture()
    from pathway_amd.engine.runtime import Runtime
    from pathway_amd.internals.rungraph import reset_all

    rt = Runtime([cap])
    reset_all(rt.nodes)
    rt.run()
    stream = [(r.values[0], r.time, r.diff) for r in cap.rows]
    assert stream == [(1, 0, 1), (2, 2, 1), (1, 4, -1)]


def test_tumbling_window():
    t = T(
        """
        t  | v
        1  | 1
        3  | 2
        7  | 3
        12 | 4
        """
    )
    res = t.windowby(pw.this.t, window=pw.temporal.tumbling(duration=5)).reduce(
        start=pw.this._pw_window_start,
        end=pw.this._pw_window_end,
        s=pw.reducers.sum(pw.this.v),
    )
    expected = T(
        """
        start | end | s
        0     | 5   | 3
        5     | 10  | 3
        10    | 15  | 4
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_sliding_window():
    t = T(
        """
        t | v
        2 | 1
        6 | 2
        """
    )
    res = t.windowby(
        pw.this.t, window=pw.temporal.sliding(hop=2, duration=4)
    ).reduce(
        start=pw.this._pw_window_start,
        s=pw.reducers.sum(pw.this.v),
    )
    # t=2 -> windows starting 0, 2; t=6 -> windows starting 4, 6
    expected = T(
        """
        start | s
        0     | 1
        2     | 1
        4     | 2
        6     | 2
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_iterate_collatz():
    t = T(
        """
        n
        3
        5
        """
    )

    def logic(t):
        return t.select(
            n=pw.if_else(
                t.n == 1,
                1,
                pw.if_else(t.n % 2 == 0, t.n // 2, 3 * t.n + 1),
            )
        )

    res = pw.iterate(logic, t=t)
    expected = T(
        """
        n
        1
        1
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_interval_join_inner():
    a = T(
        """
        t | x
        1 | a1
        5 | a5
        """
    )
    b = T(
        """
        t | y
        2 | b2
        9 | b9
        """
    )
    res = pw.temporal.interval_join(
        a, b, a.t, b.t, pw.temporal.interval(-2, 2)
    ).select(pw.left.x, pw.right.y)
    expected = T(
        """
        x  | y
        a1 | b2
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_asof_join():
    trades = T(
        """
        t  | price
        3  | 100
        7  | 101
        """
    )
    quotes = T(
        """
        t | bid
        1 | 99
        5 | 98
        6 | 97
        """
    )
    res = pw.temporal.asof_join(
        trades, quotes, trades.t, quotes.t
    ).select(pw.left.price, pw.right.bid)
    expected = T(
        """
        price | bid
        100   | 99
        101   | 97
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_window_join():
    a = T(
        """
        t | x
        1 | 1
        6 | 2
        """
    )
    b = T(
        """
        t | y
        2 | 10
        7 | 20
        """
    )
    res = pw.temporal.window_join(
        a, b, a.t, b.t, pw.temporal.tumbling(duration=5)
    ).select(pw.left.x, pw.right.y)
    expected = T(
        """
        x | y
        1 | 10
        2 | 20
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_deduplicate():
    t = T(
        """
        v | __time__ | __diff__
        1 | 0        | 1
        1 | 2        | 1
        3 | 4        | 1
        """
    )
    res = t.deduplicate(value=pw.this.v)
    expected = T(
        """
        v
        3
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_ordered_diff():
    t = T(
        """
        t | v
        1 | 10
        2 | 13
        4 | 20
        """
    )
    res = pw.ordered.diff(t, pw.this.t, pw.this.v)
    keys, cols = pw.debug.table_to_dicts(res)
    vals = sorted(v for v in cols["diff_v"].values() if v is not None)
    assert vals == [3, 7]


def test_stream_generator():
    sg = pw.debug.StreamGenerator()

    class S(pw.Schema):
        v: int

    t = sg.table_from_list_of_batches([[{"v": 1}], [{"v": 2}]], S)
    res = t.groupby().reduce(s=pw.reducers.sum(pw.this.v))
    expected = T(
        """
        s
        3
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_groupby_stream_incremental_updates():
    t = T(
        """
        w | __time__
        a | 0
        a | 2
        """
    )
    res = t.groupby(pw.this.w).reduce(pw.this.w, c=pw.reducers.count())
    cap = res._capture()
    from pathway_amd.engine.runtime import Runtime
    from pathway_amd.internals.rungraph import reset_all

    rt = Runtime([cap])
    reset_all(rt.nodes)
    rt.run()
    stream = sorted((r.time, r.diff, r.values[1]) for r in cap.rows)
    assert stream == [(0, 1, 1), (2, -1, 1), (2, 1, 2)]


def test_window_behavior_cutoff():
    # late row (t=1 arriving after watermark passed its window end + cutoff)
    t = T(
        """
        t  | v | __time__
        1  | 1 | 0
        12 | 5 | 2
        1  | 100 | 4
        """
    )
    res = t.windowby(
        pw.this.t,
        window=pw.temporal.tumbling(duration=5),
        behavior=pw.temporal.common_behavior(cutoff=2, keep_results=True),
    ).reduce(start=pw.this._pw_window_start, s=pw.reducers.sum(pw.this.v))
    # the t=1/v=100 row is late (watermark 12 > window_end 5 + cutoff 2)
    expected = T(
        """
        start | s
        0     | 1
        10    | 5
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_window_behavior_forget():
    t = T(
        """
        t  | v | __time__
        1  | 1 | 0
        12 | 5 | 2
        """
    )
    res = t.windowby(
        pw.this.t,
        window=pw.temporal.tumbling(duration=5),
        behavior=pw.temporal.common_behavior(cutoff=2, keep_results=False),
    ).reduce(start=pw.this._pw_window_start, s=pw.reducers.sum(pw.this.v))
    # first window's results are dropped once the watermark passes end+cutoff
    expected = T(
        """
        start | s
        10    | 5
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_window_behavior_delay_buffer():
    t = T(
        """
        t | v | __time__
        1 | 1 | 0
        2 | 2 | 2
        9 | 3 | 4
        """
    )
    res = t.windowby(
        pw.this.t,
        window=pw.temporal.tumbling(duration=5),
        behavior=pw.temporal.common_behavior(delay=3),
    ).reduce(start=pw.this._pw_window_start, s=pw.reducers.sum(pw.this.v))
    # rows buffered until watermark >= window_start+3; all eventually released
    expected = T(
        """
        start | s
        0     | 3
        5     | 3
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_session_window():
    t = T(
        """
        t  | v
        1  | 1
        2  | 2
        3  | 3
        10 | 4
        11 | 5
        """
    )
    res = t.windowby(
        pw.this.t, window=pw.temporal.session(max_gap=2)
    ).reduce(
        start=pw.this._pw_window_start,
        end=pw.this._pw_window_end,
        s=pw.reducers.sum(pw.this.v),
    )
    expected = T(
        """
        start | end | s
        1     | 3   | 6
        10    | 11  | 9
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_intervals_over():
    data = T(
        """
        t | v
        1 | 1
        3 | 2
        5 | 4
        """
    )
    probes = T(
        """
        pt
        2
        6
        """
    )
    res = data.windowby(
        pw.this.t,
        window=pw.temporal.intervals_over(
            at=probes.pt, lower_bound=-2, upper_bound=1
        ),
    ).reduce(
        loc=pw.this._pw_window_location,
        s=pw.reducers.sum(pw.this.v),
    )
    # at=2: t in [0,3] -> 1+2=3 ; at=6: t in [4,7] -> 4
    expected = T(
        """
        loc | s
        2   | 3
        6   | 4
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_table_sort_prev_next():
    t = T(
        """
        v
        30
        10
        20
        """
    )
    s = t.sort(pw.this.v)
    keys, cols = pw.debug.table_to_dicts(s + t if False else s)
    # row with v=10 has no prev; v=30 has no next
    tkeys, tcols = pw.debug.table_to_dicts(t)
    v_by_key = {k: tcols["v"][k] for k in tkeys}
    for k in keys:
        prev, nxt = cols["prev"][k], cols["next"][k]
        v = v_by_key[k]
        if v == 10:
            assert prev is None and v_by_key[nxt] == 20
        elif v == 20:
            assert v_by_key[prev] == 10 and v_by_key[nxt] == 30
        else:
            assert v_by_key[prev] == 20 and nxt is None


def test_interval_join_left_outer():
    a = T(
        """
        t | x
        1 | a1
        5 | a5
        """
    )
    b = T(
        """
        t | y
        2 | b2
        9 | b9
        """
    )
    res = pw.temporal.interval_join_left(
        a, b, a.t, b.t, pw.temporal.interval(-2, 2)
    ).select(pw.left.x, pw.right.y)
    expected = T(
        """
        x  | y
        a1 | b2
        a5 |
        """
    )
    assert_table_equality_wo_index(res, expected)


def test_to_stream_and_back():
    t = T(
        """
        v | __time__ | __diff__
        1 | 0        | 1
        2 | 2        | 1
        1 | 4        | -1
        """,
        id_from=["v"],
    )
    ev = t.to_stream()
    keys, cols = pw.debug.table_to_dicts(ev)
    flags = sorted(cols["is_upsert"].values())
    assert flags == [False, True, True]
    back = ev.stream_to_table()
    assert_table_equality_wo_index(
        back,
        T(
            """
            v
            2
            """
        ),
    )


def test_asof_now_join_freezes_answers():
    queries = T(
        """
        q | __time__
        1 | 2
        """
    )
    state = T(
        """
        q | v | __time__ | __diff__
        1 | 10 | 0       | 1
        1 | 10 | 4       | -1
        1 | 99 | 4       | 1
        """
    )
    res = pw.temporal.asof_now_join(queries, state, queries.q == state.q).select(
        pw.left.q, pw.right.v
    )
    keys, cols = pw.debug.table_to_dicts(res)
    # the query was answered at t=2 with v=10; the t=4 state change must
    # not retro-update the frozen answer
    assert list(cols["v"].values()) == [10]


def test_exactly_once_behavior_window():
    # windows close exactly once at (end + shift): late rows are dropped,
    # and each window emits a single final value
    t = T(
        """
        t | v | __time__
        1 | 1 |    2
        3 | 1 |    2
        6 | 1 |    8
        2 | 1 |   10
        """
    )
    res = t.windowby(
        t.t,
        window=pw.temporal.tumbling(duration=5),
        behavior=pw.temporal.exactly_once_behavior(),
    ).reduce(start=pw.this._pw_window_start, n=pw.reducers.count())
    keys, cols = pw.debug.table_to_dicts(res)
    got = sorted(zip(cols["start"].values(), cols["n"].values()))
    # window [0,5) closed when the watermark (6) passed its end: the late
    # t=2 arrival at engine-time 10 is ignored; [5,10) emits its count
    assert (0, 2) in got


def test_universe_promises():
    t1 = T(
        """
        a
        1
        2
        """
    )
    t2 = t1.filter(pw.this.a > 0)
    pw.universes.promise_is_subset_of(t2, t1)
    t3 = t2.with_universe_of(t1)
    res = t1.select(pw.this.a, b=t3.a * 10)
    _, cols = pw.debug.table_to_dicts(res)
    assert sorted(cols["b"].values()) == [10, 20]


def test_from_streams_reconstructs_state():
    # updates: id 1 gets two versions; id 2 one; deletions remove id 2
    t1 = T(
        """
        id | pet | age | __time__
         1 | cat |  3  |     2
         2 | dog | 11  |     2
         1 | cat |  4  |     4
        """
    )
    t2 = T(
        """
        id | pet | __time__
         2 | dog |     6
        """
    )
    res = t1.from_streams(t2)
    _, cols = pw.debug.table_to_dicts(res)
    rows = sorted(zip(cols["pet"].values(), cols["age"].values()))
    assert rows == [("cat", 4)]


def test_unpack_snapshots():
    t = T(
        """
        v | __time__ | __diff__
        a |    2     |    1
        b |    2     |    1
        b |    4     |   -1
        c |    4     |    1
        """,
        id_from=["v"],
    )
    snaps = t.unpack_snapshots()
    from pathway_amd.engine.runtime import Runtime
    from pathway_amd.debug import reset_all

    cap = snaps._capture()
    rt = Runtime([cap])
    reset_all(rt.nodes)
    rt.run()
    by_time = {}
    for r in cap.rows:
        by_time.setdefault(r.time, []).append((r.values[0], r.diff))
    assert sorted(v for v, d in by_time[2]) == ["a", "b"]
    assert sorted(v for v, d in by_time[4]) == ["a", "c"]
    assert all(d == 1 for vs in by_time.values() for _, vs_d in [(0, 0)] for v, d in vs)
