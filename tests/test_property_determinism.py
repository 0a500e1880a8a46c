"""Property-based correctness + determinism (hypothesis; the analog of the
reference's randomized operator tests and its deterministic-replay
guarantees)."""

import pandas as pd
import pytest
from hypothesis import given, settings, strategies as st

import pathway_amd as pw
from pathway_amd.debug import table_from_rows, table_to_dicts
from pathway_amd.internals.schema import schema_from_types


rows_strategy = st.lists(
    st.tuples(
        st.sampled_from(["a", "b", "c", "d"]),
        st.integers(min_value=-1000, max_value=1000),
    ),
    min_size=1,
    max_size=60,
)


@settings(max_examples=25, deadline=None)
@given(rows=rows_strategy)
def test_groupby_sum_matches_pandas(rows):
    pw.internals.rungraph.G.clear()
    t = table_from_rows(schema_from_types(g=str, v=int), rows)
    r = t.groupby(pw.this.g).reduce(
        pw.this.g,
        s=pw.reducers.sum(pw.this.v),
        n=pw.reducers.count(),
        mn=pw.reducers.min(pw.this.v),
        mx=pw.reducers.max(pw.this.v),
    )
    _, cols = table_to_dicts(r)
    got = sorted(
        zip(
            cols["g"].values(),
            cols["s"].values(),
            cols["n"].values(),
            cols["mn"].values(),
            cols["mx"].values(),
        )
    )
    df = pd.DataFrame(rows, columns=["g", "v"])
    ref = (
        df.groupby("g")["v"]
        .agg(["sum", "count", "min", "max"])
        .reset_index()
        .sort_values("g")
    )
    expected = [
        (r_.g, int(r_.sum), int(r_.count), int(r_.min), int(r_.max))
        for r_ in ref.itertuples()
    ]
    assert got == expected


@settings(max_examples=15, deadline=None)
@given(
    rows=st.lists(
        st.tuples(
            st.sampled_from(["x", "y", "z"]),
            st.integers(min_value=0, max_value=50),
        ),
        min_size=1,
        max_size=40,
    )
)
def test_join_matches_pandas(rows):
    pw.internals.rungraph.G.clear()
    left_rows = rows
    right_rows = [(g, v * 2) for g, v in rows[::2]]
    lt = table_from_rows(schema_from_types(g=str, v=int), left_rows)
    rt = table_from_rows(schema_from_types(g=str, w=int), right_rows)
    j = lt.join(rt, lt.g == rt.g).select(pw.this.g, pw.left.v, pw.right.w)
    _, cols = table_to_dicts(j)
    got = sorted(zip(cols["g"].values(), cols["v"].values(), cols["w"].values()))
    dl = pd.DataFrame(left_rows, columns=["g", "v"])
    dr = pd.DataFrame(right_rows, columns=["g", "w"])
    ref = dl.merge(dr, on="g")
    expected = sorted(ref.itertuples(index=False, name=None))
    assert got == expected


@settings(max_examples=10, deadline=None)
@given(rows=rows_strategy)
def test_update_stream_retractions_consolidate(rows):
    """Insert everything at t=2, retract the first half at t=4: the final
    groupby state nets out to the surviving half."""
    pw.internals.rungraph.G.clear()
    from pathway_amd.debug import table_from_markdown as T

    half = len(rows) // 2
    lines = ["g | v | __time__ | __diff__"]
    for i, (g, v) in enumerate(rows):
        lines.append(f"{g}{i} | {v} | 2 | 1")
    for i, (g, v) in enumerate(rows[:half]):
        lines.append(f"{g}{i} | {v} | 4 | -1")
    tbl = T("\n".join(lines), id_from=["g"]).select(
        g=pw.this.g.str.slice(0, 1), v=pw.this.v
    )
    r = tbl.groupby(pw.this.g).reduce(pw.this.g, s=pw.reducers.sum(pw.this.v))
    _, cols = table_to_dicts(r)
    kept = rows[half:]
    df = pd.DataFrame(kept, columns=["g", "v"]).groupby("g")["v"].sum()
    got = {g: s for g, s in zip(cols["g"].values(), cols["s"].values())}
    assert got == {g: int(s) for g, s in df.items()}


def test_two_runs_identical_outputs():
    """Determinism: rebuilding and re-running the same pipeline yields
    byte-identical keys and values (the replay guarantee)."""

    def run_once():
        pw.internals.rungraph.G.clear()
        t = table_from_rows(
            schema_from_types(g=str, v=int),
            [("a", 1), ("b", 2), ("a", 3), ("c", 4), ("b", 5)],
        )
        r = t.groupby(pw.this.g).reduce(
            pw.this.g, s=pw.reducers.sum(pw.this.v)
        )
        keys, cols = table_to_dicts(r)
        return sorted((repr(k), cols["s"][k]) for k in keys)

    assert run_once() == run_once()
