"""Checkpoint/recovery tests (reference test_persistence.py +
integration_tests/wordcount SIGKILL-recovery semantics, in-process)."""

import os

import pytest

import pathway_amd as pw
from pathway_amd.debug import table_from_rows
from pathway_amd.engine.runtime import CaptureNode, Runtime
from pathway_amd.internals.config import get_device
from pathway_amd.internals.rungraph import reset_all
from pathway_amd.internals.schema import schema_from_types
from pathway_amd.persistence.engine import PersistenceManager


def _build_pipeline():
    schema = schema_from_types(word=str)
    rows = [
        ("a", 0, 1),
        ("b", 0, 1),
        ("a", 2, 1),
        ("c", 4, 1),
        ("a", 6, 1),
        ("b", 6, -1),
    ]
    t = table_from_rows(schema, rows, is_stream=True)
    res = t.groupby(pw.this.word).reduce(pw.this.word, c=pw.reducers.count())
    cap = CaptureNode(res._node, get_device())
    return t, res, cap


def _final_counts(cap):
    from pathway_amd.internals.api import squash_updates

    state = squash_updates(cap.rows, terminate_on_error=False)
    return sorted(tuple(v) for v in state.values())


def test_crash_recovery(tmp_path):
    pdir = str(tmp_path / "snap")
    cfg = pw.persistence.Config(backend=pw.persistence.Backend.filesystem(pdir))

    # --- run 1: process only the first two timestamps, then "crash" ---
    t, res, cap = _build_pipeline()
    pm = PersistenceManager(cfg)
    rt = Runtime([cap], persistence=pm)
    reset_all(rt.nodes)
    rt.run(max_steps=2)  # times 0 and 2 processed, then crash
    pm.close()
    rows_before = list(cap.rows)
    assert rows_before  # partial results delivered before the crash

    # --- run 2: fresh process (fresh graph), recover + continue ---
    t2, res2, cap2 = _build_pipeline()
    pm2 = PersistenceManager(cfg)
    assert pm2.threshold_time == 2
    rt2 = Runtime([cap2], persistence=pm2)
    reset_all(rt2.nodes)
    rt2.run()
    pm2.close()
    rows_after = list(cap2.rows)
    # replayed outputs are suppressed — only post-crash deltas re-emitted
    assert all(r.time > 2 for r in rows_after)

    # --- combined delivered stream == uninterrupted run ---
    t3, res3, cap3 = _build_pipeline()
    rt3 = Runtime([cap3])
    reset_all(rt3.nodes)
    rt3.run()
    from pathway_amd.internals.api import squash_updates

    expected = squash_updates(cap3.rows)
    combined = squash_updates(rows_before + rows_after)
    exp = sorted(tuple(v) for v in expected.values())
    got = sorted(tuple(v) for v in combined.values())
    assert exp == got == [("a", 3), ("c", 1)]


def test_recovery_snapshot_format_truncation(tmp_path):
    """A truncated trailing block (crash mid-write) is ignored."""
    pdir = str(tmp_path / "snap2")
    cfg = pw.persistence.Config(backend=pw.persistence.Backend.filesystem(pdir))
    t, res, cap = _build_pipeline()
    pm = PersistenceManager(cfg)
    rt = Runtime([cap], persistence=pm)
    reset_all(rt.nodes)
    rt.run(max_steps=2)
    pm.close()
    # corrupt: append garbage partial block to every chunk
    for root, _, files in os.walk(pdir):
        for f in files:
            if f.isdigit():
                with open(os.path.join(root, f), "ab") as fh:
                    fh.write(b"\x40\x00\x00\x00\x00\x00\x00\x00PARTIAL")
    t2, res2, cap2 = _build_pipeline()
    pm2 = PersistenceManager(cfg)
    rt2 = Runtime([cap2], persistence=pm2)
    reset_all(rt2.nodes)
    rt2.run()
    pm2.close()
    from pathway_amd.internals.api import squash_updates

    combined = squash_updates(list(cap.rows) + list(cap2.rows))
    got = sorted(tuple(v) for v in combined.values())
    assert got == [("a", 3), ("c", 1)]


def test_pw_run_with_persistence(tmp_path):
    pdir = str(tmp_path / "snap3")
    cfg = pw.persistence.Config(backend=pw.persistence.Backend.filesystem(pdir))
    schema = schema_from_types(v=int)
    outfile = str(tmp_path / "out.csv")
    t = table_from_rows(schema, [(1, 0, 1), (2, 2, 1)], is_stream=True)
    res = t.reduce(s=pw.reducers.sum(pw.this.v))
    pw.io.csv.write(res, outfile)
    pw.run(persistence_config=cfg)
    # round-2 layout: metadata under <version>-<worker>-<rotation> keys
    mdir = os.path.join(pdir, "metadata")
    assert os.path.isdir(mdir)
    names = os.listdir(mdir)
    assert any(n.split("-")[1] == "0" for n in names), names
    with open(outfile) as f:
        assert len(f.read().strip().splitlines()) >= 2


def test_operator_persisting_recovery(tmp_path):
    pdir = str(tmp_path / "opsnap")
    cfg = pw.persistence.Config(
        backend=pw.persistence.Backend.filesystem(pdir),
        persistence_mode=pw.PersistenceMode.OPERATOR_PERSISTING,
    )
    t, res, cap = _build_pipeline()
    pm = PersistenceManager(cfg)
    assert pm.operator_persisting
    rt = Runtime([cap], persistence=pm)
    reset_all(rt.nodes)
    rt.run(max_steps=2)
    pm.close()
    rows_before = list(cap.rows)

    t2, res2, cap2 = _build_pipeline()
    pm2 = PersistenceManager(cfg)
    rt2 = Runtime([cap2], persistence=pm2)
    reset_all(rt2.nodes)
    rt2.run()
    pm2.close()
    from pathway_amd.internals.api import squash_updates

    combined = squash_updates(rows_before + list(cap2.rows))
    got = sorted(tuple(v) for v in combined.values())
    assert got == [("a", 3), ("c", 1)]


def test_operator_snapshot_asof_state(tmp_path):
    """Asof-join arrangements survive an operator-snapshot save/load."""
    import torch

    from pathway_amd.engine.nodes_asof import AsofJoinNode, _AsofSide
    from pathway_amd.persistence.operator_snapshot import (
        node_state_load,
        node_state_save,
    )
    from pathway_amd.engine.column import TensorColumn, PointerColumn
    from pathway_amd.internals import dtype as dt

    node = AsofJoinNode.__new__(AsofJoinNode)
    node.L = _AsofSide("cpu")
    node.R = _AsofSide("cpu")
    words = [torch.tensor([1, 2]), torch.tensor([0, 0]),
             torch.tensor([10, 20]), torch.tensor([5, 6]), torch.tensor([7, 8])]
    cols = {
        "v": TensorColumn(torch.tensor([100, 200]), dt.INT),
        "__rowkey__": PointerColumn(torch.tensor([[1, 2], [3, 4]])),
    }
    node.R.merge(words, torch.tensor([1, 1]), cols)
    st = node_state_save(node)
    assert st["kind"] == "asof"

    node2 = AsofJoinNode.__new__(AsofJoinNode)
    node2.L = _AsofSide("cpu")
    node2.R = _AsofSide("cpu")
    node_state_load(node2, st, torch.device("cpu"))
    assert len(node2.R) == 2
    assert node2.R.cols["v"].tensor.tolist() == [100, 200]
    assert node2.R.words[2].tolist() == [10, 20]
