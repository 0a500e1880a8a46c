"""Streaming hash-join benchmark (BASELINE config 4 shape).

Two keyed streams joined incrementally; state grows every step (the
100M×100M regime is reached by running enough steps).  Single-GPU here;
the same program runs under torchrun for the 8-GPU RCCL-shuffle variant.
"""
from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    import argparse

    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=25)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch", type=int, default=2_000_000, help="rows per side per step")
    args = p.parse_args()

    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda:0" if use_cuda else "cpu")
    os.environ["PW_DEVICE"] = str(device)

    import pathway_amd as pw
    from pathway_amd.engine import hashing
    from pathway_amd.engine.batch import DeltaBatch
    from pathway_amd.engine.column import TensorColumn
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.runtime import OutputNode, Runtime
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.api import TAG_INT
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    B = args.batch

    class Side:
        def __init__(self, lag: int, salt: int):
            self.counter = 0
            self.lag = lag
            self.salt = salt

        def next_time(self):
            return None

        def pull(self, t, dev):
            ids = torch.arange(
                self.counter - self.lag,
                self.counter - self.lag + B,
                dtype=torch.int64,
                device=dev,
            ).clamp_min(0)
            self.counter += B
            seq = ids + self.salt * (2**40)
            klo, khi = hashing.value_hash_words(seq, TAG_INT)
            keys = torch.stack([klo, khi], dim=1)
            cols = {
                "k": TensorColumn(ids, dt.INT),
                "v": TensorColumn(ids * 3 + self.salt, dt.INT),
            }
            diffs = torch.ones(B, dtype=torch.int64, device=dev)
            return DeltaBatch(keys, cols, diffs, t)

        def reset(self):
            self.counter = 0

    lsrc, rsrc = Side(0, 1), Side(B, 2)
    lt = Table(InputNode(lsrc, device), {"k": dt.INT, "v": dt.INT}, Universe())
    rt_t = Table(InputNode(rsrc, device), {"k": dt.INT, "v": dt.INT}, Universe())
    joined = lt.join(rt_t, lt.k == rt_t.k).select(
        pw.this.k, s=pw.left.v + pw.right.v
    )
    matched = [0]
    sink = OutputNode(joined._node, lambda b: matched.__setitem__(0, matched[0] + len(b)), device)
    rt = Runtime([sink], device=device)

    def sync():
        if use_cuda:
            torch.cuda.synchronize(device)

    t = 0
    for _ in range(args.warmup):
        rt.step_once(t)
        t += 2
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        rt.step_once(t)
        t += 2
    sync()
    el = time.perf_counter() - t0
    rows = 2 * B * args.steps
    state_rows = 2 * B * (args.steps + args.warmup)
    print(
        json.dumps(
            {
                "bench": "streaming_hash_join",
                "rows_per_s": rows / el,
                "ms_per_step": el / args.steps * 1000,
                "rows_per_side_per_step": B,
                "final_state_rows_total": state_rows,
                "matched_pairs_emitted": matched[0],
                "device": str(device),
            }
        )
    )


if __name__ == "__main__":
    main()
