"""Incremental asof-join throughput (tensor path, engine/nodes_asof.py).

Left events query the latest right-side quote per key as-of their time;
both sides stream and the state grows each step.
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
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch", type=int, default=1_000_000, help="rows per side per step")
    p.add_argument("--keys", type=int, default=10_000)
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
    from pathway_amd.stdlib.temporal import Direction

    B = args.batch
    gen = torch.Generator(device=device if use_cuda else "cpu").manual_seed(7)

    class Side:
        def __init__(self, salt: int):
            self.counter = 0
            self.salt = salt

        def next_time(self):
            return None

        def pull(self, t, dev):
            n = B
            base = self.counter
            self.counter += n
            k = torch.randint(0, args.keys, (n,), dtype=torch.int64,
                              generator=gen, device=gen.device).to(dev)
            ts = torch.randint(0, 1 << 30, (n,), dtype=torch.int64,
                               generator=gen, device=gen.device).to(dev)
            seq = torch.arange(base, base + n, dtype=torch.int64, device=dev)
            klo, khi = hashing.value_hash_words(seq + self.salt * (2 ** 40), TAG_INT)
            keys = torch.stack([klo, khi], dim=1)
            cols = {
                "k": TensorColumn(k, dt.INT),
                "t": TensorColumn(ts, dt.INT),
                "v": TensorColumn(seq, dt.INT),
            }
            return DeltaBatch(keys, cols, torch.ones(n, dtype=torch.int64, device=dev), t)

        def reset(self):
            self.counter = 0

    ldev, rdev = Side(1), Side(2)
    lt = Table(InputNode(ldev, device), {"k": dt.INT, "t": dt.INT, "v": dt.INT}, Universe())
    rt_t = Table(InputNode(rdev, device), {"k": dt.INT, "t": dt.INT, "v": dt.INT}, Universe())
    res = lt.asof_join(
        rt_t, lt.t, rt_t.t, lt.k == rt_t.k, how="inner",
        direction=Direction.BACKWARD,
    ).select(pw.left.v, rv=pw.right.v)
    emitted = [0]
    sink = OutputNode(res._node, lambda b: emitted.__setitem__(0, emitted[0] + len(b)), device)
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
    print(json.dumps({
        "bench": "asof_join_incremental",
        "rows_per_s": rows / el,
        "ms_per_step": el / args.steps * 1000,
        "rows_per_side_per_step": B,
        "distinct_keys": args.keys,
        "final_state_rows_per_side": B * (args.steps + args.warmup),
        "output_rows_emitted": emitted[0],
        "device": str(device),
    }))


if __name__ == "__main__":
    main()
