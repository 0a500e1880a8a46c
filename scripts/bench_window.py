"""Sliding-window aggregation benchmark (temporal operator path).

A stream of timestamped events is windowed with a sliding window
(hop h, duration d => d/h windows per event) and summed per window —
the windowby → groupby-reduce hot path with window expansion on device.
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
    p.add_argument("--batch", type=int, default=2_000_000, help="events per step")
    p.add_argument("--hop", type=int, default=10)
    p.add_argument("--duration", type=int, default=40)
    p.add_argument("--window", choices=["sliding", "session"], default="sliding")
    p.add_argument("--gap", type=int, default=3, help="session max_gap")
    p.add_argument("--instances", type=int, default=0,
                   help="session instances (0 = one global instance)")
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

    class Events:
        """Timestamps advance ~1 unit per step: old windows go quiet and
        the per-step changed-window count stays bounded."""

        def __init__(self):
            self.counter = 0

        def next_time(self):
            return None

        def pull(self, t, dev):
            base = self.counter
            self.counter += 1
            g = torch.Generator(device=dev).manual_seed(base)
            ts = (
                torch.rand(B, generator=g, device=dev) * args.duration + base
            ).to(torch.int64)
            nmod = args.instances if args.instances else 1000
            vals = torch.arange(B, dtype=torch.int64, device=dev) % nmod
            seq = torch.arange(base * B, base * B + B, dtype=torch.int64, device=dev)
            klo, khi = hashing.value_hash_words(seq, TAG_INT)
            keys = torch.stack([klo, khi], dim=1)
            cols = {
                "t": TensorColumn(ts, dt.INT),
                "v": TensorColumn(vals, dt.INT),
            }
            diffs = torch.ones(B, dtype=torch.int64, device=dev)
            return DeltaBatch(keys, cols, diffs, t)

        def reset(self):
            self.counter = 0

    src = Events()
    t_in = Table(InputNode(src, device), {"t": dt.INT, "v": dt.INT}, Universe())
    if args.window == "session":
        wspec = pw.temporal.session(max_gap=args.gap)
    else:
        wspec = pw.temporal.sliding(hop=args.hop, duration=args.duration)
    wkw = {}
    if args.instances:
        wkw["instance"] = t_in.v  # v is already `arange % 1000`-style
    win = t_in.windowby(t_in.t, window=wspec, **wkw).reduce(
        start=pw.this._pw_window_start,
        s=pw.reducers.sum(pw.this.v),
        n=pw.reducers.count(),
    )
    emitted = [0]
    sink = OutputNode(
        win._node, lambda b: emitted.__setitem__(0, emitted[0] + len(b)), device
    )
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
    ev = B * args.steps
    print(
        json.dumps(
            {
                "bench": f"{args.window}_window_sum",
                "events_per_s": ev / el,
                "ms_per_step": el / args.steps * 1000,
                "events_per_step": B,
                "windows_per_event": (args.duration // args.hop)
                if args.window == "sliding" else 1,
                "window_updates_emitted": emitted[0],
                "device": str(device),
            }
        )
    )


if __name__ == "__main__":
    main()
