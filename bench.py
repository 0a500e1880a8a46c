"""Flagship benchmark: streaming WordCount throughput (events/sec, whole node).

BASELINE.md headline: pathway's published WordCount figure is 2,000,000
messages/s with p95 end-to-end latency < 50 ms (4 CPU cores/system,
docs/.../50.how-live-data-framework-connectors-work.md:276).  This bench
runs the same shape — a stream of words → incremental groupby(word).count
→ consolidated output deltas — through the pathway_amd GPU engine:
per step one micro-batch of synthetic words is keyed (HIP xxh64-128
kernel), shuffled by key shard (RCCL all-to-all over xGMI for N>1),
segment-reduced and merged into GPU-resident state, and the (-old,+new)
count deltas are emitted.

Driver contract: --gpus N --steps K --warmup W; launched via torchrun for
N>1 (one rank per GPU, RCCL); rank 0 prints ONE json line.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch", type=int, default=4_000_000, help="events per step per GPU")
    p.add_argument("--vocab", type=int, default=50_000)
    p.add_argument("--device", type=str, default=None)
    p.add_argument(
        "--ingest",
        choices=["codes", "bytes"],
        default="bytes",
        help="codes: pre-parsed dictionary codes; bytes: raw newline-separated "
        "byte stream tokenized+hashed on-device (HIP varlen kernel)",
    )
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    use_cuda = torch.cuda.is_available()
    if args.device:
        device = torch.device(args.device)
    elif use_cuda:
        device = torch.device(f"cuda:{local_rank % torch.cuda.device_count()}")
    else:
        device = torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    os.environ["PW_DEVICE"] = str(device)

    import numpy as np

    import pathway_amd as pw
    import pathway_amd.parallel as par
    from pathway_amd.engine import hashing
    from pathway_amd.engine.batch import DeltaBatch
    from pathway_amd.engine.column import GLOBAL_STRING_POOL, StringColumn
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.runtime import OutputNode, Runtime
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.api import TAG_INT
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    comm = None
    if world > 1:
        backend = os.environ.get("PW_BACKEND") or ("nccl" if use_cuda else "gloo")
        comm = par.init(backend=backend, device=device)

    # ---- synthetic parsed word stream (identical vocab on all ranks) ----
    # fixed word width: zero-pad to however many digits the vocab needs
    DIG = max(6, len(str(args.vocab - 1)))
    WW = 4 + DIG  # b"word" + digits
    vocab = [f"word{i:0{DIG}d}" for i in range(args.vocab)]
    GLOBAL_STRING_POOL.codes(vocab)
    GLOBAL_STRING_POOL.synchronized = True
    GLOBAL_STRING_POOL.hash_tensors(device)  # pre-stage pool hashes in HBM

    # synthetic stream RNG lives ON DEVICE: a real connector DMAs messages
    # into HBM; generating on host and copying would bill a host artifact
    # to the engine. Per-rank independent streams via distinct seeds.
    gen = torch.Generator(device=device if use_cuda else "cpu")
    gen.manual_seed(1234 + rank)

    class WordStream:
        """Source: per pull, one micro-batch of `batch` word events."""

        def __init__(self, batch: int, vocab_n: int):
            self.batch = batch
            self.vocab_n = vocab_n
            self.counter = 0

        def next_time(self):
            return None  # driven manually by the bench loop

        def pull(self, t, dev):
            n = self.batch
            codes = torch.randint(
                0, self.vocab_n, (n,), dtype=torch.int64, generator=gen,
                device=gen.device,
            ).to(dev, non_blocking=True)
            seq = torch.arange(
                self.counter, self.counter + n, dtype=torch.int64, device=dev
            )
            self.counter += n
            klo, khi = hashing.value_hash_words(seq, TAG_INT)
            keys = torch.stack([klo, khi], dim=1)
            cols = {"word": StringColumn(codes, GLOBAL_STRING_POOL, dt.STR)}
            diffs = torch.ones(n, dtype=torch.int64, device=dev)
            return DeltaBatch(keys, cols, diffs, t)

        def reset(self):
            self.counter = 0

    class ByteStream:
        """Raw ingest: per pull, a newline-separated byte buffer is
        tokenized and xxh-128-keyed ON DEVICE (pw_varlen_hash) — the parse
        path a GPU-native text connector runs (message parsing included in
        the timed region)."""

        def __init__(self, batch: int, vocab_n: int):
            self.batch = batch
            self.vocab_n = vocab_n
            self.counter = 0
            # device matrix of the vocabulary's word bytes (fixed width WW)
            wb = np.frombuffer(
                "".join(f"word{i:0{DIG}d}" for i in range(vocab_n)).encode(),
                dtype=np.uint8,
            ).reshape(vocab_n, WW)
            self.word_bytes = torch.from_numpy(wb.copy()).to(device)

        def next_time(self):
            return None

        def pull(self, t, dev):
            from pathway_amd import ops
            from pathway_amd.engine.column import PointerColumn
            from pathway_amd.internals.api import TAG_STR

            n = self.batch
            codes = torch.randint(
                0, self.vocab_n, (n,), dtype=torch.int64, generator=gen,
                device=gen.device,
            ).to(dev, non_blocking=True)
            # build the wire buffer: "<word>\n" per message (device gather)
            msg = torch.empty((n, WW + 1), dtype=torch.uint8, device=dev)
            msg[:, :WW] = self.word_bytes.index_select(0, codes)
            msg[:, WW] = 10  # newline
            buf = msg.reshape(-1).contiguous()
            # PARSE on device: newline scan -> token [start, end) spans
            nl = (buf == 10).nonzero(as_tuple=True)[0]
            starts = torch.cat(
                [torch.zeros(1, dtype=torch.int64, device=dev), nl[:-1] + 1]
            )
            ends = nl
            # per-token canonical string hash == the word's group key
            if dev.type == "cuda":
                glo, ghi = ops.varlen_hash_se_gpu(buf, starts, ends, TAG_STR)
            else:
                from pathway_amd.internals.api import hash128, serialize_value

                bb = buf.cpu().numpy().tobytes()
                st = starts.cpu().tolist()
                en = ends.cpu().tolist()
                glo_l, ghi_l = [], []
                for a, b in zip(st, en):
                    l_, h_ = hash128(serialize_value(bb[a:b].decode()))
                    glo_l.append(l_ - (1 << 64) if l_ >= 1 << 63 else l_)
                    ghi_l.append(h_ - (1 << 64) if h_ >= 1 << 63 else h_)
                glo = torch.tensor(glo_l, dtype=torch.int64)
                ghi = torch.tensor(ghi_l, dtype=torch.int64)
            seq = torch.arange(
                self.counter, self.counter + n, dtype=torch.int64, device=dev
            )
            self.counter += n
            klo, khi = hashing.value_hash_words(seq, TAG_INT)
            keys = torch.stack([klo, khi], dim=1)
            cols = {
                "word": StringColumn(codes, GLOBAL_STRING_POOL, dt.STR),
                "wkey": PointerColumn(torch.stack([glo, ghi], dim=1)),
            }
            diffs = torch.ones(n, dtype=torch.int64, device=dev)
            return DeltaBatch(keys, cols, diffs, t)

        def reset(self):
            self.counter = 0

    if args.ingest == "bytes":
        source = ByteStream(args.batch, args.vocab)
        in_node = InputNode(source, device)
        words = Table(
            in_node, {"word": dt.STR, "wkey": dt.POINTER}, Universe()
        )
        counts = words.groupby(pw.this.word, id=pw.this.wkey).reduce(
            pw.this.word, count=pw.reducers.count()
        )
    else:
        source = WordStream(args.batch, args.vocab)
        in_node = InputNode(source, device)
        words = Table(in_node, {"word": dt.STR}, Universe())
        counts = words.groupby(pw.this.word).reduce(
            pw.this.word, count=pw.reducers.count()
        )

    emitted = [0]

    def counting_writer(batch):
        emitted[0] += len(batch)

    sink = OutputNode(counts._node, counting_writer, device)
    rt = Runtime([sink], device=device, comm=comm)

    def sync():
        if use_cuda:
            torch.cuda.synchronize(device)

    def barrier():
        if comm is not None:
            comm.barrier()

    # ---- warmup ----
    t = 0
    for _ in range(args.warmup):
        rt.step_once(t)
        t += 2  # even timestamps (reference alt-neu convention)
    sync()
    barrier()
    sync()

    if os.environ.get("PW_TORCH_PROF"):
        # host/device breakdown of the steady-state step (diagnostics only)
        from torch.profiler import ProfilerActivity, profile

        acts = [ProfilerActivity.CPU]
        if use_cuda:
            acts.append(ProfilerActivity.CUDA)
        with profile(activities=acts) as prof:
            for _ in range(3):
                rt.step_once(t)
                t += 2
            sync()
        ka = prof.key_averages()
        print(ka.table(sort_by="self_cpu_time_total", row_limit=30))
        if use_cuda:
            print(ka.table(sort_by="self_cuda_time_total", row_limit=20))

    # ---- timed region: exactly K steps ----
    step_times = []
    t0 = time.perf_counter()
    for _ in range(args.steps):
        s0 = time.perf_counter()
        rt.step_once(t)
        t += 2
        sync()
        step_times.append(time.perf_counter() - s0)
    sync()
    barrier()
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if comm is not None:
        import torch.distributed as dist

        et = torch.tensor([elapsed], dtype=torch.float64, device=comm._comm_device)
        dist.all_reduce(et, op=dist.ReduceOp.MAX)
        elapsed = float(et.item())

    n_gpus = world if world > 1 else 1
    total_events = args.batch * args.steps * n_gpus
    value = total_events / elapsed
    ms_per_step = elapsed / args.steps * 1000.0
    p95 = sorted(step_times)[max(0, int(len(step_times) * 0.95) - 1)] * 1000.0

    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "wordcount streaming throughput (events/sec, whole node)",
                    "value": value,
                    "unit": "events/s",
                    "n_gpus": n_gpus,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": ms_per_step,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": value / 2_000_000.0,
                    "dtype": "int64/string-dict",
                    "data": "synthetic",
                    "config": {
                        "model": "wordcount: stream -> groupby(word).count -> output deltas",
                        "global_batch": args.batch * n_gpus,
                        "seq_len": 1,
                        "parallelism": f"dp{n_gpus} (shard exchange: RCCL all-to-all)",
                        "vocab": args.vocab,
                        "ingest": args.ingest,
                        "p95_step_latency_ms": p95,
                        "emitted_delta_rows": emitted[0],
                    },
                }
            )
        )

    if comm is not None:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
