"""HIP kernel microbenchmarks: throughput vs the HBM3E roofline."""
from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench(fn, n_bytes, iters=50, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return dt, n_bytes / dt / 1e9


def main():
    from pathway_amd import ops
    from pathway_amd.internals.api import TAG_STR

    assert torch.cuda.is_available()
    dev = "cuda:0"
    n = 16_000_000

    # 1. fused 128-bit row hash (2 words in, 2 words out)
    w = [torch.randint(-2**62, 2**62, (n,), dtype=torch.int64, device=dev) for _ in range(2)]
    dt_s, gbs = bench(lambda: ops.hash128_words_gpu(w), n * 8 * 4)
    print(json.dumps({"kernel": "pw_hash128_words(w=2)", "rows": n,
                      "ms": dt_s * 1e3, "GB_s": gbs, "Mrows_s": n / dt_s / 1e6}))

    # 2. value hash (tagged single word)
    v = torch.randint(-2**62, 2**62, (n,), dtype=torch.int64, device=dev)
    dt_s, gbs = bench(lambda: ops.value_hash_gpu(v, 2), n * 8 * 3)
    print(json.dumps({"kernel": "pw_value_hash", "rows": n, "ms": dt_s * 1e3,
                      "GB_s": gbs, "Mrows_s": n / dt_s / 1e6}))

    # 3. varlen hash over 10-byte tokens
    buf = torch.randint(97, 122, (n * 10,), dtype=torch.uint8, device=dev)
    starts = torch.arange(0, n * 10, 10, dtype=torch.int64, device=dev)
    ends = starts + 10
    dt_s, gbs = bench(lambda: ops.varlen_hash_se_gpu(buf, starts, ends, TAG_STR),
                      n * (10 + 16 + 16))
    print(json.dumps({"kernel": "pw_varlen_hash_se(10B tokens)", "rows": n,
                      "ms": dt_s * 1e3, "GB_s": gbs, "Mrows_s": n / dt_s / 1e6}))

    # 4. 128-bit sorted search: 16M queries over a 64M-row sorted state
    m = 64_000_000
    sk0, _ = torch.sort(torch.randint(-2**62, 2**62, (m,), dtype=torch.int64, device=dev))
    sk1 = torch.randint(-2**62, 2**62, (m,), dtype=torch.int64, device=dev)
    q = [torch.randint(-2**62, 2**62, (n,), dtype=torch.int64, device=dev) for _ in range(2)]
    dt_s, _ = bench(lambda: ops.lookup_gpu([sk0, sk1], q), n * 16)
    print(json.dumps({"kernel": "pw_lookup (bin-search 64M state)", "queries": n,
                      "ms": dt_s * 1e3, "Mqueries_s": n / dt_s / 1e6}))

    # 5. pool hash gather
    pool = torch.randint(-2**62, 2**62, (1_000_000,), dtype=torch.int64, device=dev)
    codes = torch.randint(0, 1_000_000, (n,), dtype=torch.int64, device=dev)
    dt_s, gbs = bench(lambda: ops.pool_hash_gpu(codes, pool, pool, 1, 2), n * 8 * 3)
    print(json.dumps({"kernel": "pw_pool_hash", "rows": n, "ms": dt_s * 1e3,
                      "GB_s": gbs, "Mrows_s": n / dt_s / 1e6}))


if __name__ == "__main__":
    main()


def bench_radix_sort():
    """pw radix sort vs torch.sort (rocPRIM onesweep) on int64 keys."""
    import time

    import torch

    from pathway_amd import ops

    for n in (1 << 20, 1 << 22, 1 << 24):
        keys = torch.randint(-(2**62), 2**62, (n,), dtype=torch.int64,
                             device="cuda")
        for label, fn in [
            ("pw_radix_sort64", lambda: ops.radix_sort64_gpu(keys)),
            ("torch.sort", lambda: torch.sort(keys)),
        ]:
            for _ in range(3):
                fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(10):
                fn()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 10
            print(f"  {label:18s} n={n:>9}: {dt*1000:7.3f} ms "
                  f"({n/dt/1e9:.2f} Gkeys/s)")


if __name__ == "__main__" and __import__("os").environ.get("PW_BENCH_SORT"):
    bench_radix_sort()
