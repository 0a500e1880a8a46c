"""Embedder A/B: eager vs hipGraph capture, hipBLASLt vs pw MFMA GEMM.

Run on MI355X: python scripts/bench_encoder.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench(label, fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{label:42s} {dt*1000:8.3f} ms/iter")
    return dt


def main():
    assert torch.cuda.is_available()
    from pathway_amd.xpacks.llm._encoder import EncoderConfig, NativeEncoder

    texts = [f"document number {i} about topic {i % 17} with some words" * 3
             for i in range(1024)]

    # GEMM microbench: encoder shapes
    from pathway_amd import ops

    M = 1024 * 32
    for N, K in [(1152, 384), (384, 384), (1536, 384), (384, 1536)]:
        a = (torch.randn(M, K, device="cuda") * 0.1).to(torch.bfloat16)
        b = (torch.randn(K, N, device="cuda") * 0.1).to(torch.bfloat16)
        bt = b.T.contiguous()
        bias = torch.randn(N, device="cuda")
        t_pw = bench(f"pw_gemm_bf16 {M}x{N}x{K}",
                     lambda: ops.gemm_bias_act_gpu(a, bt, bias))
        bb = bias.to(torch.bfloat16)
        t_blas = bench(f"hipBLASLt   {M}x{N}x{K}",
                       lambda: a @ b + bb)
        flops = 2 * M * N * K
        print(f"  pw: {flops/t_pw/1e12:7.1f} TF   blas: {flops/t_blas/1e12:7.1f} TF")

    enc = NativeEncoder(EncoderConfig(), device="cuda")
    ids, mask = enc.tokenize(texts)
    print(f"token shape: {tuple(ids.shape)}")

    bench("tokenize (device)", lambda: enc.tokenize(texts))

    os.environ["PW_NO_PW_GEMM"] = "1"
    enc._f32_bias = {}
    bench("eager forward (hipBLASLt)", lambda: enc._forward_impl(ids, mask))
    del os.environ["PW_NO_PW_GEMM"]
    bench("eager forward (pw MFMA)", lambda: enc._forward_impl(ids, mask))

    os.environ["PW_NO_PW_GEMM"] = "1"
    enc._graphs = {}
    bench("hipGraph forward (hipBLASLt)", lambda: enc._forward_graphed(ids, mask))
    del os.environ["PW_NO_PW_GEMM"]
    enc._graphs = {}
    bench("hipGraph forward (pw MFMA)", lambda: enc._forward_graphed(ids, mask))

    bench("end-to-end encode 1024 texts",
          lambda: enc.encode(texts, batch_size=1024))


if __name__ == "__main__":
    main()
