"""Single-shape pw_gemm_bf16 loop for PMC counter collection.

Usage: rocprofv3 --pmc <counters> -- python scripts/prof_gemm.py [N K]
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    from pathway_amd import ops

    N = int(sys.argv[1]) if len(sys.argv) > 1 else 1536
    K = int(sys.argv[2]) if len(sys.argv) > 2 else 384
    M = 32768
    a = (torch.randn(M, K, device="cuda") * 0.1).to(torch.bfloat16)
    bt = (torch.randn(N, K, device="cuda") * 0.1).to(torch.bfloat16)
    bias = torch.randn(N, device="cuda")
    for _ in range(20):
        ops.gemm_bias_act_gpu(a, bt, bias, "gelu")
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
