#!/usr/bin/env python3
"""Small kernel-replay driver for PMC capture (run under rocprofv3 --pmc).

Runs a fixed number of dispatches of the hot kernels at the serving shapes
so per-kernel counter averages are stable: the MFMA GEMM (8-phase), torch's
GEMM on the same shape, and the three attention families.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from vilbert_multi_task_amd.ops import functional as F_ops

F_ops._load_extension()


def main():
    torch.manual_seed(0)
    reps = int(os.environ.get("PMC_REPS", "6"))

    # GEMM: the vision QKV shape
    M, N, K = 103424, 3072, 1024
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.05
    b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    for _ in range(reps):
        torch.ops.vilbert_amd.mfma_linear(x, w, b, None, False)
        torch.nn.functional.linear(x, w, b)
    torch.cuda.synchronize()

    # attention: text self / vision self / co t->v at B=1024
    B = 1024
    for H, Lq, Lk, D in ((12, 38, 38, 64), (8, 101, 101, 128), (8, 38, 101, 128)):
        q = torch.randn(B, Lq, H * D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
        keep = torch.ones(B, Lk)
        keep[:, Lk - 3:] = 0
        mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
        for _ in range(reps):
            torch.ops.vilbert_amd.attention(q, k, v, H, mask)
        torch.cuda.synchronize()
    print("pmc probe done")


if __name__ == "__main__":
    main()
