#!/usr/bin/env python3
"""A/B the hand-written MFMA GEMM (gemm_mfma.hip) vs hipBLASLt at the
serving GEMM shapes (B=1024 request batch). Interleaved rounds in one
process (guide §5.4 rule 24), medians reported with achieved TFLOP/s.
"""

import argparse
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from vilbert_multi_task_amd.ops import functional as F_ops

F_ops._load_extension()

# (name, M(batch-mult), N, K)  — M scales with --batch (rows = batch * L)
SHAPES = [
    ("t-qkv   ", 38, 2304, 768),
    ("t-out   ", 38, 768, 768),
    ("t-ffn1  ", 38, 3072, 768),
    ("t-ffn2  ", 38, 768, 3072),
    ("v-qkv   ", 101, 3072, 1024),
    ("v-ffn1  ", 101, 1024, 1024),
    ("v-out   ", 101, 1024, 1024),
    ("vocab   ", 1, 30520, 768),  # LM head (N trimmed to %8 for the kernel)
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=1024)
    ap.add_argument("--iters", type=int, default=40)
    ap.add_argument("--gelu", action="store_true")
    ap.add_argument("--res", action="store_true", help="A/B the residual-fused epilogue")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    torch.manual_seed(0)

    for name, lm, N, K in SHAPES:
        M = args.batch * lm
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.05
        bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
        res = torch.randn(M, N, device="cuda", dtype=torch.bfloat16) if args.res else None
        lnw = torch.ones(N, device="cuda", dtype=torch.bfloat16)
        lnb = torch.zeros(N, device="cuda", dtype=torch.bfloat16)
        flops = 2.0 * M * N * K
        res_t = {"mfma": [], "blaslt": []}
        for rnd in range(args.iters):
            for mode in ("mfma", "blaslt"):
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                if mode == "mfma":
                    y = torch.ops.vilbert_amd.mfma_linear(x, w, bias, res, args.gelu)
                    if args.res:
                        torch.ops.vilbert_amd.residual_layer_norm(
                            y, None, lnw, lnb, 1e-12)
                else:
                    # baseline = the shipped default: torch F.linear
                    # (hipBLASLt's own BIAS epilogue faults at large M)
                    if args.gelu:
                        torch.ops.vilbert_amd.linear_bias_gelu(x, w, bias)
                    elif args.res:
                        y = torch.nn.functional.linear(x, w, bias)
                        torch.ops.vilbert_amd.residual_layer_norm(
                            y, res, lnw, lnb, 1e-12)
                    else:
                        torch.nn.functional.linear(x, w, bias)
                torch.cuda.synchronize()
                dt = (time.perf_counter() - t0) * 1e6
                if rnd >= 5:
                    res_t[mode].append(dt)
        m0 = statistics.median(res_t["blaslt"])
        m1 = statistics.median(res_t["mfma"])
        print(
            f"{name} M={M:6d} N={N:5d} K={K:4d}  "
            f"blaslt={m0:8.1f}us ({flops / m0 / 1e6:6.0f} TF)  "
            f"mfma={m1:8.1f}us ({flops / m1 / 1e6:6.0f} TF)  ({m0 / m1:.2f}x)"
        )


if __name__ == "__main__":
    main()
