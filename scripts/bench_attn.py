#!/usr/bin/env python3
"""Within-process A/B of the attention kernel paths at the serving shapes.

Usage: python scripts/bench_attn.py [--batch 1024] [--iters 50]
Interleaves VILBERT_ATTN_BHLOOP=0 (v2 per-bh kernel) and =1 (v3 bh-loop
prefetch) rounds in ONE process (guide §5.4 rule 24) and reports per-
dispatch medians. HBM roofline per shape printed for context.
"""

import argparse
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from vilbert_multi_task_amd.ops import functional as F_ops

F_ops._load_extension()  # registers torch.ops.vilbert_amd

SHAPES = [
    # (name, H, Lq, Lk, D)
    ("text-self", 12, 38, 38, 64),
    ("vision-self", 8, 101, 101, 128),
    ("co-attn t->v", 8, 38, 101, 128),
    ("co-attn v->t", 8, 101, 38, 128),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=1024)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--probs", action="store_true", help="bench the prob-export path")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    B = args.batch
    torch.manual_seed(0)

    for name, H, Lq, Lk, D in SHAPES:
        q = torch.randn(B, Lq, H * D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
        keep = torch.ones(B, Lk)
        keep[:, Lk - 3 :] = 0
        mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
        bytes_moved = 2 * B * H * D * (Lq * 2 + Lk * 2)  # Q+O+K+V bf16
        roof_us = bytes_moved / 6.3e12 * 1e6

        res = {"0": [], "1": []}
        for rnd in range(args.iters):
            for mode in ("0", "1"):
                os.environ["VILBERT_ATTN_BHLOOP"] = mode
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                if args.probs:
                    torch.ops.vilbert_amd.attention_probs(q, k, v, H, mask)
                else:
                    torch.ops.vilbert_amd.attention(q, k, v, H, mask)
                torch.cuda.synchronize()
                dt = (time.perf_counter() - t0) * 1e6
                if rnd >= 5:
                    res[mode].append(dt)
        m0 = statistics.median(res["0"])
        m1 = statistics.median(res["1"])
        print(
            f"{name:14s} B={B} v2={m0:8.1f}us  v3={m1:8.1f}us  "
            f"({m0 / m1:.2f}x)  hbm-roofline={roof_us:6.1f}us"
        )


if __name__ == "__main__":
    main()
