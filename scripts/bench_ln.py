#!/usr/bin/env python3
"""Standalone residual_layer_norm roofline check at the serving shapes."""
import os, statistics, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from vilbert_multi_task_amd.ops import functional as F_ops
F_ops._load_extension()
torch.manual_seed(0)
for rows, dim, has_res in [(38912, 768, True), (103424, 1024, True),
                           (38912, 768, False), (103424, 1024, False)]:
    x = torch.randn(rows, dim, device="cuda", dtype=torch.bfloat16)
    r = torch.randn(rows, dim, device="cuda", dtype=torch.bfloat16) if has_res else None
    w = torch.ones(dim, device="cuda", dtype=torch.bfloat16)
    b = torch.zeros(dim, device="cuda", dtype=torch.bfloat16)
    ts = {"ours": [], "torch": []}
    for it in range(60):
        for mode in ("ours", "torch"):
            torch.cuda.synchronize(); t0 = time.perf_counter()
            if mode == "ours":
                torch.ops.vilbert_amd.residual_layer_norm(x, r, w, b, 1e-12)
            else:
                xx = x + r if has_res else x
                torch.nn.functional.layer_norm(xx, (dim,), w, b, 1e-12)
            torch.cuda.synchronize()
            if it >= 10: ts[mode].append((time.perf_counter() - t0) * 1e6)
    mo, mt = statistics.median(ts["ours"]), statistics.median(ts["torch"])
    passes = 3 if has_res else 2
    gb = rows * dim * 2 * passes / 1e9
    print(f"rows={rows} dim={dim} res={has_res}: ours={mo:7.1f}us ({gb/mo*1e6/1000:.2f} TB/s)"
          f"  torch={mt:7.1f}us  roofline={gb/8e3*1e6:.1f}us")
