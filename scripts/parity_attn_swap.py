"""Parity of the swapped-S^T attention path (VILBERT_ATTN_SWAP=1) vs fp32."""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from vilbert_multi_task_amd.ops import functional as F_ops
F_ops._load_extension()
torch.manual_seed(0)
ok = True
for (B, H, Lq, Lk, D) in [(8, 12, 38, 38, 64), (8, 8, 101, 101, 128),
                          (8, 8, 38, 101, 128), (8, 8, 101, 38, 128),
                          (3, 4, 17, 23, 64), (2, 2, 33, 64, 128)]:
    q = torch.randn(B, Lq, H * D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
    keep = torch.ones(B, Lk); keep[:, max(Lk - 3, 1):] = 0
    mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
    y = torch.ops.vilbert_amd.attention(q, k, v, H, mask)
    qf = q.float().view(B, Lq, H, D).transpose(1, 2)
    kf = k.float().view(B, Lk, H, D).transpose(1, 2)
    vf = v.float().view(B, Lk, H, D).transpose(1, 2)
    s = qf @ kf.transpose(-1, -2) / D ** 0.5 + mask.float()
    ref = (s.softmax(-1) @ vf).transpose(1, 2).reshape(B, Lq, H * D)
    err = (y.float() - ref).abs().max().item()
    stat = "OK" if err < 3e-2 else "FAIL"
    ok &= err < 3e-2
    print(f"B{B} H{H} Lq{Lq} Lk{Lk} D{D}: max_err={err:.3e} {stat}")
print("SWAP PARITY", "PASS" if ok else "FAIL")
