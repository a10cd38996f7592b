import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from vilbert_multi_task_amd.ops import functional as F_ops
F_ops._load_extension()
torch.manual_seed(0)

def onehot_map(B, H, Lq, Lk, D, jlist):
    q = torch.randn(B, Lq, H * D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
    out = []
    for j in jlist:
        m = torch.full((B, 1, 1, Lk), -1e9, device="cuda", dtype=torch.bfloat16)
        m[..., j] = 0
        y = torch.ops.vilbert_amd.attention(q, k, v, H, m)
        # per query: which V row does the output match?
        rows = []
        for qq in range(min(Lq, 8)):
            d = (v[0].float().unsqueeze(0) - y[0, qq].float().unsqueeze(0).expand(Lk, -1).unsqueeze(0)).squeeze(0)
            dist = (v[0].float() - y[0, qq].float()).abs().max(dim=1).values
            rows.append(int(dist.argmin()))
        out.append((j, rows))
    return out

print("Lk=32 D=64:", onehot_map(1, 1, 16, 32, 64, [0, 1, 4, 5, 8, 12, 17, 30]))
print("Lk=64 D=64:", onehot_map(1, 1, 16, 64, 64, [0, 5, 17, 30, 45]))
print("Lk=33 D=64:", onehot_map(1, 1, 16, 33, 64, [0, 17]))
