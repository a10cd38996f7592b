import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from vilbert_multi_task_amd.ops import functional as F_ops
F_ops._load_extension()
torch.manual_seed(0)
B, H, Lq, Lk, D = 1, 1, 16, 32, 64
q = torch.randn(B, Lq, H * D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
# V[r][d] = r  -> output reveals which V ROW fed each (q,d)
v = torch.arange(Lk, device="cuda", dtype=torch.bfloat16).view(1, Lk, 1).expand(B, Lk, H * D).contiguous()
for j in (5, 6, 13):
    m = torch.full((B, 1, 1, Lk), -1e9, device="cuda", dtype=torch.bfloat16)
    m[..., j] = 0
    y = torch.ops.vilbert_amd.attention(q, k, v, H, m)
    print(f"onehot j={j}: O[0,0,0:24] = {[int(x) for x in y[0,0,:24].float().tolist()]}")
