import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from vilbert_multi_task_amd.ops import functional as F_ops
F_ops._load_extension()
torch.manual_seed(0)
B, H, Lq, Lk, D = 1, 1, 16, 32, 64
q = torch.randn(B, Lq, H * D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
zmask = torch.zeros(B, 1, 1, Lk, device="cuda", dtype=torch.bfloat16)

# probe 1: V = all ones -> O should be exactly 1 everywhere
v1 = torch.ones(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
y = torch.ops.vilbert_amd.attention(q, k, v1, H, zmask)
print("P1 ones-V: max|y-1| =", (y - 1).abs().max().item())

# probe 2: one-hot mask at key j -> O row = V[j]
v = torch.randn(B, Lk, H * D, device="cuda", dtype=torch.bfloat16)
for j in (0, 1, 5, 17, 30):
    m = torch.full((B, 1, 1, Lk), -1e9, device="cuda", dtype=torch.bfloat16)
    m[..., j] = 0
    y = torch.ops.vilbert_amd.attention(q, k, v, H, m)
    err = (y - v[:, j:j+1, :]).abs().max().item()
    print(f"P2 onehot key {j}: max err = {err:.4f}")

# probe 3: full softmax vs ref, zero mask
y = torch.ops.vilbert_amd.attention(q, k, v, H, zmask)
qf = q.float().view(B, Lq, H, D).transpose(1, 2)
kf = k.float().view(B, Lk, H, D).transpose(1, 2)
vf = v.float().view(B, Lk, H, D).transpose(1, 2)
s = qf @ kf.transpose(-1, -2) / D ** 0.5
ref = (s.softmax(-1) @ vf).transpose(1, 2).reshape(B, Lq, H * D)
d = (y.float() - ref).abs()
print("P3 full: max err =", d.max().item())
# where is it wrong? per-query and per-dim error profile
print("per-q err:", [round(x, 3) for x in d[0].max(dim=1).values.tolist()])
print("per-d err (first 16):", [round(x, 3) for x in d[0].max(dim=0).values[:16].tolist()])

# probe 4: bpermute primitive — src[l]=l; idx = my gather patterns
idx = torch.zeros(64, dtype=torch.int32, device="cuda")
for l in range(64):
    cc, qq0 = l >> 4, l & 15
    idx[l] = ((cc & 1) * 2) * 16 + qq0   # idx0 pattern
src = torch.arange(64, dtype=torch.int32, device="cuda")
got = torch.ops.vilbert_amd.bperm_probe(idx, src)
exp = idx.clone()
print("P4 bperm idx0: match =", bool((got == exp).all().item()),
      "got[0:8]=", got[:8].tolist(), "got[16:24]=", got[16:24].tolist(),
      "got[32:40]=", got[32:40].tolist(), "got[48:56]=", got[48:56].tolist())

# probe 5: one-hot per-query pattern (which queries get wrong V rows)
j = 5
m = torch.full((B, 1, 1, Lk), -1e9, device="cuda", dtype=torch.bfloat16)
m[..., j] = 0
y = torch.ops.vilbert_amd.attention(q, k, v, H, m)
for qq in range(8):
    # which V row does query qq's output best match?
    dists = (v[0].float() - y[0, qq].float()).abs().max(dim=1).values
    print(f"P5 q={qq}: best-match V row = {int(dists.argmin())} (want {j}), dist={dists.min():.3f}")
