import os, sys, torch
sys.path.insert(0, "/root/repo")
from vilbert_multi_task_amd.ops import functional as F_ops
F_ops._load_extension()
torch.manual_seed(0)
ok = True
for (M, N, K, gelu, use_res) in [(512, 768, 768, False, False), (1000, 2304, 768, False, False),
                                  (38912, 768, 3072, True, False), (777, 1024, 1024, False, True),
                                  (1024, 30520, 768, False, False)]:
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.05
    b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    r = torch.randn(M, N, device="cuda", dtype=torch.bfloat16) if use_res else None
    y = torch.ops.vilbert_amd.mfma_linear(x, w, b, r, gelu)
    ref = torch.nn.functional.linear(x.float(), w.float(), b.float())
    if gelu: ref = torch.nn.functional.gelu(ref)
    if use_res: ref = ref + r.float()
    err = (y.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1)
    stat = "OK" if err < 2e-2 else "FAIL"
    ok &= err < 2e-2
    print(f"M={M} N={N} K={K} gelu={gelu} res={use_res}: rel_err={err:.2e} {stat}")
print("PARITY", "PASS" if ok else "FAIL")
