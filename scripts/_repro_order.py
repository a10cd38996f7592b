import sys, torch
sys.path.insert(0, "/root/repo")
from vilbert_multi_task_amd.ops import functional as F_ops
F_ops._load_extension()
order = sys.argv[1]
x = torch.randn(512, 768, device="cuda", dtype=torch.bfloat16)
w = torch.randn(768, 768, device="cuda", dtype=torch.bfloat16) * 0.05
b = torch.randn(768, device="cuda", dtype=torch.bfloat16)
def blaslt(): return torch.ops.vilbert_amd.linear_bias(x, w, b)
def mfma():  return torch.ops.vilbert_amd.mfma_linear(x, w, b, None, False)
try:
    if order == "mfma_first":
        mfma(); torch.cuda.synchronize(); print("mfma ok")
        blaslt(); torch.cuda.synchronize(); print("blaslt ok")
    else:
        blaslt(); torch.cuda.synchronize(); print("blaslt ok")
        mfma(); torch.cuda.synchronize(); print("mfma ok")
        blaslt(); torch.cuda.synchronize(); print("blaslt again ok")
except Exception as e:
    print("FAIL:", e)
