"""Fused AdamW with fp32 master weights for bf16 training.

The reference's upstream trainer uses BertAdam/AdamW on fp32 params; this
build trains bf16 (288 GB HBM sizing) — but
``torch.optim.AdamW`` on bf16 params keeps bf16 optimizer state, which
truncates the small-update tail (8 mantissa bits). ``FusedAdamW`` keeps an
fp32 master copy + fp32 moments and updates {master, m, v, bf16 param} in
ONE HIP kernel per tensor (ops/csrc/optim.hip) instead of torch's 6+
elementwise kernels. On CPU (or fp32 params) the identical math runs in
torch ops — the CPU path is the numerics oracle for the GPU kernel test.
"""

from __future__ import annotations

import math
from typing import Iterable

import torch


class FusedAdamW(torch.optim.Optimizer):
    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 1e-4,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.01,
    ):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            lr = group["lr"]
            b1, b2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["master"] = p.detach().float().clone()
                    state["exp_avg"] = torch.zeros_like(state["master"])
                    state["exp_avg_sq"] = torch.zeros_like(state["master"])
                state["step"] += 1
                t = state["step"]
                master, m, v = state["master"], state["exp_avg"], state["exp_avg_sq"]
                grad = p.grad
                use_kernel = (
                    p.is_cuda
                    and p.dtype == torch.bfloat16
                    and grad.dtype in (torch.bfloat16, torch.float32)
                )
                if use_kernel:
                    from ..ops import functional as F_ops

                    if F_ops.extension_available():
                        torch.ops.vilbert_amd.adamw_step(
                            p.data, grad.contiguous(), master, m, v,
                            lr, b1, b2, eps, wd, t,
                        )
                        continue
                # torch fallback — same math, any device/dtype
                g = grad.float()
                m.mul_(b1).add_(g, alpha=1 - b1)
                v.mul_(b2).addcmul_(g, g, value=1 - b2)
                bc1 = 1 - b1 ** t
                bc2 = 1 - b2 ** t
                update = (m / bc1) / ((v / bc2).sqrt() + eps) + wd * master
                master.add_(update, alpha=-lr)
                p.data.copy_(master.to(p.dtype))
        return loss
