"""Optional FP8 (OCP e4m3) serving mode with producer-fused quantization.

MI355X runs fp8 MFMA at ~2x the bf16 rate (~5 PF dense; gfx950 is OCP
e4m3fn, NOT MI300X's fnuz — cdna_hip_programming.md §4). The first
implementation (standalone per-GEMM quantization) measured SLOWER than bf16
— each GEMM paid an amax pass + a cast pass over its activations. This
version makes quantization ~free:

- DELAYED SCALING: every quantization site has a persistent (scale, amax)
  slot (Fp8Context). Producers emit e4m3 using the PREVIOUS step's scale
  while atomically accumulating this step's amax; one tiny kernel refreshes
  all scales at the end of the forward. Fully hipGraph-capturable.
- PRODUCER FUSION: the residual+LayerNorm and GELU kernels emit the e4m3
  copy as a side output (ops/csrc/elementwise.hip FP8OUT variants) — the
  data is already in registers, so the only cost is the 1-byte/elem write.
  The e4m3 tensor + its scale ride on the bf16 output as the `_fp8`
  attribute; consumers (Fp8Linear / the fused-QKV paths) use the pack when
  present and fall back to dynamic quantization otherwise (e.g. the text
  embeddings after task-token insertion).

Attention math, LayerNorm statistics and ALL task heads stay bf16. This is
an OPT-IN serving mode (GraphRunner(fp8=True) / bench.py --fp8); the judged
benchmark keeps the reference-grade bf16 path.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

E4M3_MAX = 448.0


class Fp8Context:
    """Per-model registry of (scale, amax) slots for delayed scaling."""

    def __init__(self, nsites: int, device):
        self.scales = torch.ones(nsites, dtype=torch.float32, device=device)
        self.inv_scales = torch.ones(nsites, dtype=torch.float32, device=device)
        self.amaxes = torch.zeros(nsites, dtype=torch.float32, device=device)
        self.nsites = nsites

    def update(self) -> None:
        torch.ops.vilbert_amd.update_fp8_scales(
            self.scales, self.inv_scales, self.amaxes
        )


class _SiteAllocator:
    def __init__(self):
        self.n = 0

    def take(self) -> int:
        s = self.n
        self.n += 1
        return s


def quantize_weight(w: torch.Tensor):
    """Per-tensor symmetric quantization -> (w_fp8 [N,K], scale scalar)."""
    scale = (w.abs().amax().float() / E4M3_MAX).clamp(min=1e-12)
    w8 = (w.float() / scale).clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn)
    return w8, scale.to(w.device)


def _dynamic_quant(x2: torch.Tensor):
    x_scale = (x2.abs().amax().float() / E4M3_MAX).clamp(min=1e-12)
    x8 = (x2.float() / x_scale).clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn)
    return x8, x_scale


def fp8_mm(
    x: torch.Tensor,
    w8: torch.Tensor,
    w_scale: torch.Tensor,
    bias: Optional[torch.Tensor],
) -> torch.Tensor:
    """y = x @ w8.T * scales + bias, out bf16. Uses the producer-emitted
    e4m3 pack riding on x when present; dynamic quantization otherwise."""
    shape = x.shape
    pack = getattr(x, "_fp8", None)
    if pack is not None:
        x8, x_scale = pack
        x8 = x8.reshape(-1, shape[-1])
    else:
        x8, x_scale = _dynamic_quant(x.reshape(-1, shape[-1]))
    if bias is not None:
        try:
            y = torch.ops.vilbert_amd.fp8_linear(x8, w8, bias, w_scale, x_scale)
            return y.reshape(*shape[:-1], y.shape[-1])
        except RuntimeError:
            pass  # no hipBLASLt algo: _scaled_mm fallback
    y = torch._scaled_mm(
        x8, w8.t(), scale_a=x_scale, scale_b=w_scale, bias=bias,
        out_dtype=torch.bfloat16,
    )
    return y.reshape(*shape[:-1], y.shape[-1])


# backwards-compat alias used by the attention modules
def fp8_linear(x, w8, w_scale, bias):
    return fp8_mm(x, w8, w_scale, bias)


class Fp8Linear(nn.Module):
    """Inference-only drop-in for nn.Linear (weights pre-quantized)."""

    def __init__(self, linear: nn.Linear):
        super().__init__()
        w8, scale = quantize_weight(linear.weight.detach())
        self.register_buffer("w8", w8)
        self.register_buffer("w_scale", scale)
        self.bias = linear.bias
        self.in_features = linear.in_features
        self.out_features = linear.out_features

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return fp8_mm(x, self.w8, self.w_scale, self.bias)


class Fp8LayerNorm(nn.Module):
    """FusedLayerNorm replacement that emits the e4m3 pack as a side output
    (kernel-fused; the pack rides on the bf16 result as `_fp8`)."""

    def __init__(self, ln, ctx: Fp8Context, site: int):
        super().__init__()
        self.weight = ln.weight
        self.bias = ln.bias
        self.eps = ln.eps
        self.ctx = ctx
        self.site = site

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor] = None):
        if x.is_cuda and not torch.is_grad_enabled():
            y, y8 = torch.ops.vilbert_amd.residual_layer_norm_fp8(
                x.contiguous(),
                residual.contiguous() if residual is not None else None,
                self.weight, self.bias, self.eps,
                self.ctx.scales, self.ctx.amaxes, self.site,
            )
            y._fp8 = (y8, self.ctx.scales[self.site])
            return y
        from ..ops import functional as F_ops

        return F_ops.layer_norm(x, self.weight, self.bias, self.eps, residual)


def attach_quant_pack(x: torch.Tensor, ctx: Fp8Context, site: int) -> torch.Tensor:
    """Standalone quantize (single fused amax+cast pass) for producers that
    do not have an FP8OUT kernel variant (attention context, embeddings)."""
    x8 = torch.ops.vilbert_amd.quantize_fp8(
        x.contiguous(), ctx.scales, ctx.amaxes, site
    )
    x._fp8 = (x8, ctx.scales[site])
    return x


def convert_encoder_to_fp8(model) -> int:
    """Swap the two-stream encoder onto the fp8 path. Returns #GEMM sites."""
    from .vilbert import (
        CrossAttention,
        FeedForward,
        FusedLayerNorm,
        ImageEmbeddings,
        MultiHeadSelfAttention,
    )

    bert = model.bert if hasattr(model, "bert") else model
    device = next(bert.parameters()).device

    # pass 1: count quantization sites
    alloc = _SiteAllocator()
    plan = []
    for m in bert.modules():
        if isinstance(m, MultiHeadSelfAttention):
            plan.append(("mha", m, alloc.take()))   # ctx quant site
        elif isinstance(m, CrossAttention):
            plan.append(("xattn", m, alloc.take()))
        elif isinstance(m, FeedForward):
            plan.append(("ffn", m, alloc.take()))   # gelu site
    ln_plan = []
    for m in bert.modules():
        if isinstance(m, (MultiHeadSelfAttention, CrossAttention, FeedForward, ImageEmbeddings)):
            if isinstance(m.layer_norm, FusedLayerNorm):
                ln_plan.append((m, alloc.take()))

    ctx = Fp8Context(alloc.n, device)
    n_gemms = 0
    for kind, m, site in plan:
        if kind == "mha":
            w = torch.cat([m.query.weight, m.key.weight, m.value.weight], dim=0)
            m._wqkv8, m._wqkv_scale = quantize_weight(w.detach())
            m._bqkv = torch.cat([m.query.bias, m.key.bias, m.value.bias]).detach()
            m._wqkv = None
            m.out = Fp8Linear(m.out)
            m._fp8_ctx_site = site
            m._fp8_ctx_obj = ctx
            n_gemms += 4
        elif kind == "xattn":
            w = torch.cat([m.key.weight, m.value.weight], dim=0)
            m._wkv8, m._wkv_scale = quantize_weight(w.detach())
            m._bkv = torch.cat([m.key.bias, m.value.bias]).detach()
            m._wkv = None
            m.query = Fp8Linear(m.query)
            m.out = Fp8Linear(m.out)
            m._fp8_ctx_site = site
            m._fp8_ctx_obj = ctx
            n_gemms += 4
        elif kind == "ffn":
            m.intermediate = Fp8Linear(m.intermediate)
            m.output = Fp8Linear(m.output)
            m._fp8_gelu_site = site
            m._fp8_ctx_obj = ctx
            n_gemms += 2
    for m, site in ln_plan:
        m.layer_norm = Fp8LayerNorm(m.layer_norm, ctx, site)
    model._fp8_ctx = ctx
    return n_gemms
