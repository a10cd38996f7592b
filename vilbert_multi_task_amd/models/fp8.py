"""Optional FP8 (OCP e4m3) serving mode for the encoder GEMMs.

MI355X runs fp8 MFMA at ~2x the bf16 rate (~5 PF dense; gfx950 uses OCP
e4m3fn, NOT MI300X's fnuz — cdna_hip_programming.md §4). This mode
quantizes the two-stream encoder's projection/FFN weights to e4m3 with
per-tensor scales and runs them through torch._scaled_mm (hipBLASLt fp8),
with dynamic per-tensor activation scales computed on-GPU (amax reduction —
hipGraph-capturable, no host sync). Attention math, LayerNorms, embeddings
and ALL task heads stay bf16.

This is an OPT-IN serving mode (GraphRunner(serving_dtype="fp8") or
bench.py --fp8): the judged benchmark keeps the reference-grade bf16 path.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

E4M3_MAX = 448.0


def quantize_weight(w: torch.Tensor):
    """Per-tensor symmetric quantization -> (w_fp8 [N,K], scale scalar)."""
    scale = (w.abs().amax().float() / E4M3_MAX).clamp(min=1e-12)
    w8 = (w.float() / scale).clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn)
    return w8, scale.to(w.device)


def fp8_linear(
    x: torch.Tensor,
    w8: torch.Tensor,
    w_scale: torch.Tensor,
    bias: Optional[torch.Tensor],
) -> torch.Tensor:
    """y = x @ w8.T * scales + bias, out bf16. x is bf16 [.., K]."""
    shape = x.shape
    x2 = x.reshape(-1, shape[-1])
    x_scale = (x2.abs().amax().float() / E4M3_MAX).clamp(min=1e-12)
    x8 = (x2.float() / x_scale).clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn)
    y = torch._scaled_mm(
        x8,
        w8.t(),
        scale_a=x_scale,
        scale_b=w_scale,
        bias=bias,
        out_dtype=torch.bfloat16,
    )
    return y.reshape(*shape[:-1], y.shape[-1])


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
        return fp8_linear(x, self.w8, self.w_scale, self.bias)


def convert_encoder_to_fp8(model) -> int:
    """Swap the two-stream encoder's Linear GEMMs (QKV/out projections and
    FFN in/out of the text, vision and connection layers) to Fp8Linear.
    Heads/poolers/embeddings stay bf16. Returns the number of conversions.

    Also clears any fused-QKV bf16 weights so the attention modules fall
    back to per-projection calls (now fp8)."""
    from .vilbert import CrossAttention, FeedForward, MultiHeadSelfAttention

    n = 0
    bert = model.bert if hasattr(model, "bert") else model
    for m in bert.modules():
        if isinstance(m, MultiHeadSelfAttention):
            # fused QKV in fp8: ONE activation quantization + ONE _scaled_mm
            w = torch.cat([m.query.weight, m.key.weight, m.value.weight], dim=0)
            m._wqkv8, m._wqkv_scale = quantize_weight(w.detach())
            m._bqkv = torch.cat([m.query.bias, m.key.bias, m.value.bias]).detach()
            m._wqkv = None  # bf16 fused path off
            m.out = Fp8Linear(m.out)
            n += 4
        elif isinstance(m, CrossAttention):
            w = torch.cat([m.key.weight, m.value.weight], dim=0)
            m._wkv8, m._wkv_scale = quantize_weight(w.detach())
            m._bkv = torch.cat([m.key.bias, m.value.bias]).detach()
            m._wkv = None
            m.query = Fp8Linear(m.query)
            m.out = Fp8Linear(m.out)
            n += 4
        elif isinstance(m, FeedForward):
            m.intermediate = Fp8Linear(m.intermediate)
            m.output = Fp8Linear(m.output)
            n += 2
    return n
