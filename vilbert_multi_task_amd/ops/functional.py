"""Dispatch layer between the plain-PyTorch reference ops and the HIP/CDNA4
extension.

Rules of the build (BASELINE.json north star):
- On a GPU box the hand-written gfx950 kernels ARE the compute path; if the
  extension is missing on a CUDA/HIP device we fail loudly instead of
  silently falling back to eager PyTorch (the driver checks which .so the
  GPU tests actually loaded).
- On CPU the plain fp32 PyTorch implementations below are the numerics
  oracle each HIP kernel is tested against (SURVEY.md §4 consequence (1)).

Set VILBERT_AMD_FORCE_EAGER=1 to force the PyTorch path on GPU (debug only).
"""

from __future__ import annotations

import math
import os
from typing import Optional, Tuple

import torch

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from . import hip_ext  # built in-tree by setup/build (vilbert_hip.so)

        _EXT = hip_ext.load()
    except Exception as e:  # pragma: no cover - exercised on GPU box only
        _EXT_ERR = f"{type(e).__name__}: {e}"
        _EXT = None
    return _EXT


def extension_available() -> bool:
    return _load_extension() is not None


def _want_hip(x: torch.Tensor) -> bool:
    if not x.is_cuda:
        return False
    if os.environ.get("VILBERT_AMD_FORCE_EAGER") == "1":
        return False
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "vilbert_multi_task_amd: tensor is on GPU but the gfx950 HIP "
            f"extension is not loaded ({_EXT_ERR}). Build it with "
            "`python -m vilbert_multi_task_amd.ops.build` (or __graft_entry__.build()). "
            "Refusing silent eager fallback."
        )
    return True


# --------------------------------------------------------------------------
# LayerNorm (+ optional fused residual add)
# --------------------------------------------------------------------------

def layer_norm(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor,
    eps: float,
    residual: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """y = LayerNorm(x + residual) over the last dim."""
    if _want_hip(x) and not (torch.is_grad_enabled() and x.requires_grad):
        ext = _load_extension()
        return ext.residual_layer_norm(x, residual, weight, bias, eps)
    if residual is not None:
        x = x + residual
    return torch.nn.functional.layer_norm(x, (x.shape[-1],), weight, bias, eps)


# --------------------------------------------------------------------------
# GELU (+ optional fused bias)
# --------------------------------------------------------------------------

def bias_gelu(x: torch.Tensor, bias: Optional[torch.Tensor]) -> torch.Tensor:
    """y = gelu(x + bias) — exact (erf) GELU, matching BERT."""
    if _want_hip(x) and not (torch.is_grad_enabled() and x.requires_grad):
        ext = _load_extension()
        return ext.bias_gelu(x, bias)
    if bias is not None:
        x = x + bias
    return torch.nn.functional.gelu(x)


def bias_tanh(x: torch.Tensor, bias: Optional[torch.Tensor]) -> torch.Tensor:
    if bias is not None:
        x = x + bias
    return torch.tanh(x)


# --------------------------------------------------------------------------
# Scaled-dot-product attention with additive mask
# --------------------------------------------------------------------------

def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    mask_bias: Optional[torch.Tensor],
    dropout_p: float = 0.0,
    training: bool = False,
    need_probs: bool = False,
) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """Multi-head SDPA.

    q: [B,H,Lq,D], k/v: [B,H,Lk,D]; mask_bias additive [B,1,1,Lk] or
    [B,1,Lq,Lk] (0 for keep, -inf-ish for masked). Returns (ctx [B,H,Lq,D],
    probs or None).
    """
    if (
        _want_hip(q)
        and not need_probs
        and not (training and dropout_p > 0.0)
        and not (torch.is_grad_enabled() and (q.requires_grad or k.requires_grad or v.requires_grad))
    ):
        ext = _load_extension()
        return ext.attention(q, k, v, mask_bias), None

    scale = 1.0 / math.sqrt(q.shape[-1])
    scores = torch.matmul(q, k.transpose(-1, -2)) * scale
    if mask_bias is not None:
        scores = scores + mask_bias
    probs = torch.softmax(scores, dim=-1)
    if training and dropout_p > 0.0:
        probs_d = torch.nn.functional.dropout(probs, p=dropout_p, training=True)
    else:
        probs_d = probs
    ctx = torch.matmul(probs_d, v)
    return ctx, (probs if need_probs else None)


# --------------------------------------------------------------------------
# Embedding lookup + sum + LayerNorm (text embeddings hot path)
# --------------------------------------------------------------------------

def embedding_ln(
    ids: torch.Tensor,
    pos_ids: torch.Tensor,
    type_ids: torch.Tensor,
    word_w: torch.Tensor,
    pos_w: torch.Tensor,
    type_w: torch.Tensor,
    ln_w: torch.Tensor,
    ln_b: torch.Tensor,
    eps: float,
) -> torch.Tensor:
    if _want_hip(word_w) and ids.is_cuda and not torch.is_grad_enabled():
        ext = _load_extension()
        return ext.embedding_ln(ids, pos_ids, type_ids, word_w, pos_w, type_w, ln_w, ln_b, eps)
    e = (
        torch.nn.functional.embedding(ids, word_w)
        + torch.nn.functional.embedding(pos_ids, pos_w)
        + torch.nn.functional.embedding(type_ids, type_w)
    )
    return torch.nn.functional.layer_norm(e, (e.shape[-1],), ln_w, ln_b, eps)
