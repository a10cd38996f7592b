"""GPU numerics: every HIP kernel vs the plain-PyTorch fp32 reference
(SURVEY.md §4 consequence (3): single-GPU numerical parity, bf16 tolerances).

All tests build reference values from the SAME bf16-rounded inputs so the
measured error is kernel error, not input quantization.
"""

import math

import pytest
import torch

from vilbert_multi_task_amd.ops import functional as F_ops

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from vilbert_multi_task_amd.ops import hip_ext

    return hip_ext.load()


def _rand_bf16(*shape, seed=0, scale=1.0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).to(torch.bfloat16).cuda()


# ---------------------------------------------------------------------------
# MFMA fragment-layout probe: asymmetric operands catch transposes
# (cdna_hip_programming.md §3 "Always A=I-check with ASYMMETRIC B")
# ---------------------------------------------------------------------------

def test_mfma_probe_layout(ext):
    torch.manual_seed(0)
    a = _rand_bf16(16, 32, seed=1)
    b = (torch.arange(32 * 16).reshape(32, 16).float() * 0.01 - 2.0).to(
        torch.bfloat16
    ).cuda()
    c = torch.ops.vilbert_amd.mfma_probe(a, b)
    ref = a.float() @ b.float()
    assert torch.allclose(c, ref, atol=1e-2, rtol=1e-2), (
        (c - ref).abs().max().item()
    )


def test_mfma_probe_identity(ext):
    a = torch.eye(16, 32).to(torch.bfloat16).cuda()
    b = _rand_bf16(32, 16, seed=3)
    c = torch.ops.vilbert_amd.mfma_probe(a, b)
    assert torch.allclose(c, b.float()[:16], atol=1e-2)


# ---------------------------------------------------------------------------
# residual + LayerNorm
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("dim", [768, 1024, 1536, 3072])
@pytest.mark.parametrize("with_res", [False, True])
def test_residual_layer_norm(ext, dim, with_res):
    rows = 37 * 8
    x = _rand_bf16(rows, dim, seed=dim)
    res = _rand_bf16(rows, dim, seed=dim + 1) if with_res else None
    w = _rand_bf16(dim, seed=dim + 2, scale=0.5)
    b = _rand_bf16(dim, seed=dim + 3, scale=0.1)
    y = torch.ops.vilbert_amd.residual_layer_norm(x, res, w, b, 1e-12)
    xf = x.float() + (res.float() if with_res else 0.0)
    ref = torch.nn.functional.layer_norm(xf, (dim,), w.float(), b.float(), 1e-12)
    assert (y.float() - ref).abs().max() < 3e-2


def test_layer_norm_f32(ext):
    x = torch.randn(64, 768, device="cuda")
    w = torch.randn(768, device="cuda")
    b = torch.randn(768, device="cuda")
    y = torch.ops.vilbert_amd.residual_layer_norm(x, None, w, b, 1e-12)
    ref = torch.nn.functional.layer_norm(x, (768,), w, b, 1e-12)
    assert (y - ref).abs().max() < 1e-5


# ---------------------------------------------------------------------------
# bias + GELU
# ---------------------------------------------------------------------------

def test_bias_gelu(ext):
    x = _rand_bf16(37 * 16, 3072, seed=7, scale=2.0)
    bias = _rand_bf16(3072, seed=8)
    y = torch.ops.vilbert_amd.bias_gelu(x, bias)
    ref = torch.nn.functional.gelu(x.float() + bias.float())
    # bf16 output ulp is |v|/128; allow rounding of large activations
    err = (y.float() - ref).abs() - ref.abs() / 128
    assert err.max() < 2e-2


def test_gelu_no_bias_odd_tail(ext):
    x = _rand_bf16(5, 97, seed=9)  # odd dim exercises scalar path
    y = torch.ops.vilbert_amd.bias_gelu(x, None)
    ref = torch.nn.functional.gelu(x.float())
    assert (y.float() - ref).abs().max() < 2e-2


# ---------------------------------------------------------------------------
# embedding + LayerNorm
# ---------------------------------------------------------------------------

def test_embedding_ln(ext):
    V, P, S, H = 30522, 512, 2, 768
    B, T = 4, 38
    g = torch.Generator().manual_seed(0)
    ids = torch.randint(0, V, (B, T), generator=g).cuda()
    pos = torch.arange(T).unsqueeze(0).expand(B, T).contiguous().cuda()
    typ = torch.zeros(B, T, dtype=torch.long).cuda()
    ww = _rand_bf16(V, H, seed=10, scale=0.02)
    pw = _rand_bf16(P, H, seed=11, scale=0.02)
    tw = _rand_bf16(2, H, seed=12, scale=0.02)
    lw = _rand_bf16(H, seed=13, scale=0.5)
    lb = _rand_bf16(H, seed=14, scale=0.1)
    y = torch.ops.vilbert_amd.embedding_ln(ids, pos, typ, ww, pw, tw, lw, lb, 1e-12)
    e = ww.float()[ids] + pw.float()[pos] + tw.float()[typ]
    ref = torch.nn.functional.layer_norm(e, (H,), lw.float(), lb.float(), 1e-12)
    assert (y.float() - ref).abs().max() < 3e-2


# ---------------------------------------------------------------------------
# fused attention: every serving shape family
# ---------------------------------------------------------------------------

def _attn_ref(q, k, v, heads, mask_bias):
    b, lq, hd = q.shape
    lk = k.shape[1]
    d = hd // heads
    qh = q.float().view(b, lq, heads, d).transpose(1, 2)
    kh = k.float().view(b, lk, heads, d).transpose(1, 2)
    vh = v.float().view(b, lk, heads, d).transpose(1, 2)
    s = qh @ kh.transpose(-1, -2) / math.sqrt(d)
    if mask_bias is not None:
        s = s + mask_bias.float()
    p = torch.softmax(s, dim=-1)
    return (p @ vh).transpose(1, 2).reshape(b, lq, hd)


@pytest.mark.parametrize(
    "B,H,Lq,Lk,D,mask_mode",
    [
        (3, 12, 38, 38, 64, 1),    # text self-attention w/ input_mask
        (2, 8, 101, 101, 128, 1),  # vision self-attention w/ image_mask
        (2, 8, 38, 101, 128, 1),   # co-attn: text queries x vision keys
        (2, 8, 101, 38, 128, 2),   # co-attn: vision queries x text keys, [B,Lq,Lk] bias
        (1, 12, 37, 37, 64, 0),    # no mask, odd length
        (5, 8, 20, 33, 128, 1),    # ragged non-multiple-of-16 lengths
        (2, 4, 16, 32, 64, 1),     # LK_PAD=32: V/P image swizzle must stay in-row
        (3, 4, 17, 23, 64, 1),     # LK_PAD=32 ragged (the r2 SWZR regression)
        (2, 2, 24, 30, 128, 2),    # LK_PAD=32 at D=128, [B,Lq,Lk] bias
    ],
)
def test_attention_vs_reference(ext, B, H, Lq, Lk, D, mask_mode):
    q = _rand_bf16(B, Lq, H * D, seed=B * 100 + Lq)
    k = _rand_bf16(B, Lk, H * D, seed=B * 100 + Lk + 1)
    v = _rand_bf16(B, Lk, H * D, seed=B * 100 + Lk + 2)
    if mask_mode == 0:
        mask = None
        ref_mask = None
    elif mask_mode == 1:
        keep = torch.ones(B, Lk)
        keep[:, Lk - 5 :] = 0  # mask the tail keys
        mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
        ref_mask = mask.float()
    else:
        keep = torch.ones(B, Lq, Lk)
        keep[:, :, Lk - 7 :] = 0
        mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, Lq, Lk)
        ref_mask = mask.float()
    out = torch.ops.vilbert_amd.attention(q, k, v, H, mask)
    ref = _attn_ref(q, k, v, H, ref_mask)
    err = (out.float() - ref).abs().max().item()
    assert err < 4e-2, f"max err {err}"


@pytest.mark.parametrize(
    "B,H,Lq,Lk,D,mask_mode",
    [
        (3, 12, 38, 38, 64, 1),    # text self (the SWAP win shape)
        (2, 8, 101, 101, 128, 1),  # vision self
        (2, 8, 38, 101, 128, 1),   # co-attn t->v
        (2, 8, 101, 38, 128, 2),   # co-attn v->t, [B,Lq,Lk] bias
        (1, 4, 16, 32, 64, 0),     # LK_PAD=32, no mask
        (3, 4, 17, 23, 64, 1),     # LK_PAD=32 ragged
    ],
)
def test_attention_swap_vs_reference(ext, monkeypatch, B, H, Lq, Lk, D, mask_mode):
    """The swapped-S^T register-transpose path (attention.hip SWAP=true):
    softmax reduced across lane groups, P normalized in-lane and gathered to
    the PV A-fragment with ds_bpermute (no P_lds)."""
    monkeypatch.setenv("VILBERT_ATTN_SWAP", "1")
    q = _rand_bf16(B, Lq, H * D, seed=B * 31 + Lq)
    k = _rand_bf16(B, Lk, H * D, seed=B * 31 + Lk + 1)
    v = _rand_bf16(B, Lk, H * D, seed=B * 31 + Lk + 2)
    if mask_mode == 0:
        mask = None
        ref_mask = None
    elif mask_mode == 1:
        keep = torch.ones(B, Lk)
        keep[:, Lk - 5 :] = 0
        mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
        ref_mask = mask.float()
    else:
        keep = torch.ones(B, Lq, Lk)
        keep[:, :, Lk - 7 :] = 0
        mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, Lq, Lk)
        ref_mask = mask.float()
    out = torch.ops.vilbert_amd.attention(q, k, v, H, mask)
    ref = _attn_ref(q, k, v, H, ref_mask)
    err = (out.float() - ref).abs().max().item()
    assert err < 4e-2, f"max err {err}"


def test_attention_masked_rows_sum_to_valid(ext):
    """Fully masked tail keys must contribute exactly zero probability."""
    B, H, Lq, Lk, D = 2, 8, 101, 101, 128
    q = _rand_bf16(B, Lq, H * D, seed=50)
    k = _rand_bf16(B, Lk, H * D, seed=51)
    v_const = torch.zeros(B, Lk, H * D)
    v_const[:, :40] = 1.0  # valid keys have value 1, masked keys 0
    v = v_const.to(torch.bfloat16).cuda()
    keep = torch.zeros(B, Lk)
    keep[:, :40] = 1
    mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
    out = torch.ops.vilbert_amd.attention(q, k, v, H, mask)
    # weighted average of ones over valid keys = 1 exactly
    assert (out.float() - 1.0).abs().max() < 2e-2


# ---------------------------------------------------------------------------
# batched multi-class NMS vs greedy CPU reference
# ---------------------------------------------------------------------------

def _nms_cpu(boxes, scores, iou_thr):
    order = scores.argsort(descending=True)
    keep = []
    suppressed = torch.zeros(len(boxes), dtype=torch.bool)
    for idx in order.tolist():
        if suppressed[idx] or scores[idx] <= 0:
            continue
        keep.append(idx)
        b = boxes[idx]
        ix = (
            torch.minimum(b[2], boxes[:, 2]) - torch.maximum(b[0], boxes[:, 0])
        ).clamp(min=0)
        iy = (
            torch.minimum(b[3], boxes[:, 3]) - torch.maximum(b[1], boxes[:, 1])
        ).clamp(min=0)
        inter = ix * iy
        area_a = (b[2] - b[0]).clamp(min=0) * (b[3] - b[1]).clamp(min=0)
        area_b = (boxes[:, 2] - boxes[:, 0]).clamp(min=0) * (
            boxes[:, 3] - boxes[:, 1]
        ).clamp(min=0)
        iou = inter / (area_a + area_b - inter)
        suppressed |= iou > iou_thr
        suppressed[idx] = False  # self-iou is 1, keep the anchor
    return sorted(keep)


def test_nms_multiclass_vs_cpu(ext):
    torch.manual_seed(0)
    R, C = 300, 25
    centers = torch.rand(R, 2) * 100
    wh = torch.rand(R, 2) * 20 + 2
    boxes = torch.cat([centers - wh / 2, centers + wh / 2], dim=1)
    scores = torch.rand(R, C)
    out = torch.ops.vilbert_amd.nms_multiclass(
        boxes.cuda(), scores.cuda(), 0.5, 0.0
    ).cpu()
    for c in range(0, C, 7):
        keep_ref = _nms_cpu(boxes, scores[:, c], 0.5)
        keep_gpu = sorted(torch.nonzero(out[:, c] > 0).flatten().tolist())
        assert keep_gpu == keep_ref, f"class {c}"


def test_linear_bias_gelu_fused(ext):
    """hipBLASLt epilogue-fused GEMM+bias+GELU vs the erf oracle (tanh-form
    GELU in the epilogue: ~3e-3 divergence, below bf16 resolution here)."""
    x = _rand_bf16(37 * 8, 768, seed=70)
    w = _rand_bf16(3072, 768, seed=71, scale=0.03)
    b = _rand_bf16(3072, seed=72, scale=0.1)
    y = torch.ops.vilbert_amd.linear_bias_gelu(x, w, b)
    ref = torch.nn.functional.gelu(x.float() @ w.float().T + b.float())
    err = (y.float() - ref).abs() - ref.abs() / 64
    assert err.max() < 3e-2, err.max().item()


def test_linear_bias_residual_fused(ext):
    """GEMM + bias + residual in one hipBLASLt call (beta=1 epilogue).
    NOT wired into the model: intermittent hipBLASLt write-faults at some
    shapes (see models/vilbert.py note); kept as the round-2 start point."""
    x = _rand_bf16(37 * 8, 3072, seed=80)
    w = _rand_bf16(768, 3072, seed=81, scale=0.02)
    b = _rand_bf16(768, seed=82, scale=0.1)
    res = _rand_bf16(37 * 8, 768, seed=83)
    y = torch.ops.vilbert_amd.linear_bias_residual(x, w, b, res)
    ref = x.float() @ w.float().T + b.float() + res.float()
    err = (y.float() - ref).abs() - ref.abs() / 64
    assert err.max() < 3e-2, err.max().item()


@pytest.mark.gpu
def test_ln_train_bwd_matches_fp32_autograd():
    """train_bwd.hip LN fwd+bwd vs a full fp32 torch.autograd reference."""
    torch.manual_seed(0)
    for rows, dim, has_res in [(300, 768, True), (300, 768, False), (257, 1024, True), (65, 52, True)]:
        xf = torch.randn(rows, dim, device="cuda")
        rf = torch.randn(rows, dim, device="cuda") if has_res else None
        wf = torch.randn(dim, device="cuda")
        bf = torch.randn(dim, device="cuda")
        gy = torch.randn(rows, dim, device="cuda")

        x32 = xf.clone().requires_grad_()
        r32 = rf.clone().requires_grad_() if has_res else None
        w32 = wf.clone().requires_grad_()
        b32 = bf.clone().requires_grad_()
        h = x32 + r32 if has_res else x32
        torch.nn.functional.layer_norm(h, (dim,), w32, b32, 1e-12).backward(gy)

        x = xf.bfloat16().requires_grad_()
        r = rf.bfloat16().requires_grad_() if has_res else None
        w = wf.bfloat16().requires_grad_()
        b = bf.bfloat16().requires_grad_()
        y = F_ops.layer_norm(x, w, b, 1e-12, residual=r)
        assert "LayerNormTrain" in type(y.grad_fn).__name__, type(y.grad_fn).__name__
        y.backward(gy.bfloat16())

        for got, ref, name in [
            (x.grad, x32.grad, "gx"),
            (w.grad, w32.grad, "gw"),
            (b.grad, b32.grad, "gb"),
        ] + ([(r.grad, r32.grad, "gres")] if has_res else []):
            got = got.float()
            tol = 0.05 * ref.abs().max().clamp(min=1.0)
            assert (got - ref).abs().max() <= tol, (rows, dim, has_res, name, (got - ref).abs().max())


@pytest.mark.gpu
def test_linear_bias_gelu_train_bwd_matches_fp32_autograd():
    torch.manual_seed(1)
    M, K, N = 300, 256, 512
    xf = torch.randn(M, K, device="cuda")
    wf = torch.randn(N, K, device="cuda") * 0.05
    bf = torch.randn(N, device="cuda") * 0.1
    gy = torch.randn(M, N, device="cuda")

    x32 = xf.clone().requires_grad_()
    w32 = wf.clone().requires_grad_()
    b32 = bf.clone().requires_grad_()
    torch.nn.functional.gelu(
        torch.nn.functional.linear(x32, w32, b32)
    ).backward(gy)

    x = xf.bfloat16().requires_grad_()
    w = wf.bfloat16().requires_grad_()
    b = bf.bfloat16().requires_grad_()
    y = F_ops.linear_bias_gelu(x, w, b)
    assert "LinearBiasGelu" in type(y.grad_fn).__name__, type(y.grad_fn).__name__
    y.backward(gy.bfloat16())

    for got, ref, name in [(x.grad, x32.grad, "gx"), (w.grad, w32.grad, "gw"), (b.grad, b32.grad, "gb")]:
        got = got.float()
        tol = 0.06 * ref.abs().max().clamp(min=1.0)
        assert (got - ref).abs().max() <= tol, (name, (got - ref).abs().max(), ref.abs().max())


@pytest.mark.gpu
def test_ln_train_bwd_deterministic():
    """fixed-chunk partials, no atomics: param grads bitwise-stable."""
    torch.manual_seed(2)
    x = torch.randn(5000, 768, device="cuda").bfloat16()
    w = torch.randn(768, device="cuda").bfloat16().requires_grad_()
    b = torch.randn(768, device="cuda").bfloat16().requires_grad_()
    gy = torch.randn(5000, 768, device="cuda").bfloat16()
    grads = []
    for _ in range(2):
        wg = w.detach().clone().requires_grad_()
        bg = b.detach().clone().requires_grad_()
        F_ops.layer_norm(x, wg, bg, 1e-12).backward(gy)
        grads.append((wg.grad.clone(), bg.grad.clone()))
    assert torch.equal(grads[0][0], grads[1][0])
    assert torch.equal(grads[0][1], grads[1][1])


# ---------------------------------------------------------------------------
# v3 bh-loop prefetch kernel (round-2): parity at every serving shape family
# ---------------------------------------------------------------------------

@pytest.mark.parametrize(
    "B,H,Lq,Lk,D,mask_mode",
    [
        (3, 12, 38, 38, 64, 1),     # text self
        (2, 8, 101, 101, 128, 1),   # vision self
        (2, 8, 38, 101, 128, 1),    # co-attn text->vision
        (96, 12, 38, 38, 64, 1),    # BH=1152: auto path, >=2 grid-stride iters
        (24, 8, 101, 101, 128, 1),  # BH=192 forced: ragged iters w/ prefetch
        (5, 8, 20, 33, 128, 0),     # no mask, ragged lengths
    ],
)
def test_attention_bhloop_vs_reference(ext, monkeypatch, B, H, Lq, Lk, D, mask_mode):
    monkeypatch.setenv("VILBERT_ATTN_BHLOOP", "1")
    q = _rand_bf16(B, Lq, H * D, seed=B * 101 + Lq)
    k = _rand_bf16(B, Lk, H * D, seed=B * 101 + Lk + 1)
    v = _rand_bf16(B, Lk, H * D, seed=B * 101 + Lk + 2)
    if mask_mode == 1:
        keep = torch.ones(B, Lk)
        keep[:, Lk - 5:] = 0
        mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
        ref_mask = mask.float()
    else:
        mask = ref_mask = None
    out = torch.ops.vilbert_amd.attention(q, k, v, H, mask)
    ref = _attn_ref(q, k, v, H, ref_mask)
    err = (out.float() - ref).abs().max().item()
    assert err < 4e-2, f"max err {err}"


def test_attention_bhloop_matches_base_kernel(ext, monkeypatch):
    """The bh-loop path must agree with the per-bh kernel to rounding (same
    MFMA order and swizzles; only the mask-add association differs)."""
    B, H, Lq, Lk, D = 16, 8, 101, 101, 128
    q = _rand_bf16(B, Lq, H * D, seed=900)
    k = _rand_bf16(B, Lk, H * D, seed=901)
    v = _rand_bf16(B, Lk, H * D, seed=902)
    keep = torch.ones(B, Lk)
    keep[:, 90:] = 0
    mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
    monkeypatch.setenv("VILBERT_ATTN_BHLOOP", "0")
    base = torch.ops.vilbert_amd.attention(q, k, v, H, mask)
    monkeypatch.setenv("VILBERT_ATTN_BHLOOP", "1")
    loop = torch.ops.vilbert_amd.attention(q, k, v, H, mask)
    assert (base.float() - loop.float()).abs().max().item() < 2e-3


# ---------------------------------------------------------------------------
# attention-prob export (output_all_attention_masks=True, worker.py:288)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize(
    "B,H,Lq,Lk,D,force_bhloop",
    [
        (3, 12, 38, 38, 64, False),
        (2, 8, 101, 101, 128, False),
        (24, 8, 101, 101, 128, True),   # prob export through the bh-loop path
    ],
)
def test_attention_probs_vs_reference(ext, monkeypatch, B, H, Lq, Lk, D, force_bhloop):
    monkeypatch.setenv("VILBERT_ATTN_BHLOOP", "1" if force_bhloop else "0")
    q = _rand_bf16(B, Lq, H * D, seed=700 + B)
    k = _rand_bf16(B, Lk, H * D, seed=701 + B)
    v = _rand_bf16(B, Lk, H * D, seed=702 + B)
    keep = torch.ones(B, Lk)
    keep[:, Lk - 5:] = 0
    mask = ((1 - keep) * -1e9).to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
    out, probs = torch.ops.vilbert_amd.attention_probs(q, k, v, H, mask)
    assert probs.shape == (B, H, Lq, Lk)
    # fp32 reference probs
    d = (H * D) // H
    qh = q.float().view(B, Lq, H, D).transpose(1, 2)
    kh = k.float().view(B, Lk, H, D).transpose(1, 2)
    vh = v.float().view(B, Lk, H, D).transpose(1, 2)
    s = qh @ kh.transpose(-1, -2) / math.sqrt(D) + mask.float()
    p_ref = torch.softmax(s, dim=-1)
    ref = (p_ref @ vh).transpose(1, 2).reshape(B, Lq, H * D)
    assert (out.float() - ref).abs().max().item() < 4e-2
    assert (probs.float() - p_ref).abs().max().item() < 2e-2
    # masked tail keys carry zero probability
    assert probs.float()[..., Lk - 5:].abs().max().item() == 0.0


def test_functional_need_probs_uses_hip(ext):
    """functional.attention(need_probs=True) must run the HIP prob-export
    kernel (round 1 silently fell back to torch math — VERDICT item 7)."""
    from vilbert_multi_task_amd.ops import functional as F_ops

    B, H, L, D = 2, 8, 101, 128
    q = _rand_bf16(B, L, H * D, seed=30)
    k = _rand_bf16(B, L, H * D, seed=31)
    v = _rand_bf16(B, L, H * D, seed=32)
    ctx, probs = F_ops.attention(q, k, v, H, None, need_probs=True)
    assert probs is not None and probs.shape == (B, H, L, L)
    ctx_ref, probs_ref = F_ops.attention(
        q.float().cpu(), k.float().cpu(), v.float().cpu(), H, None, need_probs=True
    )
    assert (ctx.float().cpu() - ctx_ref).abs().max().item() < 4e-2
    assert (probs.float().cpu() - probs_ref).abs().max().item() < 2e-2


# ---------------------------------------------------------------------------
# hand-written MFMA GEMM (gemm_mfma.hip): y = x @ W^T (+bias)(+res)(+GELU)
# ---------------------------------------------------------------------------

def _mfma_linear_ref(x, w, bias, res, gelu):
    y = x.float() @ w.float().t()
    if bias is not None:
        y = y + bias.float()
    if gelu:
        y = torch.nn.functional.gelu(y)
    if res is not None:
        y = y + res.float()
    return y


@pytest.mark.parametrize(
    "M,N,K,gelu,with_res",
    [
        (512, 768, 768, False, False),     # out-proj shape family
        (512, 3072, 768, True, False),     # FFN1 + GELU
        (512, 768, 3072, False, True),     # FFN2 + residual
        (512, 1024, 1024, False, True),    # vision out-proj + residual
        (37, 768, 768, False, False),      # ragged M < one tile
        (300, 2304, 768, False, False),    # ragged M, QKV width
        (259, 776, 1152, True, True),      # ragged M and N (N%8, not %256)
        (1024, 1032, 2048, False, False),  # ragged N just past a tile
    ],
)
def test_mfma_linear_vs_reference(ext, M, N, K, gelu, with_res):
    x = _rand_bf16(M, K, seed=M + N)
    w = _rand_bf16(N, K, seed=M + N + 1, scale=0.05)
    bias = _rand_bf16(N, seed=M + N + 2)
    res = _rand_bf16(M, N, seed=M + N + 3) if with_res else None
    y = torch.ops.vilbert_amd.mfma_linear(x, w, bias, res, gelu)
    ref = _mfma_linear_ref(x, w, bias, res, gelu)
    # bf16 inputs, f32 accumulate: error ~ bf16 rounding of the inputs
    # amplified by sqrt(K) * |x||w| — bound the max abs error empirically
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.02 * max(scale, 1.0), f"max err {err} (scale {scale})"


def test_mfma_linear_matches_hipblaslt(ext):
    """The plain-GEMM result must agree with the hipBLASLt path within bf16
    rounding (both f32-accumulate over the same K)."""
    M, N, K = 512, 768, 768
    x = _rand_bf16(M, K, seed=1)
    w = _rand_bf16(N, K, seed=2, scale=0.05)
    bias = _rand_bf16(N, seed=3)
    a = torch.ops.vilbert_amd.mfma_linear(x, w, bias, None, False)
    b = torch.ops.vilbert_amd.linear_bias(x, w, bias)
    assert (a.float() - b.float()).abs().max().item() < 2e-2


def test_mfma_linear_no_bias(ext):
    M, N, K = 128, 256, 192
    x = _rand_bf16(M, K, seed=4)
    w = _rand_bf16(N, K, seed=5, scale=0.05)
    y = torch.ops.vilbert_amd.mfma_linear(x, w, None, None, False)
    ref = x.float() @ w.float().t()
    assert (y.float() - ref).abs().max().item() < 2e-2


# ---------------------------------------------------------------------------
# attention training: fused fwd (in-kernel dropout) + hand-written backward
# ---------------------------------------------------------------------------

@pytest.mark.parametrize(
    "B,H,Lq,Lk,D",
    [
        (2, 12, 38, 38, 64),     # text self
        (2, 8, 101, 101, 128),   # vision self
        (2, 8, 38, 101, 128),    # co-attn t->v (rectangular)
        (2, 8, 101, 38, 128),    # co-attn v->t
        (2, 4, 20, 20, 64),      # LQ_PAD=LK_PAD=32 (bwd image swizzle in-row)
    ],
)
def test_attention_train_bwd_matches_fp32_autograd(ext, B, H, Lq, Lk, D):
    from vilbert_multi_task_amd.ops.functional import _AttentionTrainFn

    torch.manual_seed(B * 10 + Lq)
    qf = torch.randn(B, Lq, H * D) * 0.5
    kf = torch.randn(B, Lk, H * D) * 0.5
    vf = torch.randn(B, Lk, H * D) * 0.5
    keep = torch.ones(B, Lk)
    keep[:, Lk - 4:] = 0
    mask = ((1 - keep) * -1e9).view(B, 1, 1, Lk)
    go = torch.randn(B, Lq, H * D) * 0.1

    # fp32 autograd reference
    q32 = qf.clone().requires_grad_()
    k32 = kf.clone().requires_grad_()
    v32 = vf.clone().requires_grad_()
    qh = q32.view(B, Lq, H, D).transpose(1, 2)
    kh = k32.view(B, Lk, H, D).transpose(1, 2)
    vh = v32.view(B, Lk, H, D).transpose(1, 2)
    s = qh @ kh.transpose(-1, -2) / math.sqrt(D) + mask
    p = torch.softmax(s, dim=-1)
    ref = (p @ vh).transpose(1, 2).reshape(B, Lq, H * D)
    ref.backward(go)

    # HIP path
    q16 = qf.to(torch.bfloat16).cuda().requires_grad_()
    k16 = kf.to(torch.bfloat16).cuda().requires_grad_()
    v16 = vf.to(torch.bfloat16).cuda().requires_grad_()
    m16 = mask.to(torch.bfloat16).cuda().view(B, 1, 1, Lk)
    out = _AttentionTrainFn.apply(q16, k16, v16, H, m16.contiguous(), 0.0)
    out.backward(go.to(torch.bfloat16).cuda())

    assert (out.float().cpu() - ref.detach()).abs().max().item() < 4e-2
    for hip, refg in ((q16.grad, q32.grad), (k16.grad, k32.grad), (v16.grad, v32.grad)):
        err = (hip.float().cpu() - refg).abs().max().item()
        scale = refg.abs().max().item()
        assert err < 0.04 * max(scale, 1.0), f"grad err {err} (scale {scale})"


def test_attention_train_dropout_consistency(ext):
    """With an EXPLICIT keep-scale mask, fwd ctx and all three grads must
    match the fp32 autograd of the same dropped-probs expression."""
    B, H, Lq, Lk, D = 2, 8, 38, 101, 128
    torch.manual_seed(77)
    qf = torch.randn(B, Lq, H * D) * 0.5
    kf = torch.randn(B, Lk, H * D) * 0.5
    vf = torch.randn(B, Lk, H * D) * 0.5
    dm_f = ((torch.rand(B, H, Lq, Lk) >= 0.1).float() / 0.9)
    dm = dm_f.to(torch.bfloat16)
    go = torch.randn(B, Lq, H * D) * 0.1

    q32 = qf.clone().requires_grad_()
    k32 = kf.clone().requires_grad_()
    v32 = vf.clone().requires_grad_()
    qh = q32.view(B, Lq, H, D).transpose(1, 2)
    kh = k32.view(B, Lk, H, D).transpose(1, 2)
    vh = v32.view(B, Lk, H, D).transpose(1, 2)
    p = torch.softmax(qh @ kh.transpose(-1, -2) / math.sqrt(D), dim=-1)
    ref = ((dm.float() * p) @ vh).transpose(1, 2).reshape(B, Lq, H * D)
    ref.backward(go)

    q16 = qf.to(torch.bfloat16).cuda()
    k16 = kf.to(torch.bfloat16).cuda()
    v16 = vf.to(torch.bfloat16).cuda()
    out, probs = torch.ops.vilbert_amd.attention_train_fwd(
        q16, k16, v16, H, None, dm.cuda().contiguous()
    )
    assert (out.float().cpu() - ref.detach()).abs().max().item() < 4e-2
    assert (probs.float().cpu() - p.detach()).abs().max().item() < 2e-2
    dq, dk, dv = torch.ops.vilbert_amd.attention_bwd(
        q16, k16, v16, probs, dm.cuda().contiguous(),
        go.to(torch.bfloat16).cuda(), H
    )
    for hip, refg in ((dq, q32.grad), (dk, k32.grad), (dv, v32.grad)):
        err = (hip.float().cpu() - refg).abs().max().item()
        assert err < 0.04 * max(refg.abs().max().item(), 1.0), err


# ---------------------------------------------------------------------------
# fused AdamW kernel vs the (CPU-tested) fallback math
# ---------------------------------------------------------------------------

def test_adamw_kernel_matches_reference(ext):
    n = 12345
    torch.manual_seed(5)
    base = torch.randn(n)
    p = base.to(torch.bfloat16).cuda()
    master = p.float()
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    # fp32 reference of the same recurrence
    rm, rv2, rp = torch.zeros(n), torch.zeros(n), master.cpu().clone()
    lr, b1, b2, eps, wd = 1e-3, 0.9, 0.999, 1e-8, 0.01
    for t in range(1, 6):
        torch.manual_seed(50 + t)
        g = (torch.randn(n) * 1e-2)
        torch.ops.vilbert_amd.adamw_step(
            p, g.to(torch.bfloat16).cuda(), master, m, v, lr, b1, b2, eps, wd, t
        )
        gf = g.to(torch.bfloat16).float()  # kernel sees the bf16 grad
        rm = b1 * rm + (1 - b1) * gf
        rv2 = b2 * rv2 + (1 - b2) * gf * gf
        rp -= lr * ((rm / (1 - b1 ** t)) / ((rv2 / (1 - b2 ** t)).sqrt() + eps) + wd * rp)
    assert (master.cpu() - rp).abs().max().item() < 1e-5
    assert (p.float().cpu() - rp.to(torch.bfloat16).float()).abs().max().item() == 0.0
