"""Property-based tests (hypothesis) for the serving-side invariants."""


import pytest
import torch

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st

from vilbert_multi_task_amd.data.tokenizer import (
    CLS_ID,
    MAX_SEQ_LENGTH,
    PAD_ID,
    SEP_ID,
    BertWordPieceTokenizer,
)
from vilbert_multi_task_amd.serve.broker import Broker
from vilbert_multi_task_amd.serve.features import tensorize_regions
from vilbert_multi_task_amd.tasks import TASKS, validate_request


@settings(max_examples=200, deadline=None)
@given(st.text(max_size=300))
def test_tokenizer_total_on_any_text(text):
    """Any unicode input yields a well-formed fixed-length encoding."""
    tok = BertWordPieceTokenizer()
    ids, mask, seg = tok.encode_for_serving(text)
    assert len(ids) == len(mask) == len(seg) == MAX_SEQ_LENGTH
    assert ids[0] == CLS_ID
    assert SEP_ID in ids
    sep = ids.index(SEP_ID)
    assert all(i == PAD_ID for i in ids[sep + 1 :])
    assert mask == [1] * (sep + 1) + [0] * (MAX_SEQ_LENGTH - sep - 1)
    assert all(0 <= i < tok.vocab_size for i in ids)


@settings(max_examples=50, deadline=None)
@given(
    st.integers(min_value=-3, max_value=30),
    st.integers(min_value=0, max_value=12),
)
def test_validate_request_never_crashes(task_id, n_images):
    err = validate_request(task_id, n_images)
    spec = TASKS.get(task_id)
    if spec is None or not spec.reachable:
        assert err is not None
    elif spec.min_images <= n_images <= spec.max_images:
        assert err is None
    else:
        assert err is not None


@settings(max_examples=20, deadline=None)
@given(
    st.lists(
        st.integers(min_value=1, max_value=10), min_size=1, max_size=5
    ),
    st.integers(min_value=0, max_value=2**31 - 1),
)
def test_tensorize_regions_invariants(num_boxes_list, seed):
    g = torch.Generator().manual_seed(seed)
    infos = []
    for nb in num_boxes_list:
        w, h = 640.0, 480.0
        bbox = torch.rand(nb, 4, generator=g) * torch.tensor([w, h, w, h])
        bbox[:, 2:] = torch.maximum(bbox[:, 2:], bbox[:, :2])
        infos.append(
            {
                "features": torch.randn(nb, 8, generator=g),
                "bbox": bbox,
                "image_width": w,
                "image_height": h,
                "num_boxes": nb,
            }
        )
    reg = tensorize_regions(infos, num_regions=11)
    n = len(infos)
    assert reg["features"].shape == (n, 11, 8)
    assert reg["spatials"].shape == (n, 11, 5)
    # normalized boxes and mask consistency
    assert (reg["spatials"][..., :4] >= -1e-5).all()
    assert (reg["spatials"][..., :4] <= 1.0 + 1e-5).all()
    for i, nb in enumerate(num_boxes_list):
        k = min(nb, 10)
        assert reg["image_mask"][i, : k + 1].all()
        assert not reg["image_mask"][i, k + 1 :].any()
        # global mean region equals mean of the real region features
        assert torch.allclose(
            reg["features"][i, 0], reg["features"][i, 1 : k + 1].mean(0), atol=1e-5
        )


@settings(max_examples=15, deadline=None)
@given(st.lists(st.sampled_from(["pub", "get", "ack", "nack"]), min_size=1, max_size=40))
def test_broker_no_message_lost_or_duplicated(ops):
    """Under any publish/lease/ack/nack interleaving: every published
    message is eventually either acked (gone) or still recoverable."""
    import tempfile, os

    with tempfile.TemporaryDirectory() as td:
        _run_broker_ops(os.path.join(td, "q.sqlite3"), ops)


def _run_broker_ops(path, ops):
    broker = Broker(path, lease_timeout_s=0.0, max_attempts=1000)
    # NOTE: with lease_timeout 0 a leased message is instantly redeliverable,
    # so `get` may hand out the same msg twice (at-least-once semantics) —
    # account by id set, not by count.
    ids = set()
    acked_ids = set()
    leased = []
    for op in ops:
        if op == "pub":
            ids.add(broker.publish({"n": len(ids)}))
        elif op == "get":
            leased.extend(broker.get(max_n=2))
        elif op == "ack" and leased:
            d = leased.pop(0)
            broker.ack(d.msg_id)
            acked_ids.add(d.msg_id)
        elif op == "nack" and leased:
            broker.nack(leased.pop(0).msg_id)
    # drain: everything not acked must still be deliverable
    seen = set()
    for _ in range(len(ids) + 5):
        for d in broker.get(max_n=10):
            seen.add(d.msg_id)
            broker.ack(d.msg_id)
    assert seen | acked_ids == ids  # nothing lost
    assert seen <= ids and acked_ids <= ids  # nothing invented


@given(
    rows=st.integers(min_value=1, max_value=5),
    dim=st.sampled_from([8, 24, 52, 64]),
    has_res=st.booleans(),
    seed=st.integers(min_value=0, max_value=2**16),
)
@settings(max_examples=30, deadline=None)
def test_layer_norm_oracle_matches_torch(rows, dim, has_res, seed):
    """The CPU dispatch path of F_ops.layer_norm (the numerics oracle every
    HIP kernel is tested against) must equal torch layer_norm exactly."""
    import torch

    from vilbert_multi_task_amd.ops import functional as F_ops

    g = torch.Generator().manual_seed(seed)
    x = torch.randn(rows, dim, generator=g)
    res = torch.randn(rows, dim, generator=g) if has_res else None
    w = torch.randn(dim, generator=g)
    b = torch.randn(dim, generator=g)
    got = F_ops.layer_norm(x, w, b, 1e-12, residual=res)
    ref = torch.nn.functional.layer_norm(
        x + res if has_res else x, (dim,), w, b, 1e-12
    )
    assert torch.allclose(got, ref, atol=1e-6), (got - ref).abs().max()


@given(
    m=st.integers(min_value=1, max_value=6),
    k=st.sampled_from([4, 16, 32]),
    n=st.sampled_from([3, 8, 17]),
    seed=st.integers(min_value=0, max_value=2**16),
)
@settings(max_examples=30, deadline=None)
def test_linear_bias_gelu_oracle_matches_torch(m, k, n, seed):
    import torch

    from vilbert_multi_task_amd.ops import functional as F_ops

    g = torch.Generator().manual_seed(seed)
    x = torch.randn(m, k, generator=g)
    w = torch.randn(n, k, generator=g)
    b = torch.randn(n, generator=g)
    got = F_ops.linear_bias_gelu(x, w, b)
    ref = torch.nn.functional.gelu(torch.nn.functional.linear(x, w, b))
    assert torch.allclose(got, ref, atol=1e-5), (got - ref).abs().max()
