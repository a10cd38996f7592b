"""CPU tests of the 10-output forward contract (BASELINE.json config 1:
tiny 2-layer ViLBERT VQA forward on CPU)."""

import torch

from vilbert_multi_task_amd.config import ViLBertConfig
from vilbert_multi_task_amd.data.synthetic import forward_args, synthetic_batch
from vilbert_multi_task_amd.models import VILBertForVLTasks


def _tiny_batch(cfg, batch=2, seq=20, regions=36, task_id=1, seed=0):
    return synthetic_batch(
        batch,
        seq_len=seq,
        regions=regions,
        feat_dim=cfg.v_feature_size,
        vocab_size=cfg.vocab_size,
        task_id=task_id,
        seed=seed,
    )


def test_forward_ten_outputs(tiny_model, tiny_config):
    cfg = tiny_config
    batch = _tiny_batch(cfg)
    with torch.no_grad():
        out = tiny_model(*forward_args(batch))
    assert isinstance(out, tuple) and len(out) == 10
    (
        vil_prediction,
        vil_prediction_gqa,
        vil_logit,
        vil_binary_prediction,
        vil_tri_prediction,
        vision_prediction,
        vision_logit,
        linguisic_prediction,
        linguisic_logit,
        attn_data_list,
    ) = out
    B, R, T = 2, 36, 20
    T_task = T + 1  # task token inserted after [CLS]
    assert vil_prediction.shape == (B, cfg.num_labels_vqa)
    assert vil_prediction_gqa.shape == (B, cfg.num_labels_gqa)
    assert vil_logit.shape == (B, 1)
    assert vil_binary_prediction.shape == (B // 2, 2)
    assert vil_tri_prediction.shape == (B, 3)
    assert vision_prediction.shape == (B, R, cfg.v_target_size)
    assert vision_logit.shape == (B, R, 1)
    assert linguisic_prediction.shape == (B, T_task, cfg.vocab_size)
    assert linguisic_logit.shape == (B, T_task, 1)
    assert isinstance(attn_data_list, list) and len(attn_data_list) == 0
    for t in out[:9]:
        assert torch.isfinite(t).all()


def test_forward_attention_maps(tiny_model, tiny_config):
    batch = _tiny_batch(tiny_config)
    with torch.no_grad():
        out = tiny_model(*forward_args(batch, output_all_attention_masks=True))
    attn = out[9]
    cfg = tiny_config
    n_expected = cfg.num_hidden_layers + cfg.v_num_hidden_layers + len(cfg.t_biattention_id)
    assert len(attn) == n_expected
    co = [a for a in attn if a["type"] == "co"]
    assert len(co) == len(cfg.t_biattention_id)
    # co-attention prob shapes: text-queries x vision-keys and vice versa
    p_tv = co[0]["probs_tv"]
    p_vt = co[0]["probs_vt"]
    assert p_tv.shape[-1] == 36 and p_vt.shape[-1] == 21
    assert torch.allclose(p_tv.sum(-1), torch.ones_like(p_tv.sum(-1)), atol=1e-5)


def test_forward_deterministic_eval(tiny_model, tiny_config):
    batch = _tiny_batch(tiny_config)
    with torch.no_grad():
        a = tiny_model(*forward_args(batch))
        b = tiny_model(*forward_args(batch))
    for x, y in zip(a[:9], b[:9]):
        assert torch.equal(x, y)


def test_input_mask_respected(tiny_model, tiny_config):
    """Masked-out text tokens must not change vision-side outputs."""
    cfg = tiny_config
    batch = _tiny_batch(cfg)
    batch["input_mask"][:, 10:] = 0
    with torch.no_grad():
        base = tiny_model(*forward_args(batch))
        batch2 = {k: v.clone() for k, v in batch.items()}
        batch2["question"][:, 10:] = 777  # perturb only masked positions
        pert = tiny_model(*forward_args(batch2))
    assert torch.allclose(base[6], pert[6], atol=1e-5)  # vision_logit
    assert torch.allclose(base[0], pert[0], atol=1e-5)  # vil_prediction


def test_no_task_token_path(tiny_config):
    cfg = ViLBertConfig.tiny()
    cfg.task_specific_tokens = False
    torch.manual_seed(0)
    m = VILBertForVLTasks(cfg)
    m.eval()
    batch = _tiny_batch(cfg)
    with torch.no_grad():
        out = m(
            batch["question"], batch["features"], batch["spatials"],
            batch["segment_ids"], batch["input_mask"], batch["image_mask"],
            batch["co_attention_mask"], None, False,
        )
    assert out[7].shape[1] == 20  # no inserted token


def test_retrieval_batch_replication(tiny_model, tiny_config):
    """Task 7: same text replicated across num_images candidates
    (worker.py:278-284)."""
    cfg = tiny_config
    batch = _tiny_batch(cfg, batch=4, task_id=7)
    with torch.no_grad():
        out = tiny_model(*forward_args(batch))
    assert out[2].shape == (4, 1)  # per-pair retrieval logits


def test_grad_flows(tiny_config):
    torch.manual_seed(0)
    m = VILBertForVLTasks(tiny_config)
    m.train()
    batch = _tiny_batch(tiny_config)
    out = m(*forward_args(batch))
    loss = out[0].square().mean() + out[6].square().mean()
    loss.backward()
    grads = [p.grad for p in m.parameters() if p.grad is not None]
    assert len(grads) > 0
    assert m.bert.embeddings.word_embeddings.weight.grad is not None
    assert m.bert.c_layers[0].t_cross.query.weight.grad is not None


def test_base_model_param_count():
    """The 12-in-1 model is ~270M params (README.md:4 of the reference)."""
    cfg = ViLBertConfig.base_12in1()
    m = VILBertForVLTasks(cfg)
    n = sum(p.numel() for p in m.parameters())
    assert 200e6 < n < 340e6, f"param count {n/1e6:.1f}M out of expected range"


def test_prepare_for_serving_matches_eager(tiny_model, tiny_config):
    batch = _tiny_batch(tiny_config)
    with torch.no_grad():
        base = tiny_model(*forward_args(batch))
    tiny_model.prepare_for_serving()
    with torch.no_grad():
        fused = tiny_model(*forward_args(batch))
    for i in range(9):
        assert torch.allclose(base[i], fused[i], atol=1e-5), i


def test_dynamic_attention_flag_rejected(tiny_config):
    import dataclasses

    import pytest as _pytest

    from vilbert_multi_task_amd.models.vilbert import ViLBertModel

    cfg = dataclasses.replace(tiny_config, dynamic_attention=True)
    with _pytest.raises(NotImplementedError):
        ViLBertModel(cfg)
