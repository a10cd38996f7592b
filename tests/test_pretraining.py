"""Pretraining wrapper + data-layer API tests (worker.py:44-46 import pins)."""

import torch

from vilbert_multi_task_amd.data import (
    ConceptCapLoaderTrain,
    ConceptCapLoaderVal,
    LoadDatasetEval,
)
from vilbert_multi_task_amd.models import (
    BaseBertForVLTasks,
    BertForMultiModalPreTraining,
)


def test_pretraining_losses_and_backward(tiny_config):
    torch.manual_seed(0)
    m = BertForMultiModalPreTraining(tiny_config)
    loader = ConceptCapLoaderTrain(tiny_config, batch_size=2, num_batches=2, seq_len=20, regions=12)
    batch = next(iter(loader))
    lm, reg, align, losses = m(
        batch["question"], batch["features"], batch["spatials"],
        batch["segment_ids"], batch["input_mask"], batch["image_mask"],
        lm_labels=batch["lm_labels"], region_targets=batch["region_targets"],
        region_mask=batch["region_mask"], alignment_labels=batch["alignment_labels"],
    )
    assert set(losses) == {"masked_lm", "masked_region", "alignment"}
    total = sum(losses.values())
    assert torch.isfinite(total)
    total.backward()
    assert m.model.bert.embeddings.word_embeddings.weight.grad is not None


def test_conceptcap_loader_deterministic(tiny_config):
    a = list(ConceptCapLoaderTrain(tiny_config, batch_size=2, num_batches=2, seq_len=20, regions=12))
    b = list(ConceptCapLoaderTrain(tiny_config, batch_size=2, num_batches=2, seq_len=20, regions=12))
    assert torch.equal(a[0]["question"], b[0]["question"])
    v = next(iter(ConceptCapLoaderVal(tiny_config, batch_size=2, num_batches=1, seq_len=20, regions=12)))
    assert not torch.equal(a[0]["question"], v["question"])  # different split seed


def test_load_dataset_eval(tiny_config):
    it = LoadDatasetEval(tiny_config, "snli_ve", batch_size=4, num_batches=2)
    batch, targets = next(it)
    assert batch["question"].shape[0] == 4
    assert targets.shape == (4,)


def test_base_bert_single_stream_ten_outputs(tiny_config):
    torch.manual_seed(0)
    m = BaseBertForVLTasks(tiny_config).eval()
    from vilbert_multi_task_amd.data.synthetic import forward_args, synthetic_batch

    batch = synthetic_batch(
        2, seq_len=20, regions=12, feat_dim=tiny_config.v_feature_size,
        vocab_size=tiny_config.vocab_size,
    )
    with torch.no_grad():
        out = m(*forward_args(batch))
    assert len(out) == 10
    assert out[0].shape == (2, tiny_config.num_labels_vqa)
    assert out[5].shape == (2, 12, tiny_config.v_target_size)
    assert out[7].shape[1] == 21  # task token inserted


def test_pretrain_script_smoke(tmp_path):
    import subprocess
    import sys

    r = subprocess.run(
        [sys.executable, "scripts/pretrain.py", "--tiny", "--steps", "2",
         "--batch", "2", "--log-every", "1", "--save-every", "2",
         "--checkpoint", str(tmp_path / "ck.bin")],
        capture_output=True, text=True, timeout=240,
    )
    assert r.returncode == 0, r.stderr[-800:]
    assert '"event": "pretrain"' in r.stdout
    assert (tmp_path / "ck.bin").exists()


def test_evaluate_script_smoke():
    import subprocess
    import sys

    r = subprocess.run(
        [sys.executable, "scripts/evaluate.py", "--tiny", "--datasets",
         "vqa_v2", "snli_ve", "--batches", "1", "--batch", "4"],
        capture_output=True, text=True, timeout=240,
    )
    assert r.returncode == 0, r.stderr[-800:]
    assert '"event": "eval_summary"' in r.stdout
