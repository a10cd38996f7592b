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


def _write_corpus(root, n=6, feat_dim=64, regions_src=8):
    """Tiny on-disk corpus fixture (data/corpus.py layout)."""
    import json
    import os

    import numpy as np

    os.makedirs(os.path.join(root, "features"), exist_ok=True)
    os.makedirs(os.path.join(root, "boxes"), exist_ok=True)
    rng = np.random.default_rng(3)
    with open(os.path.join(root, "captions.jsonl"), "w") as f:
        for i in range(n):
            f.write(json.dumps({
                "image_id": f"img{i}",
                "caption": f"a dog number {i} on the grass",
                "label": i % 3,
            }) + "\n")
            np.save(os.path.join(root, "features", f"img{i}.npy"),
                    rng.standard_normal((regions_src, feat_dim)).astype("float32"))
            np.save(os.path.join(root, "boxes", f"img{i}.npy"),
                    rng.uniform(0, 1, (regions_src, 4)).astype("float32"))
    return root


def test_conceptcap_loader_file_corpus(tmp_path, tiny_config):
    """The real-corpus backend feeds the SAME pretraining batch schema:
    tokens/features come from disk, the masking machinery is unchanged."""
    import numpy as np

    from vilbert_multi_task_amd.data.loaders import ConceptCapLoaderTrain

    root = _write_corpus(str(tmp_path), feat_dim=tiny_config.v_feature_size)
    loader = ConceptCapLoaderTrain(
        tiny_config, batch_size=3, num_batches=2, seq_len=20, regions=12,
        corpus=root,
    )
    batches = list(loader)
    assert len(batches) == 2
    b = batches[0]
    assert b["features"].shape == (3, 12, tiny_config.v_feature_size)
    assert b["question"].shape == (3, 20)
    # masking applied on the REAL tokens: [MASK]=103 appears where labels set
    masked = b["lm_labels"] >= 0
    assert (b["question"][masked] == 103).all()
    # global region = mean of the on-disk features for that image
    feats0 = np.load(str(tmp_path / "features" / "img0.npy"))
    means = torch.tensor(np.stack([
        np.load(str(tmp_path / "features" / f"img{i}.npy")).mean(0)
        for i in range(6)
    ]))
    g0 = b["features"][:, 0, :]
    # every row's global region equals one of the corpus images' mean
    for r in range(3):
        d = (means - g0[r]).abs().max(dim=1).values
        assert d.min() < 1e-5


def test_load_dataset_eval_file_corpus(tmp_path, tiny_config):
    from vilbert_multi_task_amd.data.loaders import LoadDatasetEval

    root = _write_corpus(str(tmp_path), feat_dim=tiny_config.v_feature_size)
    it = LoadDatasetEval(tiny_config, "snli_ve", batch_size=3, corpus=root)
    batch, targets = next(it)
    assert targets.tolist() == [0, 1, 2]          # jsonl labels, in order
    assert int(batch["task_tokens"][0, 0]) == 13  # snli_ve -> task 13
    assert batch["question"].shape == (3, 37)
    assert batch["features"].shape[0] == 3
