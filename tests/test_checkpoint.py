"""Checkpoint round-trip: our state dict -> upstream key layout -> back."""

import torch

from vilbert_multi_task_amd.models import VILBertForVLTasks
from vilbert_multi_task_amd.models.checkpoint import (
    export_upstream_state_dict,
    load_upstream_state_dict,
    translate_key,
)


def test_text_stream_key_translation():
    # pytorch_transformers BERT naming (fixed, not inferred)
    assert (
        translate_key("bert.encoder.layer.3.attention.self.query.weight")
        == "bert.t_layers.3.attention.query.weight"
    )
    assert (
        translate_key("bert.encoder.layer.0.output.LayerNorm.bias")
        == "bert.t_layers.0.ffn.layer_norm.bias"
    )
    assert (
        translate_key("bert.embeddings.LayerNorm.weight")
        == "bert.embeddings.layer_norm.weight"
    )


def test_roundtrip_through_upstream_layout(tiny_model):
    m = tiny_model
    up = export_upstream_state_dict(m)
    # upstream layout must not contain our internal names
    assert not any(".t_layers." in k or ".ffn." in k for k in up)
    assert any(k.startswith("bert.encoder.layer.") for k in up)
    assert any(k.startswith("bert.encoder.v_layer.") for k in up)
    assert any(k.startswith("bert.encoder.c_layer.") for k in up)

    # reload into a fresh model and compare every tensor
    torch.manual_seed(123)
    from vilbert_multi_task_amd.config import ViLBertConfig

    m2 = VILBertForVLTasks(ViLBertConfig.tiny())
    report = load_upstream_state_dict(m2, up)
    assert report["unexpected"] == [], report["unexpected"][:5]
    # tied decoder weight appears as missing-by-name only if mapping broke
    assert report["missing"] == [], report["missing"][:5]
    for (k1, v1), (k2, v2) in zip(m.state_dict().items(), m2.state_dict().items()):
        assert k1 == k2
        assert torch.equal(v1, v2), k1


def test_from_pretrained_file(tmp_path, tiny_model, tiny_config):
    path = tmp_path / "pytorch_model_9.bin"
    torch.save(export_upstream_state_dict(tiny_model), str(path))
    m2 = VILBertForVLTasks.from_pretrained(str(path), tiny_config)
    for (k1, v1), (k2, v2) in zip(
        tiny_model.state_dict().items(), m2.state_dict().items()
    ):
        assert torch.equal(v1, v2), k1


def test_convert_checkpoint_cli_roundtrip(tmp_path, tiny_config):
    """CLI to-upstream then to-native reproduces the weights exactly."""
    import json as _json
    import subprocess
    import sys

    from vilbert_multi_task_amd.models.heads import VILBertForVLTasks

    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(_json.dumps(tiny_config.to_dict()))
    torch.manual_seed(3)
    m = VILBertForVLTasks(tiny_config)
    native = tmp_path / "native.pt"
    torch.save(m.state_dict(), native)

    up = tmp_path / "pytorch_model_9.bin"
    back = tmp_path / "back.pt"
    for mode, src, dst in [("to-upstream", native, up), ("to-native", up, back)]:
        r = subprocess.run(
            [sys.executable, "scripts/convert_checkpoint.py", mode, str(src),
             str(dst), "--config", str(cfg_path)],
            capture_output=True, text=True, timeout=240,
        )
        assert r.returncode == 0, r.stderr[-500:]
    sd0 = m.state_dict()
    sd1 = torch.load(back, weights_only=True)
    for k in sd0:
        assert torch.equal(sd0[k], sd1[k]), k
