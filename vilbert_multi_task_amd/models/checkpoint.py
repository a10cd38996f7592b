"""Upstream checkpoint compatibility.

The reference loads ``save/multitask_model/pytorch_model_9.bin`` through
``VILBertForVLTasks.from_pretrained`` (/root/reference/worker.py:470,530-532).
That file is a plain PyTorch state dict whose text-stream keys follow the
pytorch_transformers BERT naming and whose vision/co-attention keys follow the
upstream ``vilbert`` package naming (SURVEY.md §2.2 — [dependency — inferred]:
the package is not vendored in the reference checkout, so the mapping below is
pinned by the call-site contract and round-trip tested, not copied).

``load_upstream_state_dict`` translates upstream key names to this package's
module tree; ``export_upstream_state_dict`` is the inverse, so training
checkpoints produced here round-trip to the upstream layout
(SURVEY.md §5 checkpoint/resume obligation).
"""

from __future__ import annotations

import re
from typing import Dict, List, Tuple

import torch
import torch.nn as nn

# (upstream regex, ours template) — applied in order, first match wins.
_RULES: List[Tuple[str, str]] = [
    # ---- text embeddings (pytorch_transformers BERT naming) ----
    (r"^bert\.embeddings\.word_embeddings\.(.*)$", r"bert.embeddings.word_embeddings.\1"),
    (r"^bert\.embeddings\.position_embeddings\.(.*)$", r"bert.embeddings.position_embeddings.\1"),
    (r"^bert\.embeddings\.token_type_embeddings\.(.*)$", r"bert.embeddings.token_type_embeddings.\1"),
    (r"^bert\.embeddings\.task_embeddings\.(.*)$", r"bert.embeddings.task_embeddings.\1"),
    (r"^bert\.embeddings\.LayerNorm\.(.*)$", r"bert.embeddings.layer_norm.\1"),
    # ---- vision embeddings ----
    (r"^bert\.v_embeddings\.image_embeddings\.(.*)$", r"bert.v_embeddings.image_embeddings.\1"),
    (r"^bert\.v_embeddings\.image_location_embeddings\.(.*)$", r"bert.v_embeddings.image_location_embeddings.\1"),
    (r"^bert\.v_embeddings\.LayerNorm\.(.*)$", r"bert.v_embeddings.layer_norm.\1"),
    # ---- text self-attention layers ----
    (r"^bert\.encoder\.layer\.(\d+)\.attention\.self\.(query|key|value)\.(.*)$", r"bert.t_layers.\1.attention.\2.\3"),
    (r"^bert\.encoder\.layer\.(\d+)\.attention\.output\.dense\.(.*)$", r"bert.t_layers.\1.attention.out.\2"),
    (r"^bert\.encoder\.layer\.(\d+)\.attention\.output\.LayerNorm\.(.*)$", r"bert.t_layers.\1.attention.layer_norm.\2"),
    (r"^bert\.encoder\.layer\.(\d+)\.intermediate\.dense\.(.*)$", r"bert.t_layers.\1.ffn.intermediate.\2"),
    (r"^bert\.encoder\.layer\.(\d+)\.output\.dense\.(.*)$", r"bert.t_layers.\1.ffn.output.\2"),
    (r"^bert\.encoder\.layer\.(\d+)\.output\.LayerNorm\.(.*)$", r"bert.t_layers.\1.ffn.layer_norm.\2"),
    # ---- vision self-attention layers ----
    (r"^bert\.encoder\.v_layer\.(\d+)\.attention\.self\.(query|key|value)\.(.*)$", r"bert.v_layers.\1.attention.\2.\3"),
    (r"^bert\.encoder\.v_layer\.(\d+)\.attention\.output\.dense\.(.*)$", r"bert.v_layers.\1.attention.out.\2"),
    (r"^bert\.encoder\.v_layer\.(\d+)\.attention\.output\.LayerNorm\.(.*)$", r"bert.v_layers.\1.attention.layer_norm.\2"),
    (r"^bert\.encoder\.v_layer\.(\d+)\.intermediate\.dense\.(.*)$", r"bert.v_layers.\1.ffn.intermediate.\2"),
    (r"^bert\.encoder\.v_layer\.(\d+)\.output\.dense\.(.*)$", r"bert.v_layers.\1.ffn.output.\2"),
    (r"^bert\.encoder\.v_layer\.(\d+)\.output\.LayerNorm\.(.*)$", r"bert.v_layers.\1.ffn.layer_norm.\2"),
    # ---- co-attention ("connect") layers ----
    # Upstream BertBiAttention convention: stream 1 = vision, stream 2 = text.
    # query1 (vision) attends key2/value2 (text); query2 (text) attends
    # key1/value1 (vision).
    (r"^bert\.encoder\.c_layer\.(\d+)\.biattention\.query1\.(.*)$", r"bert.c_layers.\1.v_cross.query.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.biattention\.key1\.(.*)$", r"bert.c_layers.\1.t_cross.key.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.biattention\.value1\.(.*)$", r"bert.c_layers.\1.t_cross.value.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.biattention\.query2\.(.*)$", r"bert.c_layers.\1.t_cross.query.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.biattention\.key2\.(.*)$", r"bert.c_layers.\1.v_cross.key.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.biattention\.value2\.(.*)$", r"bert.c_layers.\1.v_cross.value.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.biOutput\.dense1\.(.*)$", r"bert.c_layers.\1.v_cross.out.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.biOutput\.LayerNorm1\.(.*)$", r"bert.c_layers.\1.v_cross.layer_norm.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.biOutput\.dense2\.(.*)$", r"bert.c_layers.\1.t_cross.out.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.biOutput\.LayerNorm2\.(.*)$", r"bert.c_layers.\1.t_cross.layer_norm.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.v_intermediate\.dense\.(.*)$", r"bert.c_layers.\1.v_ffn.intermediate.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.v_output\.dense\.(.*)$", r"bert.c_layers.\1.v_ffn.output.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.v_output\.LayerNorm\.(.*)$", r"bert.c_layers.\1.v_ffn.layer_norm.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.t_intermediate\.dense\.(.*)$", r"bert.c_layers.\1.t_ffn.intermediate.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.t_output\.dense\.(.*)$", r"bert.c_layers.\1.t_ffn.output.\2"),
    (r"^bert\.encoder\.c_layer\.(\d+)\.t_output\.LayerNorm\.(.*)$", r"bert.c_layers.\1.t_ffn.layer_norm.\2"),
    # ---- poolers ----
    (r"^bert\.t_pooler\.dense\.(.*)$", r"bert.t_pooler.dense.\1"),
    (r"^bert\.v_pooler\.dense\.(.*)$", r"bert.v_pooler.dense.\1"),
    # ---- heads ----
    # SimpleClassifier upstream is logit_fc = Sequential(Linear, GELU, LayerNorm, Linear)
    (r"^vil_prediction\.logit_fc\.0\.(.*)$", r"vil_prediction.dense.\1"),
    (r"^vil_prediction\.logit_fc\.2\.(.*)$", r"vil_prediction.layer_norm.\1"),
    (r"^vil_prediction\.logit_fc\.3\.(.*)$", r"vil_prediction.decoder.\1"),
    (r"^vil_prediction_gqa\.logit_fc\.0\.(.*)$", r"vil_prediction_gqa.dense.\1"),
    (r"^vil_prediction_gqa\.logit_fc\.2\.(.*)$", r"vil_prediction_gqa.layer_norm.\1"),
    (r"^vil_prediction_gqa\.logit_fc\.3\.(.*)$", r"vil_prediction_gqa.decoder.\1"),
    (r"^vil_binary_prediction\.logit_fc\.0\.(.*)$", r"vil_binary_prediction.dense.\1"),
    (r"^vil_binary_prediction\.logit_fc\.2\.(.*)$", r"vil_binary_prediction.layer_norm.\1"),
    (r"^vil_binary_prediction\.logit_fc\.3\.(.*)$", r"vil_binary_prediction.decoder.\1"),
    (r"^vil_logit\.(.*)$", r"vil_logit.\1"),
    (r"^vil_tri_prediction\.(.*)$", r"vil_tri_prediction.\1"),
    (r"^vision_prediction\.(.*)$", r"vision_prediction.\1"),
    (r"^vision_logit\.(.*)$", r"vision_logit.\1"),
    (r"^linguisic_prediction\.(.*)$", r"linguistic_prediction.\1"),
    (r"^linguisic_logit\.(.*)$", r"linguistic_logit.\1"),
    # ---- masked-LM cls head (pretraining layout) ----
    (r"^cls\.predictions\.transform\.dense\.(.*)$", r"linguistic_prediction.transform.\1"),
    (r"^cls\.predictions\.transform\.LayerNorm\.(.*)$", r"linguistic_prediction.layer_norm.\1"),
    (r"^cls\.predictions\.bias$", r"linguistic_prediction.decoder_bias"),
]

_COMPILED = [(re.compile(p), t) for p, t in _RULES]


def translate_key(upstream_key: str) -> str:
    for pat, tmpl in _COMPILED:
        if pat.match(upstream_key):
            return pat.sub(tmpl, upstream_key)
    return upstream_key  # pass through (our own native checkpoints)


_INVERSE = [(re.compile(t.replace("\\1", "(\\d+|weight|bias)").replace("\\2", "(.*)").replace("\\3", "(.*)")), p) for p, t in _RULES]


def load_upstream_state_dict(
    model: nn.Module, sd: Dict[str, torch.Tensor], strict: bool = False
) -> Dict[str, List[str]]:
    """Translate + load; returns {'missing': [...], 'unexpected': [...]}."""
    translated = {translate_key(k): v for k, v in sd.items()}
    own = model.state_dict()
    filtered = {}
    unexpected = []
    for k, v in translated.items():
        if k in own and own[k].shape == v.shape:
            filtered[k] = v
        else:
            unexpected.append(k)
    missing = [k for k in own if k not in filtered]
    model.load_state_dict(filtered, strict=False)
    if strict and (missing or unexpected):
        raise RuntimeError(
            f"checkpoint mismatch: missing={missing[:10]}... unexpected={unexpected[:10]}..."
        )
    return {"missing": missing, "unexpected": unexpected}


def export_upstream_state_dict(model: nn.Module) -> Dict[str, torch.Tensor]:
    """Inverse mapping: our state dict -> upstream key names (round-trip)."""
    inverse: Dict[str, str] = {}
    own = model.state_dict()
    # build inverse by forward-translating every possible upstream name is
    # impractical; instead invert rule-by-rule on our keys.
    out: Dict[str, torch.Tensor] = {}
    for ours, tensor in own.items():
        up = _ours_to_upstream(ours)
        out[up] = tensor
    return out


def _ours_to_upstream(key: str) -> str:
    subs = [
        (r"^bert\.embeddings\.layer_norm\.(.*)$", r"bert.embeddings.LayerNorm.\1"),
        (r"^bert\.v_embeddings\.layer_norm\.(.*)$", r"bert.v_embeddings.LayerNorm.\1"),
        (r"^bert\.t_layers\.(\d+)\.attention\.(query|key|value)\.(.*)$", r"bert.encoder.layer.\1.attention.self.\2.\3"),
        (r"^bert\.t_layers\.(\d+)\.attention\.out\.(.*)$", r"bert.encoder.layer.\1.attention.output.dense.\2"),
        (r"^bert\.t_layers\.(\d+)\.attention\.layer_norm\.(.*)$", r"bert.encoder.layer.\1.attention.output.LayerNorm.\2"),
        (r"^bert\.t_layers\.(\d+)\.ffn\.intermediate\.(.*)$", r"bert.encoder.layer.\1.intermediate.dense.\2"),
        (r"^bert\.t_layers\.(\d+)\.ffn\.output\.(.*)$", r"bert.encoder.layer.\1.output.dense.\2"),
        (r"^bert\.t_layers\.(\d+)\.ffn\.layer_norm\.(.*)$", r"bert.encoder.layer.\1.output.LayerNorm.\2"),
        (r"^bert\.v_layers\.(\d+)\.attention\.(query|key|value)\.(.*)$", r"bert.encoder.v_layer.\1.attention.self.\2.\3"),
        (r"^bert\.v_layers\.(\d+)\.attention\.out\.(.*)$", r"bert.encoder.v_layer.\1.attention.output.dense.\2"),
        (r"^bert\.v_layers\.(\d+)\.attention\.layer_norm\.(.*)$", r"bert.encoder.v_layer.\1.attention.output.LayerNorm.\2"),
        (r"^bert\.v_layers\.(\d+)\.ffn\.intermediate\.(.*)$", r"bert.encoder.v_layer.\1.intermediate.dense.\2"),
        (r"^bert\.v_layers\.(\d+)\.ffn\.output\.(.*)$", r"bert.encoder.v_layer.\1.output.dense.\2"),
        (r"^bert\.v_layers\.(\d+)\.ffn\.layer_norm\.(.*)$", r"bert.encoder.v_layer.\1.output.LayerNorm.\2"),
        (r"^bert\.c_layers\.(\d+)\.v_cross\.query\.(.*)$", r"bert.encoder.c_layer.\1.biattention.query1.\2"),
        (r"^bert\.c_layers\.(\d+)\.t_cross\.key\.(.*)$", r"bert.encoder.c_layer.\1.biattention.key1.\2"),
        (r"^bert\.c_layers\.(\d+)\.t_cross\.value\.(.*)$", r"bert.encoder.c_layer.\1.biattention.value1.\2"),
        (r"^bert\.c_layers\.(\d+)\.t_cross\.query\.(.*)$", r"bert.encoder.c_layer.\1.biattention.query2.\2"),
        (r"^bert\.c_layers\.(\d+)\.v_cross\.key\.(.*)$", r"bert.encoder.c_layer.\1.biattention.key2.\2"),
        (r"^bert\.c_layers\.(\d+)\.v_cross\.value\.(.*)$", r"bert.encoder.c_layer.\1.biattention.value2.\2"),
        (r"^bert\.c_layers\.(\d+)\.v_cross\.out\.(.*)$", r"bert.encoder.c_layer.\1.biOutput.dense1.\2"),
        (r"^bert\.c_layers\.(\d+)\.v_cross\.layer_norm\.(.*)$", r"bert.encoder.c_layer.\1.biOutput.LayerNorm1.\2"),
        (r"^bert\.c_layers\.(\d+)\.t_cross\.out\.(.*)$", r"bert.encoder.c_layer.\1.biOutput.dense2.\2"),
        (r"^bert\.c_layers\.(\d+)\.t_cross\.layer_norm\.(.*)$", r"bert.encoder.c_layer.\1.biOutput.LayerNorm2.\2"),
        (r"^bert\.c_layers\.(\d+)\.v_ffn\.intermediate\.(.*)$", r"bert.encoder.c_layer.\1.v_intermediate.dense.\2"),
        (r"^bert\.c_layers\.(\d+)\.v_ffn\.output\.(.*)$", r"bert.encoder.c_layer.\1.v_output.dense.\2"),
        (r"^bert\.c_layers\.(\d+)\.v_ffn\.layer_norm\.(.*)$", r"bert.encoder.c_layer.\1.v_output.LayerNorm.\2"),
        (r"^bert\.c_layers\.(\d+)\.t_ffn\.intermediate\.(.*)$", r"bert.encoder.c_layer.\1.t_intermediate.dense.\2"),
        (r"^bert\.c_layers\.(\d+)\.t_ffn\.output\.(.*)$", r"bert.encoder.c_layer.\1.t_output.dense.\2"),
        (r"^bert\.c_layers\.(\d+)\.t_ffn\.layer_norm\.(.*)$", r"bert.encoder.c_layer.\1.t_output.LayerNorm.\2"),
        (r"^(vil_prediction|vil_prediction_gqa|vil_binary_prediction)\.dense\.(.*)$", r"\1.logit_fc.0.\2"),
        (r"^(vil_prediction|vil_prediction_gqa|vil_binary_prediction)\.layer_norm\.(.*)$", r"\1.logit_fc.2.\2"),
        (r"^(vil_prediction|vil_prediction_gqa|vil_binary_prediction)\.decoder\.(.*)$", r"\1.logit_fc.3.\2"),
        (r"^linguistic_prediction\.transform\.(.*)$", r"cls.predictions.transform.dense.\1"),
        (r"^linguistic_prediction\.layer_norm\.(.*)$", r"cls.predictions.transform.LayerNorm.\1"),
        (r"^linguistic_prediction\.decoder_bias$", r"cls.predictions.bias"),
        (r"^linguistic_logit\.(.*)$", r"linguisic_logit.\1"),
    ]
    for p, t in subs:
        if re.match(p, key):
            return re.sub(p, t, key)
    return key
