"""Pretraining wrapper: masked-LM + masked-region + cross-modality alignment.

The reference imports ``BertForMultiModalPreTraining`` alongside
``VILBertForVLTasks`` (/root/reference/worker.py:45) — the class the 12-in-1
model is initialized from (Conceptual Captions pretraining). Heads:
  - masked LM over text (linguistic_prediction head, tied embeddings)
  - masked region classification over 1601 detector classes
    (vision_prediction head; soft targets = detector cls_prob, KL-div)
  - image-text alignment (binary head on the fused pooled representation)

Also provides ``BaseBertForVLTasks`` — the single-stream baseline branch the
worker can select (worker.py:525-528): same head surface, text-only trunk
with region features projected into the text stream.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..config import ViLBertConfig
from .heads import VILBertForVLTasks


class BertForMultiModalPreTraining(nn.Module):
    def __init__(self, config: ViLBertConfig):
        super().__init__()
        self.config = config
        self.model = VILBertForVLTasks(config)
        self.alignment = nn.Linear(config.bi_hidden_size, 2)

    @property
    def bert(self):
        return self.model.bert

    def forward(
        self,
        input_ids: torch.Tensor,
        features: torch.Tensor,
        spatials: torch.Tensor,
        segment_ids: torch.Tensor,
        input_mask: torch.Tensor,
        image_mask: torch.Tensor,
        lm_labels: Optional[torch.Tensor] = None,       # [B,T] (-1 = unmasked)
        region_targets: Optional[torch.Tensor] = None,  # [B,R,1601] soft labels
        region_mask: Optional[torch.Tensor] = None,     # [B,R] masked regions
        alignment_labels: Optional[torch.Tensor] = None,  # [B] 1 = aligned
    ):
        t, v, pooled_t, pooled_v, _ = self.model.bert(
            input_ids, features, spatials, segment_ids, input_mask, image_mask,
            None, None, False,
        )
        lm_logits = self.model.linguistic_prediction(t)
        region_logits = self.model.vision_prediction(v)
        align_logits = self.alignment(pooled_t * pooled_v)

        losses = {}
        if lm_labels is not None:
            losses["masked_lm"] = F.cross_entropy(
                lm_logits.reshape(-1, lm_logits.shape[-1]).float(),
                lm_labels.reshape(-1),
                ignore_index=-1,
            )
        if region_targets is not None and region_mask is not None:
            logp = F.log_softmax(region_logits.float(), dim=-1)
            kl = -(region_targets * logp).sum(-1)
            denom = region_mask.sum().clamp(min=1)
            losses["masked_region"] = (kl * region_mask).sum() / denom
        if alignment_labels is not None:
            losses["alignment"] = F.cross_entropy(
                align_logits.float(), alignment_labels
            )
        return lm_logits, region_logits, align_logits, losses


class BaseBertForVLTasks(nn.Module):
    """Single-stream baseline (worker.py:525-528 alternative branch): region
    features projected to the text hidden size and concatenated as extra
    tokens; same 10-output surface as VILBertForVLTasks."""

    def __init__(self, config: ViLBertConfig):
        super().__init__()
        c = config
        self.config = c
        from .vilbert import FusedLayerNorm, TextEmbeddings, TransformerLayer, Pooler

        self.embeddings = TextEmbeddings(c)
        self.v_proj = nn.Linear(c.v_feature_size, c.hidden_size)
        self.v_loc_proj = nn.Linear(c.v_loc_size, c.hidden_size)
        self.v_ln = FusedLayerNorm(c.hidden_size, c.layer_norm_eps)
        self.layers = nn.ModuleList(
            [
                TransformerLayer(
                    c.hidden_size, c.num_attention_heads, c.intermediate_size,
                    c.attention_probs_dropout_prob, c.hidden_dropout_prob,
                    c.layer_norm_eps,
                )
                for _ in range(c.num_hidden_layers)
            ]
        )
        self.pooler = Pooler(c.hidden_size, c.bi_hidden_size)
        bi = c.bi_hidden_size
        from .heads import SimpleClassifier, LMHead

        self.vil_prediction = SimpleClassifier(bi, c.num_labels_vqa, c.layer_norm_eps)
        self.vil_prediction_gqa = SimpleClassifier(bi, c.num_labels_gqa, c.layer_norm_eps)
        self.vil_logit = nn.Linear(bi, 1)
        self.vil_binary_prediction = SimpleClassifier(bi * 2, 2, c.layer_norm_eps)
        self.vil_tri_prediction = nn.Linear(bi, 3)
        self.vision_prediction = nn.Linear(c.hidden_size, c.v_target_size)
        self.vision_logit = nn.Linear(c.hidden_size, 1)
        self.linguistic_prediction = LMHead(c, self.embeddings.word_embeddings.weight)
        self.linguistic_logit = nn.Linear(c.hidden_size, 1)

    def forward(
        self, input_ids, features, spatials, segment_ids, input_mask, image_mask,
        co_attention_mask=None, task_ids=None, output_all_attention_masks=False,
    ):
        dtype = self.pooler.dense.weight.dtype
        t = self.embeddings(input_ids, segment_ids, task_ids)
        v = self.v_ln(self.v_proj(features.to(dtype)), residual=self.v_loc_proj(spatials.to(dtype)))
        if self.config.task_specific_tokens and task_ids is not None:
            one = torch.ones(input_mask.shape[0], 1, dtype=input_mask.dtype, device=input_mask.device)
            input_mask = torch.cat([input_mask[:, :1], one, input_mask[:, 1:]], dim=1)
        seq = torch.cat([t, v], dim=1)
        mask = torch.cat([input_mask, image_mask], dim=1)
        bias = (1.0 - mask[:, None, None, :].to(dtype)) * torch.finfo(dtype).min / 2
        attn = []
        for layer in self.layers:
            seq, p = layer(seq, bias, output_all_attention_masks)
            if output_all_attention_masks:
                attn.append({"type": "joint_self", "probs": p})
        t_len = t.shape[1]
        t_out, v_out = seq[:, :t_len], seq[:, t_len:]
        pooled = self.pooler(seq)
        fused = pooled
        b = fused.shape[0]
        if b >= 2:
            binary = self.vil_binary_prediction(fused[: 2 * (b // 2)].reshape(b // 2, -1))
        else:
            binary = fused.new_zeros(1, 2)
        return (
            self.vil_prediction(fused),
            self.vil_prediction_gqa(fused),
            self.vil_logit(fused),
            binary,
            self.vil_tri_prediction(fused),
            self.vision_prediction(v_out),
            self.vision_logit(v_out),
            self.linguistic_prediction(t_out),
            self.linguistic_logit(t_out),
            attn,
        )
