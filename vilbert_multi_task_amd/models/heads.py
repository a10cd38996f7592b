"""Task heads and the 10-output `VILBertForVLTasks` wrapper.

Output tuple order and head dimensions are fixed by the reference unpack at
/root/reference/worker.py:287-289 and the decode sites:
  vil_prediction        [B, 3129]   VQA answer logits (worker.py:295-307)
  vil_prediction_gqa    [B, 1533]   GQA answer logits (worker.py:310-323)
  vil_logit             [B, 1]      pairwise retrieval logit (worker.py:356-367)
  vil_binary_prediction [B/2, 2]    NLVR2 over an image pair (worker.py:325-338)
  vil_tri_prediction    [B, 3]      SNLI-VE 3-way (worker.py:340-354)
  vision_prediction     [B, R, 1601] masked-region class logits (v_target_size)
  vision_logit          [B, R, 1]   grounding region scores (worker.py:369-386)
  linguisic_prediction  [B, T, vocab] masked-LM logits  (sic — reference spelling)
  linguisic_logit       [B, T, 1]   per-token relevance logit
  attn_data_list        list        attention maps when output_all_attention_masks
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from ..config import ViLBertConfig
from ..ops import functional as F_ops
from .vilbert import FusedLayerNorm, ViLBertModel


class SimpleClassifier(nn.Module):
    """2-layer MLP head: in -> 2*in -> out with GELU + LayerNorm."""

    def __init__(self, in_dim: int, out_dim: int, eps: float = 1e-12):
        super().__init__()
        hid = in_dim * 2
        self.dense = nn.Linear(in_dim, hid)
        self.layer_norm = FusedLayerNorm(hid, eps)
        self.decoder = nn.Linear(hid, out_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = F_ops.linear_bias_gelu(x, self.dense.weight, self.dense.bias)
        h = self.layer_norm(h)
        return F_ops.linear_bias(h, self.decoder.weight, self.decoder.bias)


class LMHead(nn.Module):
    """Masked-LM head tied to the word embedding matrix."""

    def __init__(self, config: ViLBertConfig, word_embedding_weight: torch.Tensor):
        super().__init__()
        self.transform = nn.Linear(config.hidden_size, config.hidden_size)
        self.layer_norm = FusedLayerNorm(config.hidden_size, config.layer_norm_eps)
        self.decoder_weight = word_embedding_weight  # tied
        self.decoder_bias = nn.Parameter(torch.zeros(config.vocab_size))

    def forward(self, t: torch.Tensor) -> torch.Tensor:
        h = F_ops.linear_bias_gelu(t, self.transform.weight, self.transform.bias)
        h = self.layer_norm(h)
        # 30522-way vocab GEMM — the single biggest head GEMM; autotuned path
        return F_ops.linear_bias(h, self.decoder_weight, self.decoder_bias)


class VILBertForVLTasks(nn.Module):
    def __init__(self, config: ViLBertConfig):
        super().__init__()
        self.config = config
        c = config
        self.bert = ViLBertModel(c)
        bi = c.bi_hidden_size
        self.dropout = nn.Dropout(c.hidden_dropout_prob)
        # fused pooled representation = pooled_t * pooled_v  (fusion_method="mul")
        self.vil_prediction = SimpleClassifier(bi, c.num_labels_vqa, c.layer_norm_eps)
        self.vil_prediction_gqa = SimpleClassifier(bi, c.num_labels_gqa, c.layer_norm_eps)
        self.vil_logit = nn.Linear(bi, 1)
        self.vil_binary_prediction = SimpleClassifier(bi * 2, 2, c.layer_norm_eps)
        self.vil_tri_prediction = nn.Linear(bi, 3)
        self.vision_prediction = nn.Linear(c.v_hidden_size, c.v_target_size)
        self.vision_logit = nn.Linear(c.v_hidden_size, 1)
        self.linguistic_prediction = LMHead(c, self.bert.embeddings.word_embeddings.weight)
        self.linguistic_logit = nn.Linear(c.hidden_size, 1)

    def forward(
        self,
        input_ids: torch.Tensor,
        features: torch.Tensor,
        spatials: torch.Tensor,
        segment_ids: torch.Tensor,
        input_mask: torch.Tensor,
        image_mask: torch.Tensor,
        co_attention_mask: Optional[torch.Tensor] = None,
        task_ids: Optional[torch.Tensor] = None,
        output_all_attention_masks: bool = False,
    ):
        t, v, pooled_t, pooled_v, attn_data_list = self.bert(
            input_ids,
            features,
            spatials,
            segment_ids,
            input_mask,
            image_mask,
            co_attention_mask,
            task_ids,
            output_all_attention_masks,
        )
        fused = self.dropout(pooled_t * pooled_v)

        vil_prediction = self.vil_prediction(fused)
        vil_prediction_gqa = self.vil_prediction_gqa(fused)
        vil_logit = self.vil_logit(fused)
        # NLVR2: batch is interleaved image pairs (worker.py:266-276 replicates
        # the text x2); concatenate consecutive pair representations.
        b = fused.shape[0]
        if b >= 2:
            pair = fused[: 2 * (b // 2)].reshape(b // 2, -1)
            vil_binary_prediction = self.vil_binary_prediction(pair)
        else:
            vil_binary_prediction = fused.new_zeros(1, 2)
        vil_tri_prediction = self.vil_tri_prediction(fused)
        vision_logit = self.vision_logit(self.dropout(v))
        if getattr(self, "skip_unused_heads", False) and not self.training:
            # Serving fast path (worker opt-in, NOT the benchmark default):
            # the demo decode (worker.py:295-386) never reads the masked-LM,
            # masked-region or per-token-relevance heads — skip their GEMMs
            # (the LM head alone is rows x 30522 x 768 and a >1 GB logits
            # write at large batch). Placeholders keep the 10-tuple arity.
            vision_prediction = v.new_zeros(b, v.shape[1], 0)
            linguisic_prediction = t.new_zeros(b, t.shape[1], 0)
            linguisic_logit = t.new_zeros(b, t.shape[1], 0)
        else:
            vp_in = self.dropout(v)
            vision_prediction = F_ops.linear_bias(
                vp_in, self.vision_prediction.weight, self.vision_prediction.bias
            )
            linguisic_prediction = self.linguistic_prediction(self.dropout(t))
            linguisic_logit = self.linguistic_logit(self.dropout(t))

        if getattr(self, "_fp8_ctx", None) is not None and not self.training:
            self._fp8_ctx.update()  # delayed-scaling refresh (capturable)

        return (
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
        )

    def prepare_for_serving(self) -> None:
        """Fuse projection GEMMs for the inference path (idempotent).
        Call after weights are final (post-load, post-.to(device/dtype))."""
        for m in self.modules():
            if hasattr(m, "prepare_serving"):
                m.prepare_serving()

    # ---- factory matching the reference loader contract -----------------
    @classmethod
    def from_pretrained(
        cls, checkpoint_path: str, config: ViLBertConfig, strict: bool = False
    ) -> "VILBertForVLTasks":
        """Load an upstream ``pytorch_model_*.bin`` state dict
        (worker.py:470,530-532 contract) via the key-translation table in
        ``checkpoint.py``."""
        from .checkpoint import load_upstream_state_dict

        model = cls(config)
        sd = torch.load(checkpoint_path, map_location="cpu", weights_only=True)
        load_upstream_state_dict(model, sd, strict=strict)
        return model
