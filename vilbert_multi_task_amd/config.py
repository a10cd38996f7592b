"""Typed configuration for the MI355X-native ViLBERT 12-in-1 engine.

Single source of truth replacing the reference's four ad-hoc config layers
(Django settings, frozen SimpleNamespace arg blocks at
/root/reference/worker.py:68-75,470-493, post-load BertConfig mutation at
worker.py:509-523, and the vilbert_tasks.yml EasyDict at worker.py:496-497).

The model geometry mirrors the upstream ``bert_base_6layer_6conect.json``
contract pinned at /root/reference/worker.py:472,495 (text stream = BERT-base,
vision stream = 6 layers of hidden 1024, 6 interleaved co-attention layers).
"""

from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from typing import List


@dataclass
class ViLBertConfig:
    # ---- text stream (BERT-base) ----
    vocab_size: int = 30522
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    intermediate_size: int = 3072
    hidden_act: str = "gelu"
    hidden_dropout_prob: float = 0.1
    attention_probs_dropout_prob: float = 0.1
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    initializer_range: float = 0.02
    layer_norm_eps: float = 1e-12

    # ---- vision stream ----
    v_feature_size: int = 2048
    v_loc_size: int = 5
    v_hidden_size: int = 1024
    v_num_hidden_layers: int = 6
    v_num_attention_heads: int = 8
    v_intermediate_size: int = 1024
    v_hidden_dropout_prob: float = 0.1
    v_attention_probs_dropout_prob: float = 0.1
    v_target_size: int = 1601  # worker.py:513 sets 1601 class targets

    # ---- co-attention ("connect") layers ----
    bi_hidden_size: int = 1024
    bi_num_attention_heads: int = 8
    bi_intermediate_size: int = 1024
    bi_attention_type: int = 1
    t_biattention_id: List[int] = field(default_factory=lambda: [6, 7, 8, 9, 10, 11])
    v_biattention_id: List[int] = field(default_factory=lambda: [0, 1, 2, 3, 4, 5])

    # ---- behavioral flags used by the reference worker ----
    task_specific_tokens: bool = True   # worker.py:516-517
    dynamic_attention: bool = False     # worker.py:519 — dead flag in the demo
                                        # path; model raises if set (vilbert.py)
    visualization: bool = False         # worker.py:522 (True in demo path).
                                        # Advisory here: the native path returns
                                        # attention maps whenever the caller
                                        # passes output_all_attention_masks=True
                                        # (a superset of the upstream gate).
    predict_feature: bool = False       # worker.py:509-514 — demo forces False
                                        # (1601-way masked-region classification;
                                        # heads.py vision_prediction)
    fast_mode: bool = False
    fusion_method: str = "mul"          # pooled fusion: t_pooled * v_pooled
    in_batch_pairs: bool = False
    num_task_tokens: int = 20           # task-token embedding table size

    # ---- head dims (worker.py:295-386,523) ----
    num_labels_vqa: int = 3129
    num_labels_gqa: int = 1533

    def __post_init__(self) -> None:
        if len(self.t_biattention_id) != len(self.v_biattention_id):
            raise ValueError("t_biattention_id and v_biattention_id must align")
        if self.hidden_size % self.num_attention_heads:
            raise ValueError("hidden_size not divisible by heads")
        if self.v_hidden_size % self.v_num_attention_heads:
            raise ValueError("v_hidden_size not divisible by v heads")
        if self.bi_hidden_size % self.bi_num_attention_heads:
            raise ValueError("bi_hidden_size not divisible by bi heads")

    # -- io ---------------------------------------------------------------
    @classmethod
    def from_json_file(cls, path: str) -> "ViLBertConfig":
        with open(path) as f:
            raw = json.load(f)
        return cls.from_dict(raw)

    @classmethod
    def from_file(cls, path: str) -> "ViLBertConfig":
        """Load a model config from .json or .yaml/.yml (one
        typed config format replacing the reference's four ad-hoc layers)."""
        if str(path).endswith((".yaml", ".yml")):
            import yaml

            with open(path) as f:
                return cls.from_dict(yaml.safe_load(f) or {})
        return cls.from_json_file(path)

    @classmethod
    def from_dict(cls, raw: dict) -> "ViLBertConfig":
        known = {f.name for f in dataclasses.fields(cls)}
        # accept upstream key aliases
        aliases = {
            "v_attention_probs_dropout_prob": "v_attention_probs_dropout_prob",
            "pooling_method": None,
        }
        kwargs = {}
        for k, v in raw.items():
            if k in known:
                kwargs[k] = v
            elif k in aliases and aliases[k]:
                kwargs[aliases[k]] = v
        return cls(**kwargs)

    def to_dict(self) -> dict:
        return dataclasses.asdict(self)

    def to_json_file(self, path: str) -> None:
        with open(path, "w") as f:
            json.dump(self.to_dict(), f, indent=2, sort_keys=True)

    # -- canonical variants ------------------------------------------------
    @classmethod
    def base_12in1(cls) -> "ViLBertConfig":
        """The 270M-parameter demo model (bert_base_6layer_6conect)."""
        return cls()

    @classmethod
    def tiny(cls) -> "ViLBertConfig":
        """2-layer tiny model for CPU tests (BASELINE.json config 1)."""
        return cls(
            vocab_size=1024,
            hidden_size=64,
            num_hidden_layers=2,
            num_attention_heads=4,
            intermediate_size=128,
            v_feature_size=128,
            v_hidden_size=96,
            v_num_hidden_layers=2,
            v_num_attention_heads=4,
            v_intermediate_size=96,
            v_target_size=16,
            bi_hidden_size=96,
            bi_num_attention_heads=4,
            bi_intermediate_size=96,
            t_biattention_id=[1],
            v_biattention_id=[0],
            num_labels_vqa=32,
            num_labels_gqa=16,
            max_position_embeddings=64,
        )
