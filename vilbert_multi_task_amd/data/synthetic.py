"""Synthetic request batches of the exact serving shapes.

Shapes fixed by the reference worker (/root/reference/worker.py:408-455):
question [B,37] int64, features [B,101,2048] f32, spatials [B,101,5] f32,
segment_ids/input_mask [B,37], image_mask [B,101], co_attention_mask
[B,101,37], task_tokens [B,1]. BASELINE.md: benchmarks run on synthetic
region-features/tokens of this shape with random-init weights (no network
for datasets/checkpoints).
"""

from __future__ import annotations

from typing import Dict

import torch

from ..tasks import FEATURE_DIM, MAX_SEQ_LENGTH, NUM_REGIONS


def synthetic_batch(
    batch: int,
    seq_len: int = MAX_SEQ_LENGTH,
    regions: int = NUM_REGIONS,
    feat_dim: int = FEATURE_DIM,
    vocab_size: int = 30522,
    task_id: int = 1,
    device: str = "cpu",
    seed: int = 0,
    dtype: torch.dtype = torch.float32,
) -> Dict[str, torch.Tensor]:
    g = torch.Generator(device="cpu").manual_seed(seed)
    question = torch.randint(1, vocab_size, (batch, seq_len), generator=g)
    question[:, 0] = 101  # [CLS]
    question[:, -1] = 102  # [SEP]
    features = torch.randn(batch, regions, feat_dim, generator=g)
    boxes = torch.rand(batch, regions, 4, generator=g)
    x1 = torch.minimum(boxes[..., 0], boxes[..., 2])
    x2 = torch.maximum(boxes[..., 0], boxes[..., 2])
    y1 = torch.minimum(boxes[..., 1], boxes[..., 3])
    y2 = torch.maximum(boxes[..., 1], boxes[..., 3])
    area = (x2 - x1) * (y2 - y1)
    spatials = torch.stack([x1, y1, x2, y2, area], dim=-1)
    spatials[:, 0] = torch.tensor([0.0, 0.0, 1.0, 1.0, 1.0])  # global box (worker.py:443)
    out = {
        "question": question,
        "features": features.to(dtype),
        "spatials": spatials.to(dtype),
        "segment_ids": torch.zeros(batch, seq_len, dtype=torch.long),
        "input_mask": torch.ones(batch, seq_len, dtype=torch.long),
        "image_mask": torch.ones(batch, regions, dtype=torch.long),
        "co_attention_mask": torch.zeros(batch, regions, seq_len),
        "task_tokens": torch.full((batch, 1), int(task_id), dtype=torch.long),
    }
    return {k: v.to(device) for k, v in out.items()}


def forward_args(batch: Dict[str, torch.Tensor], output_all_attention_masks: bool = False):
    """Order matching VILBertForVLTasks.forward / worker.py:287-289."""
    return (
        batch["question"],
        batch["features"],
        batch["spatials"],
        batch["segment_ids"],
        batch["input_mask"],
        batch["image_mask"],
        batch["co_attention_mask"],
        batch["task_tokens"],
        output_all_attention_masks,
    )
