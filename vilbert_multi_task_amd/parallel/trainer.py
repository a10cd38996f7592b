"""12-task round-robin multi-task trainer with overlapped DP all-reduce.

The engine the reference links to but does not vendor (SURVEY.md §0
obligation 2: the 12-in-1 round-robin training engine), built MI355X-first:
one process per GPU over RCCL (`nccl` backend on ROCm), bucketed all-reduce
overlapped with backward (parallel/ddp.py), bf16 params with fp32-master
fused AdamW (parallel/optim.py)
master state, checkpoints that round-trip to the upstream .bin layout
(models/checkpoint.py).

Per-dataset losses (12-in-1 recipe, pinned by the head dims the serving
worker decodes — worker.py:295-386):
  vqa_v2 / visual_genome_qa -> BCE on vil_prediction (3129 soft labels)
  gqa                       -> BCE on vil_prediction_gqa (1533)
  refcoco/refcoco+/refcocog/visual7w/guesswhat
                            -> BCE on per-region vision_logit (grounding)
  coco/flickr30k retrieval  -> CE over G=4 candidate images per caption
  snli_ve                   -> CE 3-way on vil_tri_prediction
  nlvr2                     -> CE 2-way on vil_binary_prediction (pairs)
"""

from __future__ import annotations

import os
from typing import Tuple

import torch
import torch.nn.functional as F

from ..config import ViLBertConfig
from ..data.synthetic import synthetic_batch, forward_args
from .ddp import BucketedDataParallel
from .sampler import RoundRobinTaskSampler

RETRIEVAL_GROUP = 4  # candidates per caption (1 positive + 3 distractors)

_TASK_ID_FOR_DATASET = {
    "vqa_v2": 1,
    "visual_genome_qa": 2,
    "gqa": 15,
    "refcoco": 11,
    "refcoco_plus": 11,
    "refcocog": 11,
    "visual7w": 4,
    "guesswhat": 16,
    "coco_retrieval": 7,
    "flickr30k_retrieval": 7,
    "snli_ve": 13,
    "nlvr2": 12,
}


def make_training_batch(
    dataset: str, batch: int, cfg: ViLBertConfig, seed: int, device: str = "cpu",
    seq_len: int = 37, regions: int = 101,
) -> Tuple[dict, torch.Tensor]:
    """Synthetic supervised batch for `dataset` (no network for real data —
    BASELINE.md). Returns (forward batch, targets)."""
    if dataset == "nlvr2" and batch % 2:
        batch += 1
    b = synthetic_batch(
        batch, seq_len=seq_len, regions=regions,
        feat_dim=cfg.v_feature_size, vocab_size=cfg.vocab_size,
        task_id=_TASK_ID_FOR_DATASET[dataset], seed=seed, device=device,
    )
    g = torch.Generator().manual_seed(seed ^ 0x5EED)
    if dataset in ("vqa_v2", "visual_genome_qa"):
        t = torch.zeros(batch, cfg.num_labels_vqa)
        idx = torch.randint(0, cfg.num_labels_vqa, (batch,), generator=g)
        t[torch.arange(batch), idx] = 1.0
    elif dataset == "gqa":
        t = torch.zeros(batch, cfg.num_labels_gqa)
        idx = torch.randint(0, cfg.num_labels_gqa, (batch,), generator=g)
        t[torch.arange(batch), idx] = 1.0
    elif dataset in ("refcoco", "refcoco_plus", "refcocog", "visual7w", "guesswhat"):
        t = torch.zeros(batch, regions)
        idx = torch.randint(1, regions, (batch,), generator=g)  # not the global box
        t[torch.arange(batch), idx] = 1.0
    elif dataset in ("coco_retrieval", "flickr30k_retrieval"):
        assert batch % RETRIEVAL_GROUP == 0
        t = torch.randint(0, RETRIEVAL_GROUP, (batch // RETRIEVAL_GROUP,), generator=g)
    elif dataset == "snli_ve":
        t = torch.randint(0, 3, (batch,), generator=g)
    elif dataset == "nlvr2":
        t = torch.randint(0, 2, (batch // 2,), generator=g)
    else:
        raise KeyError(dataset)
    return b, t.to(device)


def task_loss(dataset: str, outputs, targets: torch.Tensor) -> torch.Tensor:
    if dataset in ("vqa_v2", "visual_genome_qa"):
        return F.binary_cross_entropy_with_logits(outputs[0].float(), targets)
    if dataset == "gqa":
        return F.binary_cross_entropy_with_logits(outputs[1].float(), targets)
    if dataset in ("refcoco", "refcoco_plus", "refcocog", "visual7w", "guesswhat"):
        return F.binary_cross_entropy_with_logits(
            outputs[6].squeeze(-1).float(), targets
        )
    if dataset in ("coco_retrieval", "flickr30k_retrieval"):
        logits = outputs[2].float().view(-1, RETRIEVAL_GROUP)
        return F.cross_entropy(logits, targets)
    if dataset == "snli_ve":
        return F.cross_entropy(outputs[4].float(), targets)
    if dataset == "nlvr2":
        return F.cross_entropy(outputs[3].float(), targets)
    raise KeyError(dataset)


class MultiTaskTrainer:
    def __init__(
        self,
        model: torch.nn.Module,
        cfg: ViLBertConfig,
        lr: float = 4e-5,
        batch_size: int = 8,
        device: str = "cpu",
        rank: int = 0,
        world_size: int = 1,
        bucket_bytes: int = 50 * 1024 * 1024,
        seq_len: int = 37,
        regions: int = 101,
        grad_clip: float = 1.0,
        warmup_steps: int = 0,
        total_steps: int = 0,
        grad_accum: int = 1,
    ):
        self.cfg = cfg
        self.device = device
        self.batch_size = batch_size
        self.seq_len = seq_len
        self.regions = regions
        self.grad_clip = grad_clip
        self.model = model.to(device)
        self.ddp = BucketedDataParallel(self.model, bucket_bytes=bucket_bytes)
        # fused AdamW with fp32 master weights (parallel/optim.py): one HIP
        # kernel per tensor on GPU; identical-math torch fallback elsewhere
        from .optim import FusedAdamW

        self.opt = FusedAdamW(self.model.parameters(), lr=lr, weight_decay=0.01)
        self.sampler = RoundRobinTaskSampler(rank=rank, world_size=world_size)
        self.base_lr = lr
        self.warmup_steps = warmup_steps
        self.total_steps = total_steps
        self.grad_accum = max(grad_accum, 1)

    def _lr_at(self, step: int) -> float:
        """Warmup-linear schedule (the 12-in-1 recipe): linear ramp over
        warmup_steps, then linear decay to 0 at total_steps (constant when
        total_steps == 0)."""
        if self.warmup_steps and step < self.warmup_steps:
            return self.base_lr * (step + 1) / self.warmup_steps
        if self.total_steps:
            frac = max(0.0, 1.0 - (step - self.warmup_steps) / max(
                self.total_steps - self.warmup_steps, 1))
            return self.base_lr * frac
        return self.base_lr

    def train_step(self) -> Tuple[str, float]:
        """One optimizer step = grad_accum micro-batches of ONE task
        (accumulation stays within a task so the gradient layout is stable
        for the bucketed all-reduce)."""
        self.model.train()
        dataset = self.sampler.next_task()
        self.ddp.zero_grad()
        self.ddp.defer_reduction = self.grad_accum > 1
        total = 0.0
        for micro in range(self.grad_accum):
            seed = self.sampler.shard_seed(dataset) + micro * 1000003
            batch, targets = make_training_batch(
                dataset, self.batch_size, self.cfg, seed, self.device,
                self.seq_len, self.regions,
            )
            out = self.ddp(*forward_args(batch))
            loss = task_loss(dataset, out, targets) / self.grad_accum
            if micro < self.grad_accum - 1:
                # defer the all-reduce hooks' finalize to the last micro-batch
                loss.backward()
            else:
                loss.backward()
                self.ddp.finalize_backward()
            total += float(loss.detach())
        if self.grad_clip:
            torch.nn.utils.clip_grad_norm_(self.model.parameters(), self.grad_clip)
        lr = self._lr_at(self.sampler.state.step)
        for g in self.opt.param_groups:
            g["lr"] = lr
        self.opt.step()
        return dataset, total

    # -- checkpoint (model + optimizer + sampler: SURVEY.md §5) -----------
    def save_checkpoint(self, path: str, upstream_layout: bool = True) -> None:
        from ..models.checkpoint import export_upstream_state_dict

        state = {
            "model": (
                export_upstream_state_dict(self.model)
                if upstream_layout
                else self.model.state_dict()
            ),
            "model_layout": "upstream" if upstream_layout else "native",
            "optimizer": self.opt.state_dict(),
            "sampler": self.sampler.state_dict(),
            "config": self.cfg.to_dict(),
        }
        tmp = path + ".tmp"
        torch.save(state, tmp)
        os.replace(tmp, path)

    def load_checkpoint(self, path: str) -> None:
        from ..models.checkpoint import load_upstream_state_dict

        state = torch.load(path, map_location="cpu", weights_only=False)
        if state.get("model_layout") == "upstream":
            load_upstream_state_dict(self.model, state["model"])
        else:
            self.model.load_state_dict(state["model"])
        self.model.to(self.device)
        self.opt.load_state_dict(state["optimizer"])
        self.sampler.load_state_dict(state["sampler"])
