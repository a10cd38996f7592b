"""Data-layer API pinned by the reference imports:
ConceptCapLoaderTrain / ConceptCapLoaderVal (/root/reference/worker.py:44)
and LoadDatasetEval (worker.py:46).

Offline build (no network for datasets): the loaders are
synthetic-backed iterators producing batches of the exact pretraining /
eval tensor shapes; a real-corpus backend can be slotted in by replacing
``_sample`` (the batch schema is the API).
"""

from __future__ import annotations

from typing import Dict, Iterator

import torch

from ..config import ViLBertConfig
from ..tasks import MAX_SEQ_LENGTH, NUM_REGIONS
from .synthetic import synthetic_batch


class ConceptCapLoaderBase:
    """Iterator of pretraining batches: masked text + masked regions +
    alignment labels (the Conceptual Captions pretraining recipe)."""

    def __init__(
        self,
        cfg: ViLBertConfig,
        batch_size: int = 32,
        num_batches: int = 100,
        seq_len: int = MAX_SEQ_LENGTH,
        regions: int = NUM_REGIONS,
        mask_prob: float = 0.15,
        seed: int = 0,
    ):
        self.cfg = cfg
        self.batch_size = batch_size
        self.num_batches = num_batches
        self.seq_len = seq_len
        self.regions = regions
        self.mask_prob = mask_prob
        self.seed = seed

    def __len__(self) -> int:
        return self.num_batches

    def _sample(self, i: int) -> Dict[str, torch.Tensor]:
        g = torch.Generator().manual_seed(self.seed * 100003 + i)
        b = synthetic_batch(
            self.batch_size, seq_len=self.seq_len, regions=self.regions,
            feat_dim=self.cfg.v_feature_size, vocab_size=self.cfg.vocab_size,
            seed=self.seed * 100003 + i,
        )
        ids = b["question"]
        lm_labels = torch.full_like(ids, -1)
        mask = torch.rand(ids.shape, generator=g) < self.mask_prob
        mask[:, 0] = False  # never mask [CLS]
        lm_labels[mask] = ids[mask]
        ids = ids.clone()
        ids[mask] = 103  # [MASK]
        region_mask = (torch.rand(self.batch_size, self.regions, generator=g) < self.mask_prob).float()
        region_mask[:, 0] = 0  # global region unmasked
        region_targets = torch.softmax(
            torch.randn(self.batch_size, self.regions, self.cfg.v_target_size, generator=g), dim=-1
        )
        alignment = torch.randint(0, 2, (self.batch_size,), generator=g)
        b.update(
            question=ids,
            lm_labels=lm_labels,
            region_targets=region_targets,
            region_mask=region_mask,
            alignment_labels=alignment,
        )
        return b

    def __iter__(self) -> Iterator[Dict[str, torch.Tensor]]:
        for i in range(self.num_batches):
            yield self._sample(i)


class ConceptCapLoaderTrain(ConceptCapLoaderBase):
    pass


class ConceptCapLoaderVal(ConceptCapLoaderBase):
    def __init__(self, cfg: ViLBertConfig, batch_size: int = 32, num_batches: int = 10, **kw):
        super().__init__(cfg, batch_size, num_batches, seed=999, **kw)


def LoadDatasetEval(
    cfg: ViLBertConfig,
    dataset: str,
    batch_size: int = 32,
    num_batches: int = 10,
    seed: int = 7,
):
    """Eval-split loader factory (worker.py:46 import contract): yields
    (batch, targets) pairs for the named task dataset."""
    from ..parallel.trainer import make_training_batch

    def gen():
        for i in range(num_batches):
            yield make_training_batch(dataset, batch_size, cfg, seed * 7919 + i)

    return gen()
