"""FPN box pooler + fc6/fc7 box head.

The fc6 activations ARE the 2048-d ViLBERT region features (SURVEY.md §2.3:
the reference configures feature_name="fc6", /root/reference/worker.py:69,
reads them at worker.py:130). Pooling uses the HIP RoIAlign kernel with the
standard FPN level assignment k = floor(4 + log2(sqrt(area)/224)).
"""

from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as F_ops


class FPNPooler(nn.Module):
    def __init__(self, out_size: int = 7, strides=(4, 8, 16, 32), sampling_ratio: int = 2):
        super().__init__()
        self.out_size = out_size
        self.strides = strides
        self.sampling_ratio = sampling_ratio

    def forward(self, feats: List[torch.Tensor], rois: torch.Tensor) -> torch.Tensor:
        """feats: P2..P5; rois [R,5] (batch_idx,x1,y1,x2,y2). -> [R,C,s,s]"""
        if rois.numel() == 0:
            c = feats[0].shape[1]
            return feats[0].new_zeros(0, c, self.out_size, self.out_size)
        area = ((rois[:, 3] - rois[:, 1]) * (rois[:, 4] - rois[:, 2])).clamp(min=1e-6)
        lvl = torch.floor(4 + torch.log2(area.sqrt() / 224 + 1e-6)) - 2
        lvl = lvl.clamp(0, len(self.strides) - 1).long()
        c = feats[0].shape[1]
        out = feats[0].new_zeros(rois.shape[0], c, self.out_size, self.out_size)
        for li in range(len(self.strides)):
            idx = torch.nonzero(lvl == li).flatten()
            if idx.numel() == 0:
                continue
            pooled = F_ops.roi_align(
                feats[li], rois[idx], self.out_size,
                1.0 / self.strides[li], self.sampling_ratio,
            )
            out[idx] = pooled.to(out.dtype)
        return out


class BoxFeatureHead(nn.Module):
    """fc6 (flattened pool -> 2048) + fc7 + classifier over 1601 classes."""

    def __init__(self, in_ch: int, pool_size: int = 7, rep_dim: int = 2048, num_classes: int = 1601):
        super().__init__()
        in_dim = in_ch * pool_size * pool_size
        self.fc6 = nn.Linear(in_dim, rep_dim)
        self.fc7 = nn.Linear(rep_dim, rep_dim)
        self.cls_score = nn.Linear(rep_dim, num_classes)

    def forward(self, pooled: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """-> (fc6_features [R, rep_dim], class_logits [R, num_classes])"""
        x = pooled.flatten(1)
        fc6 = F.relu(self.fc6(x))
        fc7 = F.relu(self.fc7(fc6))
        return fc6, self.cls_score(fc7)
