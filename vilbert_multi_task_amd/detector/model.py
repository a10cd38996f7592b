"""The detection model: backbone + RPN + pooler + box head.

Serving-path surface identical to what the reference exercises of
maskrcnn_benchmark: forward(images) -> per-image
{"proposals" boxes, "scores" [R,1601] softmax, "fc6" [R,2048]}, from which
the extractor builds the ViLBERT region features. The X-152 geometry is the
default; `DetectorConfig.tiny()` shrinks everything for CPU tests.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Tuple

import torch
import torch.nn as nn

from .backbone import ResNeXtFPN
from .roi_heads import BoxFeatureHead, FPNPooler
from .rpn import RPN


@dataclass
class DetectorConfig:
    stage_blocks: Tuple[int, ...] = (3, 8, 36, 3)  # X-152
    groups: int = 32
    base_width: int = 8
    fpn_channels: int = 256
    width_mult: float = 1.0
    pool_size: int = 7
    rep_dim: int = 2048
    num_classes: int = 1601
    pre_nms_top_n: int = 1000
    post_nms_top_n: int = 1000
    rpn_nms_thresh: float = 0.7

    @classmethod
    def x152(cls) -> "DetectorConfig":
        return cls()

    @classmethod
    def tiny(cls) -> "DetectorConfig":
        return cls(
            stage_blocks=(1, 1, 1, 1),
            groups=4,
            base_width=4,
            fpn_channels=32,
            width_mult=0.125,
            rep_dim=64,
            num_classes=16,
            pre_nms_top_n=50,
            post_nms_top_n=20,
        )


class DetectionModel(nn.Module):
    def __init__(self, cfg: DetectorConfig = DetectorConfig.x152()):
        super().__init__()
        self.cfg = cfg
        self.backbone = ResNeXtFPN(
            stage_blocks=cfg.stage_blocks,
            groups=cfg.groups,
            base_width=cfg.base_width,
            fpn_ch=cfg.fpn_channels,
            width_mult=cfg.width_mult,
        )
        self.rpn = RPN(
            cfg.fpn_channels,
            pre_nms_top_n=cfg.pre_nms_top_n,
            post_nms_top_n=cfg.post_nms_top_n,
            nms_thresh=cfg.rpn_nms_thresh,
        )
        self.pooler = FPNPooler(out_size=cfg.pool_size)
        self.box_head = BoxFeatureHead(
            cfg.fpn_channels, cfg.pool_size, cfg.rep_dim, cfg.num_classes
        )

    def to_bf16(self) -> "DetectionModel":
        """Serving-precision mode: convs (MIOpen) + fc6/cls GEMMs in bf16;
        box arithmetic stays fp32 (anchor coords up to 1333 px exceed bf16's
        8 mantissa bits — a bf16 anchor is off by multiple pixels, and the
        RPN decode/NMS math is latency-trivial anyway). The RPN conv head
        emits bf16 which decode upcasts (rpn.py .float() calls)."""
        self.backbone.to(torch.bfloat16)
        self.box_head.to(torch.bfloat16)
        self.rpn.to(torch.bfloat16)
        for m in self.rpn.modules():  # anchors and friends back to fp32
            for k, b in m._buffers.items():
                if b is not None and b.dtype == torch.bfloat16:
                    m._buffers[k] = b.float()
        return self

    @torch.no_grad()
    def forward(
        self, images: torch.Tensor, image_sizes: List[Tuple[int, int]]
    ) -> List[Dict[str, torch.Tensor]]:
        images = images.to(next(self.backbone.parameters()).dtype)
        feats = self.backbone(images)
        proposals = self.rpn(feats, image_sizes)
        results = []
        for i, (boxes, obj_scores) in enumerate(proposals):
            rois = torch.cat(
                [torch.full_like(boxes[:, :1], i), boxes], dim=1
            )
            pooled = self.pooler(feats[:4], rois)
            fc6, logits = self.box_head(pooled)
            results.append(
                {
                    "proposals": boxes,
                    "objectness": obj_scores,
                    "scores": torch.softmax(logits.float(), dim=-1),
                    "fc6": fc6,
                }
            )
        return results
