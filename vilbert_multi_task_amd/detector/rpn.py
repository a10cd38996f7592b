"""Region Proposal Network over the FPN pyramid.

MI355X-native rebuild of the anchor/proposal machinery the reference's
detector exercises (SURVEY.md §2.3 "anchor/proposal machinery"): anchor
grids per level, objectness + box-delta head, top-k pre-NMS, box decode,
clip, per-level NMS (HIP nms kernel, single class), cross-level top-1000.
"""

from __future__ import annotations

import math
from typing import List, Sequence, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


def generate_anchors(
    base_size: int, aspect_ratios: Sequence[float] = (0.5, 1.0, 2.0)
) -> torch.Tensor:
    """[A,4] anchors centered at origin for one level."""
    anchors = []
    area = float(base_size * base_size)
    for ar in aspect_ratios:
        w = math.sqrt(area / ar)
        h = w * ar
        anchors.append([-w / 2, -h / 2, w / 2, h / 2])
    return torch.tensor(anchors, dtype=torch.float32)


def shift_anchors(cell: torch.Tensor, stride: int, h: int, w: int) -> torch.Tensor:
    """Tile [A,4] cell anchors over an h x w grid -> [h*w*A, 4]."""
    xs = torch.arange(w, dtype=torch.float32, device=cell.device) * stride
    ys = torch.arange(h, dtype=torch.float32, device=cell.device) * stride
    yy, xx = torch.meshgrid(ys, xs, indexing="ij")
    shifts = torch.stack([xx, yy, xx, yy], dim=-1).reshape(-1, 1, 4)
    return (shifts + cell.view(1, -1, 4)).reshape(-1, 4)


def decode_boxes(deltas: torch.Tensor, anchors: torch.Tensor) -> torch.Tensor:
    """Standard (dx,dy,dw,dh) box decode."""
    aw = anchors[:, 2] - anchors[:, 0]
    ah = anchors[:, 3] - anchors[:, 1]
    ax = anchors[:, 0] + aw / 2
    ay = anchors[:, 1] + ah / 2
    dx, dy, dw, dh = deltas.unbind(-1)
    dw = dw.clamp(max=math.log(1000.0 / 16))
    dh = dh.clamp(max=math.log(1000.0 / 16))
    cx = dx * aw + ax
    cy = dy * ah + ay
    w = dw.exp() * aw
    h = dh.exp() * ah
    return torch.stack([cx - w / 2, cy - h / 2, cx + w / 2, cy + h / 2], dim=-1)


def nms_single(boxes: torch.Tensor, scores: torch.Tensor, thr: float) -> torch.Tensor:
    """Greedy NMS -> kept indices (descending score). Dispatches to the HIP
    multiclass kernel (C=1) on GPU, plain torch on CPU."""
    if boxes.numel() == 0:
        return torch.empty(0, dtype=torch.long, device=boxes.device)
    if boxes.is_cuda:
        from ..ops import functional as F_ops

        if F_ops.extension_available():
            ext = F_ops._load_extension()
            n = min(boxes.shape[0], 1024)
            order0 = scores.argsort(descending=True)[:n]
            b = boxes[order0].contiguous()
            s = scores[order0].contiguous().view(-1, 1)
            # suppressed entries come back as 0; sigmoid scores are > 0
            kept_scores = ext.nms_multiclass(b, s, thr, 0.0)
            keep_local = torch.nonzero(kept_scores[:, 0] > 0).flatten()
            keep = order0[keep_local]
            return keep[scores[keep].argsort(descending=True)]
    # CPU reference
    order = scores.argsort(descending=True)
    keep = []
    suppressed = torch.zeros(boxes.shape[0], dtype=torch.bool)
    area = (boxes[:, 2] - boxes[:, 0]).clamp(min=0) * (boxes[:, 3] - boxes[:, 1]).clamp(min=0)
    for i in order.tolist():
        if suppressed[i]:
            continue
        keep.append(i)
        ix = (torch.minimum(boxes[i, 2], boxes[:, 2]) - torch.maximum(boxes[i, 0], boxes[:, 0])).clamp(min=0)
        iy = (torch.minimum(boxes[i, 3], boxes[:, 3]) - torch.maximum(boxes[i, 1], boxes[:, 1])).clamp(min=0)
        inter = ix * iy
        iou = inter / (area[i] + area - inter)
        suppressed |= iou > thr
        suppressed[i] = False
    return torch.tensor(keep, dtype=torch.long, device=boxes.device)


class RPN(nn.Module):
    def __init__(
        self,
        in_ch: int,
        strides: Sequence[int] = (4, 8, 16, 32, 64),
        anchor_sizes: Sequence[int] = (32, 64, 128, 256, 512),
        aspect_ratios: Sequence[float] = (0.5, 1.0, 2.0),
        pre_nms_top_n: int = 1000,
        post_nms_top_n: int = 1000,
        nms_thresh: float = 0.7,
    ):
        super().__init__()
        self.strides = list(strides)
        self.cell_anchors = [generate_anchors(s, aspect_ratios) for s in anchor_sizes]
        A = len(aspect_ratios)
        self.conv = nn.Conv2d(in_ch, in_ch, 3, padding=1)
        self.objectness = nn.Conv2d(in_ch, A, 1)
        self.deltas = nn.Conv2d(in_ch, A * 4, 1)
        self.pre_nms_top_n = pre_nms_top_n
        self.post_nms_top_n = post_nms_top_n
        self.nms_thresh = nms_thresh

    @torch.no_grad()
    def forward(
        self, feats: List[torch.Tensor], image_sizes: List[Tuple[int, int]]
    ) -> List[Tuple[torch.Tensor, torch.Tensor]]:
        """Returns per image: (boxes [P,4], scores [P])."""
        n = feats[0].shape[0]
        per_level = []
        for lvl, f in enumerate(feats):
            t = F.relu(self.conv(f))
            obj = self.objectness(t)  # [N,A,h,w]
            dlt = self.deltas(t)  # [N,A*4,h,w]
            h, w = f.shape[-2:]
            anchors = shift_anchors(
                self.cell_anchors[lvl].to(f.device), self.strides[lvl], h, w
            )
            obj = obj.permute(0, 2, 3, 1).reshape(n, -1)
            dlt = dlt.permute(0, 2, 3, 1).reshape(n, -1, 4)
            per_level.append((obj, dlt, anchors))

        results = []
        for i in range(n):
            boxes_all, scores_all = [], []
            ih, iw = image_sizes[i]
            for obj, dlt, anchors in per_level:
                scores = obj[i].float().sigmoid()
                k = min(self.pre_nms_top_n, scores.shape[0])
                top, idx = scores.topk(k)
                boxes = decode_boxes(dlt[i][idx].float(), anchors[idx])
                boxes[:, 0].clamp_(0, iw)
                boxes[:, 1].clamp_(0, ih)
                boxes[:, 2].clamp_(0, iw)
                boxes[:, 3].clamp_(0, ih)
                wide = (boxes[:, 2] - boxes[:, 0] >= 1) & (boxes[:, 3] - boxes[:, 1] >= 1)
                boxes, top = boxes[wide], top[wide]
                keep = nms_single(boxes, top, self.nms_thresh)
                boxes_all.append(boxes[keep])
                scores_all.append(top[keep])
            boxes = torch.cat(boxes_all)
            scores = torch.cat(scores_all)
            k = min(self.post_nms_top_n, scores.shape[0])
            top, idx = scores.topk(k)
            results.append((boxes[idx], top))
        return results
