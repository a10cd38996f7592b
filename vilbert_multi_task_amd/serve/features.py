"""Region-feature providers + the region tensorization of the reference.

Tensorization is behavior-identical to /root/reference/worker.py:421-457:
  - mean-pooled global feature PREPENDED -> 101 regions x 2048
  - 5-d spatial = normalized x1,y1,x2,y2 + fractional area; global box
    [0,0,1,1,1] (worker.py:436-444)
  - image_mask all ones, co_attention_mask zeros

Providers:
  SyntheticFeatureProvider  — deterministic per-path features (demo/bench
                              without the detector; BASELINE.md synthetic mode)
  PrecomputedFeatureProvider — .npz files {features, bbox, image_w, image_h}
  DetectorFeatureProvider    — full Faster R-CNN stack (detector module)
"""

from __future__ import annotations

import hashlib
import os
from typing import Dict, List, Sequence

import torch

from ..tasks import FEATURE_DIM, NUM_REGIONS


class SyntheticFeatureProvider:
    """Deterministic pseudo-features keyed by image path (no detector)."""

    def __init__(self, num_boxes: int = 100, feat_dim: int = FEATURE_DIM,
                 cache_size: int = 1024, device: str = "cpu"):
        self.num_boxes = num_boxes
        self.feat_dim = feat_dim
        self._cache: Dict[str, Dict] = {}
        self._cache_size = cache_size
        self.device = device  # "cuda": features cached on-GPU, so batch
        # assembly (tensorize_regions) runs on the GPU — the serving boxes
        # have weak CPUs and the 60+ MB host-side stack dominated the batch

    def extract(self, image_paths: Sequence[str]) -> List[Dict]:
        out = []
        for p in image_paths:
            hit = self._cache.get(p)
            if hit is not None:
                out.append(hit)
                continue
            seed = int.from_bytes(hashlib.sha1(p.encode()).digest()[:4], "little")
            g = torch.Generator().manual_seed(seed)
            feats = torch.randn(self.num_boxes, self.feat_dim, generator=g).abs()
            w, h = 640.0, 480.0
            c = torch.rand(self.num_boxes, 2, generator=g) * torch.tensor([w, h])
            wh = torch.rand(self.num_boxes, 2, generator=g) * torch.tensor([w / 3, h / 3]) + 8
            bbox = torch.cat([c - wh / 2, c + wh / 2], dim=1).clamp(min=0)
            bbox[:, 2].clamp_(max=w)
            bbox[:, 3].clamp_(max=h)
            if self.device != "cpu":
                feats = feats.to(self.device)
                bbox = bbox.to(self.device)
            info = {
                "features": feats,
                "bbox": bbox,
                "image_width": w,
                "image_height": h,
                "num_boxes": self.num_boxes,
            }
            if len(self._cache) < self._cache_size:
                self._cache[p] = info
            out.append(info)
        return out


class PrecomputedFeatureProvider:
    def __init__(self, root: str = ""):
        self.root = root

    def extract(self, image_paths: Sequence[str]) -> List[Dict]:
        import numpy as np

        out = []
        for p in image_paths:
            fp = p if p.endswith(".npz") else os.path.join(self.root, os.path.basename(p) + ".npz")
            z = np.load(fp)
            out.append(
                {
                    "features": torch.from_numpy(z["features"]).float(),
                    "bbox": torch.from_numpy(z["bbox"]).float(),
                    "image_width": float(z["image_w"]),
                    "image_height": float(z["image_h"]),
                    "num_boxes": int(z["features"].shape[0]),
                }
            )
        return out


def tensorize_regions(
    infos: Sequence[Dict], num_regions: int = NUM_REGIONS
) -> Dict[str, torch.Tensor]:
    """Per-image region tensors (worker.py:421-457): returns features
    [N,101,2048], spatials [N,101,5], image_mask [N,101]. Images with fewer
    detector boxes than num_regions-1 are zero-padded and masked (the
    reference always has exactly 100 boxes -> all-ones mask, same behavior)."""
    feat_dim = infos[0]["features"].shape[1]
    n = len(infos)
    nb = {int(i["num_boxes"]) for i in infos}
    if nb == {num_regions - 1}:
        # fast path (the common serving case: every image has exactly 100
        # boxes) — batched tensor ops on the features' own device (GPU when
        # the provider caches there; detector features are already on-GPU)
        dev = infos[0]["features"].device
        f = torch.stack([i["features"] for i in infos]).float()  # [N,100,F]
        bbox = torch.stack([i["bbox"].to(dev) for i in infos]).float()
        wh = torch.tensor(
            [[i["image_width"], i["image_height"]] for i in infos], device=dev
        ).view(n, 1, 2)
        features = torch.empty(n, num_regions, feat_dim, device=dev)
        features[:, 0] = f.mean(dim=1)
        features[:, 1:] = f
        norm = bbox / wh.repeat(1, 1, 2)
        area = (norm[..., 2] - norm[..., 0]) * (norm[..., 3] - norm[..., 1])
        spatials = torch.empty(n, num_regions, 5, device=dev)
        spatials[:, 0] = torch.tensor([0.0, 0.0, 1.0, 1.0, 1.0], device=dev)
        spatials[:, 1:, :4] = norm
        spatials[:, 1:, 4] = area
        return {
            "features": features,
            "spatials": spatials,
            "image_mask": torch.ones(n, num_regions, dtype=torch.long),
        }
    dev = infos[0]["features"].device  # keep assembly on the features' device
    features = torch.zeros(n, num_regions, feat_dim, device=dev)
    spatials = torch.zeros(n, num_regions, 5, device=dev)
    image_mask = torch.zeros(n, num_regions, dtype=torch.long)
    for i, info in enumerate(infos):
        f = info["features"].to(dev).float()
        num = min(int(info["num_boxes"]), num_regions - 1)
        w, h = float(info["image_width"]), float(info["image_height"])
        features[i, 0] = f[:num].mean(dim=0)  # global mean-pooled feature
        features[i, 1 : num + 1] = f[:num]
        bbox = info["bbox"].to(dev).float()[:num]
        x1, y1 = bbox[:, 0] / w, bbox[:, 1] / h
        x2, y2 = bbox[:, 2] / w, bbox[:, 3] / h
        area = (x2 - x1) * (y2 - y1)
        spatials[i, 0] = torch.tensor([0.0, 0.0, 1.0, 1.0, 1.0], device=dev)  # worker.py:443
        spatials[i, 1 : num + 1] = torch.stack([x1, y1, x2, y2, area], dim=1)
        image_mask[i, : num + 1] = 1
    return {"features": features, "spatials": spatials, "image_mask": image_mask}
