"""Image preprocessing: behavior-identical to the reference's
FeatureExtractor._image_transform (/root/reference/worker.py:91-121):
  - PIL load, grayscale -> 3-channel repeat (worker.py:95-96)
  - RGB -> BGR channel order
  - mean subtract [102.9801, 115.9465, 122.7717] (BGR means)
  - resize shorter side to 800, cap longer side at 1333 (worker.py:60-61,107-112)
Batch padding to /32 mirrors to_image_list(tensors, size_divisible=32)
(worker.py:189).
"""

from __future__ import annotations

from typing import List, Sequence, Tuple

import torch

BGR_MEANS = (102.9801, 115.9465, 122.7717)
MIN_SIZE = 800
MAX_SIZE = 1333


def image_to_tensor(img) -> torch.Tensor:
    """PIL image -> float [3,H,W] BGR mean-subtracted."""
    import numpy as np

    if img.mode != "RGB":
        img = img.convert("RGB")  # grayscale -> 3ch repeat
    arr = torch.from_numpy(np.asarray(img).copy()).float()  # [H,W,3] RGB
    arr = arr.flip(-1)  # RGB -> BGR
    arr = arr - torch.tensor(BGR_MEANS)
    return arr.permute(2, 0, 1).contiguous()


def resize_shorter_side(
    x: torch.Tensor, min_size: int = MIN_SIZE, max_size: int = MAX_SIZE
) -> Tuple[torch.Tensor, float]:
    """Resize [3,H,W] so shorter side == min_size unless the longer side
    would exceed max_size (then scale to max_size). Returns (image, scale)."""
    h, w = x.shape[-2:]
    short, long_ = min(h, w), max(h, w)
    scale = min_size / short
    if long_ * scale > max_size:
        scale = max_size / long_
    nh, nw = int(round(h * scale)), int(round(w * scale))
    y = torch.nn.functional.interpolate(
        x.unsqueeze(0), size=(nh, nw), mode="bilinear", align_corners=False
    ).squeeze(0)
    return y, scale


def to_image_batch(
    images: Sequence[torch.Tensor], size_divisible: int = 32
) -> Tuple[torch.Tensor, List[Tuple[int, int]]]:
    """Pad a list of [3,H,W] to a common /32 size -> [N,3,Hp,Wp]."""
    hs = [int(i.shape[-2]) for i in images]
    ws = [int(i.shape[-1]) for i in images]
    hp = (max(hs) + size_divisible - 1) // size_divisible * size_divisible
    wp = (max(ws) + size_divisible - 1) // size_divisible * size_divisible
    out = images[0].new_zeros(len(images), 3, hp, wp)
    for i, img in enumerate(images):
        out[i, :, : img.shape[-2], : img.shape[-1]] = img
    return out, list(zip(hs, ws))
