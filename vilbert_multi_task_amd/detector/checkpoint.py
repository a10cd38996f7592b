"""Detector checkpoint load/save (worker.py:83-89 contract).

The reference does `torch.load(model_final.pth, cpu)` then a name-matched
`load_state_dict` (/root/reference/worker.py:83-85). Offline there is no
real X-152 checkpoint to pin the exact upstream names against, so the
loader is layered:
  1. native: exact-name load of checkpoints saved by save_checkpoint()
  2. containerized dicts: unwraps {"model": ...} (detectron-style) and
     strips "module." prefixes
  3. foreign names: shape-and-order greedy matching with a report — the
     deterministic module order of ResNeXtFPN mirrors the upstream stage
     ordering, so same-architecture checkpoints map 1:1 by shape sequence.
"""

from __future__ import annotations

from collections import defaultdict
from typing import Dict, List

import torch
import torch.nn as nn


def save_checkpoint(model: nn.Module, path: str) -> None:
    torch.save({"model": model.state_dict()}, path)


def load_detectron_checkpoint(
    model: nn.Module, path: str, strict: bool = False
) -> Dict[str, List[str]]:
    raw = torch.load(path, map_location="cpu", weights_only=False)
    if isinstance(raw, dict) and "model" in raw and isinstance(raw["model"], dict):
        raw = raw["model"]
    sd = {k[len("module."):] if k.startswith("module.") else k: v for k, v in raw.items()}

    own = model.state_dict()
    # pass 1: exact names with matching shapes
    hit = {k: v for k, v in sd.items() if k in own and own[k].shape == v.shape}
    missing = [k for k in own if k not in hit]
    unexpected = [k for k in sd if k not in hit]

    # pass 2: shape-sequence matching for foreign names
    if missing and unexpected:
        by_shape: Dict[tuple, List[str]] = defaultdict(list)
        for k in unexpected:
            by_shape[tuple(sd[k].shape)].append(k)
        matched_src = set()
        for k in list(missing):
            cands = by_shape.get(tuple(own[k].shape), [])
            src = next((c for c in cands if c not in matched_src), None)
            if src is not None:
                hit[k] = sd[src]
                matched_src.add(src)
                missing.remove(k)
        unexpected = [k for k in unexpected if k not in matched_src]

    model.load_state_dict(hit, strict=False)
    if strict and (missing or unexpected):
        raise RuntimeError(
            f"detector checkpoint mismatch: missing={missing[:5]} unexpected={unexpected[:5]}"
        )
    return {"missing": missing, "unexpected": unexpected}
