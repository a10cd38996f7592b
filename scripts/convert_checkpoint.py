#!/usr/bin/env python3
"""Convert between native trainer checkpoints and the upstream
``pytorch_model_*.bin`` layout (worker.py:470,530-532 format).

  # native trainer checkpoint -> upstream-layout .bin (weights only)
  python scripts/convert_checkpoint.py to-upstream ckpt.bin pytorch_model_9.bin

  # upstream .bin -> native state dict (our module names)
  python scripts/convert_checkpoint.py to-native pytorch_model_9.bin native.pt \
      --config configs/bert_base_6layer_6conect.json
"""

from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from vilbert_multi_task_amd.config import ViLBertConfig
from vilbert_multi_task_amd.models import VILBertForVLTasks
from vilbert_multi_task_amd.models.checkpoint import (
    export_upstream_state_dict,
    load_upstream_state_dict,
)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("mode", choices=["to-upstream", "to-native"])
    ap.add_argument("src")
    ap.add_argument("dst")
    ap.add_argument("--config", default="configs/bert_base_6layer_6conect.json")
    args = ap.parse_args()
    cfg = (
        ViLBertConfig.from_json_file(args.config)
        if os.path.exists(args.config)
        else ViLBertConfig.base_12in1()
    )
    model = VILBertForVLTasks(cfg)
    if args.mode == "to-upstream":
        state = torch.load(args.src, map_location="cpu", weights_only=False)
        sd = state["model"] if isinstance(state, dict) and "model" in state else state
        if isinstance(state, dict) and state.get("model_layout") == "upstream":
            torch.save(sd, args.dst)
        else:
            model.load_state_dict(sd, strict=False)
            torch.save(export_upstream_state_dict(model), args.dst)
    else:
        sd = torch.load(args.src, map_location="cpu", weights_only=True)
        report = load_upstream_state_dict(model, sd)
        if report["missing"] or report["unexpected"]:
            print(
                f"warning: missing={len(report['missing'])} "
                f"unexpected={len(report['unexpected'])} keys", file=sys.stderr,
            )
        torch.save(model.state_dict(), args.dst)
    print(f"wrote {args.dst}")


if __name__ == "__main__":
    main()
