#!/usr/bin/env python3
"""Per-task evaluation over the LoadDatasetEval API (worker.py:46 contract).

Runs the 10-output forward on eval batches of each requested dataset and
reports the per-task metric (accuracy for classification families, R@1 for
retrieval groups). Synthetic-backed offline: with random-init weights the
numbers are chance-level — the value of this script is the eval
*machinery* (loader -> forward -> metric), which accepts a real checkpoint
via --checkpoint.

  python scripts/evaluate.py --tiny --datasets vqa_v2 nlvr2
  python scripts/evaluate.py --checkpoint save/multitask_model/pytorch_model_9.bin
"""

from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from vilbert_multi_task_amd.config import ViLBertConfig
from vilbert_multi_task_amd.data.loaders import LoadDatasetEval
from vilbert_multi_task_amd.models import VILBertForVLTasks
from vilbert_multi_task_amd.parallel.trainer import RETRIEVAL_GROUP, forward_args
from vilbert_multi_task_amd.tasks import TRAINING_DATASETS


@torch.no_grad()
def evaluate_dataset(model, cfg, dataset, batches, batch_size, device):
    correct = total = 0
    for batch, targets in LoadDatasetEval(
        cfg, dataset, batch_size=batch_size, num_batches=batches
    ):
        batch = {k: (v.to(device) if torch.is_tensor(v) else v) for k, v in batch.items()}
        targets = targets.to(device)
        out = model(*forward_args(batch))
        if dataset in ("vqa_v2", "visual_genome_qa"):
            pred = out[0].float().argmax(-1)
            correct += int((targets[torch.arange(len(pred)), pred] > 0).sum())
            total += len(pred)
        elif dataset == "gqa":
            pred = out[1].float().argmax(-1)
            correct += int((targets[torch.arange(len(pred)), pred] > 0).sum())
            total += len(pred)
        elif dataset in ("refcoco", "refcoco_plus", "refcocog", "visual7w", "guesswhat"):
            pred = out[6].squeeze(-1).float().argmax(-1)
            correct += int((targets[torch.arange(len(pred)), pred] > 0).sum())
            total += len(pred)
        elif dataset in ("coco_retrieval", "flickr30k_retrieval"):
            logits = out[2].float().view(-1, RETRIEVAL_GROUP)
            correct += int((logits.argmax(-1) == targets).sum())
            total += logits.shape[0]
        elif dataset == "snli_ve":
            correct += int((out[4].float().argmax(-1) == targets).sum())
            total += len(targets)
        elif dataset == "nlvr2":
            correct += int((out[3].float().argmax(-1) == targets).sum())
            total += len(targets)
    return correct / max(total, 1), total


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="")
    ap.add_argument("--tiny", action="store_true")
    ap.add_argument("--checkpoint", default="")
    ap.add_argument("--datasets", nargs="*", default=list(TRAINING_DATASETS))
    ap.add_argument("--batches", type=int, default=10)
    ap.add_argument("--batch", type=int, default=32)
    args = ap.parse_args()

    cfg = (
        ViLBertConfig.tiny() if args.tiny
        else ViLBertConfig.from_file(args.config) if args.config
        else ViLBertConfig.base_12in1()
    )

    torch.manual_seed(0)
    if args.checkpoint and os.path.exists(args.checkpoint):
        model = VILBertForVLTasks.from_pretrained(args.checkpoint, cfg)
    else:
        model = VILBertForVLTasks(cfg)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if device == "cuda":
        model = model.to(device=device, dtype=torch.bfloat16)
    model.eval()

    results = {}
    for ds in args.datasets:
        acc, n = evaluate_dataset(model, cfg, ds, args.batches, args.batch, device)
        results[ds] = {"accuracy": round(acc, 4), "n": n}
        print(json.dumps({"dataset": ds, **results[ds]}), flush=True)
    print(json.dumps({"event": "eval_summary", "results": results}))


if __name__ == "__main__":
    main()
