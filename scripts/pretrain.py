#!/usr/bin/env python3
"""Conceptual-Captions-style pretraining driver.

Runs ``BertForMultiModalPreTraining`` (masked LM + masked-region KL +
alignment losses — models/pretraining.py) over the ConceptCap loader API
the reference imports at /root/reference/worker.py:44. Synthetic-backed
offline; DP-capable the same way scripts/train.py is (one
process per GPU over RCCL, torchrun-launchable).

  python scripts/pretrain.py --steps 100 --batch 32
  python -m torch.distributed.run --nproc-per-node 8 --master-addr \
      127.0.0.1 scripts/pretrain.py --steps 1000
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from vilbert_multi_task_amd.config import ViLBertConfig
from vilbert_multi_task_amd.data.loaders import ConceptCapLoaderTrain, ConceptCapLoaderVal
from vilbert_multi_task_amd.models.pretraining import BertForMultiModalPreTraining
from vilbert_multi_task_amd.parallel.ddp import BucketedDataParallel
from vilbert_multi_task_amd.utils.trace import log_json


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="")
    ap.add_argument("--tiny", action="store_true")
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--lr", type=float, default=1e-4)
    ap.add_argument("--checkpoint", default="save/pretrain/checkpoint.bin")
    ap.add_argument("--save-every", type=int, default=500)
    ap.add_argument("--eval-every", type=int, default=0)
    ap.add_argument("--log-every", type=int, default=10)
    ap.add_argument("--seq-len", type=int, default=37)
    ap.add_argument("--regions", type=int, default=101)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
        torch.distributed.init_process_group(backend=backend)

    if args.tiny:
        cfg = ViLBertConfig.tiny()
        args.seq_len, args.regions = 20, 12
    elif args.config:
        cfg = ViLBertConfig.from_file(args.config)
    else:
        cfg = ViLBertConfig.base_12in1()

    device = f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(0)
    model = BertForMultiModalPreTraining(cfg)
    if device.startswith("cuda"):
        model = model.to(device=device, dtype=torch.bfloat16)
    model.train()
    ddp = BucketedDataParallel(model)
    opt = torch.optim.AdamW(model.parameters(), lr=args.lr, weight_decay=0.01)

    loader = ConceptCapLoaderTrain(
        cfg, batch_size=args.batch, num_batches=args.steps,
        seq_len=args.seq_len, regions=args.regions,
        seed=rank,  # disjoint shards per rank
    )

    step = 0
    t_last = time.perf_counter()
    for b in loader:
        b = {
            k: (v.to(device) if torch.is_tensor(v) else v) for k, v in b.items()
        }
        ddp.zero_grad()
        _, _, _, losses = ddp(
            b["question"],
            b["features"].to(next(model.parameters()).dtype),
            b["spatials"].to(next(model.parameters()).dtype),
            b["segment_ids"], b["input_mask"], b["image_mask"],
            lm_labels=b["lm_labels"], region_targets=b["region_targets"],
            region_mask=b["region_mask"], alignment_labels=b["alignment_labels"],
        )
        loss = sum(losses.values())
        loss.backward()
        ddp.finalize_backward()
        torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
        opt.step()
        step += 1
        if rank == 0 and step % args.log_every == 0:
            dt = (time.perf_counter() - t_last) / args.log_every
            t_last = time.perf_counter()
            log_json(
                "pretrain", step=step, loss=round(float(loss), 4),
                **{k: round(float(v), 4) for k, v in losses.items()},
                ms_per_step=round(dt * 1e3, 1),
                samples_per_s=round(world * args.batch / dt, 1),
            )
        if rank == 0 and args.save_every and step % args.save_every == 0:
            os.makedirs(os.path.dirname(args.checkpoint), exist_ok=True)
            torch.save(
                {"model": model.state_dict(), "optimizer": opt.state_dict(),
                 "step": step},
                args.checkpoint,
            )
            log_json("checkpoint", path=args.checkpoint, step=step)
        if rank == 0 and args.eval_every and step % args.eval_every == 0:
            model.eval()
            with torch.no_grad():
                vb = next(iter(ConceptCapLoaderVal(
                    cfg, batch_size=args.batch, num_batches=1,
                    seq_len=args.seq_len, regions=args.regions)))
                vb = {k: (v.to(device) if torch.is_tensor(v) else v)
                      for k, v in vb.items()}
                _, _, _, vl = model(
                    vb["question"],
                    vb["features"].to(next(model.parameters()).dtype),
                    vb["spatials"].to(next(model.parameters()).dtype),
                    vb["segment_ids"], vb["input_mask"], vb["image_mask"],
                    lm_labels=vb["lm_labels"],
                    region_targets=vb["region_targets"],
                    region_mask=vb["region_mask"],
                    alignment_labels=vb["alignment_labels"],
                )
                log_json("eval", step=step,
                         **{k: round(float(v), 4) for k, v in vl.items()})
            model.train()

    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
