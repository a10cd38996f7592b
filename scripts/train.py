#!/usr/bin/env python3
"""12-task round-robin multi-task training CLI.

Single GPU:   python scripts/train.py --steps 100
8-GPU DP:     python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
                --master-addr 127.0.0.1 scripts/train.py --steps 1000

One process per GPU over RCCL (xGMI); gloo on CPU for smoke tests.
Checkpoints (model in upstream .bin layout + optimizer + sampler state) are
written every --save-every steps and auto-resumed from --checkpoint.
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from vilbert_multi_task_amd.config import ViLBertConfig
from vilbert_multi_task_amd.models import VILBertForVLTasks
from vilbert_multi_task_amd.parallel.trainer import MultiTaskTrainer
from vilbert_multi_task_amd.utils.trace import log_json


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="")
    ap.add_argument("--tiny", action="store_true")
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--lr", type=float, default=4e-5)
    ap.add_argument("--checkpoint", default="save/multitask_model/checkpoint.bin")
    ap.add_argument("--save-every", type=int, default=500)
    ap.add_argument("--seq-len", type=int, default=37)
    ap.add_argument("--regions", type=int, default=101)
    ap.add_argument("--log-every", type=int, default=10)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            torch.cuda.set_device(local_rank)
        torch.distributed.init_process_group(backend=backend)

    if args.tiny:
        cfg = ViLBertConfig.tiny()
        args.seq_len, args.regions = 20, 36
    elif args.config:
        cfg = ViLBertConfig.from_file(args.config)
    else:
        cfg = ViLBertConfig.base_12in1()

    device = f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(1234)  # same init on every rank (broadcast re-syncs too)
    model = VILBertForVLTasks(cfg)
    if device.startswith("cuda"):
        model = model.to(device=device, dtype=torch.bfloat16)

    trainer = MultiTaskTrainer(
        model, cfg, lr=args.lr, batch_size=args.batch, device=device,
        rank=rank, world_size=world, seq_len=args.seq_len, regions=args.regions,
    )
    if os.path.exists(args.checkpoint):
        trainer.load_checkpoint(args.checkpoint)
        if rank == 0:
            log_json("resume", step=trainer.sampler.state.step, path=args.checkpoint)

    t0 = time.perf_counter()
    window = []
    for i in range(args.steps):
        task, loss = trainer.train_step()
        window.append(loss)
        step = trainer.sampler.state.step
        if rank == 0 and (i + 1) % args.log_every == 0:
            dt = (time.perf_counter() - t0) / len(window)
            log_json(
                "train", step=step, task=task, loss=round(sum(window) / len(window), 4),
                ms_per_step=round(dt * 1e3, 1),
                samples_per_s=round(world * args.batch / dt, 1),
            )
            window = []
            t0 = time.perf_counter()
        if rank == 0 and args.save_every and step % args.save_every == 0:
            os.makedirs(os.path.dirname(args.checkpoint) or ".", exist_ok=True)
            trainer.save_checkpoint(args.checkpoint)
    if rank == 0 and args.save_every:
        os.makedirs(os.path.dirname(args.checkpoint) or ".", exist_ok=True)
        trainer.save_checkpoint(args.checkpoint)
        log_json("checkpoint", path=args.checkpoint, step=trainer.sampler.state.step)
    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
