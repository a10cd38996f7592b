#!/usr/bin/env python3
"""Secondary benchmark configs from BASELINE.json (the flagship VQA serving
bench is /root/repo/bench.py — this script covers the rest):

  config 3 (1-GPU slice): multi-task round-robin training step time
  config 4: caption-based retrieval, batch = 2048 image-text pairs
  config 5: mixed-task concurrent serving through the dynamic batcher

Usage: python benchmarks/bench_configs.py [--which retrieval|training|mixed|all]
Prints one JSON line per config. Synthetic data, random-init weights
(no network for datasets/checkpoints).
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from vilbert_multi_task_amd.config import ViLBertConfig
from vilbert_multi_task_amd.data.synthetic import synthetic_batch
from vilbert_multi_task_amd.engine.runner import GraphRunner
from vilbert_multi_task_amd.models import VILBertForVLTasks


def _model(device, dtype=torch.bfloat16):
    cfg = ViLBertConfig.base_12in1()
    torch.manual_seed(0)
    m = VILBertForVLTasks(cfg)
    if device.startswith("cuda"):
        m = m.to(device=device, dtype=dtype)
    return m, cfg


def bench_retrieval(device: str, pairs: int = 2048, steps: int = 10, warmup: int = 3, fp8: bool = False):
    """Config 4: one caption scored against `pairs` candidate images — the
    co-attention MFMA saturation config. Runs as `pairs` rows per forward."""
    m, cfg = _model(device)
    runner = GraphRunner(m, device=device, use_graphs=device.startswith("cuda"), fp8=fp8)
    batch = synthetic_batch(pairs, task_id=7, seed=1)
    for _ in range(warmup):
        runner.run(batch)
    torch.cuda.synchronize() if device.startswith("cuda") else None
    t0 = time.perf_counter()
    for _ in range(steps):
        runner.run(batch)
    torch.cuda.synchronize() if device.startswith("cuda") else None
    dt = (time.perf_counter() - t0) / steps
    return {
        "metric": "retrieval pairs/sec (batch=2048 image-text pairs)",
        "value": round(pairs / dt, 2),
        "unit": "pairs/s",
        "ms_per_step": round(dt * 1e3, 3),
        "config": {"pairs": pairs, "task": "retrieval (7)"},
        "dtype": ("fp8-mixed" if fp8 else "bf16") if device.startswith("cuda") else "fp32",
        "data": "synthetic",
    }


def bench_training(device: str, batch: int = 32, steps: int = 12, warmup: int = 3):
    """Config 3 single-GPU slice: round-robin multi-task training step."""
    from vilbert_multi_task_amd.parallel.trainer import MultiTaskTrainer

    cfg = ViLBertConfig.base_12in1()
    torch.manual_seed(0)
    model = VILBertForVLTasks(cfg)
    if device.startswith("cuda"):
        model = model.to(device=device, dtype=torch.bfloat16)
    tr = MultiTaskTrainer(model, cfg, batch_size=batch, device=device)
    for _ in range(warmup):
        tr.train_step()
    torch.cuda.synchronize() if device.startswith("cuda") else None
    t0 = time.perf_counter()
    for _ in range(steps):
        tr.train_step()
    torch.cuda.synchronize() if device.startswith("cuda") else None
    dt = (time.perf_counter() - t0) / steps
    return {
        "metric": "multi-task training samples/sec (round-robin 12 datasets)",
        "value": round(batch / dt, 2),
        "unit": "samples/s",
        "ms_per_step": round(dt * 1e3, 3),
        "config": {"batch": batch, "optimizer": "AdamW"},
        "dtype": "bf16" if device.startswith("cuda") else "fp32",
        "data": "synthetic",
    }


def bench_mixed_serving(device: str, requests: int = 512, max_batch: int = 64, fp8: bool = False):
    """Config 5: RefCOCO + NLVR2 + VQA + GQA concurrently through the
    dynamic batcher (queue -> batched hipGraph forwards -> decode)."""
    import tempfile

    from vilbert_multi_task_amd.serve.broker import Broker, vilbert_task
    from vilbert_multi_task_amd.serve.db import Database
    from vilbert_multi_task_amd.serve.push import NullPush
    from vilbert_multi_task_amd.serve.worker import ServingWorker

    m, cfg = _model(device)
    runner = GraphRunner(m, device=device, use_graphs=device.startswith("cuda"),
                         serving_fast=True, fp8=fp8)
    with tempfile.TemporaryDirectory() as td:
        from vilbert_multi_task_amd.serve.features import SyntheticFeatureProvider

        broker = Broker(os.path.join(td, "q.sqlite3"))
        db = Database(os.path.join(td, "db.sqlite3"))
        worker = ServingWorker(
            runner, broker, db, NullPush(), max_batch_rows=max_batch,
            provider=SyntheticFeatureProvider(device=device),
        )
        tasks = [1, 15, 13, 11, 12]

        def enqueue(count, tag):
            i = 0
            while i < count:
                t = tasks[i % len(tasks)]
                imgs = ["/a.jpg", "/b.jpg"] if t == 12 else ["/a.jpg"]
                vilbert_task(broker, imgs, f"question {i}", t, f"{tag}{i}")
                i += 1

        # warmup: captures the hipGraph buckets before the timed window
        enqueue(2 * max_batch, "w")
        while broker.depth() > 0:
            worker.process_once()
        enqueue(requests, "s")
        lat = []
        t0 = time.perf_counter()
        served = 0
        while served < requests:
            ts = time.perf_counter()
            got = worker.process_once()
            if got:
                lat.append((time.perf_counter() - ts) * 1e3)
            served += got
        dt = time.perf_counter() - t0
    return {
        "metric": "mixed-task serving requests/sec (dynamic batching)",
        "value": round(requests / dt, 2),
        "unit": "requests/s",
        "p50_batch_ms": round(statistics.median(lat), 2),
        "config": {"tasks": tasks, "max_batch_rows": max_batch, "requests": requests},
        "dtype": "bf16" if device.startswith("cuda") else "fp32",
        "data": "synthetic",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--which", default="all", choices=["retrieval", "training", "mixed", "all"])
    ap.add_argument("--pairs", type=int, default=2048)
    ap.add_argument("--fp8", action="store_true")
    args = ap.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if device == "cpu":
        print("WARNING: CPU — shrinking configs", file=sys.stderr)
        args.pairs = 8
    runs = {
        "retrieval": lambda: bench_retrieval(device, args.pairs, fp8=args.fp8),
        "training": lambda: bench_training(device, batch=4 if device == "cpu" else 32,
                                           steps=3 if device == "cpu" else 12,
                                           warmup=1 if device == "cpu" else 3),
        "mixed": lambda: bench_mixed_serving(device, requests=20 if device == "cpu" else 512, fp8=args.fp8),
    }
    for name, fn in runs.items():
        if args.which in (name, "all"):
            out = fn()
            out["n_gpus"] = 1 if device.startswith("cuda") else 0
            print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
