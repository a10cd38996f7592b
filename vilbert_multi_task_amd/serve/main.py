"""Service entry points.

  python -m vilbert_multi_task_amd.serve.main app     # HTTP+WS front (uvicorn)
  python -m vilbert_multi_task_amd.serve.main worker  # GPU inference worker
  python -m vilbert_multi_task_amd.serve.main all     # both (demo box)

Replaces the reference's `python manage.py runserver` + `python worker.py`
pair (/root/reference/README, worker.py:661-676) with the same split: any
number of workers compete on the durable queue; the app pushes results over
websockets via the embedded hub.
"""

from __future__ import annotations

import argparse
import os
import threading

import torch


def run_app(args) -> None:
    import uvicorn

    from .app import create_app

    app = create_app(
        db_path=args.db, queue_path=args.queue, media_root=args.media,
        hub_port=args.hub_port,
    )
    uvicorn.run(
        app, host=args.host, port=args.port, log_level="info",
        ws="vilbert_multi_task_amd.serve.ws_protocol:MinimalWebSocketProtocol",
    )


def build_worker(args):
    from ..config import ViLBertConfig
    from ..engine.runner import GraphRunner
    from ..models import VILBertForVLTasks
    from .broker import Broker
    from .db import Database
    from .decode import AnswerVocab
    from .features import SyntheticFeatureProvider
    from .push import PushClient
    from .worker import ServingWorker

    cfg = (
        ViLBertConfig.from_file(args.config)
        if args.config
        else ViLBertConfig.base_12in1()
    )
    if args.checkpoint and os.path.exists(args.checkpoint):
        model = VILBertForVLTasks.from_pretrained(args.checkpoint, cfg)
    else:
        torch.manual_seed(0)
        model = VILBertForVLTasks(cfg)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if device == "cuda" and args.device >= 0:
        # multi-GPU serving = one worker process per GPU, all competing on
        # the same durable queue (SURVEY.md §2.4 scale-out model):
        #   for i in $(seq 0 7); do
        #     python -m vilbert_multi_task_amd.serve.main worker --device $i &
        #   done
        torch.cuda.set_device(args.device)
        device = f"cuda:{args.device}"
    if device.startswith("cuda"):
        model = model.to(device=device, dtype=torch.bfloat16)
    runner = GraphRunner(
        model, device=device, use_graphs=device.startswith("cuda"),
        serving_fast=True, fp8=args.fp8, feat_dim=cfg.v_feature_size,
    )

    provider = None
    if args.detector:
        from ..detector import DetectorFeatureProvider

        provider = DetectorFeatureProvider(device=device)
    else:
        # feature dim must match the CONFIG (the tiny smoke config is 128-d)
        provider = SyntheticFeatureProvider(device=device, feat_dim=cfg.v_feature_size)

    from ..data.tokenizer import BertWordPieceTokenizer

    return ServingWorker(
        runner,
        Broker(args.queue),
        Database(args.db),
        PushClient(port=args.hub_port),
        provider=provider,
        # tokenizer ids must stay inside the CONFIG's vocab (a tiny config
        # with the default 30522-sized hash fallback crashed the embedding)
        tokenizer=BertWordPieceTokenizer(vocab_size=cfg.vocab_size),
        vqa_vocab=AnswerVocab(cfg.num_labels_vqa, args.vqa_answers),
        gqa_vocab=AnswerVocab(cfg.num_labels_gqa, args.gqa_answers),
        max_batch_rows=args.max_batch,
    )


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("role", choices=["app", "worker", "all"])
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--hub-port", type=int, default=6381)
    ap.add_argument("--db", default="vilbert_demo.sqlite3")
    ap.add_argument("--queue", default="vilbert_queue.sqlite3")
    ap.add_argument("--media", default="media")
    ap.add_argument("--config", default="configs/bert_base_6layer_6conect.json")
    ap.add_argument("--checkpoint", default="save/multitask_model/pytorch_model_9.bin")
    ap.add_argument("--vqa-answers", default="save/VQA/cache/trainval_label2ans.pkl")
    ap.add_argument("--gqa-answers", default="save/gqa/cache/trainval_label2ans.pkl")
    ap.add_argument("--detector", action="store_true", help="full Faster R-CNN features")
    ap.add_argument("--fp8", action="store_true",
                    help="fp8 (e4m3) encoder GEMMs: +9-14%% throughput, "
                    "logit cosine > 0.97 vs bf16 (docs/PERFORMANCE.md)")
    ap.add_argument("--max-batch", type=int, default=64)
    ap.add_argument("--device", type=int, default=-1,
                    help="GPU ordinal for this worker (-1: default device); "
                    "run one worker per GPU against the same --queue")
    ap.add_argument("--metrics-port", type=int, default=0,
                    help="Prometheus /metrics port for the worker (0 = off)")
    args = ap.parse_args()

    if args.role == "app":
        run_app(args)
    elif args.role == "worker":
        if args.metrics_port:
            from ..utils.trace import start_metrics_server

            start_metrics_server(args.metrics_port)
        w = build_worker(args)
        if hasattr(w.runner, "warmup_buckets"):
            w.runner.warmup_buckets(args.max_batch)
        w.run_forever()
    else:
        t = threading.Thread(target=lambda: build_worker(args).run_forever(), daemon=True)
        t.start()
        run_app(args)


if __name__ == "__main__":
    main()
