"""GPU detector-stack tests: RoIAlign kernel vs CPU oracle, tiny detector
forward on GPU, end-to-end serving worker on the hipGraph path."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_roi_align_gpu_vs_cpu_oracle():
    from vilbert_multi_task_amd.ops.functional import _roi_align_ref, roi_align

    torch.manual_seed(0)
    x = torch.randn(2, 8, 24, 32)
    rois = torch.tensor(
        [
            [0, 2.0, 3.0, 20.0, 18.0],
            [1, 0.0, 0.0, 31.0, 23.0],
            [0, 5.5, 7.25, 9.0, 12.5],
            [1, 30.0, 22.0, 31.5, 23.5],  # edge box
        ]
    )
    ref = _roi_align_ref(x, rois, 7, 0.5, 2)
    out = roi_align(x.cuda(), rois.cuda(), 7, 0.5, 2).cpu()
    assert (out - ref).abs().max() < 1e-4


def test_roi_align_gpu_bf16():
    from vilbert_multi_task_amd.ops.functional import _roi_align_ref, roi_align

    torch.manual_seed(1)
    x = torch.randn(1, 16, 16, 16)
    rois = torch.tensor([[0, 1.0, 1.0, 14.0, 14.0]])
    ref = _roi_align_ref(x, rois, 7, 1.0, 2)
    out = roi_align(x.cuda().to(torch.bfloat16), rois.cuda(), 7, 1.0, 2).float().cpu()
    assert (out - ref).abs().max() < 3e-2


def test_tiny_detector_forward_gpu():
    from vilbert_multi_task_amd.detector import DetectionModel, DetectorConfig

    torch.manual_seed(0)
    m = DetectionModel(DetectorConfig.tiny()).eval().cuda()
    imgs = torch.randn(2, 3, 96, 128, device="cuda")
    outs = m(imgs, [(96, 128), (80, 100)])
    for out in outs:
        assert out["fc6"].shape[1] == 64
        assert torch.isfinite(out["scores"]).all()


def test_serving_worker_gpu_end_to_end(tmp_path):
    """Queue message -> dynamic batch -> hipGraph forward on the HIP kernels
    -> decoded answer + DB row (BASELINE.json config 2 end-to-end slice)."""
    import json

    from vilbert_multi_task_amd.config import ViLBertConfig
    from vilbert_multi_task_amd.engine.runner import GraphRunner
    from vilbert_multi_task_amd.models import VILBertForVLTasks
    from vilbert_multi_task_amd.serve.broker import Broker, vilbert_task
    from vilbert_multi_task_amd.serve.db import Database
    from vilbert_multi_task_amd.serve.push import NullPush
    from vilbert_multi_task_amd.serve.worker import ServingWorker

    torch.manual_seed(0)
    model = VILBertForVLTasks(ViLBertConfig.base_12in1()).to("cuda", torch.bfloat16)
    runner = GraphRunner(model, use_graphs=True)
    broker = Broker(str(tmp_path / "q.sqlite3"))
    db = Database(str(tmp_path / "db.sqlite3"))
    push = NullPush()
    worker = ServingWorker(runner, broker, db, push)
    vilbert_task(broker, ["/img/cat.jpg"], "What animal is this?", 1, "g1")
    vilbert_task(broker, ["/a.jpg", "/b.jpg"], "both show dogs", 12, "g2")
    assert worker.process_once() == 2
    results = [json.loads(p["result"]) for s, p in push.messages if "result" in p]
    assert {r["task_id"] for r in results} == {"1", "12"}
    vqa = next(r for r in results if r["task_id"] == "1")
    assert len(vqa["result"]) == 3
    assert all(0 <= e["confidence"] <= 100 for e in vqa["result"])


def test_x152_detector_bf16_matches_fp32():
    """bf16 serving mode vs fp32 on FIXED proposals. The full pipeline
    cannot be compared end-to-end with random-init weights: RPN objectness
    of a random network is near-uniform, so any numeric perturbation
    reshuffles the kept top-100 (measured: only 11% box overlap). Instead
    feed both precisions the same rois and compare the fc6 features —
    which is what ViLBERT actually consumes."""
    import copy

    from vilbert_multi_task_amd.detector import DetectionModel, DetectorConfig

    torch.manual_seed(0)
    m32 = DetectionModel(DetectorConfig.x152()).eval().cuda()
    mbf = copy.deepcopy(m32).to_bf16()
    images = torch.randn(1, 3, 480, 640, device="cuda")
    with torch.no_grad():
        f32 = m32.backbone(images)
        fbf = mbf.backbone(images.bfloat16())
        boxes = torch.rand(100, 4, device="cuda") * 300
        boxes[:, 2:] += boxes[:, :2] + 32
        rois = torch.cat([torch.zeros(100, 1, device="cuda"), boxes], dim=1)
        fc32 = m32.box_head(m32.pooler(f32[:4], rois))[0].float()
        fcbf = mbf.box_head(mbf.pooler(fbf[:4], rois))[0].float()
    cos = torch.nn.functional.cosine_similarity(fc32, fcbf, dim=1)
    # random-init X-152 is the bf16 worst case (no trained scale structure)
    assert cos.mean() > 0.9, float(cos.mean())
    assert cos.min() > 0.7, float(cos.min())
