"""Detector stack tests (CPU, tiny geometry): transforms contract, backbone
shapes, RPN proposals, RoIAlign oracle, end-to-end feature extraction."""

import numpy as np
import pytest
import torch
from PIL import Image

from vilbert_multi_task_amd.detector import (
    DetectionModel,
    DetectorConfig,
    DetectorFeatureProvider,
)
from vilbert_multi_task_amd.detector.rpn import (
    decode_boxes,
    generate_anchors,
    nms_single,
    shift_anchors,
)
from vilbert_multi_task_amd.detector.transforms import (
    image_to_tensor,
    resize_shorter_side,
    to_image_batch,
)
from vilbert_multi_task_amd.ops.functional import _roi_align_ref
from vilbert_multi_task_amd.serve.features import tensorize_regions


def test_image_transform_contract():
    # RGB->BGR + mean subtract (worker.py:91-121)
    arr = np.zeros((10, 12, 3), dtype=np.uint8)
    arr[..., 0] = 200  # R
    arr[..., 2] = 50  # B
    img = Image.fromarray(arr)
    t = image_to_tensor(img)
    assert t.shape == (3, 10, 12)
    assert abs(t[0, 0, 0].item() - (50 - 102.9801)) < 1e-4  # BGR ch0 = B
    assert abs(t[2, 0, 0].item() - (200 - 122.7717)) < 1e-4  # ch2 = R


def test_grayscale_to_3ch():
    img = Image.fromarray(np.full((8, 8), 128, dtype=np.uint8), mode="L")
    t = image_to_tensor(img)
    assert t.shape == (3, 8, 8)


def test_resize_rules():
    # shorter side -> 800 (worker.py:107-112)
    t = torch.zeros(3, 400, 600)
    y, s = resize_shorter_side(t)
    assert y.shape[-2] == 800 and s == 2.0
    # longer side cap at 1333
    t = torch.zeros(3, 400, 1000)
    y, s = resize_shorter_side(t)
    assert max(y.shape[-2:]) <= 1333
    assert abs(s - 1.333) < 1e-3


def test_batch_padding_div32():
    a = torch.ones(3, 37, 50)
    b = torch.ones(3, 64, 33)
    batch, sizes = to_image_batch([a, b])
    assert batch.shape == (2, 3, 64, 64)
    assert sizes == [(37, 50), (64, 33)]


def test_anchor_decode_identity():
    cell = generate_anchors(64)
    assert cell.shape == (3, 4)
    anchors = shift_anchors(cell, 16, 2, 3)
    assert anchors.shape == (2 * 3 * 3, 4)
    boxes = decode_boxes(torch.zeros_like(anchors), anchors)
    assert torch.allclose(boxes, anchors, atol=1e-4)


def test_nms_single_cpu():
    boxes = torch.tensor(
        [[0, 0, 10, 10], [1, 1, 11, 11], [20, 20, 30, 30]], dtype=torch.float32
    )
    scores = torch.tensor([0.9, 0.8, 0.7])
    keep = nms_single(boxes, scores, 0.5)
    assert keep.tolist() == [0, 2]


def test_roi_align_ref_constant():
    x = torch.full((1, 2, 16, 16), 3.0)
    rois = torch.tensor([[0, 2.0, 2.0, 10.0, 10.0]])
    out = _roi_align_ref(x, rois, 7, 1.0, 2)
    assert out.shape == (1, 2, 7, 7)
    assert torch.allclose(out, torch.full_like(out, 3.0), atol=1e-5)


def test_roi_align_ref_gradient_field():
    # linear field f(y,x) = x  -> pooled values increase along pw
    H = W = 16
    x = torch.arange(W, dtype=torch.float32).repeat(H, 1).view(1, 1, H, W)
    rois = torch.tensor([[0, 0.0, 0.0, 15.0, 15.0]])
    out = _roi_align_ref(x, rois, 4, 1.0, 2)[0, 0]
    col_means = out.mean(dim=0)
    assert (col_means[1:] > col_means[:-1]).all()


@pytest.fixture(scope="module")
def tiny_detector():
    torch.manual_seed(0)
    return DetectionModel(DetectorConfig.tiny()).eval()


def test_detection_model_output_contract(tiny_detector):
    imgs = torch.randn(2, 3, 96, 128)
    outs = tiny_detector(imgs, [(96, 128), (80, 100)])
    assert len(outs) == 2
    for out in outs:
        r = out["proposals"].shape[0]
        assert out["proposals"].shape == (r, 4)
        assert out["scores"].shape == (r, 16)
        assert out["fc6"].shape == (r, 64)
        assert torch.allclose(
            out["scores"].sum(-1), torch.ones(r), atol=1e-4
        )


def test_feature_provider_end_to_end(tmp_path, tiny_detector):
    # two small images through load->detect->nms->top-k->tensorize
    for name in ("a.jpg", "b.jpg"):
        arr = (np.random.RandomState(42).rand(60, 80, 3) * 255).astype(np.uint8)
        Image.fromarray(arr).save(tmp_path / name)
    provider = DetectorFeatureProvider(
        tiny_detector, num_features=10, min_size=64, max_size=100
    )
    infos = provider.extract([str(tmp_path / "a.jpg"), str(tmp_path / "b.jpg")])
    assert len(infos) == 2
    for info in infos:
        k = info["num_boxes"]
        assert 0 < k <= 10
        assert info["features"].shape == (k, 64)
        assert info["bbox"].shape == (k, 4)
        assert info["cls_prob"].shape == (k, 16)
        # bbox in ORIGINAL image coords (unscaled: worker.py:165-174)
        assert info["bbox"][:, 2].max() <= info["image_width"] + 1e-3
    reg = tensorize_regions(infos, num_regions=11)
    assert reg["features"].shape == (2, 11, 64)
    assert reg["spatials"].shape == (2, 11, 5)
    assert (reg["spatials"][:, :, :4] <= 1.001).all()
    assert reg["image_mask"][0, 0] == 1  # global region always valid


def test_detector_checkpoint_roundtrip_and_foreign_names(tmp_path, tiny_detector):
    import torch
    from vilbert_multi_task_amd.detector.checkpoint import (
        load_detectron_checkpoint,
        save_checkpoint,
    )
    from vilbert_multi_task_amd.detector import DetectionModel, DetectorConfig

    path = str(tmp_path / "model_final.pth")
    save_checkpoint(tiny_detector, path)
    torch.manual_seed(99)
    m2 = DetectionModel(DetectorConfig.tiny())
    rep = load_detectron_checkpoint(m2, path)
    assert rep["missing"] == [] and rep["unexpected"] == []
    for (k1, v1), (k2, v2) in zip(
        tiny_detector.state_dict().items(), m2.state_dict().items()
    ):
        assert torch.equal(v1, v2), k1

    # foreign naming (module. prefixes + renamed keys) -> shape matching
    sd = {"module.some.foreign." + k: v for k, v in tiny_detector.state_dict().items()}
    torch.save({"model": sd}, path)
    torch.manual_seed(100)
    m3 = DetectionModel(DetectorConfig.tiny())
    rep = load_detectron_checkpoint(m3, path)
    assert rep["missing"] == []  # every param found a same-shape source
