import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_config():
    from vilbert_multi_task_amd.config import ViLBertConfig

    return ViLBertConfig.tiny()


@pytest.fixture
def tiny_model(tiny_config):
    from vilbert_multi_task_amd.models import VILBertForVLTasks

    torch.manual_seed(0)
    m = VILBertForVLTasks(tiny_config)
    m.eval()
    return m
