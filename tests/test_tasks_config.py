import json

import pytest

from vilbert_multi_task_amd.config import ViLBertConfig
from vilbert_multi_task_amd.tasks import (
    MAX_SEQ_LENGTH,
    NUM_REGIONS,
    TASKS,
    get_task,
    validate_request,
)


def test_task_ids_match_reference():
    # IDs from result.html:320-336 + worker.py:256-263
    assert set(TASKS) == {1, 2, 4, 7, 11, 12, 13, 15, 16}


def test_image_arity_matches_worker():
    # worker.py:256-263
    for tid in (1, 15, 13, 11, 4, 16):
        assert validate_request(tid, 1) is None
        assert validate_request(tid, 2) is not None
    assert validate_request(12, 2) is None
    assert validate_request(12, 1) is not None
    assert validate_request(7, 2) is None
    assert validate_request(7, 10) is None
    assert validate_request(7, 1) is not None
    assert validate_request(7, 11) is not None


def test_task2_dead_path_preserved():
    # worker.py:295 decodes task 2 but worker.py:256-263 rejects it
    assert validate_request(2, 1) is not None
    assert get_task(2).name == "VG QA"


def test_serving_shape_constants():
    assert MAX_SEQ_LENGTH == 37 and NUM_REGIONS == 101


def test_config_json_roundtrip(tmp_path):
    cfg = ViLBertConfig.base_12in1()
    p = tmp_path / "cfg.json"
    cfg.to_json_file(str(p))
    cfg2 = ViLBertConfig.from_json_file(str(p))
    assert cfg == cfg2


def test_config_ignores_unknown_keys(tmp_path):
    p = tmp_path / "cfg.json"
    with open(p, "w") as f:
        json.dump({"hidden_size": 96, "num_attention_heads": 4, "some_upstream_flag": True}, f)
    cfg = ViLBertConfig.from_json_file(str(p))
    assert cfg.hidden_size == 96


def test_config_validation():
    with pytest.raises(ValueError):
        ViLBertConfig(hidden_size=100, num_attention_heads=7)


def test_tasks_yaml_round_trip():
    """configs/vilbert_tasks.yml (the reference's worker.py:496 registry
    file, rebuilt) must load to exactly the built-in TASKS registry."""
    from vilbert_multi_task_amd.tasks import TASKS, load_tasks_yaml

    reg = load_tasks_yaml("configs/vilbert_tasks.yml")
    assert reg == TASKS


def test_config_from_yaml(tmp_path):
    import json

    import yaml

    from vilbert_multi_task_amd.config import ViLBertConfig

    raw = json.load(open("configs/bert_base_6layer_6conect.json"))
    p = tmp_path / "cfg.yaml"
    p.write_text(yaml.safe_dump(raw))
    assert ViLBertConfig.from_file(str(p)) == ViLBertConfig.from_json_file(
        "configs/bert_base_6layer_6conect.json"
    )
