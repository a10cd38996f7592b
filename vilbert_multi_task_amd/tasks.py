"""Task registry for the 12-in-1 demo task families.

Task IDs, names and image arities are behavior-identical to the reference
(IDs from /root/reference/demo/templates/vilbert_multitask/result.html:320-336,
arity check from /root/reference/worker.py:256-263, decode families from
worker.py:295-386).

Note the reference's task 2 (VG QA) quirk: it has a decode branch at
worker.py:295 but is rejected by the image-count check at worker.py:256-263
(dead path). We preserve the ID and mark it ``reachable=False`` so the serving
layer mirrors the observable behavior (a task-2 request is refused) while the
training engine can still train on VG QA.
"""

from __future__ import annotations

from dataclasses import dataclass
from enum import Enum
from typing import Dict, Optional, Tuple


class DecodeFamily(str, Enum):
    """How the model outputs are decoded into a user-facing answer."""

    VQA = "vqa"                  # softmax over answer vocab, top-3
    GQA = "gqa"                  # softmax over GQA answer vocab, top-3
    BINARY = "binary"            # NLVR2 True/False over image pair
    ENTAILMENT = "entailment"    # SNLI-VE 3-way
    RETRIEVAL = "retrieval"      # softmax over candidate images
    GROUNDING = "grounding"      # top-3 region boxes from vision_logit


@dataclass(frozen=True)
class TaskSpec:
    task_id: int
    name: str
    decode: DecodeFamily
    min_images: int
    max_images: int
    reachable: bool = True       # servable through the demo worker path
    dataset: str = ""

    def validate_num_images(self, n: int) -> bool:
        return self.min_images <= n <= self.max_images


# worker.py:256-263: {1,15,13,11,4,16}->1 image, {12}->2, {7}->2..10
TASKS: Dict[int, TaskSpec] = {
    t.task_id: t
    for t in [
        TaskSpec(1, "VQA", DecodeFamily.VQA, 1, 1, dataset="vqa_v2"),
        TaskSpec(2, "VG QA", DecodeFamily.VQA, 1, 1, reachable=False, dataset="visual_genome_qa"),
        TaskSpec(4, "Visual7W Pointing", DecodeFamily.GROUNDING, 1, 1, dataset="visual7w"),
        TaskSpec(7, "Caption-Image Retrieval", DecodeFamily.RETRIEVAL, 2, 10, dataset="coco+flickr30k"),
        TaskSpec(11, "Referring Expressions", DecodeFamily.GROUNDING, 1, 1, dataset="refcoco+refcoco+​refcocog"),
        TaskSpec(12, "NLVR2", DecodeFamily.BINARY, 2, 2, dataset="nlvr2"),
        TaskSpec(13, "Visual Entailment", DecodeFamily.ENTAILMENT, 1, 1, dataset="snli_ve"),
        TaskSpec(15, "GQA", DecodeFamily.GQA, 1, 1, dataset="gqa"),
        TaskSpec(16, "GuessWhat", DecodeFamily.GROUNDING, 1, 1, dataset="guesswhat"),
    ]
}

# The 12 training datasets of the 12-in-1 setup (README.md:4 + header_content.html)
TRAINING_DATASETS: Tuple[str, ...] = (
    "vqa_v2",
    "gqa",
    "visual_genome_qa",
    "refcoco",
    "refcoco_plus",
    "refcocog",
    "visual7w",
    "guesswhat",
    "coco_retrieval",
    "flickr30k_retrieval",
    "snli_ve",
    "nlvr2",
)

# Demo defaults fixed by the reference worker
MAX_SEQ_LENGTH = 37       # worker.py:408-414
NUM_REGIONS = 101         # 100 detector boxes + 1 global mean-pooled (worker.py:432-434)
FEATURE_DIM = 2048        # fc6 features (worker.py:69)
SPATIAL_DIM = 5           # normalized x1,y1,x2,y2,area (worker.py:436-444)
NUM_DETECTION_CLASSES = 1601


def get_task(task_id: int) -> TaskSpec:
    try:
        return TASKS[int(task_id)]
    except (KeyError, ValueError) as e:
        raise KeyError(f"unknown task id {task_id!r}") from e


def validate_request(task_id: int, num_images: int) -> Optional[str]:
    """Mirror the reference worker's image-count validation.

    Returns None if acceptable, else a human-readable error string
    (worker.py:256-263 returns an error dict; we return the message).
    """
    spec = TASKS.get(int(task_id))
    if spec is None or not spec.reachable:
        return f"Task {task_id} is not available"
    if not spec.validate_num_images(num_images):
        if spec.min_images == spec.max_images:
            want = f"{spec.min_images}"
        else:
            want = f"{spec.min_images}-{spec.max_images}"
        return (
            f"Task {spec.name} expects {want} image(s), got {num_images}"
        )
    return None

def load_tasks_yaml(path: str) -> Dict[int, TaskSpec]:
    """Load a task registry from a ``vilbert_tasks.yml``-style file
    (upstream shape: top-level ``TASK<N>`` sections — worker.py:496-497
    loads the same file into an EasyDict). Unknown per-task keys are
    ignored so upstream training fields (dataroot, lr, ...) pass through.
    Returns a registry dict; the built-in ``TASKS`` stays the default."""
    import yaml

    with open(path) as f:
        raw = yaml.safe_load(f) or {}
    fields = {"task_id", "name", "decode", "min_images", "max_images", "reachable", "dataset"}
    reg: Dict[int, TaskSpec] = {}
    for key, spec in raw.items():
        if not str(key).upper().startswith("TASK") or not isinstance(spec, dict):
            continue
        tid = int(spec.get("task_id", str(key)[4:]))
        base = TASKS.get(tid)
        kw = {k: v for k, v in spec.items() if k in fields}
        kw["task_id"] = tid
        if "decode" in kw:
            kw["decode"] = DecodeFamily(kw["decode"])
        elif base is not None:
            kw["decode"] = base.decode
        else:
            raise ValueError(f"task {tid}: 'decode' family required for new tasks")
        if base is not None:  # fill unspecified fields from the built-in spec
            for k in fields - set(kw):
                kw[k] = getattr(base, k)
        reg[tid] = TaskSpec(**{k: kw[k] for k in fields if k in kw})
    return reg
