"""Per-task-family output decoding.

Behavior-identical to the reference's decode branches
(/root/reference/worker.py:295-386) and result formatting
(worker.py:564-645), with the documented perf fixes:
  - answer vocab pickles are loaded ONCE and cached (the reference reloads
    trainval_label2ans.pkl from disk on every request — worker.py:300,315)
  - task_id parsed with int(), not eval() (worker.py:562)

Result JSON schemas (consumed by result.html:96-262):
  tasks 1/15/13 -> {"task_id": t, "result": [{"answer": a, "confidence": c} x3]}
  task 12       -> same with 2 entries (True/False)
  tasks 4/11/16/7 -> {"task_id": t, "image_name_list": [...], "confidence_list": [...]}
"""

from __future__ import annotations

import os
import pickle
from typing import Any, Dict, List, Optional, Sequence

import torch

from ..tasks import DecodeFamily, get_task


class AnswerVocab:
    """label -> answer-string table; pickle-compatible with the reference's
    save/VQA/cache/trainval_label2ans.pkl (a plain list of strings)."""

    def __init__(self, size: int, path: Optional[str] = None, prefix: str = "answer"):
        self.labels: List[str]
        if path and os.path.exists(path):
            with open(path, "rb") as f:
                self.labels = pickle.load(f)
        else:
            self.labels = [f"{prefix}_{i}" for i in range(size)]

    def __getitem__(self, i: int) -> str:
        return self.labels[i]

    def __len__(self) -> int:
        return len(self.labels)


_BINARY_ANSWERS = ["False", "True"]  # worker.py:325-338 (index 1 = True)
_TRI_ANSWERS = [  # exact reference label strings (worker.py:342)
    "contradiction (false)",
    "neutral",
    "entailment (true)",
]


def _topk_answers(logits: torch.Tensor, vocab: AnswerVocab, k: int = 3) -> List[Dict[str, Any]]:
    probs = torch.softmax(logits.float(), dim=-1)
    conf, idx = probs.topk(min(k, probs.shape[-1]))
    return [
        {"answer": vocab[int(i)], "confidence": float(c)}
        for c, i in zip(conf.tolist(), idx.tolist())
    ]


def decode_answer_task(
    task_id: int,
    outputs: Sequence[torch.Tensor],
    row: int,
    vqa_vocab: AnswerVocab,
    gqa_vocab: AnswerVocab,
) -> Dict[str, Any]:
    """Tasks 1/2/15 (answer vocab), 13 (entailment), 12 (binary pair)."""
    spec = get_task(task_id)
    if spec.decode == DecodeFamily.VQA:
        result = _topk_answers(outputs[0][row], vqa_vocab)
    elif spec.decode == DecodeFamily.GQA:
        result = _topk_answers(outputs[1][row], gqa_vocab)
    elif spec.decode == DecodeFamily.ENTAILMENT:
        probs = torch.softmax(outputs[4][row].float(), dim=-1)
        conf, idx = probs.sort(descending=True)
        result = [
            {"answer": _TRI_ANSWERS[int(i)], "confidence": float(c)}
            for c, i in zip(conf.tolist(), idx.tolist())
        ]
    elif spec.decode == DecodeFamily.BINARY:
        # row here indexes the PAIR dimension of vil_binary_prediction
        probs = torch.softmax(outputs[3][row].float(), dim=-1)
        conf, idx = probs.sort(descending=True)
        result = [
            {"answer": _BINARY_ANSWERS[int(i)], "confidence": float(c)}
            for c, i in zip(conf.tolist(), idx.tolist())
        ]
    else:
        raise ValueError(f"task {task_id} is not an answer task")
    return {"task_id": task_id, "result": result}


def decode_retrieval(
    task_id: int,
    outputs: Sequence[torch.Tensor],
    rows: Sequence[int],
    image_names: Sequence[str],
) -> Dict[str, Any]:
    """Task 7: softmax over the candidate images' vil_logit (worker.py:356-367)."""
    logits = outputs[2][list(rows), 0].float()
    probs = torch.softmax(logits, dim=0)
    conf, order = probs.sort(descending=True)
    return {
        "task_id": task_id,
        "image_name_list": [image_names[int(i)] for i in order.tolist()],
        "confidence_list": [float(c) for c in conf.tolist()],
    }


def decode_grounding(
    task_id: int,
    outputs: Sequence[torch.Tensor],
    row: int,
    spatials: torch.Tensor,
    image_width: float,
    image_height: float,
    k: int = 3,
) -> Dict[str, Any]:
    """Tasks 4/11/16: top-k region boxes from vision_logit, denormalized to
    pixel coords (worker.py:369-386). Returns boxes; the worker renders them
    into result images (worker.py:591-600 equivalent)."""
    scores = outputs[6][row, :, 0].float()
    # softmax across the 101 regions (worker.py:374) — NOT sigmoid: the
    # reference normalizes region confidence over the whole image.
    probs = torch.softmax(scores, dim=0)
    conf, idx = probs.topk(min(k, scores.shape[0]))
    boxes = []
    for i in idx.tolist():
        x1, y1, x2, y2 = spatials[row, i, :4].float().tolist()
        boxes.append(
            [x1 * image_width, y1 * image_height, x2 * image_width, y2 * image_height]
        )
    return {
        "task_id": task_id,
        "boxes": boxes,
        "confidence_list": [float(c) for c in conf.tolist()],
    }
