"""Per-task-family output decoding.

Behavior-identical to the reference's decode branches
(/root/reference/worker.py:295-386) AND its wire formatting
(worker.py:564-645), with the documented perf fixes:
  - answer vocab pickles are loaded ONCE and cached (the reference reloads
    trainval_label2ans.pkl from disk on every request — worker.py:300,315)
  - task_id parsed with int(), not eval() (worker.py:562)

Wire contract (exactly what the reference pushes over the websocket and
stores in the DB; consumed by result.html:96-262):
  - task_id is a STRING (the reference round-trips the message's string id)
  - confidence is scaled to 0-100 and rounded to 2 decimals
    (worker.py:569,585,597,...: ``round(conf*100, 2)``; result.html renders
    ``confidence + "%"``)
  tasks 1/15/13 -> {"task_id": "t", "result": [{"answer": a, "confidence": c} x3]}
  task 12       -> same with 2 entries (True/False)
  tasks 4/11/16 -> {"task_id": "t", "image_name_list": [uuid x3],
                    "confidence_list": [c x3]}   (bare uuids, files written to
                    media/refer_expressions_task/<uuid>.jpg — worker.py:591-600)
  task 7        -> {"task_id": "7", "image_name_list": ["demo/x.jpg" |
                    "test2014/x.jpg" ...], "confidence_list": [...]}
                   (worker.py:625-639 name format)
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


def _pct(p: float) -> float:
    """worker.py:569 etc.: round(confidence * 100, 2)."""
    return round(p * 100, 2)


def _topk_answers(logits: torch.Tensor, vocab: AnswerVocab, k: int = 3) -> List[Dict[str, Any]]:
    probs = torch.softmax(logits.float(), dim=-1)
    conf, idx = probs.topk(min(k, probs.shape[-1]))
    return [
        {"answer": vocab[int(i)], "confidence": _pct(float(c))}
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
            {"answer": _TRI_ANSWERS[int(i)], "confidence": _pct(float(c))}
            for c, i in zip(conf.tolist(), idx.tolist())
        ]
    elif spec.decode == DecodeFamily.BINARY:
        # row here indexes the PAIR dimension of vil_binary_prediction
        probs = torch.softmax(outputs[3][row].float(), dim=-1)
        conf, idx = probs.sort(descending=True)
        result = [
            {"answer": _BINARY_ANSWERS[int(i)], "confidence": _pct(float(c))}
            for c, i in zip(conf.tolist(), idx.tolist())
        ]
    else:
        raise ValueError(f"task {task_id} is not an answer task")
    return {"task_id": str(task_id), "result": result}


def retrieval_image_name(ranked_path: str, first_path: str) -> str:
    """worker.py:631-635 name format: '<demo|test2014>/<basename-no-ext>.<ext>'
    where the prefix comes from whether 'demo' appears in the FIRST image's
    path components and the extension is the FIRST image's (reference quirk:
    every entry carries image_path[0]'s extension)."""
    prefix = "demo" if "demo" in first_path.split("/") else "test2014"
    base = os.path.split(ranked_path)[1].split(".")[0]
    ext = first_path.split("/")[-1].split(".")[1] if "." in first_path.split("/")[-1] else "jpg"
    return f"{prefix}/{base}.{ext}"


def decode_retrieval(
    task_id: int,
    outputs: Sequence[torch.Tensor],
    rows: Sequence[int],
    image_paths: Sequence[str],
) -> Dict[str, Any]:
    """Task 7: softmax over the candidate images' vil_logit (worker.py:356-367),
    names formatted per worker.py:625-639."""
    logits = outputs[2][list(rows), 0].float()
    probs = torch.softmax(logits, dim=0)
    conf, order = probs.sort(descending=True)
    return {
        "task_id": str(task_id),
        "image_name_list": [
            retrieval_image_name(image_paths[int(i)], image_paths[0])
            for i in order.tolist()
        ],
        "confidence_list": [_pct(float(c)) for c in conf.tolist()],
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
    INT pixel coords (worker.py:369-386). Returns boxes for the renderer; the
    worker draws them into result images, fills image_name_list with the bare
    uuid names (worker.py:591-600) and strips "boxes" from the wire payload
    (the reference result carries only task_id/image_name_list/confidence_list)."""
    scores = outputs[6][row, :, 0].float()
    # softmax across the 101 regions (worker.py:374) — NOT sigmoid: the
    # reference normalizes region confidence over the whole image.
    probs = torch.softmax(scores, dim=0)
    conf, idx = probs.topk(min(k, scores.shape[0]))
    boxes = []
    for i in idx.tolist():
        x1, y1, x2, y2 = spatials[row, i, :4].float().tolist()
        # worker.py:381-384: int() truncation of the denormalized coords
        boxes.append(
            [int(x1 * image_width), int(y1 * image_height),
             int(x2 * image_width), int(y2 * image_height)]
        )
    return {
        "task_id": str(task_id),
        "boxes": boxes,
        "confidence_list": [_pct(float(c)) for c in conf.tolist()],
    }
