"""The serving worker: queue consumer with dynamic cross-task batching.

Replaces the reference's one-message-at-a-time blocking consumer
(/root/reference/worker.py:542-673) with a polling consumer that drains up
to ``max_batch_rows`` of queued requests per cycle and runs them through ONE
hipGraph-replayed forward (BASELINE.json config 5: mixed-task dynamic
batching). Everything user-observable is contract-identical:

  message in : {"image_path": [...], "question": str, "socket_id": str,
                "task_id": str}                     (sender.py:19-24)
  push out   : {"terminal": json}, {"result": json}, {"terminal": "Completed..."}
                                                    (worker.py:647-649)
  DB rows    : questionanswer insert before inference, answer update after
                                                    (worker.py:548-552,579-645)
  ack        : only after success; failures redeliver (worker.py:650,653-655)
                + dead-letter cap (broker.py) fixing the poison-message loop.

Batch layout rule: NLVR2 (task 12) messages are placed FIRST so their two
rows land at an even offset — the pair head concatenates rows (2i, 2i+1)
(models/heads.py), matching the reference's pair batching (worker.py:266-276).
"""

from __future__ import annotations

import json
import time
import traceback
import uuid
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence

import torch

from ..data.tokenizer import BertWordPieceTokenizer
from ..tasks import MAX_SEQ_LENGTH, get_task, validate_request, DecodeFamily
from .broker import Broker, Delivery, QUEUE_NAME
from .db import Database
from .decode import (
    AnswerVocab,
    decode_answer_task,
    decode_grounding,
    decode_retrieval,
)
from .features import SyntheticFeatureProvider, tensorize_regions
from .push import PushClient, log_to_terminal


def guesswhat_rewrite_enabled() -> bool:
    """The reference computes the GuessWhat ``q:/a:`` rewrite and then
    discards it (worker.py:391-402 overwrites ``tokens`` with the raw
    query), so its OBSERVABLE behavior feeds the raw query to the model.
    Default here is reference-exact (raw query); set
    ``VILBERT_GUESSWHAT_REWRITE=1`` to apply the rewrite the reference
    clearly intended (opt-in fix, both behaviors tested)."""
    import os

    return os.environ.get("VILBERT_GUESSWHAT_REWRITE", "0") == "1"


def guesswhat_dialog_rewrite(query: str) -> str:
    """Rewrite a GuessWhat dialog ``q: ..? a: ..`` into the
    ``start <q> answer <a> stop`` token stream the model expects.

    This is the string the reference builds (worker.py:391-400) and then
    accidentally overwrites with the raw query (worker.py:402) — dead code /
    latent bug flagged in SURVEY.md §2.1. Applied only when
    ``guesswhat_rewrite_enabled()`` (default: reference-exact, i.e. NOT
    applied). Queries without a ``q:`` marker pass unchanged either way.
    """
    if "q:" not in query:
        return query
    parts = []
    for dialog in query.split("q:")[1:]:
        qa = dialog.split("a:")
        a = qa[1] if len(qa) > 1 else ""
        parts.append("start " + qa[0].strip() + " answer " + a.strip() + " stop")
    return " ".join(parts)


@dataclass
class _Request:
    delivery: Delivery
    task_id: int
    question: str
    image_paths: List[str]
    socket_id: str
    qa_id: int = -1
    row_start: int = 0
    num_rows: int = 0
    infos: List[Dict] = field(default_factory=list)


class ServingWorker:
    def __init__(
        self,
        runner,
        broker: Broker,
        db: Database,
        push: PushClient,
        provider=None,
        tokenizer: Optional[BertWordPieceTokenizer] = None,
        vqa_vocab: Optional[AnswerVocab] = None,
        gqa_vocab: Optional[AnswerVocab] = None,
        max_batch_rows: int = 64,
        queue: str = QUEUE_NAME,
    ):
        self.runner = runner
        self.broker = broker
        self.db = db
        self.push = push
        self.provider = provider or SyntheticFeatureProvider()
        self.tokenizer = tokenizer or BertWordPieceTokenizer()
        self.vqa_vocab = vqa_vocab or AnswerVocab(3129)
        self.gqa_vocab = gqa_vocab or AnswerVocab(1533)
        self.max_batch_rows = max_batch_rows
        self.queue = queue
        self._tok_cache: Dict[str, tuple] = {}

    # ------------------------------------------------------------------
    def _parse(self, d: Delivery) -> Optional[_Request]:
        body = d.body
        try:
            task_id = int(body["task_id"])  # int(), not eval() (worker.py:562)
            req = _Request(
                delivery=d,
                task_id=task_id,
                question=str(body["question"]).lower(),
                image_paths=list(body["image_path"]),
                socket_id=str(body["socket_id"]),
            )
        except (KeyError, ValueError, TypeError):
            self.broker.ack(d.msg_id)  # malformed: drop
            return None
        err = validate_request(req.task_id, len(req.image_paths))
        if err:
            log_to_terminal(self.push, req.socket_id, {"terminal": f"Error: {err}"})
            self.broker.ack(d.msg_id)
            return None
        return req

    # ------------------------------------------------------------------
    def gather_batch(self, max_wait_s: float = 0.0) -> List[_Request]:
        deadline = time.time() + max_wait_s
        reqs: List[_Request] = []
        rows = 0
        while rows < self.max_batch_rows:
            deliveries = self.broker.get(self.queue, max_n=self.max_batch_rows - rows)
            for d in deliveries:
                r = self._parse(d)
                if r is None:
                    continue
                r.num_rows = len(r.image_paths)
                reqs.append(r)
                rows += r.num_rows
            if deliveries or time.time() >= deadline:
                break
            time.sleep(0.005)
        # NLVR2 first for even pair alignment
        reqs.sort(key=lambda r: 0 if r.task_id == 12 else 1)
        off = 0
        for r in reqs:
            r.row_start = off
            off += r.num_rows
        return reqs

    # ------------------------------------------------------------------
    def build_batch(self, reqs: Sequence[_Request]) -> Dict[str, torch.Tensor]:
        """Tokenize + tensorize a drained request group into one model batch.

        Text follows the reference serving contract (worker.py:402-414):
        wordpiece + [CLS]/[SEP], END-padded to 37 (the reference's comment
        claims front-padding but its code pads at the end — we mirror the
        code). GuessWhat dialogs pass through RAW by default (the reference
        computes the ``q:/a:`` rewrite but discards it — worker.py:391-402);
        VILBERT_GUESSWHAT_REWRITE=1 opts into the intended rewrite.
        """
        q_rows, mask_rows, seg_rows, task_rows = [], [], [], []
        infos_all: List[Dict] = []
        rewrite_gw = guesswhat_rewrite_enabled()
        for r in reqs:
            text = r.question
            if r.task_id == 16 and rewrite_gw:
                text = guesswhat_dialog_rewrite(text)
            cached = self._tok_cache.get(text)
            if cached is None:
                cached = self.tokenizer.encode_for_serving(text, MAX_SEQ_LENGTH)
                if len(self._tok_cache) < 8192:
                    self._tok_cache[text] = cached
            ids, mask, seg = cached
            r.infos = self.provider.extract(r.image_paths)
            infos_all.extend(r.infos)
            for _ in range(r.num_rows):  # text replicated per image row
                q_rows.append(ids)
                mask_rows.append(mask)
                seg_rows.append(seg)
                task_rows.append([r.task_id])
        reg = tensorize_regions(infos_all)
        n = len(q_rows)
        return {
            "question": torch.tensor(q_rows, dtype=torch.long),
            "input_mask": torch.tensor(mask_rows, dtype=torch.long),
            "segment_ids": torch.tensor(seg_rows, dtype=torch.long),
            "task_tokens": torch.tensor(task_rows, dtype=torch.long),
            "features": reg["features"],
            "spatials": reg["spatials"],
            "image_mask": reg["image_mask"],
            "co_attention_mask": torch.zeros(n, reg["features"].shape[1], MAX_SEQ_LENGTH),
        }

    # ------------------------------------------------------------------
    def decode_request(self, r: _Request, outputs, batch) -> Dict[str, Any]:
        spec = get_task(r.task_id)
        row = r.row_start
        if spec.decode == DecodeFamily.RETRIEVAL:
            return decode_retrieval(
                r.task_id, outputs, range(row, row + r.num_rows), r.image_paths
            )
        if spec.decode == DecodeFamily.GROUNDING:
            info = r.infos[0]
            result = decode_grounding(
                r.task_id, outputs, row, batch["spatials"],
                info["image_width"], info["image_height"],
            )
            # worker.py:591-600 contract: image_name_list carries BARE uuid
            # strings; files land at media/refer_expressions_task/<uuid>.jpg.
            result["image_name_list"] = self.render_grounding_images(
                r.image_paths[0], result.pop("boxes")
            )
            return result
        if spec.decode == DecodeFamily.BINARY:
            return decode_answer_task(
                r.task_id, outputs, row // 2, self.vqa_vocab, self.gqa_vocab
            )
        return decode_answer_task(r.task_id, outputs, row, self.vqa_vocab, self.gqa_vocab)

    # ------------------------------------------------------------------
    def process_once(self, max_wait_s: float = 0.0) -> int:
        """One drain-batch-infer-respond cycle; returns #requests served."""
        from ..utils.trace import RequestTrace, get_metrics

        trace = RequestTrace()
        with trace.stage("gather"):
            reqs = self.gather_batch(max_wait_s)
        if not reqs:
            return 0
        m = get_metrics()
        with trace.stage("db_insert"):
            qa_ids = self.db.create_questions(
                [
                    (r.task_id, r.question, json.dumps(r.image_paths), r.socket_id)
                    for r in reqs
                ]
            )
        for r, qa in zip(reqs, qa_ids):
            r.qa_id = qa
        try:
            with trace.stage("build_batch"):
                batch = self.build_batch(reqs)
            with trace.stage("forward"):
                outputs = self.runner.run(batch)
        except Exception:
            traceback.print_exc()
            for r in reqs:
                self.broker.nack(r.delivery.msg_id)
                m.requests_total.labels(str(r.task_id), "error").inc()
            return 0
        served = 0
        answers = []
        acks = []
        with trace.stage("decode"):
            # ONE device->host sync for the whole batch (per-request .tolist()
            # was 12 ms of a 15 ms cycle: a GPU round trip per request)
            outputs = tuple(
                o.float().cpu() if torch.is_tensor(o) and o.is_cuda else o
                for o in outputs
            )
            if torch.is_tensor(batch.get("spatials")) and batch["spatials"].is_cuda:
                batch = dict(batch, spatials=batch["spatials"].float().cpu())
            for r in reqs:
                try:
                    result = self.decode_request(r, outputs, batch)
                    payload = json.dumps(result)
                    answers.append((r.qa_id, payload))
                    log_to_terminal(self.push, r.socket_id, {"terminal": payload})
                    log_to_terminal(self.push, r.socket_id, {"result": payload})
                    # exact reference completion string (worker.py:649)
                    log_to_terminal(
                        self.push, r.socket_id, {"terminal": "Completed Task"}
                    )
                    acks.append(r.delivery.msg_id)
                    m.requests_total.labels(str(r.task_id), "ok").inc()
                    served += 1
                except Exception:
                    traceback.print_exc()
                    self.broker.nack(r.delivery.msg_id)
                    m.requests_total.labels(str(r.task_id), "error").inc()
        with trace.stage("db_commit"):
            self.db.save_answers(answers)
            self.broker.ack_many(acks)
        rows = sum(r.num_rows for r in reqs)
        m.batch_rows.observe(rows)
        m.request_latency.observe(trace.total_ms() / 1e3)
        m.queue_depth.set(self.broker.depth(self.queue))
        trace.report(
            "serving_batch", requests=len(reqs), rows=rows, served=served,
            tasks=[r.task_id for r in reqs],
            request_traces=[r.delivery.body.get("trace_id") for r in reqs],
        )
        return served

    # per-box outline colors: the reference's cv2 BGR list
    # [(0,0,255),(0,255,0),(255,0,0)] (worker.py:589) = red/green/blue in RGB
    _BOX_COLORS = [(255, 0, 0), (0, 255, 0), (0, 0, 255)]

    def render_grounding_images(self, image_path: str, boxes) -> List[str]:
        """Draw top-k grounding boxes into result JPEGs
        (worker.py:591-600 contract: one image per box at
        media/refer_expressions_task/<uuid>.jpg; the returned names are the
        BARE uuid strings the reference puts in image_name_list; PIL instead
        of cv2, line width 4 like the reference). A missing source image
        still yields the uuid name list (the file write is skipped) so the
        wire contract shape is independent of disk state."""
        import os

        names = [str(uuid.uuid4()) for _ in boxes]
        if not os.path.exists(image_path):
            return names
        try:
            from PIL import Image, ImageDraw

            out_dir = os.path.join("media", "refer_expressions_task")
            os.makedirs(out_dir, exist_ok=True)
            base = Image.open(image_path).convert("RGB")
            for name, box, color in zip(names, boxes, self._BOX_COLORS):
                img = base.copy()
                ImageDraw.Draw(img).rectangle(
                    [box[0], box[1], box[2], box[3]], outline=color, width=4
                )
                img.save(os.path.join(out_dir, f"{name}.jpg"), "JPEG")
        except Exception:
            traceback.print_exc()
        return names

    def run_forever(self, poll_s: float = 0.02) -> None:
        """Blocking consume loop; SIGTERM/SIGINT finish the in-flight batch
        then exit (unacked messages redeliver — at-least-once). Transient
        errors (e.g. sqlite contention under competing consumers) are logged
        and retried instead of killing the worker process — leased messages
        redeliver after the lease timeout either way."""
        import signal

        stop = {"flag": False}

        def _sig(_s, _f):
            stop["flag"] = True

        try:
            signal.signal(signal.SIGTERM, _sig)
            signal.signal(signal.SIGINT, _sig)
        except ValueError:
            pass  # not the main thread
        while not stop["flag"]:
            try:
                served = self.process_once(max_wait_s=poll_s)
            except Exception:
                traceback.print_exc()
                served = 0
                time.sleep(poll_s * 5)
            if served == 0:
                time.sleep(poll_s)
