"""Golden-output tests of the serving contract (SURVEY.md §4 consequence (2)):
queue message in -> websocket messages + DB rows out, per-task result schemas
from worker.py:574-645."""

import json

import pytest
import torch

from vilbert_multi_task_amd.engine.runner import GraphRunner
from vilbert_multi_task_amd.serve.broker import Broker, vilbert_task
from vilbert_multi_task_amd.serve.db import Database
from vilbert_multi_task_amd.serve.features import SyntheticFeatureProvider
from vilbert_multi_task_amd.serve.push import NullPush
from vilbert_multi_task_amd.serve.worker import ServingWorker
from vilbert_multi_task_amd.data.tokenizer import BertWordPieceTokenizer
from vilbert_multi_task_amd.serve.decode import AnswerVocab


@pytest.fixture
def serving(tmp_path, tiny_model, tiny_config):
    broker = Broker(str(tmp_path / "q.sqlite3"), max_attempts=3)
    db = Database(str(tmp_path / "db.sqlite3"))
    db.seed_tasks()
    push = NullPush()
    runner = GraphRunner(
        tiny_model, device="cpu", use_graphs=False,
        feat_dim=tiny_config.v_feature_size,
    )
    worker = ServingWorker(
        runner, broker, db, push,
        provider=SyntheticFeatureProvider(feat_dim=tiny_config.v_feature_size),
        tokenizer=BertWordPieceTokenizer(vocab_size=tiny_config.vocab_size),
        vqa_vocab=AnswerVocab(tiny_config.num_labels_vqa),
        gqa_vocab=AnswerVocab(tiny_config.num_labels_gqa),
    )
    return broker, db, push, worker


def _result_messages(push, socket_id):
    return [
        p for s, p in push.messages if s == socket_id and "result" in p
    ]


def test_vqa_roundtrip(serving):
    broker, db, push, worker = serving
    vilbert_task(broker, ["/img/cat.jpg"], "What animal is this?", 1, "sock1")
    assert worker.process_once() == 1
    assert broker.depth() == 0
    res = _result_messages(push, "sock1")
    assert len(res) == 1
    payload = json.loads(res[0]["result"])
    # reference wire contract (worker.py:564-579): STRING task_id,
    # confidence scaled 0-100 and rounded to 2 decimals
    assert payload["task_id"] == "1"
    assert len(payload["result"]) == 3
    for entry in payload["result"]:
        assert set(entry) == {"answer", "confidence"}
        assert 0.0 <= entry["confidence"] <= 100.0
        assert entry["confidence"] == round(entry["confidence"], 2)
    # DB row saved (worker.py:548-552,579-645 contract)
    qa = db.get_question(1)
    assert qa["input_text"] == "what animal is this?"
    assert json.loads(qa["answer_text"])["task_id"] == "1"
    # terminal messages bracket the result (worker.py:647-649)
    terms = [p for s, p in push.messages if s == "sock1" and "terminal" in p]
    # exact reference completion push (worker.py:649)
    assert terms[-1]["terminal"] == "Completed Task"


def test_nlvr2_pair(serving):
    broker, db, push, worker = serving
    vilbert_task(broker, ["/a.jpg", "/b.jpg"], "both images contain dogs", 12, "s12")
    assert worker.process_once() == 1
    payload = json.loads(_result_messages(push, "s12")[0]["result"])
    answers = {e["answer"] for e in payload["result"]}
    assert answers == {"True", "False"}
    assert abs(sum(e["confidence"] for e in payload["result"]) - 100.0) < 1e-2


def test_retrieval_orders_all_images(serving):
    broker, db, push, worker = serving
    imgs = [f"/imgs/{i}.jpg" for i in range(4)]
    vilbert_task(broker, imgs, "a dog on a beach", 7, "s7")
    assert worker.process_once() == 1
    payload = json.loads(_result_messages(push, "s7")[0]["result"])
    assert payload["task_id"] == "7"
    # worker.py:631-635 name format: test2014/<basename>.<ext of first image>
    assert sorted(payload["image_name_list"]) == sorted(
        f"test2014/{i}.jpg" for i in range(4)
    )
    confs = payload["confidence_list"]
    assert confs == sorted(confs, reverse=True)
    assert abs(sum(confs) - 100.0) < 1e-2


def test_grounding_boxes(serving):
    broker, db, push, worker = serving
    vilbert_task(broker, ["/img/street.jpg"], "the red car", 11, "s11")
    assert worker.process_once() == 1
    payload = json.loads(_result_messages(push, "s11")[0]["result"])
    # reference grounding wire schema (worker.py:599-604): task_id +
    # image_name_list (bare uuids, no path/extension) + confidence_list
    assert payload["task_id"] == "11"
    assert set(payload) == {"task_id", "image_name_list", "confidence_list"}
    assert len(payload["image_name_list"]) == 3
    for name in payload["image_name_list"]:
        assert "/" not in name and not name.endswith(".jpg")
    assert all(0.0 <= c <= 100.0 for c in payload["confidence_list"])


def test_mixed_task_batch(serving):
    """BASELINE.json config 5: concurrent mixed-task dynamic batching."""
    broker, db, push, worker = serving
    vilbert_task(broker, ["/1.jpg"], "what is this?", 1, "m1")
    vilbert_task(broker, ["/2.jpg", "/3.jpg"], "both have cats", 12, "m2")
    vilbert_task(broker, ["/4.jpg"], "the blue ball", 11, "m3")
    vilbert_task(broker, ["/5.jpg"], "is it raining?", 15, "m4")
    assert worker.process_once() == 4
    assert broker.depth() == 0
    for sid, tid in [("m1", 1), ("m2", 12), ("m3", 11), ("m4", 15)]:
        payload = json.loads(_result_messages(push, sid)[0]["result"])
        assert payload["task_id"] == str(tid)


def test_invalid_image_count_rejected(serving):
    """worker.py:256-263 arity check, incl. the task-2 dead path."""
    broker, db, push, worker = serving
    vilbert_task(broker, ["/a.jpg", "/b.jpg"], "what is this?", 1, "bad1")
    vilbert_task(broker, ["/a.jpg"], "vg question", 2, "bad2")  # dead path
    assert worker.process_once() == 0
    assert broker.depth() == 0  # refused + acked, no redelivery
    errs = [p for s, p in push.messages if "Error" in str(p.get("terminal", ""))]
    assert len(errs) == 2


def test_broker_redelivery_and_dead_letter(tmp_path):
    broker = Broker(str(tmp_path / "q.sqlite3"), lease_timeout_s=0.0, max_attempts=2)
    broker.publish({"x": 1})
    d1 = broker.get()[0]
    broker.nack(d1.msg_id)  # crash analogue -> redelivered (worker.py:653-655)
    d2 = broker.get()[0]
    assert d2.msg_id == d1.msg_id and d2.attempts == 2
    broker.nack(d2.msg_id)
    assert broker.get() == []  # attempt cap -> dead letter, not a poison loop
    assert broker.dead_count() == 1


def test_broker_lease_expiry(tmp_path):
    broker = Broker(str(tmp_path / "q.sqlite3"), lease_timeout_s=0.0)
    broker.publish({"x": 1})
    assert len(broker.get()) == 1
    # consumer died without ack; lease expired -> ready again
    assert len(broker.get()) == 1


def test_tokenizer_padding_contract():
    tok = BertWordPieceTokenizer()
    ids, mask, seg = tok.encode_for_serving("what color is the sky?")
    assert len(ids) == 37 and len(mask) == 37 and len(seg) == 37
    assert ids[0] == 101  # [CLS]
    sep_pos = ids.index(102)
    # END padding after [SEP] (worker.py:408-414 code, not its comment)
    assert all(i == 0 for i in ids[sep_pos + 1 :])
    assert all(m == 0 for m in mask[sep_pos + 1 :])
    assert all(m == 1 for m in mask[: sep_pos + 1])
    assert seg == [0] * 37


def test_serving_fast_mode_same_answers(tmp_path, tiny_model, tiny_config):
    """skip_unused_heads must not change any decodable output."""
    import torch
    from vilbert_multi_task_amd.data.synthetic import forward_args, synthetic_batch

    batch = synthetic_batch(
        2, seq_len=20, regions=36, feat_dim=tiny_config.v_feature_size,
        vocab_size=tiny_config.vocab_size,
    )
    with torch.no_grad():
        full = tiny_model(*forward_args(batch))
    tiny_model.skip_unused_heads = True
    try:
        with torch.no_grad():
            fast = tiny_model(*forward_args(batch))
    finally:
        tiny_model.skip_unused_heads = False
    for i in (0, 1, 2, 3, 4, 6):  # every head the decode reads
        assert torch.equal(full[i], fast[i]), i
    assert fast[7].shape[-1] == 0  # placeholders


def test_multi_worker_competing_consumers(tmp_path, tiny_model, tiny_config):
    """Scale-out model: N workers share one durable queue;
    every request is served exactly once."""
    broker_a = Broker(str(tmp_path / "shared.sqlite3"))
    broker_b = Broker(str(tmp_path / "shared.sqlite3"))
    db = Database(str(tmp_path / "db2.sqlite3"))
    push = NullPush()

    def make_worker(br):
        runner = GraphRunner(
            tiny_model, device="cpu", use_graphs=False,
            feat_dim=tiny_config.v_feature_size,
        )
        return ServingWorker(
            runner, br, db, push,
            provider=SyntheticFeatureProvider(feat_dim=tiny_config.v_feature_size),
            tokenizer=BertWordPieceTokenizer(vocab_size=tiny_config.vocab_size),
            vqa_vocab=AnswerVocab(tiny_config.num_labels_vqa),
            gqa_vocab=AnswerVocab(tiny_config.num_labels_gqa),
            max_batch_rows=3,
        )

    wa, wb = make_worker(broker_a), make_worker(broker_b)
    for i in range(8):
        vilbert_task(broker_a, [f"/i{i}.jpg"], f"q {i}", 1, f"mw{i}")
    served = 0
    for _ in range(10):
        served += wa.process_once() + wb.process_once()
        if served >= 8:
            break
    assert served == 8
    assert broker_a.depth() == 0
    assert len([p for s, p in push.messages if "result" in p]) == 8


def test_guesswhat_dialog_rewrite():
    from vilbert_multi_task_amd.serve.worker import guesswhat_dialog_rewrite

    # the rewrite the reference computes (then discards) at worker.py:391-400
    q = "q: is it a person? a: yes q: on the left? a: no"
    assert guesswhat_dialog_rewrite(q) == (
        "start is it a person? answer yes stop "
        "start on the left? answer no stop"
    )
    # no dialog markers -> unchanged (plain questions pass through)
    assert guesswhat_dialog_rewrite("what is this") == "what is this"
    # unanswered trailing question
    assert guesswhat_dialog_rewrite("q: is it red?") == "start is it red? answer  stop"


def test_tensorize_regions_mixed_box_counts():
    """Slow path: images with fewer boxes are zero-padded and masked, and
    assembly stays on the features' device (detector output case)."""
    from vilbert_multi_task_amd.serve.features import tensorize_regions

    def info(nb):
        return {
            "features": torch.randn(nb, 64).abs(),
            "bbox": torch.tensor([[0.0, 0.0, 320.0, 240.0]] * nb),
            "image_width": 640.0,
            "image_height": 480.0,
            "num_boxes": nb,
        }

    out = tensorize_regions([info(10), info(5)], num_regions=12)
    assert out["features"].shape == (2, 12, 64)
    assert out["image_mask"][0].sum() == 11 and out["image_mask"][1].sum() == 6
    assert torch.all(out["features"][1, 7:] == 0)
    # global box row
    assert torch.allclose(out["spatials"][0, 0], torch.tensor([0.0, 0, 1, 1, 1]))
    assert torch.allclose(out["spatials"][1, 1, :4], torch.tensor([0.0, 0.0, 0.5, 0.5]))


def test_mixed_batch_nlvr2_first_even_alignment(serving):
    """A drained batch mixing NLVR2 (2-row) and single-row tasks places the
    pair requests first at even row offsets (worker gather contract — the
    pair head reads consecutive even/odd rows)."""
    broker, db, push, worker = serving
    from vilbert_multi_task_amd.serve.broker import vilbert_task

    vilbert_task(broker, ["/x.jpg"], "what is this", 1, "s1")
    vilbert_task(broker, ["/a.jpg", "/b.jpg"], "true or false", 12, "s2")
    vilbert_task(broker, ["/y.jpg"], "entailment?", 13, "s3")
    vilbert_task(broker, ["/c.jpg", "/d.jpg"], "left image has", 12, "s4")
    reqs = worker.gather_batch(0.0)
    assert len(reqs) == 4
    # NLVR2 requests first, each starting on an even row
    kinds = [r.task_id for r in reqs]
    assert kinds[:2] == [12, 12]
    for r in reqs:
        if r.task_id == 12:
            assert r.row_start % 2 == 0 and r.num_rows == 2
    # the gathered requests were leased (unacked); nack them back so the
    # full-path check below serves a fresh equivalent mix
    for r in reqs:
        broker.nack(r.delivery.msg_id)
    n = 0
    while n < 4:
        n += worker.process_once()


def test_prometheus_metrics_exposition(serving):
    """Metrics endpoint serves the serving counters (SURVEY.md §5 build
    obligation: Prometheus observability the reference lacks)."""
    import urllib.request

    from vilbert_multi_task_amd.serve.broker import vilbert_task
    from vilbert_multi_task_amd.utils.trace import start_metrics_server

    broker, db, push, worker = serving
    port = start_metrics_server(0)
    if port is None:
        import pytest as _pytest

        _pytest.skip("prometheus_client not importable")
    vilbert_task(broker, ["/m.jpg"], "metrics?", 1, "sm")
    assert worker.process_once() == 1
    body = urllib.request.urlopen(
        f"http://127.0.0.1:{port}/metrics", timeout=5
    ).read().decode()
    assert "vilbert_requests_total" in body
    assert "vilbert_batch_rows" in body


def test_runner_chunked_replay_concatenation(tiny_model, tiny_config):
    """Batches above max_bucket split into even chunks whose outputs are
    torch.cat-ed, with the pair head (index 3) carrying n_chunk/2 rows
    (the fp8 bucket-cap path in engine/runner.py)."""
    from vilbert_multi_task_amd.data.synthetic import synthetic_batch
    from vilbert_multi_task_amd.engine.runner import GraphRunner

    runner = GraphRunner(
        tiny_model, device="cpu", use_graphs=False,
        feat_dim=tiny_config.v_feature_size, seq_len=20, regions=12,
    )
    runner.use_graphs = True  # force the chunk branch of run()
    runner.max_bucket = 4
    seen = []
    orig_run = GraphRunner.run

    def spy(self, batch):
        n = batch["question"].shape[0]
        if n <= self.max_bucket:  # leaf chunk: serve eagerly, record size
            seen.append(n)
            self.use_graphs = False
            try:
                return orig_run(self, batch)
            finally:
                self.use_graphs = True
        return orig_run(self, batch)

    b = synthetic_batch(
        10, seq_len=20, regions=12, feat_dim=tiny_config.v_feature_size,
        vocab_size=tiny_config.vocab_size,
    )
    import unittest.mock as mock

    with mock.patch.object(GraphRunner, "run", spy):
        out = GraphRunner.run(runner, b)
    assert seen == [4, 4, 2]
    assert out[0].shape[0] == 10          # row-wise heads concatenated
    assert out[3].shape[0] == 5           # pair head: 2+2+1 rows
    # chunked result equals the unchunked eager forward
    runner.use_graphs = False
    ref = runner.run(b)
    assert torch.allclose(out[0], ref[0], atol=1e-5)


def test_guesswhat_strict_parity_default(serving, monkeypatch):
    """Default behavior matches the reference's OBSERVABLE semantics: the
    q:/a: rewrite is computed-then-discarded there (worker.py:391-402), so
    the RAW query reaches the tokenizer; VILBERT_GUESSWHAT_REWRITE=1 opts
    into the intended rewrite."""
    import vilbert_multi_task_amd.serve.worker as worker_mod

    broker, db, push, worker = serving
    seen = []
    orig = worker.tokenizer.encode_for_serving

    def spy(text, max_len):
        seen.append(text)
        return orig(text, max_len)

    worker.tokenizer.encode_for_serving = spy
    q = "q: is it a person? a: yes"

    monkeypatch.delenv("VILBERT_GUESSWHAT_REWRITE", raising=False)
    vilbert_task(broker, ["/g.jpg"], q, 16, "gw1")
    assert worker.process_once() == 1
    assert seen[-1] == q  # raw query (reference-exact)

    monkeypatch.setenv("VILBERT_GUESSWHAT_REWRITE", "1")
    worker._tok_cache.clear()
    vilbert_task(broker, ["/g.jpg"], q, 16, "gw2")
    assert worker.process_once() == 1
    assert seen[-1] == "start is it a person? answer yes stop"
