"""Full-topology integration: uvicorn app (with TCP PushHub) + a worker in
ANOTHER process publishing through PushClient — the production layout of
serve/main.py `app` + `worker`, covering the cross-process push path that
in-process tests bypass."""

import base64
import json
import multiprocessing as mp
import os
import socket
import struct
import threading
import time
import urllib.request

import pytest

uvicorn = pytest.importorskip("uvicorn")


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _worker_proc(queue_path, db_path, hub_port):
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch

    from vilbert_multi_task_amd.config import ViLBertConfig
    from vilbert_multi_task_amd.data.tokenizer import BertWordPieceTokenizer
    from vilbert_multi_task_amd.engine.runner import GraphRunner
    from vilbert_multi_task_amd.models import VILBertForVLTasks
    from vilbert_multi_task_amd.serve.broker import Broker
    from vilbert_multi_task_amd.serve.db import Database
    from vilbert_multi_task_amd.serve.decode import AnswerVocab
    from vilbert_multi_task_amd.serve.features import SyntheticFeatureProvider
    from vilbert_multi_task_amd.serve.push import PushClient
    from vilbert_multi_task_amd.serve.worker import ServingWorker

    cfg = ViLBertConfig.tiny()
    torch.manual_seed(0)
    model = VILBertForVLTasks(cfg).eval()
    runner = GraphRunner(model, device="cpu", use_graphs=False, feat_dim=cfg.v_feature_size)
    worker = ServingWorker(
        runner,
        Broker(queue_path),
        Database(db_path),
        PushClient(port=hub_port),
        provider=SyntheticFeatureProvider(feat_dim=cfg.v_feature_size),
        tokenizer=BertWordPieceTokenizer(vocab_size=cfg.vocab_size),
        vqa_vocab=AnswerVocab(cfg.num_labels_vqa),
        gqa_vocab=AnswerVocab(cfg.num_labels_gqa),
    )
    deadline = time.time() + 30
    while time.time() < deadline:
        if worker.process_once():
            return
        time.sleep(0.05)


def _ws_connect(port):
    s = socket.create_connection(("127.0.0.1", port), timeout=10)
    key = base64.b64encode(os.urandom(16)).decode()
    s.sendall(
        (f"GET /chat/ HTTP/1.1\r\nHost: x\r\nUpgrade: websocket\r\n"
         f"Connection: Upgrade\r\nSec-WebSocket-Key: {key}\r\n"
         f"Sec-WebSocket-Version: 13\r\n\r\n").encode()
    )
    assert s.recv(4096).startswith(b"HTTP/1.1 101")
    return s


def _ws_send(sock, msg):
    p = msg.encode()
    mask = os.urandom(4)
    sock.sendall(bytes([0x81, 0x80 | len(p)]) + mask +
                 bytes(b ^ mask[i % 4] for i, b in enumerate(p)))


def _ws_recv(sock):
    h = sock.recv(2)
    ln = h[1] & 0x7F
    if ln == 126:
        ln = struct.unpack(">H", sock.recv(2))[0]
    d = b""
    while len(d) < ln:
        d += sock.recv(ln - len(d))
    return d.decode()


@pytest.mark.timeout(180)
def test_http_worker_ws_across_processes(tmp_path):
    from vilbert_multi_task_amd.serve.app import create_app

    http_port, hub_port = _free_port(), _free_port()
    queue_path = str(tmp_path / "q.sqlite3")
    db_path = str(tmp_path / "db.sqlite3")
    app = create_app(
        db_path=db_path, queue_path=queue_path,
        media_root=str(tmp_path / "media"), hub_port=hub_port,
    )
    config = uvicorn.Config(
        app, host="127.0.0.1", port=http_port, log_level="error",
        ws="vilbert_multi_task_amd.serve.ws_protocol:MinimalWebSocketProtocol",
    )
    srv = uvicorn.Server(config)
    th = threading.Thread(target=srv.run, daemon=True)
    th.start()
    for _ in range(100):
        if srv.started:
            break
        time.sleep(0.05)

    try:
        # full demo loop: take the socketid the SERVED PAGE injects
        # (views.py:39-42 context contract; app.js sends it on ws open)
        import re

        page = urllib.request.urlopen(
            f"http://127.0.0.1:{http_port}/", timeout=10
        ).read().decode()
        m = re.search(r'data-socketid="([0-9a-f-]{36})"', page)
        assert m, "served page must carry a fresh socketid"
        sock_id = m.group(1)
        assert 'id="selected-task"' in page  # frontend really served

        ws = _ws_connect(http_port)
        _ws_send(ws, sock_id)
        time.sleep(0.3)

        ctx = mp.get_context("spawn")
        proc = ctx.Process(target=_worker_proc, args=(queue_path, db_path, hub_port))
        proc.start()

        body = (f"socket_id={sock_id}&task_id=1&question=what+is+this"
                "&image_list%5B%5D=demo/z.jpg").encode()
        urllib.request.urlopen(urllib.request.Request(
            f"http://127.0.0.1:{http_port}/", data=body,
            headers={"Content-Type": "application/x-www-form-urlencoded"},
        ), timeout=10)

        ws.settimeout(60)
        result = None
        # submit pushes 3 messages (publish before/after + info) and the
        # worker 3 more per request — scan a wide window for the result
        for _ in range(14):
            msg = json.loads(_ws_recv(ws))
            if "result" in msg:
                result = json.loads(msg["result"])
                break
        assert result is not None
        assert result["task_id"] == "1" and len(result["result"]) == 3
        proc.join(30)
        assert proc.exitcode == 0
        ws.close()
    finally:
        srv.should_exit = True
        th.join(5)
