"""HTTP/WS API contract tests (urls fixed by demo/urls.py:7-11)."""

import io
import json

import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from vilbert_multi_task_amd.serve.app import create_app


@pytest.fixture
def client(tmp_path):
    app = create_app(
        db_path=str(tmp_path / "db.sqlite3"),
        queue_path=str(tmp_path / "q.sqlite3"),
        media_root=str(tmp_path / "media"),
        hub_port=0,
    )
    with TestClient(app) as c:
        yield c, app


def test_index(client):
    c, _ = client
    r = c.get("/")
    assert r.status_code == 200 and "ViLBERT" in r.text


def test_submit_enqueues_message(client):
    c, app = client
    r = c.post(
        "/",
        data={
            "socket_id": "sockA",
            "task_id": "1",
            "question": "What COLOR is the sky?",
            "image_list[]": ["demo/x.jpg"],
        },
    )
    assert r.status_code == 200
    assert app.state.broker.depth() == 1
    d = app.state.broker.get()[0]
    # message schema fixed by sender.py:19-24 (+ our additive trace_id)
    assert {"image_path", "question", "socket_id", "task_id"} <= set(d.body)
    assert set(d.body) - {"image_path", "question", "socket_id", "task_id", "trace_id"} == set()
    assert d.body["question"] == "what color is the sky?"  # lowercased (views.py:28)
    assert d.body["task_id"] == "1"
    assert d.body["socket_id"] == "sockA"


def test_get_task_details(client):
    c, _ = client
    r = c.get("/get_task_details/1/")
    assert r.status_code == 200
    body = r.json()
    assert body["unique_id"] == 1 and body["name"] == "VQA"
    assert c.get("/get_task_details/99/").status_code == 404


def test_upload_image(client):
    c, _ = client
    files = {"file": ("cat.jpg", io.BytesIO(b"\xff\xd8fakejpeg"), "image/jpeg")}
    r = c.post("/upload_image/", files=files)
    assert r.status_code == 200
    paths = r.json()["file_paths"]
    assert len(paths) == 1 and paths[0].endswith(".jpg")


def test_websocket_push_roundtrip(client):
    c, app = client
    with c.websocket_connect("/chat/") as ws:
        ws.send_text("sockWS")
        import time

        time.sleep(0.05)
        app.state.push.publish("sockWS", {"terminal": "hello"})
        msg = json.loads(ws.receive_text())
        assert msg == {"terminal": "hello"}


def test_demo_images_endpoint(client, tmp_path):
    c, _ = client
    r = c.get("/demo_images/")
    assert r.status_code == 200
    assert "images" in r.json()


def test_admin_surface(client):
    c, _ = client
    c.post("/", data={"socket_id": "adm", "task_id": "1", "question": "Q?",
                      "image_list[]": ["demo/x.jpg"]})
    r = c.get("/admin/")
    assert r.status_code == 200
    body = r.json()
    assert len(body["tasks"]) == 9  # the 9 registry task ids
    assert body["queue"]["ready"] >= 1
    assert body["recent_questions"] == []  # worker inserts the row, not the app


def test_media_path_traversal_blocked(client):
    c, _ = client
    r = c.get("/media/../../etc/passwd")
    assert r.status_code == 404
    r = c.get("/media/%2e%2e/%2e%2e/etc/passwd")
    assert r.status_code == 404


def test_index_renders_frontend(client):
    """GET / serves the full demo page with a fresh socketid and the demo
    image list substituted (views.py:39-42 template-context contract)."""
    client, _app = client
    r = client.get("/")
    assert r.status_code == 200
    body = r.text
    assert "__SOCKET_ID__" not in body and "__DEMO_IMAGES__" not in body
    assert 'id="selected-task"' in body       # the 8-task dropdown
    assert 'option value="16"' in body        # GuessWhat entry
    assert 'id="Console"' in body             # terminal panel
    assert 'id="ReferExpressionsTaskResultImage_1"' in body
    assert "/static/app.js" in body


def test_static_assets_served(client):
    client, _app = client
    js = client.get("/static/app.js")
    assert js.status_code == 200
    assert "addImagesToSampleList" in js.text
    assert "Only a maximum of 4 files are allowed!" in js.text  # upload cap
    css = client.get("/static/style.css")
    assert css.status_code == 200
    r = client.get("/static/../app.py")
    assert r.status_code == 404  # traversal guarded


def test_admin_crud(client):
    """Admin CRUD surface (demo/admin.py equivalent): edit a Tasks row,
    delete a QuestionAnswer row."""
    client, app = client
    r = client.post("/admin/tasks/1/", json={"placeholder": "edited!", "bogus": 1})
    assert r.status_code == 200 and r.json()["placeholder"] == "edited!"
    assert client.get("/get_task_details/1/").json()["placeholder"] == "edited!"
    # no editable fields -> 400
    assert client.post("/admin/tasks/1/", json={"bogus": 1}).status_code == 400
    # question delete
    qa = app.state.db.create_question(1, "q", "[]", "s")
    assert client.delete(f"/admin/questions/{qa}/").json() == {"deleted": qa}
    assert client.delete(f"/admin/questions/{qa}/").status_code == 404
