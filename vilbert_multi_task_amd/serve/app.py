"""HTTP + websocket API, contract-identical to the reference's Django app.

URL surface (from /root/reference/demo/urls.py:7-11 and
vilbert_multitask/urls.py:21-31):
  GET  /                        demo page (views.py:39-42)
  POST /                        submit job: socket_id, task_id, question,
                                image_list[] (views.py:20-38)
  GET  /get_task_details/<id>/  JSON of the Tasks row (views.py:45-61)
  POST /upload_image/           multi-file upload -> media/demo/<uuid>,
                                {"file_paths": [...]} (views.py:89-106)
  WS   /chat/                   client sends its socket_id first, then
                                receives {terminal|result|info} pushes
                                (routers.py:4-7, consumers.py:4-12)

Implementation: FastAPI + the embedded PushHub (offline image has no
Django/channels/Redis; the wire contracts are what the frontend consumes).
"""

from __future__ import annotations

import asyncio
import json
import os
import uuid
from typing import List, Optional

from fastapi import FastAPI, Request, WebSocket, WebSocketDisconnect
from fastapi.responses import HTMLResponse, JSONResponse


async def _parse_form(request: Request):
    """stdlib form parsing (offline image lacks python-multipart):
    returns (fields: {name: [values]}, files: [(name, filename, bytes)])."""
    import email
    import email.policy
    from urllib.parse import parse_qs

    ctype = request.headers.get("content-type", "")
    body = await request.body()
    fields: dict = {}
    files: list = []
    if ctype.startswith("multipart/form-data"):
        raw = (
            b"Content-Type: " + ctype.encode() + b"\r\nMIME-Version: 1.0\r\n\r\n" + body
        )
        msg = email.message_from_bytes(raw, policy=email.policy.HTTP)
        for part in msg.iter_parts():
            name = part.get_param("name", header="content-disposition")
            filename = part.get_filename()
            payload = part.get_payload(decode=True) or b""
            if filename:
                files.append((name, filename, payload))
            else:
                fields.setdefault(name, []).append(payload.decode(errors="replace"))
    else:
        for k, vs in parse_qs(body.decode(errors="replace")).items():
            fields[k] = vs
    return fields, files

from .broker import Broker, vilbert_task
from .db import Database
from .push import PushHub, log_to_terminal

_INDEX_HTML = """<!doctype html>
<html><head><title>ViLBERT Multi-Task Demo (MI355X)</title></head>
<body>
<h2>ViLBERT 12-in-1 — MI355X-native demo</h2>
<p>POST / with socket_id, task_id, question, image_list[] to submit a job;
open a websocket to /chat/ and send your socket_id to receive results.</p>
</body></html>"""


def create_app(
    db_path: str = "vilbert_demo.sqlite3",
    queue_path: str = "vilbert_queue.sqlite3",
    media_root: str = "media",
    hub_port: int = 0,
) -> FastAPI:
    app = FastAPI(title="vilbert-multitask-amd")
    db = Database(db_path)
    db.seed_tasks()
    broker = Broker(queue_path)
    hub = PushHub(port=hub_port) if hub_port else None
    app.state.db = db
    app.state.broker = broker
    app.state.hub = hub
    app.state.push = _LocalPush(None)  # in-process fanout (single-proc mode/tests)
    os.makedirs(os.path.join(media_root, "demo"), exist_ok=True)

    if hub is not None:

        @app.on_event("startup")
        async def _start_hub():
            await hub.start()

        @app.on_event("shutdown")
        async def _stop_hub():
            await hub.stop()

    static_dir = os.path.join(os.path.dirname(__file__), "static")
    static_index = os.path.join(static_dir, "index.html")

    def _demo_image_urls():
        """views.py:64-81: 6 random COCO test2014 images, falling back to the
        uploaded-demo directory, then to an empty gallery."""
        import random

        coco_dir = os.path.join(media_root, "test2014")
        if os.path.isdir(coco_dir):
            pool = [f for f in os.listdir(coco_dir) if f.lower().endswith((".jpg", ".png"))]
            picks = random.sample(pool, min(6, len(pool)))
            return [f"/media/test2014/{p}" for p in picks]
        demo_dir = os.path.join(media_root, "demo")
        pool = sorted(os.listdir(demo_dir)) if os.path.isdir(demo_dir) else []
        return [f"/media/demo/{p}" for p in pool[:6]]

    def _render_index() -> str:
        """Server-side substitution standing in for the Django template
        context {demo_images, socketid} (views.py:39-42)."""
        if not os.path.exists(static_index):
            return _INDEX_HTML
        with open(static_index, encoding="utf-8") as f:
            page = f.read()
        sock = str(uuid.uuid4())
        return page.replace("__SOCKET_ID__", sock).replace(
            "__DEMO_IMAGES__", json.dumps(_demo_image_urls())
        )

    @app.get("/", response_class=HTMLResponse)
    async def index() -> str:
        return _render_index()

    @app.get("/static/{name}")
    async def static_file(name: str):
        from fastapi.responses import FileResponse

        full = os.path.normpath(os.path.join(static_dir, name))
        if os.path.commonpath([full, static_dir]) != static_dir or not os.path.isfile(full):
            return JSONResponse({"error": "not found"}, status_code=404)
        media_types = {".js": "text/javascript", ".css": "text/css", ".html": "text/html"}
        return FileResponse(full, media_type=media_types.get(os.path.splitext(name)[1]))

    @app.get("/demo_images/")
    async def demo_images():
        """Random demo gallery as JSON (views.py:64-81 equivalent for
        API consumers; the index page gets the list inlined)."""
        return JSONResponse({"images": [p[len("/media/"):] for p in _demo_image_urls()]})

    @app.get("/admin/")
    async def admin_index():
        """Admin surface (the reference exposes Django admin CRUD for Tasks
        and QuestionAnswer — demo/admin.py:1-34). List view here; edits via
        POST /admin/tasks/{id}/ and DELETE /admin/questions/{id}/."""
        return JSONResponse(
            {
                "tasks": app.state.db.list_tasks(),
                "recent_questions": app.state.db.recent_questions(25),
                "queue": {
                    "ready": app.state.broker.depth(),
                    "dead": app.state.broker.dead_count(),
                },
            }
        )

    @app.post("/admin/tasks/{task_id}/")
    async def admin_update_task(task_id: int, request: Request):
        """Edit a Tasks row (Django admin change-view equivalent): JSON body
        with any of name/placeholder/description/num_of_images/example."""
        try:
            fields = await request.json()
        except Exception:
            return JSONResponse({"error": "json body required"}, status_code=400)
        if not isinstance(fields, dict):
            return JSONResponse({"error": "json object required"}, status_code=400)
        if not app.state.db.update_task(task_id, fields):
            return JSONResponse({"error": "no editable fields or unknown task"},
                                status_code=400)
        return JSONResponse(app.state.db.get_task(task_id))

    @app.delete("/admin/questions/{qa_id}/")
    async def admin_delete_question(qa_id: int):
        if not app.state.db.delete_question(qa_id):
            return JSONResponse({"error": "not found"}, status_code=404)
        return JSONResponse({"deleted": qa_id})

    @app.get("/media/{path:path}")
    async def media(path: str):
        from fastapi.responses import FileResponse

        full = os.path.normpath(os.path.join(media_root, path))
        root = os.path.normpath(media_root)
        # commonpath (not startswith): "/mediaX" must not pass for "/media"
        if os.path.commonpath([full, root]) != root or not os.path.isfile(full):
            return JSONResponse({"error": "not found"}, status_code=404)
        return FileResponse(full)

    @app.post("/")
    async def submit(request: Request):
        fields, _ = await _parse_form(request)
        socket_id = fields.get("socket_id", [""])[0]
        task_id = fields.get("task_id", [""])[0]
        question = fields.get("question", [""])[0].lower()  # views.py:28
        image_list: List[str] = fields.get("image_list[]") or fields.get("image_list") or []
        # absolute-path resolution analogue of views.py:30-32: the frontend
        # POSTs /media/... URL paths (upload response + <img> pathname) which
        # resolve under media_root; anything else absolute passes through
        def _resolve(p: str) -> str:
            if p.startswith("/media/"):
                return os.path.join(media_root, p[len("/media/"):])
            return p if os.path.isabs(p) else os.path.join(media_root, p.lstrip("/"))

        paths = [_resolve(p) for p in image_list]
        from ..utils.trace import log_json, new_trace_id

        trace_id = new_trace_id()
        # exact submit-time terminal sequence of the reference:
        # views.py:35 before dispatch, then sender.py:25 / sender.py:34
        # around the publish
        _push_local(app, socket_id, {"terminal": "Starting Vilbert Multitask Job..."})
        _push_local(app, socket_id, {"terminal": "Publishing job to ViLBERT Queue"})
        vilbert_task(
            app.state.broker, paths, question, task_id or "1", socket_id,
            trace_id=trace_id,
        )
        _push_local(app, socket_id, {"terminal": "Job published successfully"})
        log_json("submit", trace_id=trace_id, task_id=task_id, socket_id=socket_id)
        return HTMLResponse(_INDEX_HTML)

    @app.get("/get_task_details/{task_id}/")
    async def get_task_details(task_id: int):
        row = app.state.db.get_task(task_id)
        if row is None:
            return JSONResponse({"error": f"task {task_id} not found"}, status_code=404)
        return JSONResponse(row)

    @app.post("/upload_image/")
    async def upload_image(request: Request):
        _, files = await _parse_form(request)
        file_paths = []
        for _name, filename, payload in files[:4]:  # <=4 files (demo_images.html)
            ext = os.path.splitext(filename or "img.jpg")[1] or ".jpg"
            name = f"{uuid.uuid4().hex}{ext}"
            dst = os.path.join(media_root, "demo", name)
            with open(dst, "wb") as out:
                out.write(payload)
            app.state.db.add_attachment(dst)  # Attachment row (models.py:45-46)
            # reference returns URL paths the frontend uses as <img> src and
            # later POSTs back (views.py:106 + demo_images.html done handler);
            # the submit handler resolves them back under media_root
            file_paths.append(f"/media/demo/{name}")
        return JSONResponse({"file_paths": file_paths})

    @app.websocket("/chat/")
    async def chat(ws: WebSocket):
        await ws.accept()
        socket_id: Optional[str] = None
        q: asyncio.Queue = asyncio.Queue()

        def _send(payload: dict):
            q.put_nowait(payload)

        try:
            # first client message is the socket id (consumers.py:8-11)
            socket_id = (await ws.receive_text()).strip()
            _register(app, socket_id, _send)
            while True:
                getter = asyncio.create_task(q.get())
                recv = asyncio.create_task(ws.receive_text())
                done, pending = await asyncio.wait(
                    {getter, recv}, return_when=asyncio.FIRST_COMPLETED
                )
                for t in pending:
                    t.cancel()
                if getter in done:
                    await ws.send_text(json.dumps(getter.result()))
                if recv in done:
                    recv.result()  # raises on disconnect; content ignored
        except (WebSocketDisconnect, RuntimeError):
            pass
        finally:
            if socket_id:
                _unregister(app, socket_id, _send)

    return app


class _LocalPush:
    """In-process push used when hub_port=0 (single-process app+worker, and
    tests): delivers straight to registered WS queues."""

    def __init__(self, app):
        self.app = app
        self.groups = {}

    def publish(self, socket_id: str, payload: dict) -> bool:
        for send in list(self.groups.get(socket_id, [])):
            send(payload)
        return True


def _push_local(app: FastAPI, socket_id: str, payload: dict) -> None:
    p = app.state.push
    if isinstance(p, _LocalPush):
        p.publish(socket_id, payload)
    else:
        log_to_terminal(p, socket_id, payload)
    if app.state.hub is not None:
        try:
            loop = asyncio.get_event_loop()
            loop.create_task(app.state.hub.dispatch(socket_id, payload))
        except RuntimeError:
            pass


def _register(app: FastAPI, socket_id: str, send) -> None:
    if app.state.hub is not None:
        app.state.hub.join(socket_id, send)
    if isinstance(app.state.push, _LocalPush):
        app.state.push.groups.setdefault(socket_id, []).append(send)


def _unregister(app: FastAPI, socket_id: str, send) -> None:
    if app.state.hub is not None:
        app.state.hub.leave(socket_id, send)
    if isinstance(app.state.push, _LocalPush):
        g = app.state.push.groups.get(socket_id, [])
        if send in g:
            g.remove(send)
