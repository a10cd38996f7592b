"""Realtime push: the reference's Redis-channel websocket fanout
(demo/utils.py:5-6 `log_to_terminal`, demo/consumers.py group-per-socket-id)
rebuilt as a lightweight TCP pub/sub hub embedded in the web app (offline
image has no Redis).

Topology: the FastAPI app runs `PushHub` (asyncio TCP server). Workers (other
processes) connect with `PushClient` and send JSON lines
{"socket_id": ..., "payload": {...}}; the hub forwards each payload to every
websocket registered for that socket_id — same per-socket-id group semantics
as channels' Group(socketid) (consumers.py:10-11).
"""

from __future__ import annotations

import asyncio
import json
import socket
import threading
from collections import defaultdict
from typing import Any, Callable, Dict, List, Optional

DEFAULT_PORT = 6381


class PushHub:
    """Runs inside the web app's event loop."""

    def __init__(self, host: str = "127.0.0.1", port: int = DEFAULT_PORT):
        self.host = host
        self.port = port
        self._groups: Dict[str, List[Callable[[dict], Any]]] = defaultdict(list)
        self._server: Optional[asyncio.AbstractServer] = None
        self._lock = threading.Lock()

    # -- group management (called by the WS endpoint) ----------------------
    def join(self, socket_id: str, send: Callable[[dict], Any]) -> None:
        with self._lock:
            self._groups[socket_id].append(send)

    def leave(self, socket_id: str, send: Callable[[dict], Any]) -> None:
        with self._lock:
            if send in self._groups.get(socket_id, []):
                self._groups[socket_id].remove(send)

    async def dispatch(self, socket_id: str, payload: dict) -> None:
        with self._lock:
            targets = list(self._groups.get(socket_id, []))
        for send in targets:
            try:
                r = send(payload)
                if asyncio.iscoroutine(r):
                    await r
            except Exception:
                self.leave(socket_id, send)

    # -- TCP server for cross-process publishers ---------------------------
    async def start(self) -> None:
        self._server = await asyncio.start_server(self._handle, self.host, self.port)

    async def stop(self) -> None:
        if self._server:
            self._server.close()
            await self._server.wait_closed()

    async def _handle(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        try:
            while True:
                line = await reader.readline()
                if not line:
                    break
                try:
                    msg = json.loads(line)
                    await self.dispatch(msg["socket_id"], msg["payload"])
                except (json.JSONDecodeError, KeyError):
                    continue
        finally:
            writer.close()


class PushClient:
    """Synchronous publisher used by worker processes (and the HTTP view's
    progress pushes, views.py:35)."""

    def __init__(self, host: str = "127.0.0.1", port: int = DEFAULT_PORT):
        self.host = host
        self.port = port
        self._sock: Optional[socket.socket] = None

    def _ensure(self) -> Optional[socket.socket]:
        if self._sock is None:
            try:
                self._sock = socket.create_connection((self.host, self.port), timeout=2.0)
            except OSError:
                self._sock = None
        return self._sock

    def publish(self, socket_id: str, payload: dict) -> bool:
        s = self._ensure()
        if s is None:
            return False
        try:
            s.sendall((json.dumps({"socket_id": socket_id, "payload": payload}) + "\n").encode())
            return True
        except OSError:
            try:
                s.close()
            finally:
                self._sock = None
            return False

    def close(self) -> None:
        if self._sock:
            self._sock.close()
            self._sock = None


def log_to_terminal(push: PushClient, socket_id: str, message: dict) -> None:
    """demo/utils.py:5-6 contract."""
    push.publish(socket_id, message)


class NullPush(PushClient):
    """Push sink for tests/offline runs; records messages instead."""

    def __init__(self):  # noqa: D401
        super().__init__()
        self.messages: List[tuple] = []

    def publish(self, socket_id: str, payload: dict) -> bool:
        self.messages.append((socket_id, payload))
        return True
