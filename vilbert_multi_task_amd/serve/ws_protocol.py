"""Minimal RFC6455 websocket protocol for uvicorn.

The offline image has neither `websockets` nor `wsproto`, so stock uvicorn
rejects upgrade requests — but the realtime push channel is a core layer of
the serving stack (L2, SURVEY.md §1: ws `/chat/`, consumers.py contract).
This is a self-contained ASGI websocket server protocol implementing the
subset the demo needs: handshake, text/binary/ping/pong/close frames,
client-masked parsing, server-unmasked writes, single-frame messages
(fragmentation handled by buffering continuation frames).

Select it with: uvicorn.run(app, ws="vilbert_multi_task_amd.serve.ws_protocol:MinimalWebSocketProtocol")
"""

from __future__ import annotations

import asyncio
import base64
import hashlib
from typing import Any, Dict, List, Optional, Tuple
from urllib.parse import unquote, urlsplit

_GUID = b"258EAFA5-E914-47DA-95CA-C5AB0DC85B11"

# DoS bounds: a client frame/assembled message (or the pre-handshake header
# block) above this is rejected with close 1009 (message too big) instead of
# growing self.buf/_frag_data without limit. The demo's messages are small
# JSON; 1 MiB is generous.
MAX_MESSAGE_BYTES = 1 << 20
MAX_HANDSHAKE_BYTES = 64 << 10


def _accept_key(key: str) -> str:
    return base64.b64encode(hashlib.sha1(key.encode() + _GUID).digest()).decode()


class MinimalWebSocketProtocol(asyncio.Protocol):
    def __init__(self, config, server_state, app_state=None, _loop=None):
        self.config = config
        if not getattr(config, "loaded", False):
            config.load()
        self.app = config.loaded_app
        self.server_state = server_state
        self.app_state = app_state if app_state is not None else {}
        self.loop = _loop or asyncio.get_event_loop()
        self.transport: Optional[asyncio.Transport] = None
        self.buf = b""
        self.handshaken = False
        self.accepted = False
        self.closed = False
        self.recv_q: asyncio.Queue = asyncio.Queue()
        self.task: Optional[asyncio.Task] = None
        self._frag_op = 0
        self._frag_data = b""

    # ---- asyncio.Protocol -------------------------------------------------
    def connection_made(self, transport) -> None:
        self.transport = transport
        self.server_state.connections.add(self)

    def connection_lost(self, exc) -> None:
        self.server_state.connections.discard(self)
        if not self.closed:
            self.closed = True
            self.recv_q.put_nowait({"type": "websocket.disconnect", "code": 1006})

    def shutdown(self) -> None:  # called by uvicorn on server shutdown
        if self.transport and not self.transport.is_closing():
            self._send_close(1012)
            self.transport.close()

    def data_received(self, data: bytes) -> None:
        self.buf += data
        if not self.handshaken:
            if b"\r\n\r\n" not in self.buf:
                if len(self.buf) > MAX_HANDSHAKE_BYTES:
                    self.transport.write(
                        b"HTTP/1.1 431 Request Header Fields Too Large\r\n"
                        b"content-length: 0\r\n\r\n"
                    )
                    self.transport.close()
                return
            head, self.buf = self.buf.split(b"\r\n\r\n", 1)
            self._handshake(head)
        self._parse_frames()

    def _protocol_error(self, code: int) -> None:
        """Close the connection with the given code (1009 too-big /
        1007 invalid-utf8) and drop buffered input."""
        if not self.closed:
            self._send_close(code)  # before the closed flag gates _send_frame
            self.closed = True
            self.recv_q.put_nowait({"type": "websocket.disconnect", "code": code})
        self.buf = b""
        self._frag_op, self._frag_data = 0, b""
        self.transport.close()

    # ---- handshake --------------------------------------------------------
    def _handshake(self, head: bytes) -> None:
        lines = head.decode("latin1").split("\r\n")
        method, target, _ = lines[0].split(" ", 2)
        headers: List[Tuple[bytes, bytes]] = []
        hmap: Dict[str, str] = {}
        for line in lines[1:]:
            if ":" in line:
                k, v = line.split(":", 1)
                headers.append((k.strip().lower().encode(), v.strip().encode()))
                hmap[k.strip().lower()] = v.strip()
        key = hmap.get("sec-websocket-key", "")
        if method != "GET" or not key:
            self.transport.write(b"HTTP/1.1 400 Bad Request\r\ncontent-length: 0\r\n\r\n")
            self.transport.close()
            return
        self.handshaken = True
        self._pending_accept_headers = (
            b"HTTP/1.1 101 Switching Protocols\r\nupgrade: websocket\r\n"
            b"connection: Upgrade\r\nsec-websocket-accept: "
            + _accept_key(key).encode()
            + b"\r\n\r\n"
        )
        split = urlsplit(target)
        scope: Dict[str, Any] = {
            "type": "websocket",
            "asgi": {"version": "3.0", "spec_version": "2.3"},
            "http_version": "1.1",
            "scheme": "ws",
            "server": self.transport.get_extra_info("sockname"),
            "client": self.transport.get_extra_info("peername"),
            "root_path": getattr(self.config, "root_path", ""),
            "path": unquote(split.path),
            "raw_path": split.path.encode(),
            "query_string": split.query.encode(),
            "headers": headers,
            "subprotocols": [
                p.strip()
                for p in hmap.get("sec-websocket-protocol", "").split(",")
                if p.strip()
            ],
            "state": self.app_state,
        }
        self.recv_q.put_nowait({"type": "websocket.connect"})
        self.task = self.loop.create_task(self._run_app(scope))
        self.server_state.tasks.add(self.task)
        self.task.add_done_callback(self.server_state.tasks.discard)

    async def _run_app(self, scope) -> None:
        try:
            await self.app(scope, self._asgi_receive, self._asgi_send)
        except Exception:
            if not self.closed and self.transport:
                if not self.accepted:
                    self.transport.write(
                        b"HTTP/1.1 500 Internal Server Error\r\ncontent-length: 0\r\n\r\n"
                    )
                else:
                    self._send_close(1011)
                self.transport.close()

    # ---- ASGI bridge ------------------------------------------------------
    async def _asgi_receive(self):
        return await self.recv_q.get()

    async def _asgi_send(self, message) -> None:
        t = message["type"]
        if t == "websocket.accept":
            self.accepted = True
            self.transport.write(self._pending_accept_headers)
        elif t == "websocket.send":
            if "text" in message and message["text"] is not None:
                self._send_frame(0x1, message["text"].encode())
            elif message.get("bytes") is not None:
                self._send_frame(0x2, message["bytes"])
        elif t == "websocket.close":
            if not self.accepted:
                self.transport.write(b"HTTP/1.1 403 Forbidden\r\ncontent-length: 0\r\n\r\n")
            else:
                self._send_close(message.get("code", 1000))
            self.closed = True
            self.transport.close()

    # ---- frames -----------------------------------------------------------
    def _send_frame(self, opcode: int, payload: bytes) -> None:
        if self.closed or self.transport.is_closing():
            return
        n = len(payload)
        if n < 126:
            header = bytes([0x80 | opcode, n])
        elif n < 65536:
            header = bytes([0x80 | opcode, 126]) + n.to_bytes(2, "big")
        else:
            header = bytes([0x80 | opcode, 127]) + n.to_bytes(8, "big")
        self.transport.write(header + payload)

    def _send_close(self, code: int) -> None:
        self._send_frame(0x8, code.to_bytes(2, "big"))

    def _parse_frames(self) -> None:
        while True:
            if len(self.buf) < 2:
                return
            b0, b1 = self.buf[0], self.buf[1]
            fin = b0 & 0x80
            opcode = b0 & 0x0F
            masked = b1 & 0x80
            ln = b1 & 0x7F
            off = 2
            if ln == 126:
                if len(self.buf) < 4:
                    return
                ln = int.from_bytes(self.buf[2:4], "big")
                off = 4
            elif ln == 127:
                if len(self.buf) < 10:
                    return
                ln = int.from_bytes(self.buf[2:10], "big")
                off = 10
            if ln > MAX_MESSAGE_BYTES:
                self._protocol_error(1009)
                return
            mask = b""
            if masked:
                if len(self.buf) < off + 4:
                    return
                mask = self.buf[off : off + 4]
                off += 4
            if len(self.buf) < off + ln:
                return
            payload = self.buf[off : off + ln]
            self.buf = self.buf[off + ln :]
            if masked:
                payload = bytes(b ^ mask[i % 4] for i, b in enumerate(payload))
            self._on_frame(fin, opcode, payload)

    def _on_frame(self, fin: int, opcode: int, payload: bytes) -> None:
        if opcode == 0x0:  # continuation
            if len(self._frag_data) + len(payload) > MAX_MESSAGE_BYTES:
                self._protocol_error(1009)  # endless-continuation DoS guard
                return
            self._frag_data += payload
            if fin:
                opcode, payload = self._frag_op, self._frag_data
                self._frag_op, self._frag_data = 0, b""
            else:
                return
        elif not fin:
            self._frag_op, self._frag_data = opcode, payload
            return
        if opcode == 0x1:
            try:
                text = payload.decode()
            except UnicodeDecodeError:
                self._protocol_error(1007)  # RFC6455: invalid UTF-8 in text
                return
            self.recv_q.put_nowait({"type": "websocket.receive", "text": text})
        elif opcode == 0x2:
            self.recv_q.put_nowait({"type": "websocket.receive", "bytes": payload})
        elif opcode == 0x8:
            code = int.from_bytes(payload[:2], "big") if len(payload) >= 2 else 1000
            if not self.closed:
                self._send_close(code)  # echo close before gating _send_frame
                self.closed = True
                self.recv_q.put_nowait({"type": "websocket.disconnect", "code": code})
            self.transport.close()
        elif opcode == 0x9:
            self._send_frame(0xA, payload)  # ping -> pong
