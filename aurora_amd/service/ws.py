"""WebSocket chat endpoint: hand-rolled RFC 6455 over asyncio.

The reference runs a dedicated websockets server on :5006 (reference:
main_chatbot.py:909,1665) with per-connection JWT auth, a rate limiter,
init/query/control:cancel message types, a 20k-token input cap, and the
workflow streamed as JSON frames.  Same protocol here; the frame layer is
implemented directly (no ws library in the offline wheelhouse).
"""
from __future__ import annotations

import asyncio
import base64
import hashlib
import json
import struct
import time
import urllib.parse
import uuid
from typing import Dict, Optional

from ..agent.state import State
from ..agent.workflow import Workflow

_GUID = "258EAFA5-E914-47DA-95CA-C5AB0DC85B11"


class AgentConn:
    """Server-side handle of a paired kubectl-agent (reference:
    utils/kubectl/agent_ws_handler.py)."""

    def __init__(self, writer):
        self.writer = writer
        self.pending: Dict[str, asyncio.Future] = {}
        self._n = 0

    async def exec(self, args: str, timeout_s: float = 90.0) -> str:
        import uuid as _uuid
        rid = _uuid.uuid4().hex[:12]
        fut = asyncio.get_running_loop().create_future()
        self.pending[rid] = fut
        try:
            self.writer.write(encode_frame(json.dumps(
                {"type": "exec", "id": rid, "args": args}).encode()))
            await self.writer.drain()
            return await asyncio.wait_for(fut, timeout=timeout_s)
        except asyncio.TimeoutError:
            return "ERROR: on-prem agent did not respond in time"
        finally:
            self.pending.pop(rid, None)

    def resolve(self, rid: str, output: str) -> None:
        fut = self.pending.get(rid)
        if fut is not None and not fut.done():
            fut.set_result(output)


# ------------------------------------------------------------ frame layer

def _accept_key(key: str) -> str:
    return base64.b64encode(hashlib.sha1((key + _GUID).encode()).digest()).decode()


def encode_frame(data: bytes, opcode: int = 0x1) -> bytes:
    head = bytes([0x80 | opcode])
    n = len(data)
    if n < 126:
        head += bytes([n])
    elif n < 65536:
        head += bytes([126]) + struct.pack(">H", n)
    else:
        head += bytes([127]) + struct.pack(">Q", n)
    return head + data


def mask_frame(data: bytes, opcode: int = 0x1) -> bytes:
    """Client-side frame (masked) — used by tests and the kubectl agent."""
    import os as _os
    mask = _os.urandom(4)
    head = bytes([0x80 | opcode])
    n = len(data)
    if n < 126:
        head += bytes([0x80 | n])
    elif n < 65536:
        head += bytes([0x80 | 126]) + struct.pack(">H", n)
    else:
        head += bytes([0x80 | 127]) + struct.pack(">Q", n)
    return head + mask + bytes(b ^ mask[i % 4] for i, b in enumerate(data))


MAX_FRAME_BYTES = 16 * 1024 * 1024  # refuse absurd length fields up front


async def read_frame(reader: asyncio.StreamReader):
    """Returns (opcode, payload) or (0x8, b'') on close/EOF/oversize.
    Any mid-frame EOF and any length beyond MAX_FRAME_BYTES map to a
    close frame — a hostile peer must not be able to raise out of here
    or make the server buffer a 2**63-byte frame."""
    try:
        h = await reader.readexactly(2)
        opcode = h[0] & 0x0F
        masked = h[1] & 0x80
        n = h[1] & 0x7F
        if n == 126:
            n = struct.unpack(">H", await reader.readexactly(2))[0]
        elif n == 127:
            n = struct.unpack(">Q", await reader.readexactly(8))[0]
        if n > MAX_FRAME_BYTES:
            return 0x8, b""
        mask = await reader.readexactly(4) if masked else b"\x00" * 4
        data = await reader.readexactly(n) if n else b""
    except (asyncio.IncompleteReadError, ConnectionResetError):
        return 0x8, b""
    if masked:
        data = bytes(b ^ mask[i % 4] for i, b in enumerate(data))
    return opcode, data


# ----------------------------------------------------------- chat server


class RateLimiter:
    """N messages / 60 s / client (reference: main_chatbot.py:68-90)."""

    def __init__(self, limit: int = 5, window_s: float = 60.0):
        self.limit = limit
        self.window = window_s
        self.hits: Dict[str, list] = {}

    def allow(self, client: str) -> bool:
        now = time.time()
        h = self.hits.setdefault(client, [])
        h[:] = [t for t in h if now - t < self.window]
        if len(h) >= self.limit:
            return False
        h.append(now)
        return True


class ChatServer:
    def __init__(self, rt):
        self.rt = rt
        self.rate = RateLimiter(rt.settings.ws_rate_limit)
        self._seen_tokens: Dict[str, float] = {}  # dev-mode nonce replay cache
        from ..utils.auth_tokens import ReplayCache
        self._replay = ReplayCache()  # signed-token jti replay (Redis analog)
        self._server: Optional[asyncio.AbstractServer] = None
        self.port: Optional[int] = None

    async def start(self, host: str = "127.0.0.1", port: int = 0) -> int:
        self._server = await asyncio.start_server(self._handle, host, port)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.port

    async def stop(self) -> None:
        if self._server:
            self._server.close()
            await self._server.wait_closed()

    # ---- auth: token resolves to a user; one-shot (replay-rejected) ----
    def _auth(self, query: Dict[str, str]) -> Optional[dict]:
        token = query.get("token", "")
        if not token:
            return None
        secret = self.rt.settings.internal_api_secret
        if secret:
            # authenticated mode: HMAC-signed one-shot token, jti replay
            # rejected (reference: signed JWT + Redis jti,
            # main_chatbot.py:107-158). Forged/expired/replayed => refuse.
            from ..utils.auth_tokens import verify_token
            claims = verify_token(token, secret)
            if claims is None:
                return None
            if not self._replay.check_and_store(claims["jti"]):
                return None
            return self.rt.admin.get_user(claims["user_id"])
        # dev mode (no secret configured): "<user_id>.<nonce>" identity
        # assertion with nonce replay guard
        parts = token.split(".", 1)
        user = self.rt.admin.get_user(parts[0])
        if user is None:
            return None
        if len(parts) == 2:
            now = time.time()
            self._seen_tokens = {k: v for k, v in self._seen_tokens.items()
                                 if now - v < 600}
            if token in self._seen_tokens:
                return None  # replay
            self._seen_tokens[token] = now
        return user

    async def _handle(self, reader: asyncio.StreamReader,
                      writer: asyncio.StreamWriter) -> None:
        try:
            await self._handle_inner(reader, writer)
        except Exception:  # noqa: BLE001
            pass
        finally:
            try:
                writer.close()
                await writer.wait_closed()
            except Exception:  # noqa: BLE001
                pass

    async def _handle_inner(self, reader, writer) -> None:
        # HTTP upgrade
        req = await reader.readuntil(b"\r\n\r\n")
        lines = req.decode("latin-1").split("\r\n")
        path = lines[0].split(" ")[1] if " " in lines[0] else "/"
        headers = {}
        for ln in lines[1:]:
            if ":" in ln:
                k, v = ln.split(":", 1)
                headers[k.strip().lower()] = v.strip()
        key = headers.get("sec-websocket-key")
        if not key:
            writer.write(b"HTTP/1.1 400 Bad Request\r\n\r\n")
            await writer.drain()
            return
        query = dict(urllib.parse.parse_qsl(urllib.parse.urlsplit(path).query))
        if "agent_token" in query:  # kubectl-agent pairing (reference:
            # main_chatbot.py:911-914 routes agents off the same socket)
            await self._agent_session(reader, writer, key, query["agent_token"])
            return
        user = self._auth(query)
        if user is None:
            writer.write(b"HTTP/1.1 401 Unauthorized\r\n\r\n")
            await writer.drain()
            return
        writer.write(
            b"HTTP/1.1 101 Switching Protocols\r\nUpgrade: websocket\r\n"
            b"Connection: Upgrade\r\nSec-WebSocket-Accept: "
            + _accept_key(key).encode() + b"\r\n\r\n")
        await writer.drain()
        await self._session(reader, writer, user)

    async def _agent_session(self, reader, writer, key: str, token: str) -> None:
        from ..kubectl_agent.agent import cluster_id_from_token
        from .kubectl_registry import resolve_agent_token
        org_id = resolve_agent_token(self.rt, token)
        if org_id is None:
            writer.write(b"HTTP/1.1 401 Unauthorized\r\n\r\n")
            await writer.drain()
            return
        writer.write(
            b"HTTP/1.1 101 Switching Protocols\r\nUpgrade: websocket\r\n"
            b"Connection: Upgrade\r\nSec-WebSocket-Accept: "
            + _accept_key(key).encode() + b"\r\n\r\n")
        await writer.drain()
        cid = cluster_id_from_token(token)
        conn = AgentConn(writer)
        self.rt.kubectl_agents[cid] = conn
        self.rt.org(org_id).audit("kubectl_agent_connected", cid)
        try:
            while True:
                opcode, data = await read_frame(reader)
                if opcode == 0x8:
                    return
                if opcode == 0x9:
                    writer.write(encode_frame(data, 0xA))
                    await writer.drain()
                    continue
                if opcode not in (0x1, 0x2):
                    continue
                try:
                    msg = json.loads(data)
                except json.JSONDecodeError:
                    continue
                if msg.get("type") == "result":
                    conn.resolve(msg.get("id"), msg.get("output", ""))
        finally:
            if self.rt.kubectl_agents.get(cid) is conn:
                del self.rt.kubectl_agents[cid]

    async def _send(self, writer, obj: dict) -> None:
        writer.write(encode_frame(json.dumps(obj).encode()))
        await writer.drain()

    async def _session(self, reader, writer, user: dict) -> None:
        rt = self.rt
        org_id = user["org_id"]
        store = rt.org(org_id)
        session_id: Optional[str] = None
        incident_id: Optional[str] = None
        mode = "agent"
        confirm_tools = False
        pending_confirm: Dict[str, asyncio.Future] = {}
        current: Optional[asyncio.Task] = None
        tok = rt.engine.tokenizer()

        async def send(obj):
            await self._send(writer, obj)

        await send({"type": "ready", "user_id": user["id"]})
        while True:
            opcode, data = await read_frame(reader)
            if opcode == 0x8:  # close / EOF
                if current and not current.done():
                    current.cancel()
                return
            if opcode == 0x9:  # ping
                writer.write(encode_frame(data, 0xA))
                await writer.drain()
                continue
            if opcode not in (0x1, 0x2):
                continue
            try:
                msg = json.loads(data)
            except json.JSONDecodeError:
                await send({"type": "error", "error": "invalid JSON"})
                continue
            mtype = msg.get("type")
            if mtype == "init":
                session_id = msg.get("session_id") or store.create_session(
                    user_id=user["id"])
                if store.get_session(session_id) is None:
                    session_id = store.create_session(user_id=user["id"],
                                                      session_id=session_id)
                incident_id = msg.get("incident_id")
                mode = msg.get("mode", "agent")
                confirm_tools = bool(msg.get("confirm_tools", False))
                await send({"type": "initialized", "session_id": session_id})
                continue
            if mtype == "confirmation_response":
                fut = pending_confirm.pop(msg.get("id", ""), None)
                if fut is not None and not fut.done():
                    fut.set_result(bool(msg.get("approved", False)))
                continue
            if mtype == "control:cancel":
                # reference: main_chatbot.py:1041-1157 — cancel, consolidate
                if current and not current.done():
                    current.cancel()
                    await send({"type": "cancelled"})
                continue
            if mtype != "query":
                await send({"type": "error", "error": f"unknown type {mtype}"})
                continue
            if not self.rate.allow(user["id"]):
                await send({"type": "error", "error": "rate limited"})
                continue
            question = msg.get("content", "")
            if tok.count(question) > rt.settings.max_input_tokens:
                await send({"type": "error",
                            "error": "input exceeds token cap"})  # :1445-1474
                continue
            if session_id is None:
                session_id = store.create_session(user_id=user["id"])
            if current and not current.done():
                await send({"type": "error", "error": "busy"})
                continue
            confirmer = None
            if confirm_tools:
                async def confirmer(tool, targs):
                    import uuid as _u
                    cid = _u.uuid4().hex[:10]
                    fut = asyncio.get_running_loop().create_future()
                    pending_confirm[cid] = fut
                    await send({"type": "confirmation_request", "id": cid,
                                "tool": tool, "args": targs})
                    return await fut
            current = asyncio.create_task(self._run_query(
                send, org_id, user["id"], session_id, incident_id, mode,
                question, confirmer))

    async def _run_query(self, send, org_id, user_id, session_id, incident_id,
                         mode, question, confirmer=None) -> None:
        rt = self.rt
        try:
            ctx = rt.tool_ctx(org_id, user_id=user_id, session_id=session_id,
                              incident_id=incident_id, mode=mode, notifier=send)
            ctx.confirmer = confirmer
            wf = Workflow(rt.llm(org_id), ctx)
            state = State(question=question, user_id=user_id, org_id=org_id,
                          session_id=session_id, incident_id=incident_id, mode=mode)
            # 30-min workflow ceiling (reference: main_chatbot.py:337)
            async def _pump():
                async for ev in wf.stream(state):
                    await send(ev)
            await asyncio.wait_for(_pump(), timeout=rt.settings.workflow_timeout_s)
        except asyncio.TimeoutError:
            await self._try_send(send, {"type": "error", "error": "workflow timeout"})
        except asyncio.CancelledError:
            await self._try_send(send, {"type": "final", "content": "(cancelled)"})
        except (ConnectionResetError, BrokenPipeError):
            pass  # client went away mid-stream
        except Exception as e:  # noqa: BLE001
            await self._try_send(send, {"type": "error", "error": str(e)[:500]})

    @staticmethod
    async def _try_send(send, obj) -> None:
        try:
            await send(obj)
        except (ConnectionResetError, BrokenPipeError):
            pass
