"""kubectl-agent: outbound-WebSocket tunnel deployed into customer clusters.

Rebuild of the reference's kubectl-agent (reference: kubectl-agent/src/
agent.py — 211-LoC Python agent, Helm-deployed, token-derived cluster_id
at agent.py:33-37, reconnect/heartbeat loop).  The agent dials OUT to the
chat WS port with its token, so clusters behind NAT need no inbound access;
the server pairs it by token (service/ws.py) and the on_prem_kubectl tool
proxies commands through it.
"""
from __future__ import annotations

import asyncio
import hashlib
import json
import shlex
import subprocess
from typing import Callable, Optional

from ..service.ws import mask_frame, read_frame


def cluster_id_from_token(token: str) -> str:
    # token-derived cluster id (reference: agent.py:33-37)
    return hashlib.sha256(token.encode()).hexdigest()[:12]


def default_runner(args: str, timeout: int = 60) -> str:
    try:
        out = subprocess.run(["kubectl", *shlex.split(args)], capture_output=True,
                             text=True, timeout=timeout)
        return out.stdout + (("\n[stderr]\n" + out.stderr) if out.stderr else "")
    except FileNotFoundError:
        return "ERROR: kubectl binary not found in the agent pod"
    except subprocess.TimeoutExpired:
        return f"ERROR: kubectl timed out after {timeout}s"


class KubectlAgent:
    def __init__(self, host: str, port: int, token: str,
                 runner: Optional[Callable[[str], str]] = None,
                 heartbeat_s: float = 20.0, reconnect_s: float = 3.0):
        self.host = host
        self.port = port
        self.token = token
        self.cluster_id = cluster_id_from_token(token)
        self.runner = runner or default_runner
        self.heartbeat_s = heartbeat_s
        self.reconnect_s = reconnect_s
        self._stop = asyncio.Event()

    async def run_forever(self) -> None:
        while not self._stop.is_set():
            try:
                await self._session()
            except (OSError, asyncio.IncompleteReadError, ConnectionError):
                pass
            if not self._stop.is_set():
                await asyncio.sleep(self.reconnect_s)

    def stop(self) -> None:
        self._stop.set()

    async def _session(self) -> None:
        reader, writer = await asyncio.open_connection(self.host, self.port)
        writer.write((
            f"GET /?agent_token={self.token} HTTP/1.1\r\nHost: {self.host}\r\n"
            f"Upgrade: websocket\r\nConnection: Upgrade\r\n"
            f"Sec-WebSocket-Key: a2ViY3RsLWFnZW50LWtleQ==\r\n"
            f"Sec-WebSocket-Version: 13\r\n\r\n").encode())
        await writer.drain()
        resp = await reader.readuntil(b"\r\n\r\n")
        if b"101" not in resp.split(b"\r\n")[0]:
            writer.close()
            return
        loop = asyncio.get_running_loop()
        last_beat = loop.time()
        while not self._stop.is_set():
            try:
                opcode, data = await asyncio.wait_for(
                    read_frame(reader), timeout=self.heartbeat_s)
            except asyncio.TimeoutError:
                writer.write(mask_frame(b"", 0x9))  # ping keepalive
                await writer.drain()
                continue
            if opcode == 0x8:
                return
            if opcode == 0xA:  # pong
                continue
            if opcode == 0x9:
                writer.write(mask_frame(data, 0xA))
                await writer.drain()
                continue
            try:
                msg = json.loads(data)
            except json.JSONDecodeError:
                continue
            if msg.get("type") == "exec":
                out = await loop.run_in_executor(
                    None, self.runner, msg.get("args", ""))
                writer.write(mask_frame(json.dumps(
                    {"type": "result", "id": msg.get("id"),
                     "output": out[:200000]}).encode()))
                await writer.drain()


def main():  # pragma: no cover - the in-cluster entrypoint
    import os
    agent = KubectlAgent(
        os.environ.get("AURORA_HOST", "127.0.0.1"),
        int(os.environ.get("AURORA_WS_PORT", "5006")),
        os.environ["AURORA_AGENT_TOKEN"],
    )
    asyncio.run(agent.run_forever())


if __name__ == "__main__":  # pragma: no cover
    main()
