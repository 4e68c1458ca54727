"""Service entrypoint: run the full stack in one process.

    python -m aurora_amd.service.serve [--api-port 5080] [--ws-port 5006]

Starts the FastAPI REST API (webhooks/incidents/SSE/metrics/...), the
WebSocket chatbot (+ kubectl-agent pairing on the same port), the
background worker pool with beats, and the shared engine — the
one-process-per-GPU deployment unit.  For DP serving, launch one of these
per GPU (torchrun or systemd template) behind any HTTP load balancer;
webhook fan-in shards across them exactly like the reference's Celery
workers sharded investigations.
"""
from __future__ import annotations

import argparse
import asyncio
import logging

import uvicorn

from ..utils.config import settings as get_settings
from ..utils.log_sanitizer import configure_logging
from .app import create_app
from .runtime import Runtime
from .ws import ChatServer

log = logging.getLogger("aurora.serve")


async def amain(args) -> None:
    configure_logging()
    cfg = get_settings()
    rt = Runtime(cfg)
    # bootstrap org/user for fresh installs
    if not rt.admin.orgs():
        oid = rt.admin.create_org("default")
        uid = rt.admin.create_user(oid, "admin@localhost", role="admin")
        log.info("bootstrapped org=%s admin user=%s", oid, uid)

    chat = ChatServer(rt)
    ws_port = await chat.start("0.0.0.0", args.ws_port)
    log.info("chat websocket on :%d", ws_port)

    app = create_app(rt)
    config = uvicorn.Config(app, host="0.0.0.0", port=args.api_port,
                            log_level="warning", loop="asyncio")
    server = uvicorn.Server(config)
    log.info("REST API on :%d", args.api_port)
    try:
        await server.serve()
    finally:
        await chat.stop()
        await rt.stop()
        rt.engine.shutdown()


def main() -> None:
    cfg = get_settings()
    ap = argparse.ArgumentParser()
    ap.add_argument("--api-port", type=int, default=cfg.api_port)
    ap.add_argument("--ws-port", type=int, default=cfg.ws_port)
    args = ap.parse_args()
    asyncio.run(amain(args))


if __name__ == "__main__":
    main()
