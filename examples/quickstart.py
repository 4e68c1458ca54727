"""End-to-end quickstart on CPU (no GPU needed): starts the full stack
with the tiny CPU model, sends a PagerDuty-style webhook, waits for the
background RCA to complete, and prints the RCA + postmortem.

    python examples/quickstart.py

On an MI355X box set AURORA_MODEL=llama3-8b and AURORA_CHECKPOINT to an
HF safetensors dir for real output quality; the pipeline is identical.
"""
import asyncio
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from fastapi.testclient import TestClient

from aurora_amd.engine.engine import EngineConfig
from aurora_amd.engine.server import AsyncEngineServer
from aurora_amd.service.app import create_app
from aurora_amd.service.runtime import Runtime
from aurora_amd.utils.config import Settings


def main() -> None:
    model = os.environ.get("AURORA_MODEL", "tiny-llama")
    device = "cuda" if os.environ.get("AURORA_CHECKPOINT") else "cpu"
    cfg = Settings(db_path=":memory:", secrets_path="/tmp/qs_secrets.json",
                   rca_max_turns=int(os.environ.get("QS_TURNS", "3")))
    server = AsyncEngineServer(EngineConfig(
        model=model, device=device, seed=0,
        checkpoint_path=os.environ.get("AURORA_CHECKPOINT")))
    rt = Runtime(cfg, engine_server=server)
    org = rt.admin.create_org("demo")
    rt.admin.create_user(org, "oncall@demo.io", user_id="oncall")
    app = create_app(rt)

    with TestClient(app) as client:
        h = {"X-User-Id": "oncall", "X-Org-Id": org}
        # 1. alert arrives (PagerDuty v3 shape; unsigned since no secret set)
        r = client.post(f"/webhook/pagerduty/{org}", json={
            "event": {"eventType": "incident.triggered", "data": {
                "id": "PD-1", "title": "checkout-api 5xx rate > 5%",
                "urgency": "high", "service": {"summary": "checkout-api"},
            }}})
        r.raise_for_status()
        # 2. wait for the background RCA task to finish
        iid = None
        for _ in range(600):
            time.sleep(0.2)
            incidents = client.get("/incidents", headers=h).json()
            if incidents:
                iid = incidents[0]["id"]
                if incidents[0]["aurora_status"] in ("completed", "failed"):
                    break
        inc = client.get(f"/incidents/{iid}", headers=h).json()
        print("== status:", inc["aurora_status"])
        print("== summary:\n", inc.get("summary") or "(none)")
        print("== root cause:\n", (inc.get("root_cause") or "")[:800])
        pm = client.get(f"/postmortems/{iid}", headers=h)
        if pm.status_code == 200:
            print("== postmortem:\n", json.dumps(pm.json(), indent=1)[:800])
        print("== dashboard: run `python -m aurora_amd.service.serve` "
              "and open http://localhost:8000/ui")
    server.shutdown()


if __name__ == "__main__":
    main()
