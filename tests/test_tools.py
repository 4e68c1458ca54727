"""Direct tests for the file/IaC/right-sizing tools (tools/core.py)."""
import asyncio
import base64
import io
import zipfile

import pytest

from aurora_amd.agent.tools.registry import GLOBAL_REGISTRY, ToolContext
import aurora_amd.agent.tools.core  # noqa: F401 - registers tools
from aurora_amd.service.store import AdminStore, Database


@pytest.fixture()
def ctx():
    db = Database(":memory:")
    admin = AdminStore(db)
    store = admin.org_store(admin.create_org("tools"))
    return ToolContext(org_store=store, user_id="u1")


def _run(tool_name, ctx, **kw):
    tool = GLOBAL_REGISTRY.get(tool_name)
    assert tool is not None, tool_name
    return asyncio.run(tool.func(ctx, **kw))


def test_artifact_roundtrip(ctx):
    out = _run("artifact_write", ctx, name="notes.md", content="# findings")
    assert "saved" in out
    assert "notes.md" in _run("artifact_list", ctx)
    assert "# findings" in _run("artifact_read", ctx, name="notes.md")
    assert "ERROR" in _run("artifact_read", ctx, name="nope.md")


def test_analyze_zip(ctx):
    buf = io.BytesIO()
    with zipfile.ZipFile(buf, "w") as z:
        z.writestr("app.log", "ERROR timeout connecting to db\nINFO ok\n" * 5)
        z.writestr("conf/settings.yaml", "retries: 3")
    ctx.org_store.save_artifact(
        "dump.zip", base64.b64encode(buf.getvalue()).decode())
    out = _run("analyze_zip_file", ctx, artifact="dump.zip")
    assert "app.log" in out and "settings.yaml" in out
    assert "ERROR: no artifact" in _run("analyze_zip_file", ctx, artifact="x")


def test_iac_tool_saves_config(ctx):
    out = _run("iac_tool", ctx, name="alerting",
               hcl='resource "x" "y" {}')
    assert "iac:alerting.tf" in out
    assert ctx.org_store.get_artifact("iac:alerting.tf")


def test_rightsizing_analysis(ctx):
    usage = "api-7f9 480m 900Mi\nworker-2b 120m 3Gi\n"
    out = _run("rightsizing_analysis", ctx, usage=usage,
               limits="api-7f9 500m 1Gi")
    assert "api-7f9" in out
