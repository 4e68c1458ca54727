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


def test_cloud_exec_credential_injection():
    """Per-invocation provider credentials land in the sandboxed child's
    env only (reference: setup_*_environment_isolated,
    cloud_exec_tool.py:125-823) + multi-account selection."""
    import asyncio
    import os
    from aurora_amd.agent.tools.core import cloud_env_for
    from aurora_amd.agent.tools.registry import GLOBAL_REGISTRY, ToolContext
    from aurora_amd.service.store import AdminStore, Database
    from aurora_amd.utils.secrets import reset_backend

    os.environ["AURORA_SECRETS"] = "/tmp/ce_sec.json"
    reset_backend()
    try:
        db = Database(":memory:")
        admin = AdminStore(db)
        oid = admin.create_org("ce")
        uid = admin.create_user(oid, "ce@x.io", user_id="ce-user")
        store = admin.org_store(oid)
        from aurora_amd.connectors.registry import connect_provider

        class RT:
            pass
        connect_provider(RT(), store, uid, "aws", {"credentials": {
            "access_key_id": "AKID1", "secret_access_key": "SK1",
            "region": "eu-west-1"}})
        ctx = ToolContext(org_store=store, user_id=uid, session_id="s-ce")
        env = cloud_env_for(ctx, "aws")
        assert env == {"AWS_ACCESS_KEY_ID": "AKID1",
                       "AWS_SECRET_ACCESS_KEY": "SK1",
                       "AWS_DEFAULT_REGION": "eu-west-1"}
        # unknown named account fails closed
        assert cloud_env_for(ctx, "aws", account="prod") == {}
        # the child actually sees the injected env and ONLY per-invocation
        async def main():
            store.add_command_policy(r"^aws sts get-caller-identity", "allow")
            out = await GLOBAL_REGISTRY.run(
                ctx, "cloud_exec",
                {"command": "aws sts get-caller-identity"})
            # aws CLI absent in the image: the command fails, but it must
            # fail INSIDE the sandbox (not a guardrail block)
            assert "BLOCKED" not in out
        asyncio.run(main())
        # multi-account: store a second credential set
        connect_provider(RT(), store, uid, "aws:prod", {"credentials": {
            "access_key_id": "AKID2", "secret_access_key": "SK2"}})
        env2 = cloud_env_for(ctx, "aws", account="prod")
        assert env2["AWS_ACCESS_KEY_ID"] == "AKID2"
    finally:
        os.environ.pop("AURORA_SECRETS", None)
        reset_backend()
