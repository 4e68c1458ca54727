"""Per-session terminal sandbox (reference: terminal_pod_manager.py:68 —
isolation, resource limits, idle cleanup)."""
import asyncio
import os
import time

from aurora_amd.agent.tools.sandbox import SandboxManager


def test_sandbox_isolation_and_persistence(tmp_path):
    async def main():
        m = SandboxManager(base_dir=str(tmp_path), isolate_network=False)
        # scratch persists across commands of one session
        r = await m.run("sess-a", ["sh", "-c", "echo hello > f.txt"])
        assert r["returncode"] == 0
        r = await m.run("sess-a", ["cat", "f.txt"])
        assert r["stdout"].strip() == "hello"
        # other sessions can't see it
        r = await m.run("sess-b", ["cat", "f.txt"])
        assert r["returncode"] != 0
        # env is stripped: no tokens leak into the child
        os.environ["SECRET_TOKEN_X"] = "leakme"
        try:
            r = await m.run("sess-a", ["sh", "-c", "env"])
            assert "leakme" not in r["stdout"]
        finally:
            del os.environ["SECRET_TOKEN_X"]
        # unknown binary: explicit 127, no exception
        r = await m.run("sess-a", ["definitely-not-a-binary-xyz"])
        assert r["returncode"] == 127
        m.destroy_all()

    asyncio.run(main())


def test_sandbox_timeout_kills_process_tree(tmp_path):
    async def main():
        m = SandboxManager(base_dir=str(tmp_path), isolate_network=False)
        t0 = time.monotonic()
        r = await m.run("s", ["sh", "-c", "sleep 30 & sleep 30"],
                        timeout_s=1.0)
        assert r["timed_out"] and time.monotonic() - t0 < 10
        m.destroy_all()

    asyncio.run(main())


def test_sandbox_rlimits(tmp_path):
    async def main():
        m = SandboxManager(base_dir=str(tmp_path), isolate_network=False,
                           limits={"fsize_bytes": 1 << 20})
        # file-size limit: writing 8 MB must fail
        r = await m.run("s", ["sh", "-c",
                              "dd if=/dev/zero of=big bs=1M count=8 2>&1; "
                              "ls -la big"])
        assert "8388608" not in r["stdout"]
        m.destroy_all()

    asyncio.run(main())


def test_sandbox_idle_cleanup(tmp_path):
    async def main():
        m = SandboxManager(base_dir=str(tmp_path), idle_ttl_s=100.0,
                           isolate_network=False)
        await m.run("old", ["true"])
        await m.run("fresh", ["true"])
        m._boxes["old"].last_used = time.time() - 1000
        root_old = m._boxes["old"].root
        assert m.cleanup_idle() == 1
        assert not os.path.isdir(root_old)
        assert "fresh" in m._boxes
        m.destroy_all()

    asyncio.run(main())


def test_network_isolation_probe():
    m = SandboxManager(isolate_network=None)
    assert isinstance(m.network_isolation, bool)
    if m.network_isolation:
        async def main():
            # with -rn there is no route to anywhere but loopback
            r = await m.run("n", ["sh", "-c", "ip route 2>/dev/null; true"])
            assert "default" not in r["stdout"]
        asyncio.run(main())
    m.destroy_all()
