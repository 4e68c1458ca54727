"""Per-session terminal sandboxes with resource limits, network
isolation and idle cleanup.

The reference runs agent commands in per-user K8s pods with a
NetworkPolicy and an idle-cleanup Celery beat (reference:
utils/terminal/terminal_pod_manager.py:68,298; terminal_pod_cleanup.py:85
every 10 min). This deployment has no cluster, so the same isolation
contract is built from host primitives:

* per-session working directory (scratch survives across the session's
  commands, like the pod's filesystem);
* rlimits in the child (CPU seconds, address space, file size, nproc,
  open files) — a runaway diagnostic cannot take the host down;
* network isolation via ``unshare -rn`` (user+net namespaces: loopback
  only) when the host allows unprivileged namespaces — probed once, and
  the sandbox records whether commands run network-isolated;
* idle cleanup: sandboxes idle past the TTL are removed by the runtime
  beat.
"""
from __future__ import annotations

import asyncio
import os
import shutil
import subprocess
import tempfile
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

#: rlimits applied in the child (soft, hard)
_DEFAULT_LIMITS = {
    "cpu_s": 30,              # RLIMIT_CPU
    "as_bytes": 2 << 30,      # RLIMIT_AS: 2 GB
    "fsize_bytes": 256 << 20,  # RLIMIT_FSIZE: 256 MB
    "nproc": 64,
    "nofile": 256,
}


def _probe_unshare() -> bool:
    """Can this host create unprivileged user+net namespaces?"""
    try:
        r = subprocess.run(["unshare", "-rn", "true"], capture_output=True,
                           timeout=5)
        return r.returncode == 0
    except (OSError, subprocess.TimeoutExpired):
        return False


@dataclass
class Sandbox:
    session_id: str
    root: str
    created_at: float = field(default_factory=time.time)
    last_used: float = field(default_factory=time.time)
    commands_run: int = 0


class SandboxManager:
    def __init__(self, base_dir: Optional[str] = None,
                 idle_ttl_s: float = 1800.0, limits: Optional[dict] = None,
                 isolate_network: Optional[bool] = None):
        self.base_dir = base_dir or os.path.join(tempfile.gettempdir(),
                                                 "aurora_sandboxes")
        os.makedirs(self.base_dir, exist_ok=True)
        self.idle_ttl_s = idle_ttl_s
        self.limits = dict(_DEFAULT_LIMITS, **(limits or {}))
        self._boxes: Dict[str, Sandbox] = {}
        self.network_isolation = (_probe_unshare() if isolate_network is None
                                  else isolate_network)

    def get(self, session_id: str) -> Sandbox:
        sid = (session_id or "default").replace("/", "_")[:64]
        box = self._boxes.get(sid)
        if box is None or not os.path.isdir(box.root):
            root = os.path.join(self.base_dir, f"sb-{sid}")
            os.makedirs(root, mode=0o700, exist_ok=True)
            box = Sandbox(session_id=sid, root=root)
            self._boxes[sid] = box
        return box

    def _preexec(self):
        import resource
        lim = self.limits

        def fn():
            os.setsid()  # own process group: timeouts kill the whole tree
            resource.setrlimit(resource.RLIMIT_CPU,
                               (lim["cpu_s"], lim["cpu_s"] + 5))
            resource.setrlimit(resource.RLIMIT_AS,
                               (lim["as_bytes"], lim["as_bytes"]))
            resource.setrlimit(resource.RLIMIT_FSIZE,
                               (lim["fsize_bytes"], lim["fsize_bytes"]))
            try:
                resource.setrlimit(resource.RLIMIT_NPROC,
                                   (lim["nproc"], lim["nproc"]))
            except (ValueError, OSError):
                pass  # host soft limit may already be lower
            resource.setrlimit(resource.RLIMIT_NOFILE,
                               (lim["nofile"], lim["nofile"]))
        return fn

    async def run(self, session_id: str, argv: Optional[List[str]],
                  shell_command: Optional[str] = None,
                  timeout_s: float = 60.0,
                  extra_env: Optional[Dict[str, str]] = None) -> dict:
        """Run argv (exec, preferred) or a shell command inside the
        session's sandbox. Returns {stdout, stderr, returncode,
        timed_out, isolated}."""
        box = self.get(session_id)
        box.last_used = time.time()
        box.commands_run += 1
        env = {"PATH": "/usr/bin:/bin:/usr/sbin:/sbin", "HOME": box.root,
               "TMPDIR": box.root, "LANG": "C.UTF-8"}
        if extra_env:
            env.update({str(k): str(v) for k, v in extra_env.items()})
        if argv is not None:
            cmd = list(argv)
        else:
            cmd = ["/bin/sh", "-c", shell_command or ""]
        if self.network_isolation:
            cmd = ["unshare", "-rn"] + cmd
        loop = asyncio.get_running_loop()

        def _spawn():
            return subprocess.Popen(
                cmd, cwd=box.root, env=env, stdout=subprocess.PIPE,
                stderr=subprocess.PIPE, preexec_fn=self._preexec())
        try:
            proc = await loop.run_in_executor(None, _spawn)
        except FileNotFoundError as e:
            return {"stdout": "", "stderr": f"command not found: {e}",
                    "returncode": 127, "timed_out": False,
                    "isolated": self.network_isolation}

        def _wait():
            return proc.communicate(timeout=timeout_s)
        try:
            out, err = await loop.run_in_executor(None, _wait)
            timed_out = False
        except subprocess.TimeoutExpired:
            try:
                os.killpg(proc.pid, 9)
            except (ProcessLookupError, PermissionError):
                proc.kill()
            out, err = proc.communicate()
            timed_out = True
        return {"stdout": out.decode("utf-8", "replace"),
                "stderr": err.decode("utf-8", "replace"),
                "returncode": proc.returncode, "timed_out": timed_out,
                "isolated": self.network_isolation}

    # ------------------------------------------------------------ cleanup

    def cleanup_idle(self, now: Optional[float] = None) -> int:
        """Remove sandboxes idle past the TTL (reference:
        terminal_pod_cleanup.py beat)."""
        t = now if now is not None else time.time()
        removed = 0
        for sid, box in list(self._boxes.items()):
            if t - box.last_used > self.idle_ttl_s:
                shutil.rmtree(box.root, ignore_errors=True)
                del self._boxes[sid]
                removed += 1
        return removed

    def destroy_all(self) -> None:
        for sid in list(self._boxes):
            shutil.rmtree(self._boxes.pop(sid).root, ignore_errors=True)


_manager: Optional[SandboxManager] = None


def sandbox_manager() -> SandboxManager:
    global _manager
    if _manager is None:
        _manager = SandboxManager()
    return _manager
