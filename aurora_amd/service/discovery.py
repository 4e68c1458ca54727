"""Dependency inference: turn environment evidence into graph edges.

Rebuilds the reference's discovery inference modules (reference:
services/discovery/inference/ — dns_inference.py, env_var_inference.py,
load_balancer_inference.py, connection_inference.py among 12) as pure
parsers over collected evidence, so each is testable against recorded
fixtures; ``run_inference`` feeds the results into the org's infra graph
(reference: services/discovery/graph_writer.py). Collection itself comes
from prediscovery's probes (local env, kubectl) or provider payloads.
"""
from __future__ import annotations

import json
import re
from typing import Dict, Iterable, List, Tuple

Edge = Tuple[str, str, str]  # (service, depends_on, via)

# ----------------------------------------------------------- env vars

#: VAR name shapes that name a dependency endpoint
_ENV_DEP = re.compile(
    r"^(?P<name>[A-Z0-9_]+?)_?(HOST|HOSTNAME|URL|URI|ADDR|ADDRESS|ENDPOINT|DSN)$")
#: generic url schemes -> dependency kind
_SCHEME_KIND = {
    "postgres": "database", "postgresql": "database", "mysql": "database",
    "redis": "cache", "rediss": "cache", "amqp": "queue", "kafka": "queue",
    "mongodb": "database", "http": "service", "https": "service",
    "grpc": "service", "nats": "queue", "memcached": "cache",
}

_SKIP_HOSTS = {"localhost", "127.0.0.1", "0.0.0.0", "::1", ""}


def _host_of(value: str) -> str:
    v = value.strip().strip('"\'')
    m = re.match(r"^(?P<scheme>[a-z][a-z0-9+.-]*)://", v)
    if m:
        rest = v[m.end():]
        host = re.split(r"[/:@?]", rest.split("@")[-1], 1)[0]
        return host.lower()
    # bare host[:port]
    return re.split(r"[:/]", v, 1)[0].lower()


def infer_env_dependencies(service: str,
                           env: Dict[str, str]) -> List[Edge]:
    """DATABASE_URL / REDIS_HOST / PAYMENTS_ENDPOINT ... -> edges
    (reference: env_var_inference.py)."""
    edges: List[Edge] = []
    for key, value in env.items():
        if not value:
            continue
        m = _ENV_DEP.match(key.upper())
        if not m:
            continue
        host = _host_of(value)
        if host in _SKIP_HOSTS or not host:
            continue
        scheme = (re.match(r"^([a-z][a-z0-9+.-]*)://", value.strip()) or
                  [None, ""])[1] if "://" in value else ""
        kind = _SCHEME_KIND.get(scheme, "service")
        target = host.split(".")[0] or host
        edges.append((service, target, f"env:{key}"))
        _ = kind
    return edges


# ---------------------------------------------------------------- dns

def infer_dns_dependencies(service: str, hosts_file: str,
                           known_services: Iterable[str]) -> List[Edge]:
    """/etc/hosts + resolver evidence: hostnames that resolve locally and
    match known service names become edges (reference: dns_inference.py
    — offline subset: no live lookups without egress)."""
    known = {k.lower() for k in known_services}
    edges: List[Edge] = []
    for line in hosts_file.splitlines():
        line = line.split("#", 1)[0].strip()
        if not line:
            continue
        parts = line.split()
        for name in parts[1:]:
            base = name.split(".")[0].lower()
            if base in known and base != service:
                edges.append((service, base, "dns:hosts"))
    return edges


# -------------------------------------------------------- k8s ingress/LB

def infer_lb_dependencies(ingress_json: str) -> List[Edge]:
    """Ingress -> backend service edges from `kubectl get ingress -o json`
    (reference: load_balancer_inference.py)."""
    try:
        doc = json.loads(ingress_json)
    except json.JSONDecodeError:
        return []
    edges: List[Edge] = []
    for item in doc.get("items", []):
        ing = (item.get("metadata") or {}).get("name", "ingress")
        spec = item.get("spec") or {}
        for rule in spec.get("rules", []):
            for path in ((rule.get("http") or {}).get("paths") or []):
                be = (path.get("backend") or {})
                svc = ((be.get("service") or {}).get("name")
                       or be.get("serviceName"))
                if svc:
                    host = rule.get("host", "*")
                    edges.append((ing, svc, f"ingress:{host}"))
        # default backend
        be = (spec.get("defaultBackend") or {}).get("service") or {}
        if be.get("name"):
            edges.append((ing, be["name"], "ingress:default"))
    return edges


# ------------------------------------------------------- live connections

def infer_connection_dependencies(service_ports: Dict[str, int],
                                  ss_output: str) -> List[Edge]:
    """Established TCP peers: a process whose local port belongs to
    service A talking to the listen port of service B => A depends on B
    (reference: connection_inference.py / network_proximity)."""
    port_to_svc = {p: s for s, p in service_ports.items()}
    edges: set = set()
    for line in ss_output.splitlines():
        if "ESTAB" not in line:
            continue
        m = re.findall(r"[\[\]0-9a-f.:*]+:(\d+)\b", line)
        if len(m) < 2:
            continue
        lp, rp = int(m[0]), int(m[1])
        src = port_to_svc.get(lp)
        dst = port_to_svc.get(rp)
        if src and dst and src != dst:
            edges.add((src, dst, f"tcp:{rp}"))
    return sorted(edges)


# ------------------------------------------------------------- writer

def write_edges(graph, edges: List[Edge]) -> int:
    """Feed inferred edges into the infra graph (reference:
    graph_writer.py): nodes upserted, duplicate edges collapsed."""
    n = 0
    seen = set()
    for svc, dep, via in edges:
        if (svc, dep) in seen:
            continue
        seen.add((svc, dep))
        graph.upsert_service(svc)
        graph.upsert_service(dep)
        graph.add_dependency(svc, dep, via=via)
        n += 1
    return n


async def run_inference(graph, service: str = "host",
                        env: Dict[str, str] = None) -> int:
    """Collect local evidence and run every inference module."""
    import asyncio
    import os

    async def _sh(cmd: str) -> str:
        try:
            proc = await asyncio.create_subprocess_shell(
                cmd, stdout=asyncio.subprocess.PIPE,
                stderr=asyncio.subprocess.DEVNULL)
            out, _ = await asyncio.wait_for(proc.communicate(), timeout=10)
            return out.decode("utf-8", "replace")
        except Exception:  # noqa: BLE001
            return ""

    edges: List[Edge] = []
    edges += infer_env_dependencies(service, env if env is not None
                                    else dict(os.environ))
    try:
        with open("/etc/hosts") as f:
            hosts = f.read()
    except OSError:
        hosts = ""
    known = [s.get("name", "") for s in graph.list_services()]
    edges += infer_dns_dependencies(service, hosts, known)
    ing = await _sh("kubectl get ingress -A -o json 2>/dev/null")
    if ing:
        edges += infer_lb_dependencies(ing)
    return write_edges(graph, edges)
