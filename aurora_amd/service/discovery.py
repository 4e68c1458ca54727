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


# ----------------------------------------------------- provider assets
# (reference: services/discovery/providers/ — asset discovery per cloud;
# parsers are pure over the CLI/API JSON so they fixture-test without
# credentials; collection rides cloud_exec's per-invocation creds)


def parse_aws_instances(describe_json: str) -> List[dict]:
    """`aws ec2 describe-instances` -> asset nodes."""
    try:
        doc = json.loads(describe_json)
    except json.JSONDecodeError:
        return []
    out = []
    for res in doc.get("Reservations", []):
        for inst in res.get("Instances", []):
            name = next((t.get("Value") for t in inst.get("Tags", [])
                         if t.get("Key") == "Name"), None)
            out.append({
                "name": name or inst.get("InstanceId", "?"),
                "kind": "aws_ec2",
                "id": inst.get("InstanceId", ""),
                "state": (inst.get("State") or {}).get("Name", ""),
                "type": inst.get("InstanceType", ""),
                "private_ip": inst.get("PrivateIpAddress", ""),
            })
    return out


def parse_aws_rds(describe_json: str) -> List[dict]:
    """`aws rds describe-db-instances` -> asset nodes."""
    try:
        doc = json.loads(describe_json)
    except json.JSONDecodeError:
        return []
    return [{
        "name": db.get("DBInstanceIdentifier", "?"), "kind": "aws_rds",
        "id": db.get("DbiResourceId", ""),
        "state": db.get("DBInstanceStatus", ""),
        "engine": db.get("Engine", ""),
        "endpoint": (db.get("Endpoint") or {}).get("Address", ""),
    } for db in doc.get("DBInstances", [])]


def parse_aws_lambda(list_json: str) -> List[dict]:
    try:
        doc = json.loads(list_json)
    except json.JSONDecodeError:
        return []
    return [{"name": f.get("FunctionName", "?"), "kind": "aws_lambda",
             "id": f.get("FunctionArn", ""), "runtime": f.get("Runtime", "")}
            for f in doc.get("Functions", [])]


def parse_gcp_instances(list_json: str) -> List[dict]:
    """`gcloud compute instances list --format json`."""
    try:
        doc = json.loads(list_json)
    except json.JSONDecodeError:
        return []
    if not isinstance(doc, list):
        return []
    return [{"name": i.get("name", "?"), "kind": "gcp_instance",
             "id": str(i.get("id", "")), "state": i.get("status", ""),
             "zone": (i.get("zone", "").rsplit("/", 1) or [""])[-1]}
            for i in doc if isinstance(i, dict)]


def write_assets(graph, assets: List[dict],
                 endpoints_to_services: bool = True) -> int:
    """Asset nodes into the graph; RDS/LB endpoints become dependency
    targets discoverable by the env/dns inference (resource_mapper.py
    analog)."""
    n = 0
    for a in assets:
        props = {k: v for k, v in a.items() if k not in ("name", "kind") and v}
        graph.upsert_service(a["name"], kind=a["kind"], **props)
        n += 1
        ep = a.get("endpoint", "")
        if endpoints_to_services and ep:
            base = ep.split(".")[0]
            if base and base != a["name"]:
                graph.upsert_service(base, kind="endpoint")
                graph.add_dependency(base, a["name"], via="endpoint")
    return n


async def discover_cloud_assets(ctx, graph) -> int:
    """Collect via the guarded cloud_exec path when provider credentials
    are connected; every CLI absence degrades to zero assets."""
    from ..agent.tools.registry import GLOBAL_REGISTRY
    total = 0
    cmds = [
        ("aws", "aws ec2 describe-instances --output json",
         parse_aws_instances),
        ("aws", "aws rds describe-db-instances --output json", parse_aws_rds),
        ("aws", "aws lambda list-functions --output json", parse_aws_lambda),
        ("gcp", "gcloud compute instances list --format json",
         parse_gcp_instances),
    ]
    connected = []
    if ctx.org_store is not None and ctx.user_id:
        connected = ctx.org_store.connected_providers(ctx.user_id)
    for provider, cmd, parser in cmds:
        if provider not in connected:
            continue
        out = await GLOBAL_REGISTRY.run(ctx, "cloud_exec", {"command": cmd})
        if out.startswith(("ERROR", "BLOCKED", "CANCELLED")):
            continue
        total += write_assets(graph, parser(out))
    return total


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
