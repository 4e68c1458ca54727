"""Infrastructure dependency graph.

API surface mirrors the reference's MemgraphClient (reference:
services/graph/memgraph_client.py:38,127,318,456,470,518,555,573,606,661,833):
service/dependency upserts, downstream/upstream traversals, impact radius,
SPOF detection, incident<->service links with root-cause marking, change
tracking and staleness.  Storage is the org-scoped SQLite tables (no Bolt
server in this environment); when MEMGRAPH_URL is set and the neo4j driver
is importable, the same operations can be mirrored to a real Memgraph —
keeping the reference's graph API compatible (BASELINE.json).
"""
from __future__ import annotations

import json
import time
from collections import defaultdict, deque
from typing import Dict, List, Set


class InfraGraph:
    def __init__(self, org_store):
        self.store = org_store
        self._nodes: Dict[str, dict] = {}
        self._out: Dict[str, Set[str]] = defaultdict(set)  # svc -> dependencies
        self._in: Dict[str, Set[str]] = defaultdict(set)   # svc -> dependents
        self._load()

    def _load(self) -> None:
        for n in self.store.graph_nodes():
            self._nodes[n["id"]] = dict(json.loads(n["props"] or "{}"),
                                        kind=n["kind"], updated_at=n["updated_at"])
        for e in self.store.graph_edges():
            if e["kind"] == "depends_on":
                self._out[e["src"]].add(e["dst"])
                self._in[e["dst"]].add(e["src"])

    # ---- writes (discovery graph_writer analog) ----
    def upsert_service(self, name: str, kind: str = "service", **props) -> None:
        self._nodes[name] = dict(props, kind=kind, updated_at=time.time())
        self.store.upsert_graph_node(name, kind, props)

    def add_dependency(self, src: str, dst: str, **props) -> None:
        self._out[src].add(dst)
        self._in[dst].add(src)
        self.store.upsert_graph_edge(src, dst, "depends_on", props)

    def link_incident(self, incident_id: str, service: str,
                      root_cause: bool = False) -> None:
        self.store.upsert_graph_edge(incident_id, service, "affects",
                                     {"root_cause": root_cause})

    def mark_root_cause(self, incident_id: str, service: str) -> None:
        self.link_incident(incident_id, service, root_cause=True)

    def record_change(self, service: str, description: str) -> None:
        cid = f"change:{service}:{int(time.time()*1000)}"
        self.store.upsert_graph_node(cid, "change",
                                     {"service": service, "description": description})

    # ---- reads ----
    def list_services(self) -> List[dict]:
        return [dict(name=n, **p) for n, p in self._nodes.items()
                if p.get("kind") != "change"]

    def _bfs(self, start: str, adj: Dict[str, Set[str]], depth: int = 10) -> List[str]:
        seen = {start}
        order: List[str] = []
        q = deque([(start, 0)])
        while q:
            cur, d = q.popleft()
            if d >= depth:
                continue
            for nxt in sorted(adj.get(cur, ())):
                if nxt not in seen:
                    seen.add(nxt)
                    order.append(nxt)
                    q.append((nxt, d + 1))
        return order

    def downstream(self, service: str) -> List[str]:
        """Services `service` depends on (transitively)."""
        return self._bfs(service, self._out)

    def upstream(self, service: str) -> List[str]:
        """Services that depend on `service` (transitively)."""
        return self._bfs(service, self._in)

    def impact_radius(self, service: str) -> dict:
        up = self.upstream(service)
        return {"service": service, "impacted_dependents": up,
                "impact_count": len(up)}

    def single_points_of_failure(self) -> List[str]:
        """Services with >=2 transitive dependents and no sibling redundancy
        (every dependent reaches them on all paths — approximated as: the
        node is the only provider of its kind among its direct dependents'
        dependencies)."""
        out = []
        for svc in self._nodes:
            dependents = self._in.get(svc, set())
            if len(self.upstream(svc)) >= 2 and all(
                    len(self._out.get(d, set())) == 1 or
                    svc in self._out.get(d, set()) and
                    not (self._out[d] - {svc}) for d in dependents):
                if dependents:
                    out.append(svc)
        return sorted(out)

    def incident_services(self, incident_id: str) -> List[dict]:
        return [
            {"service": e["dst"],
             "root_cause": json.loads(e["props"] or "{}").get("root_cause", False)}
            for e in self.store.graph_edges()
            if e["src"] == incident_id and e["kind"] == "affects"]

    def stale_services(self, older_than_s: float = 86400.0) -> List[str]:
        cut = time.time() - older_than_s
        return [n for n, p in self._nodes.items()
                if p.get("kind") != "change" and p.get("updated_at", 0) < cut]

    def overview(self) -> dict:
        svcs = self.list_services()
        return {
            "services": len(svcs),
            "dependencies": sum(len(v) for v in self._out.values()),
            "spof": self.single_points_of_failure(),
            "names": sorted(s["name"] for s in svcs)[:50],
        }
