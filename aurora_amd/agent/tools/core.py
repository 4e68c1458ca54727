"""Built-in agent tools (reference: chat/backend/agent/tools/* — the core
execution, introspection, knowledge and artifact surface rebuilt)."""
from __future__ import annotations

import json
import shlex
import tempfile
from typing import Any

from ...guardrails.command_safety import evaluate_command
from .registry import GLOBAL_REGISTRY as R
from .registry import ToolContext

# --------------------------------------------------------------- execution

#: shell metacharacters that make a command string more than a single argv
_SHELL_META = ("|", ";", "&", ">", "<", "`", "$", "\n", "\r", "*", "?", "~")


def _exec_argv(command: str) -> list[str] | None:
    """Tokenized argv when `command` is a single plain command, else None."""
    if any(ch in command for ch in _SHELL_META):
        return None
    try:
        argv = shlex.split(command)
    except ValueError:
        return None
    return argv or None


@R.add("terminal_exec",
       "Run a shell command in the sandboxed diagnostics environment "
       "(read-only infra commands; every command is safety-gated).",
       {"command": "the shell command to run"}, read_only=False,
       needs_confirmation=True)
async def terminal_exec(ctx: ToolContext, command: str, timeout: int = 60,
                        extra_env: dict = None) -> str:
    verdict = await evaluate_command(command, llm=ctx.llm, org_store=ctx.org_store,
                                     user_id=ctx.user_id)
    if not verdict.safe:
        return (f"BLOCKED by guardrail ({verdict.source}"
                f"{':' + verdict.rule_id if verdict.rule_id else ''}): {verdict.reason}")
    return await _run_sandboxed(ctx, command, extra_env or {}, timeout)


async def _run_sandboxed(ctx: ToolContext, command: str, extra_env: dict,
                         timeout: int = 60) -> str:
    """Post-judge execution: per-session sandbox (scratch dir, rlimits,
    unshare network isolation, idle cleanup — the reference's per-user K8s
    pods, terminal_pod_manager.py:68; sandbox.py is the host-primitive
    rebuild). extra_env injects per-invocation provider credentials into
    the child only."""
    from .sandbox import sandbox_manager
    argv = _exec_argv(command)
    # argv path: the judged string IS the executed command (no shell
    # re-interpretation); pipelines reach the shell path only after an
    # explicit judge or org-policy allow
    res = await sandbox_manager().run(
        ctx.session_id or ctx.user_id or "anon",
        argv, shell_command=None if argv is not None else command,
        timeout_s=timeout, extra_env=extra_env)
    if res["timed_out"]:
        return f"ERROR: command timed out after {timeout}s"
    text = res["stdout"]
    if res["stderr"]:
        text += "\n[stderr]\n" + res["stderr"]
    return text or f"(exit {res['returncode']}, no output)"


@R.add("kubectl",
       "Run a kubectl command against the connected cluster (safety-gated).",
       {"args": "kubectl arguments, e.g. 'get pods -n default'"}, read_only=False)
async def kubectl(ctx: ToolContext, args: str) -> str:
    return await terminal_exec(ctx, f"kubectl {args}")


#: CLI binary -> (connector name, credential -> env-var map). Mirrors the
#: reference's per-provider isolated env setup (reference:
#: cloud_exec_tool.py:125-823 setup_*_environment_isolated): credentials
#: are injected per invocation into the sandboxed child only, never into
#: the service process environment.
_CLOUD_ENV = {
    "aws": ("aws", {"access_key_id": "AWS_ACCESS_KEY_ID",
                    "secret_access_key": "AWS_SECRET_ACCESS_KEY",
                    "session_token": "AWS_SESSION_TOKEN",
                    "region": "AWS_DEFAULT_REGION"}),
    "az": ("azure", {"access_token": "AZURE_ACCESS_TOKEN",
                     "subscription": "AZURE_SUBSCRIPTION_ID"}),
    "gcloud": ("gcp", {"access_token": "CLOUDSDK_AUTH_ACCESS_TOKEN",
                       "project": "CLOUDSDK_CORE_PROJECT"}),
    "scw": ("scaleway", {"access_key": "SCW_ACCESS_KEY",
                         "secret_key": "SCW_SECRET_KEY",
                         "project_id": "SCW_DEFAULT_PROJECT_ID"}),
    "ovhcli": ("ovh", {"app_key": "OVH_APPLICATION_KEY",
                       "app_secret": "OVH_APPLICATION_SECRET",
                       "consumer_key": "OVH_CONSUMER_KEY"}),
    "tailscale": ("tailscale", {"api_key": "TS_API_KEY"}),
    "flyctl": ("flyio", {"token": "FLY_API_TOKEN"}),
}


def cloud_env_for(ctx: ToolContext, binname: str,
                  account: str = "") -> dict:
    """Per-invocation credential env for the sandboxed child. ``account``
    selects an alternate credential set stored under
    '<provider>:<account>' (the reference's AWS multi-account fan-out,
    cloud_exec_tool.py:1199)."""
    spec = _CLOUD_ENV.get(binname)
    if spec is None:
        return {}
    provider, mapping = spec
    from ...connectors.registry import get_credentials
    creds = None
    if ctx.org_store is not None and ctx.user_id:
        name = f"{provider}:{account}" if account else provider
        creds = get_credentials(ctx.org_store, ctx.user_id, name)
        if creds is None and account:
            return {}
    if not creds:
        return {}
    return {env: str(creds[k]) for k, env in mapping.items() if creds.get(k)}


@R.add("cloud_exec",
       "Run a cloud-provider CLI command (aws/az/gcloud...) with isolated "
       "per-invocation credentials (safety-gated).",
       {"command": "full CLI command starting with the provider binary",
        "account": "optional named account/profile (multi-account orgs)"},
       read_only=False)
async def cloud_exec(ctx: ToolContext, command: str, account: str = "") -> str:
    binname = shlex.split(command)[0] if command.strip() else ""
    if binname not in _CLOUD_ENV:
        return ("ERROR: cloud_exec only accepts "
                + "/".join(sorted(_CLOUD_ENV)))
    env = cloud_env_for(ctx, binname, account)
    if account and not env:
        return f"ERROR: no credentials stored for {binname} account {account!r}"
    return await terminal_exec(ctx, command, extra_env=env)


@R.add("aws_exec_all_accounts",
       "Run a read-only aws CLI command across every connected AWS "
       "account and aggregate the outputs (safety-gated once).",
       {"command": "aws CLI command"}, read_only=False)
async def aws_exec_all_accounts(ctx: ToolContext, command: str) -> str:
    """Reference: AWS multi-account fan-out (cloud_exec_tool.py:1199)."""
    binname = shlex.split(command)[0] if command.strip() else ""
    if binname != "aws":
        return "ERROR: this tool only accepts aws commands"
    if ctx.org_store is None or not ctx.user_id:
        return "ERROR: no credential store bound"
    accounts = [p.split(":", 1)[1] for p
                in ctx.org_store.connected_providers(ctx.user_id)
                if p.startswith("aws:")]
    targets = [""] + accounts  # default account + named ones
    verdict = await evaluate_command(command, llm=ctx.llm,
                                     org_store=ctx.org_store,
                                     user_id=ctx.user_id)
    if not verdict.safe:
        return f"BLOCKED by guardrail ({verdict.source}): {verdict.reason}"
    outs = []
    for acct in targets:
        env = cloud_env_for(ctx, "aws", acct)
        if acct and not env:
            continue
        out = await _run_sandboxed(ctx, command, env)
        outs.append(f"## account: {acct or 'default'}\n{out[:4000]}")
    return "\n\n".join(outs) if outs else "ERROR: no AWS accounts connected"


@R.add("on_prem_kubectl",
       "Run kubectl on a connected on-prem cluster through its outbound "
       "agent tunnel (safety-gated).",
       {"cluster": "cluster id (empty = the only connected cluster)",
        "args": "kubectl arguments"}, read_only=False, needs_confirmation=True)
async def on_prem_kubectl(ctx: ToolContext, args: str, cluster: str = "") -> str:
    agents = ctx.extra.get("kubectl_agents")
    if agents is None and ctx.queue is not None:
        agents = getattr(ctx.extra.get("runtime"), "kubectl_agents", None)
    if not agents:
        return "ERROR: no on-prem kubectl agent connected"
    if not cluster:
        if len(agents) != 1:
            return f"ERROR: specify cluster; connected: {sorted(agents)}"
        cluster = next(iter(agents))
    conn = agents.get(cluster)
    if conn is None:
        return f"ERROR: cluster '{cluster}' not connected"
    verdict = await evaluate_command(f"kubectl {args}", llm=ctx.llm,
                                     org_store=ctx.org_store, user_id=ctx.user_id)
    if not verdict.safe:
        return f"BLOCKED by guardrail ({verdict.source}): {verdict.reason}"
    return await conn.exec(args)


# ------------------------------------------------------------ introspection
# (reference: tools/introspection_tools.py:101-341)


@R.add("list_incidents", "List recent incidents for this org.",
       {"status": "optional filter: open|investigating|resolved"})
async def list_incidents(ctx: ToolContext, status: str = "") -> str:
    if ctx.org_store is None:
        return "ERROR: no data store bound"
    rows = ctx.org_store.list_incidents(status or None, limit=20)
    return json.dumps([
        {k: r[k] for k in ("id", "title", "source", "severity", "status",
                           "aurora_status", "created_at")} for r in rows], indent=1)


@R.add("get_incident", "Full record of one incident incl. summary and root cause.",
       {"incident_id": "the incident id"})
async def get_incident(ctx: ToolContext, incident_id: str) -> str:
    row = ctx.org_store.get_incident(incident_id)
    return json.dumps(row, indent=1) if row else f"ERROR: no incident {incident_id}"


@R.add("get_incident_timeline", "Lifecycle events + tool steps of an incident.",
       {"incident_id": "the incident id"})
async def get_incident_timeline(ctx: ToolContext, incident_id: str) -> str:
    ev = ctx.org_store.lifecycle_events(incident_id)
    steps = ctx.org_store.steps(incident_id)
    return json.dumps({"events": ev, "steps": [
        {k: s[k] for k in ("tool", "status", "started_at", "finished_at")}
        for s in steps]}, indent=1)


@R.add("get_alert_field", "Read a field from the triggering alert payload.",
       {"field": "dot-path into the alert payload, e.g. 'alert.severity'"})
async def get_alert_field(ctx: ToolContext, field: str = "") -> str:
    if not ctx.incident_id:
        return "ERROR: no incident bound to this session"
    inc = ctx.org_store.get_incident(ctx.incident_id)
    if not inc:
        return "ERROR: incident not found"
    doc: Any = json.loads(inc.get("payload") or "{}")
    if not field:
        return json.dumps(doc, indent=1)[:8000]
    for part in field.split("."):
        if isinstance(doc, dict) and part in doc:
            doc = doc[part]
        else:
            return f"ERROR: field '{field}' not present"
    return json.dumps(doc) if isinstance(doc, (dict, list)) else str(doc)


@R.add("list_services", "List services known to the infrastructure graph.")
async def list_services(ctx: ToolContext) -> str:
    if ctx.graph is None:
        return "ERROR: no infrastructure graph available"
    return json.dumps(ctx.graph.list_services(), indent=1)


@R.add("get_service_impact",
       "Downstream impact radius of a service from the dependency graph.",
       {"service": "service name"})
async def get_service_impact(ctx: ToolContext, service: str) -> str:
    if ctx.graph is None:
        return "ERROR: no infrastructure graph available"
    return json.dumps(ctx.graph.impact_radius(service), indent=1)


@R.add("get_infrastructure_context",
       "Dependency-graph context around a service (up/downstream, SPOFs).",
       {"service": "service name (empty = whole-graph overview)"})
async def get_infrastructure_context(ctx: ToolContext, service: str = "") -> str:
    if ctx.graph is None:
        return "ERROR: no infrastructure graph available"
    if not service:
        return json.dumps(ctx.graph.overview(), indent=1)
    return json.dumps({
        "upstream": ctx.graph.upstream(service),
        "downstream": ctx.graph.downstream(service),
        "spof": ctx.graph.single_points_of_failure(),
    }, indent=1)


@R.add("save_infrastructure_context",
       "Record a discovered service or dependency into the infra graph.",
       {"service": "service name", "depends_on": "optional dependency name",
        "kind": "node kind, e.g. service/database/queue"}, read_only=False)
async def save_infrastructure_context(ctx: ToolContext, service: str,
                                      depends_on: str = "", kind: str = "service") -> str:
    if ctx.graph is None:
        return "ERROR: no infrastructure graph available"
    ctx.graph.upsert_service(service, kind=kind)
    if depends_on:
        ctx.graph.upsert_service(depends_on)
        ctx.graph.add_dependency(service, depends_on)
    return f"saved: {service}" + (f" -> {depends_on}" if depends_on else "")


# ----------------------------------------------------------- knowledge base


@R.add("knowledge_base_search",
       "Hybrid (semantic + keyword) search over the org knowledge base / runbooks.",
       {"query": "search query", "k": "number of results (default 5)"})
async def knowledge_base_search(ctx: ToolContext, query: str, k: int = 5) -> str:
    if ctx.rag is None:
        return "ERROR: knowledge base not available"
    hits = ctx.rag.search(query, k=int(k))
    if not hits:
        return "No matching knowledge-base content."
    # excerpt per hit (full chunks quadruple the observation for no
    # extra signal; the agent can re-query with a narrower question)
    return "\n\n".join(
        f"[{i+1}] (score {h['score']:.3f}) {h['doc_name']}:\n{h['text'][:400]}"
        for i, h in enumerate(hits))


@R.add("web_search", "Search the web for relevant context.",
       {"query": "search query"})
async def web_search(ctx: ToolContext, query: str) -> str:
    # Self-hosted SearXNG when configured (reference:
    # web_search_service.py:162); else fall back to the knowledge base and
    # say so explicitly rather than fabricate results.
    import os
    base = os.environ.get("AURORA_SEARXNG_URL", "")
    if base:
        from ...connectors.clients import NoEgressError, ProviderAPIError
        from ...connectors.clients.websearch import compose_query, web_search as _ws
        try:
            return await _ws(_transport(ctx), base, compose_query(query))
        except NoEgressError as e:
            return f"ERROR: SearXNG unreachable (no egress): {e}"
        except ProviderAPIError as e:
            return f"ERROR: web search: {e}"
    if ctx.rag is not None:
        hits = ctx.rag.search(query, k=3)
        if hits:
            return ("(no internet egress; nearest knowledge-base matches)\n\n"
                    + "\n\n".join(h["text"][:500] for h in hits))
    return "ERROR: web search unavailable (no egress) and knowledge base empty"


# -------------------------------------------------------------- RCA / cards


@R.add("trigger_rca",
       "Open a new incident and dispatch a background RCA investigation for it.",
       {"title": "incident title", "details": "what happened"}, read_only=False)
async def trigger_rca(ctx: ToolContext, title: str, details: str = "") -> str:
    if ctx.depth > 0:
        return "ERROR: nested RCA triggers are blocked"  # trigger_rca_tool.py:38
    if ctx.org_store is None or ctx.queue is None:
        return "ERROR: background pipeline unavailable"
    rt = (ctx.extra or {}).get("runtime")
    if rt is not None and not rt.rca_allowed(
            f"tool:{ctx.user_id or ctx.org_store.org_id}"):
        return ("ERROR: RCA trigger rate limit — an investigation was "
                "already dispatched recently")  # trigger_rca_tool.py rate limit
    iid = ctx.org_store.create_incident(
        title=title, source="agent", payload={"details": details},
        user_id=ctx.user_id)
    ctx.queue.enqueue("run_background_chat",
                      {"org_id": ctx.org_store.org_id, "incident_id": iid,
                       "prompt": f"{title}\n{details}", "user_id": ctx.user_id},
                      dedup_key=f"rca:{iid}")
    return f"RCA dispatched for incident {iid}"


@R.add("get_postmortem", "Fetch the postmortem document of an incident.",
       {"incident_id": "the incident id"})
async def get_postmortem(ctx: ToolContext, incident_id: str) -> str:
    pm = ctx.org_store.get_postmortem(incident_id)
    return pm["content"] if pm else f"No postmortem for {incident_id} yet."


@R.add("save_postmortem", "Save/overwrite the postmortem document of an incident.",
       {"incident_id": "the incident id", "content": "markdown postmortem"},
       read_only=False)
async def save_postmortem(ctx: ToolContext, incident_id: str, content: str) -> str:
    pid = ctx.org_store.save_postmortem(incident_id, content)
    return f"postmortem saved ({pid})"


# ---------------------------------------------------------------- artifacts


@R.add("artifact_list", "List persistent investigation documents.")
async def artifact_list(ctx: ToolContext) -> str:
    rows = ctx.org_store.artifacts(ctx.session_id)
    return json.dumps([{"name": r["name"], "updated_at": r["updated_at"]}
                       for r in rows], indent=1)


@R.add("artifact_read", "Read a persistent investigation document.",
       {"name": "artifact name"})
async def artifact_read(ctx: ToolContext, name: str) -> str:
    row = ctx.org_store.get_artifact(name)
    return row["content"] if row else f"ERROR: no artifact '{name}'"


@R.add("artifact_write", "Create or update a persistent investigation document.",
       {"name": "artifact name", "content": "full document content"},
       read_only=False)
async def artifact_write(ctx: ToolContext, name: str, content: str) -> str:
    ctx.org_store.save_artifact(name, content, session_id=ctx.session_id)
    return f"artifact '{name}' saved ({len(content)} chars)"


# -------------------------------------------------------------------- skills


@R.add("load_skill", "Load an investigation skill/playbook by name.",
       {"name": "skill name from the skill catalog"})
async def load_skill(ctx: ToolContext, name: str) -> str:
    from ..skills.registry import get_skill_registry
    sk = get_skill_registry().get(name)
    return sk.body if sk else f"ERROR: unknown skill '{name}'"


# ------------------------------------------------ connector-gated API tools
# Real query construction + response parsing per provider
# (connectors/clients/); only the socket itself degrades when the
# deployment has no egress. Transport is swappable via
# ctx.extra["provider_transport"] (tests inject recorded fixtures).


def _transport(ctx: ToolContext):
    t = (ctx.extra or {}).get("provider_transport")
    if t is not None:
        return t
    from ...connectors.clients import http_transport
    return http_transport()


def _creds(ctx: ToolContext, provider: str) -> dict:
    from ...connectors.registry import get_credentials
    creds = None
    if ctx.org_store is not None and ctx.user_id:
        creds = get_credentials(ctx.org_store, ctx.user_id, provider)
    return creds or {}


def _provider_tool(name: str, provider: str, desc: str,
                   params: dict | None = None):
    from ...connectors.clients import NoEgressError, ProviderAPIError, QUERY_FNS

    @R.add(name, desc, params or {
        "query": "provider query",
        "time_from": "window start: 'now-1h', '-30m', unix or ISO (optional)",
        "time_to": "window end (optional, default now)",
    }, requires_connection=provider)
    async def _tool(ctx: ToolContext, query: str, time_from: str = "",
                    time_to: str = "", query_type: str = "",
                    _p=provider) -> str:
        fn = QUERY_FNS[_p]
        try:
            kwargs = {"time_from": time_from or None,
                      "time_to": time_to or None}
            if _p == "datadog":
                return await fn(_transport(ctx), _creds(ctx, _p),
                                query_type or "logs", query, **kwargs)
            return await fn(_transport(ctx), _creds(ctx, _p), query, **kwargs)
        except NoEgressError as e:
            return (f"ERROR: {_p} API unreachable from this deployment "
                    f"(no egress): {e}")
        except ProviderAPIError as e:
            return f"ERROR: {_p} API: {e}"
    return _tool


_provider_tool("query_datadog", "datadog",
               "Query Datadog: logs (default), metrics, monitors or events "
               "(set query_type).",
               {"query": "log filter / metric query / monitor search",
                "query_type": "logs|metrics|monitors|events",
                "time_from": "window start (e.g. 'now-1h')",
                "time_to": "window end (default now)"})
_provider_tool("query_newrelic", "newrelic",
               "Run a NRQL query against New Relic (NerdGraph).",
               {"query": "NRQL, e.g. SELECT count(*) FROM TransactionError "
                         "SINCE 1 hour ago"})
_provider_tool("query_sentry", "sentry",
               "Search Sentry issues (issue search syntax).",
               {"query": "e.g. is:unresolved level:error release:1.2"})
_provider_tool("search_splunk", "splunk",
               "Run a Splunk search (SPL; 'indexes' lists indexes).")
_provider_tool("query_dynatrace", "dynatrace",
               "Dynatrace problems in the window ('problems') or a metric "
               "selector query.")
_provider_tool("query_grafana", "grafana",
               "Search Grafana dashboards/folders.",
               {"query": "dashboard title text"})
_provider_tool("query_slack", "slack",
               "Slack: search messages; 'channels' lists channels; "
               "'history <channel_id>' reads a channel.")
_provider_tool("query_jira", "jira", "Search Jira issues with JQL.",
               {"query": "JQL, e.g. project=OPS AND status=Open"})
_provider_tool("notion_search", "notion", "Search Notion workspace pages.")
_provider_tool("sharepoint_search", "sharepoint",
               "Search SharePoint/OneDrive content (Microsoft Graph).")
_provider_tool("coroot", "coroot", "Coroot application health overview.")
_provider_tool("thousandeyes", "thousandeyes",
               "ThousandEyes: list tests or fetch a test's network results "
               "(query = test id).")
_provider_tool("netdata", "netdata",
               "Netdata chart data with stats (query = chart id, e.g. "
               "system.cpu).")
_provider_tool("cloudflare", "cloudflare",
               "Cloudflare zone HTTP analytics (requests/cache/5xx/threats; "
               "query = zone tag).")
_provider_tool("query_opsgenie", "opsgenie",
               "Opsgenie alert search (query syntax, default status:open).")
_provider_tool("list_incidentio_incidents", "incidentio",
               "incident.io incidents ('list') or an incident's timeline "
               "(query = incident id).")
_provider_tool("query_flyio_metrics", "flyio",
               "Fly.io Prometheus metrics (query = PromQL; default HTTP "
               "status rates).")
_provider_tool("confluence_search", "confluence",
               "Search Confluence pages/runbooks (text match).")


@R.add("query_cloudwatch",
       "CloudWatch metric statistics for a namespace/metric over a window.",
       {"namespace": "e.g. AWS/EC2", "metric": "e.g. CPUUtilization",
        "dimensions": "k=v,k=v (optional)",
        "time_from": "window start", "time_to": "window end",
        "stat": "Average|Sum|Maximum"},
       requires_connection="aws")
async def query_cloudwatch(ctx: ToolContext, namespace: str, metric: str,
                           dimensions: str = "", time_from: str = "",
                           time_to: str = "", stat: str = "Average") -> str:
    """CloudWatch GetMetricStatistics (SigV4 signing requires the aws
    connector's credentials; construction/parsing is provider-exact)."""
    from ...connectors.clients import NoEgressError, ProviderAPIError
    from ...connectors.clients.observability import (build_cloudwatch_metric,
                                                     parse_cloudwatch_metric)
    dims = {}
    for part in (dimensions or "").split(","):
        if "=" in part:
            k, v = part.split("=", 1)
            dims[k.strip()] = v.strip()
    try:
        req = build_cloudwatch_metric(_creds(ctx, "aws"), namespace, metric,
                                      dims, time_from or None, time_to or None,
                                      stat=stat)
        return parse_cloudwatch_metric(await _transport(ctx)(req))
    except NoEgressError as e:
        return f"ERROR: aws API unreachable (no egress): {e}"
    except ProviderAPIError as e:
        return f"ERROR: cloudwatch: {e}"


@R.add("github_rca", "Correlate the incident window with recent commits and "
       "merged PRs of a repository (risk-ranked).",
       {"repo": "owner/name", "incident_time":
        "incident start: unix, ISO, or 'now-2h' (defaults to this incident)"},
       requires_connection="github")
async def github_rca_tool(ctx: ToolContext, repo: str,
                          incident_time: str = "") -> str:
    from ...connectors.clients import NoEgressError, ProviderAPIError
    from ...connectors.clients.devops import github_rca
    if not incident_time and ctx.org_store is not None and ctx.incident_id:
        inc = ctx.org_store.get_incident(ctx.incident_id)
        if inc:
            incident_time = str(inc.get("created_at", ""))
    try:
        return await github_rca(_transport(ctx), _creds(ctx, "github"),
                                repo, incident_time or "now")
    except NoEgressError as e:
        return f"ERROR: github API unreachable (no egress): {e}"
    except ProviderAPIError as e:
        return f"ERROR: github: {e}"


@R.add("jenkins_rca", "Analyze a Jenkins job's recent builds and extract "
       "failure lines from failing consoles.",
       {"job": "job path, e.g. folder/jobname"},
       requires_connection="jenkins")
async def jenkins_rca_tool(ctx: ToolContext, job: str) -> str:
    from ...connectors.clients import NoEgressError, ProviderAPIError
    from ...connectors.clients.devops import jenkins_rca
    try:
        return await jenkins_rca(_transport(ctx), _creds(ctx, "jenkins"), job)
    except NoEgressError as e:
        return f"ERROR: jenkins API unreachable (no egress): {e}"
    except ProviderAPIError as e:
        return f"ERROR: jenkins: {e}"


@R.add("spinnaker_rca", "Recent Spinnaker pipeline executions of an "
       "application with failed stages.",
       {"application": "Spinnaker application name"},
       requires_connection="spinnaker")
async def spinnaker_rca_tool(ctx: ToolContext, application: str) -> str:
    from ...connectors.clients import NoEgressError, ProviderAPIError
    from ...connectors.clients.devops import spinnaker_rca
    try:
        return await spinnaker_rca(_transport(ctx), _creds(ctx, "spinnaker"),
                                   application)
    except NoEgressError as e:
        return f"ERROR: spinnaker API unreachable (no egress): {e}"
    except ProviderAPIError as e:
        return f"ERROR: spinnaker: {e}"


@R.add("cloudbees_rca", "Analyze a CloudBees (Jenkins-based) job's recent "
       "builds and extract failure lines.",
       {"job": "job path"}, requires_connection="cloudbees")
async def cloudbees_rca_tool(ctx: ToolContext, job: str) -> str:
    # CloudBees CI speaks the Jenkins remote API (reference: cloudbees_rca
    # reuses the Jenkins analysis against the CloudBees endpoint)
    from ...connectors.clients import NoEgressError, ProviderAPIError
    from ...connectors.clients.devops import jenkins_rca
    try:
        return await jenkins_rca(_transport(ctx), _creds(ctx, "cloudbees"),
                                 job)
    except NoEgressError as e:
        return f"ERROR: cloudbees API unreachable (no egress): {e}"
    except ProviderAPIError as e:
        return f"ERROR: cloudbees: {e}"


@R.add("github_fix", "Draft a fix suggestion (branch + patch description) "
       "for the incident from a suspect commit.",
       {"repo": "owner/name", "commit_sha": "suspect commit",
        "description": "what to change and why"},
       read_only=False, requires_connection="github", needs_confirmation=True)
async def github_fix_tool(ctx: ToolContext, repo: str, commit_sha: str = "",
                          description: str = "") -> str:
    # The reference's github_fix persists a structured fix suggestion on the
    # incident card (github_fix_tool.py); PR creation is a separate,
    # confirmed step. Persist the suggestion here.
    if ctx.org_store is None or not ctx.incident_id:
        return "ERROR: github_fix needs an incident context"
    ctx.org_store.add_suggestion(
        ctx.incident_id, title=f"fix in {repo}"
        + (f" (suspect {commit_sha[:10]})" if commit_sha else ""),
        command="", rationale=description[:2000], safe=True)
    return f"fix suggestion saved to incident {ctx.incident_id}: {repo}@{commit_sha}"


# ------------------------------------------------------- files / artefacts


@R.add("publish_postmortem", "Publish this incident's postmortem to a "
       "Notion page.", {"parent_page_id": "Notion parent page id"},
       read_only=False, requires_connection="notion",
       needs_confirmation=True)
async def publish_postmortem(ctx: ToolContext, parent_page_id: str) -> str:
    from ...connectors.clients import NoEgressError, ProviderAPIError
    from ...connectors.clients.comms import notion_publish_postmortem
    if ctx.org_store is None or not ctx.incident_id:
        return "ERROR: no incident bound"
    pm = ctx.org_store.get_postmortem(ctx.incident_id)
    if not pm:
        return "ERROR: this incident has no postmortem yet"
    inc = ctx.org_store.get_incident(ctx.incident_id) or {}
    try:
        return await notion_publish_postmortem(
            _transport(ctx), _creds(ctx, "notion"), parent_page_id,
            f"Postmortem: {inc.get('title', ctx.incident_id)}",
            pm["content"] if isinstance(pm, dict) else str(pm))
    except NoEgressError as e:
        return f"ERROR: notion unreachable (no egress): {e}"
    except ProviderAPIError as e:
        return f"ERROR: notion: {e}"


@R.add("analyze_zip_file",
       "List and summarise the contents of an uploaded zip archive.",
       {"artifact": "name of the zip artifact (base64 content)"})
async def analyze_zip_file(ctx: ToolContext, artifact: str) -> str:
    import base64
    import io
    import zipfile
    row = ctx.org_store.get_artifact(artifact)
    if not row:
        return f"ERROR: no artifact '{artifact}'"
    try:
        data = base64.b64decode(row["content"])
        zf = zipfile.ZipFile(io.BytesIO(data))
    except Exception as e:  # noqa: BLE001
        return f"ERROR: not a readable zip: {e}"
    lines = []
    for info in zf.infolist()[:200]:
        lines.append(f"{info.filename} ({info.file_size} B)")
    return "Archive contents:\n" + "\n".join(lines)


@R.add("rag_index_zip",
       "Index the text files of an uploaded zip archive into the knowledge base.",
       {"artifact": "name of the zip artifact (base64 content)"}, read_only=False)
async def rag_index_zip(ctx: ToolContext, artifact: str) -> str:
    import base64
    import io
    import zipfile
    row = ctx.org_store.get_artifact(artifact)
    if not row:
        return f"ERROR: no artifact '{artifact}'"
    try:
        zf = zipfile.ZipFile(io.BytesIO(base64.b64decode(row["content"])))
    except Exception as e:  # noqa: BLE001
        return f"ERROR: not a readable zip: {e}"
    n = 0
    for info in zf.infolist():
        if info.file_size > 2_000_000 or info.is_dir():
            continue
        if not info.filename.endswith((".txt", ".md", ".log", ".yaml", ".yml",
                                       ".json", ".conf", ".cfg", ".py", ".sh")):
            continue
        try:
            text = zf.read(info).decode("utf-8", "replace")
        except Exception:  # noqa: BLE001
            continue
        did = ctx.org_store.add_kb_document(f"{artifact}:{info.filename}", text)
        if ctx.rag is not None:
            ctx.rag.add_document(did, info.filename, text)
        n += 1
    return f"indexed {n} files from {artifact} into the knowledge base"


# -------------------------------------------------------------- automation


@R.add("iac_tool",
       "Write and plan Terraform for a proposed infrastructure change "
       "(plan only; apply requires human confirmation).",
       {"name": "change name", "hcl": "terraform configuration"},
       read_only=False, needs_confirmation=True)
async def iac_tool(ctx: ToolContext, name: str, hcl: str) -> str:
    import shutil
    import tempfile
    ctx.org_store.save_artifact(f"iac:{name}.tf", hcl, session_id=ctx.session_id)
    if shutil.which("terraform") is None:
        return (f"terraform configuration saved as artifact 'iac:{name}.tf'. "
                f"(terraform binary unavailable in this deployment; plan/apply "
                f"must run from a workstation)")
    with tempfile.TemporaryDirectory() as d:
        with open(f"{d}/main.tf", "w") as f:
            f.write(hcl)
        out = await terminal_exec(ctx, f"cd {d} && terraform init -backend=false "
                                       f"-input=false && terraform plan -input=false")
        return out


@R.add("rightsizing_analysis",
       "Analyze pod resource usage vs limits and propose HPA/VPA right-sizing.",
       {"usage": "kubectl top / metrics output (name cpu memory per line)",
        "limits": "current requests/limits description"})
async def rightsizing_analysis(ctx: ToolContext, usage: str, limits: str = "") -> str:
    # deterministic analysis core of the reference's hpa_vpa right-sizing
    # cards (reference: services/actions/hpa_vpa_* 958 LoC)
    import re
    rows = []
    for line in usage.splitlines():
        m = re.match(r"\s*(\S+)\s+(\d+)m?\s+(\d+)(Mi|Gi)", line)
        if m:
            mem = int(m.group(3)) * (1024 if m.group(4) == "Gi" else 1)
            rows.append((m.group(1), int(m.group(2)), mem))
    if not rows:
        return "ERROR: no parsable usage rows (expected 'name cpu mem' lines)"
    recs = []
    for name, cpu_m, mem_mi in rows:
        rec_cpu = max(int(cpu_m * 1.4), 10)
        rec_mem = max(int(mem_mi * 1.3), 32)
        recs.append(f"- {name}: request cpu={rec_cpu}m memory={rec_mem}Mi "
                    f"(observed {cpu_m}m / {mem_mi}Mi + headroom)")
    return ("Right-sizing recommendations (1.4x cpu / 1.3x memory headroom "
            "over observed usage):\n" + "\n".join(recs))
