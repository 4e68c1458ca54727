"""Provider API clients against recorded fixtures:
request construction (URLs, auth headers, time windows, pagination) and
response parsing for datadog/newrelic/splunk/sentry/dynatrace/grafana +
slack/jira/confluence/notion + github_rca/jenkins_rca. No sockets: the
transport seam is a fixture function."""
import asyncio
import json

import pytest

from aurora_amd.connectors.clients import (ProviderAPIError, comms, datadog,
                                           devops, observability)
from aurora_amd.connectors.clients.base import (parse_time, series_stats,
                                                window)

NOW = 1_700_000_000.0


def run(coro):
    return asyncio.run(coro)


def transport_of(handler):
    calls = []

    async def t(req):
        calls.append(req)
        return handler(req)
    t.calls = calls
    return t


# ------------------------------------------------------------------ base

def test_parse_time_forms():
    assert parse_time("now", now=NOW) == NOW
    assert parse_time("now-15m", now=NOW) == NOW - 900
    assert parse_time("-2h", now=NOW) == NOW - 7200
    assert parse_time("1700000000", now=NOW) == 1_700_000_000
    assert parse_time("1700000000000", now=NOW) == 1_700_000_000  # ms epoch
    assert abs(parse_time("2023-11-14T22:13:20Z") - 1_700_000_000) < 1
    with pytest.raises(ProviderAPIError):
        parse_time("yesterday-ish")
    with pytest.raises(ProviderAPIError):
        window("now", "now-1h", now=NOW)  # empty window


def test_series_stats():
    st = series_stats([1.0, 2.0, 3.0, None, 4.0])
    assert st["points"] == 4 and st["min"] == 1.0 and st["max"] == 4.0
    assert st["avg"] == 2.5 and st["last"] == 4.0
    assert series_stats([]) == {"points": 0}


# --------------------------------------------------------------- datadog

DD_CREDS = {"api_key": "k", "app_key": "a"}


def test_datadog_logs_pagination_and_parse():
    pages = [
        {"data": [{"attributes": {"timestamp": "t1", "service": "api",
                                  "status": "error", "message": "boom 1"}}],
         "meta": {"page": {"after": "cur1"}}},
        {"data": [{"attributes": {"timestamp": "t2", "service": "api",
                                  "status": "error", "message": "boom 2"}}],
         "meta": {}},
    ]
    state = {"i": 0}

    def handler(req):
        assert req.url.endswith("/api/v2/logs/events/search")
        assert req.headers["DD-API-KEY"] == "k"
        body = req.json_body
        assert body["filter"]["query"] == "service:api status:error"
        if state["i"] == 1:
            assert body["page"]["cursor"] == "cur1"
        out = pages[state["i"]]
        state["i"] += 1
        return out

    t = transport_of(handler)
    out = run(datadog.query_logs(t, DD_CREDS, "service:api status:error",
                                 "now-1h", "now", now=NOW))
    assert "2 events" in out and "boom 1" in out and "boom 2" in out
    assert len(t.calls) == 2  # followed the cursor once, stopped at end


def test_datadog_metrics_window_and_summary():
    def handler(req):
        assert req.url.endswith("/api/v1/query")
        assert req.params["from"] == int(NOW - 3600)
        assert req.params["to"] == int(NOW)
        return {"status": "ok", "series": [
            {"metric": "system.cpu.user", "scope": "host:web-1",
             "unit": [{"short_name": "%"}],
             "pointlist": [[1, 10.0], [2, 30.0], [3, 20.0]]}]}

    out = run(datadog.query_metrics(transport_of(handler), DD_CREDS,
                                    "avg:system.cpu.user{*}",
                                    "now-1h", "now", now=NOW))
    assert "system.cpu.user{host:web-1} [%]" in out
    assert '"max": 30.0' in out and '"last": 20.0' in out


def test_datadog_monitors_and_errors():
    def handler(req):
        return {"monitors": [{"id": 7, "status": "Alert", "name": "cpu high",
                              "type": "metric alert"}],
                "metadata": {"total_count": 1}}
    out = run(datadog.query(transport_of(handler), DD_CREDS, "monitors",
                            "status:alert", now=NOW))
    assert "#7 [Alert] cpu high" in out
    with pytest.raises(ProviderAPIError):
        datadog.build_logs_request({}, "q", None, None)  # missing creds
    with pytest.raises(ProviderAPIError):
        run(datadog.query(transport_of(lambda r: {}), DD_CREDS, "nope", "q"))


def test_datadog_interval_clamp():
    assert datadog.pick_interval(300) == 1
    assert datadog.pick_interval(3600 * 24) == 300
    assert datadog.pick_interval(3600, caller_interval=60) == 60
    # caller interval too fine for the window: clamped up
    assert datadog.pick_interval(MAX := 3600 * 24 * 30,
                                 caller_interval=1) >= MAX // 300


# -------------------------------------------------------------- newrelic

def test_newrelic_nrql():
    def handler(req):
        assert req.url == "https://api.newrelic.com/graphql"
        assert req.headers["API-Key"] == "nrk"
        q = req.json_body["query"]
        assert "account(id: 42)" in q and "SELECT count(*)" in q
        return {"data": {"actor": {"account": {"nrql": {"results": [
            {"count": 123}, {"count": 4}]}}}}}

    out = run(observability.newrelic_query(
        transport_of(handler), {"api_key": "nrk", "account_id": "42"},
        "SELECT count(*) FROM Transaction SINCE 1 hour ago"))
    assert "2 rows" in out and '"count": 123' in out
    # graphql errors surface
    with pytest.raises(ProviderAPIError):
        run(observability.newrelic_query(
            transport_of(lambda r: {"errors": [{"message": "bad nrql"}]}),
            {"api_key": "nrk", "account_id": "42"}, "SELECT"))


# ---------------------------------------------------------------- splunk

def test_splunk_normalize_and_search():
    assert observability.normalize_spl("index=web error") == \
        "search index=web error"
    assert observability.normalize_spl("| tstats count") == "| tstats count"
    assert observability.normalize_spl("search x") == "search x"

    def handler(req):
        assert req.url.endswith("/services/search/jobs")
        assert req.json_body["search"].startswith("search ")
        assert req.json_body["exec_mode"] == "oneshot"
        return {"results": [{"_raw": "ERROR db timeout"},
                            {"field": "v"}]}

    out = run(observability.splunk_query(
        transport_of(handler),
        {"base_url": "https://sp:8089", "token": "t"},
        "index=web error", "now-1h", "now", now=NOW))
    assert "2 results" in out and "ERROR db timeout" in out


def test_splunk_indexes():
    def handler(req):
        assert req.url.endswith("/services/data/indexes")
        return {"entry": [{"name": "main"}, {"name": "web"}]}
    out = run(observability.splunk_query(
        transport_of(handler), {"base_url": "https://sp", "token": "t"},
        "indexes"))
    assert out == "splunk indexes: main, web"


# ---------------------------------------------------------------- sentry

def test_sentry_issue_search():
    def handler(req):
        assert "/api/0/organizations/acme/issues/" in req.url
        assert req.params["query"] == "is:unresolved level:error"
        return [{"shortId": "API-1", "level": "error", "count": "42",
                 "title": "TypeError in checkout", "lastSeen": "2026-01-01"}]
    out = run(observability.sentry_query(
        transport_of(handler), {"token": "t", "org": "acme"},
        "is:unresolved level:error"))
    assert "API-1 [error] x42 TypeError in checkout" in out


# ------------------------------------------------------------- dynatrace

def test_dynatrace_problems_and_metrics():
    def handler(req):
        if "/problems" in req.url:
            assert req.params["from"] == int((NOW - 7200) * 1000)
            return {"totalCount": 1, "problems": [
                {"displayId": "P-1", "status": "OPEN",
                 "severityLevel": "ERROR", "title": "Response time spike",
                 "impactedEntities": [{"name": "svc-a"}]}]}
        assert "/metrics/query" in req.url
        return {"result": [{"metricId": "builtin:cpu",
                            "data": [{"dimensions": ["host-1"],
                                      "values": [1.0, None, 3.0]}]}]}
    creds = {"base_url": "https://dt", "token": "t"}
    out = run(observability.dynatrace_query(transport_of(handler), creds,
                                            "problems", now=NOW))
    assert "P-1 [OPEN/ERROR]" in out and "svc-a" in out
    out = run(observability.dynatrace_query(transport_of(handler), creds,
                                            "builtin:cpu", now=NOW))
    assert "builtin:cpu{host-1}" in out


# ------------------------------------------------------------------ slack

def test_slack_modes_and_error():
    def handler(req):
        if req.url.endswith("conversations.list"):
            return {"ok": True, "channels": [
                {"name": "ops", "id": "C1", "num_members": 7}]}
        if req.url.endswith("conversations.history"):
            assert req.params["channel"] == "C1"
            return {"ok": True, "messages": [
                {"ts": "1.0", "user": "U1", "text": "deploy started"}]}
        return {"ok": True, "messages": {"total": 1, "matches": [
            {"channel": {"name": "ops"}, "username": "al",
             "ts": "2.0", "text": "db down"}]}}
    creds = {"token": "xoxb"}
    assert "#ops (C1" in run(comms.slack_query(transport_of(handler), creds,
                                               "channels"))
    assert "deploy started" in run(comms.slack_query(transport_of(handler),
                                                     creds, "history C1"))
    assert "db down" in run(comms.slack_query(transport_of(handler), creds,
                                              "db down"))
    with pytest.raises(ProviderAPIError):
        run(comms.slack_query(
            transport_of(lambda r: {"ok": False, "error": "invalid_auth"}),
            creds, "x"))


# ------------------------------------------------------------------- jira

def test_jira_jql_pagination():
    def handler(req):
        assert req.url.endswith("/rest/api/2/search")
        assert req.headers["Authorization"].startswith("Basic ")
        start = req.json_body["startAt"]
        if start == 0:
            return {"total": 2, "issues": [{"key": "OPS-1", "fields": {
                "status": {"name": "Open"}, "priority": {"name": "P1"},
                "summary": "db latency", "assignee": None,
                "updated": "2026-01-01"}}]}
        return {"total": 2, "issues": [{"key": "OPS-2", "fields": {
            "status": {"name": "Done"}, "priority": None,
            "summary": "retro", "assignee": {"displayName": "sam"},
            "updated": "2026-01-02"}}]}

    t = transport_of(handler)
    out = run(comms.jira_query(
        t, {"base_url": "https://j", "token": "t", "email": "e@x.io"},
        "project=OPS"))
    assert "OPS-1 [Open/P1]" in out and "OPS-2 [Done/-]" in out
    assert "assignee=unassigned" in out and "assignee=sam" in out
    assert len(t.calls) == 2  # paged until total reached


# ------------------------------------------------------------- confluence

def test_confluence_cql_and_html_strip():
    def handler(req):
        assert 'text ~ "db runbook"' in req.params["cql"]
        return {"size": 1, "results": [
            {"title": "DB Runbook", "space": {"key": "OPS"},
             "body": {"view": {"value": "<p>Step <b>one</b>: check\npool</p>"}}}]}
    out = run(comms.confluence_query(
        transport_of(handler), {"base_url": "https://c", "token": "t"},
        "db runbook"))
    assert "[OPS] DB Runbook: Step one : check pool" in out.replace("  ", " ")


def test_notion_search_title_extraction():
    def handler(req):
        assert req.headers["Notion-Version"]
        return {"results": [{"object": "page", "last_edited_time": "2026",
                             "properties": {"Name": {
                                 "type": "title",
                                 "title": [{"plain_text": "Postmortem X"}]}}}]}
    out = run(comms.notion_query(transport_of(handler), {"token": "t"}, "x"))
    assert "[page] Postmortem X" in out


# ----------------------------------------------------------------- github

def _commit(sha, date, msg):
    return {"sha": sha, "commit": {"message": msg,
                                   "author": {"date": date}}}


def test_github_rca_correlation():
    t_inc = "2023-11-14T22:13:20Z"  # == NOW

    def handler(req):
        if req.url.endswith("/commits"):
            assert req.params["until"] == "2023-11-14T22:13:20Z"
            return [
                _commit("a" * 40, "2023-11-14T21:30:20Z",
                        "bump db pool config"),
                _commit("b" * 40, "2023-11-13T10:00:00Z", "docs typo"),
            ]
        if "/commits/" in req.url:
            return {"files": [{"filename": "deploy/helm/values.yaml",
                               "additions": 3, "deletions": 1}]}
        if req.url.endswith("/pulls"):
            return [{"number": 12, "title": "Tune connection pool",
                     "merged_at": "2023-11-14T21:00:00Z"}]
        raise AssertionError(req.url)

    out = run(devops.github_rca(transport_of(handler), {"token": "t"},
                                "acme/api", t_inc, now=NOW))
    assert "commits in the 24h before the incident: 2" in out
    assert "#12 Tune connection pool" in out
    # the risky config commit ranks above the docs commit
    lines = [ln for ln in out.splitlines() if ln.strip().startswith(("aaaa", "bbbb"))]
    assert lines and lines[0].startswith("  aaaaaaaaaa")
    assert "risky message" in lines[0] and "43min before incident" in lines[0]


def test_github_score_commit():
    s, reasons = devops.score_commit(
        _commit("c", "", "hotfix: rollback timeout change"),
        [{"filename": "terraform/main.tf", "additions": 600, "deletions": 10}])
    assert s >= 6 and any("terraform" in r for r in reasons)
    s2, _ = devops.score_commit(_commit("d", "", "update readme"), [])
    assert s2 <= 2


# ---------------------------------------------------------------- jenkins

def test_jenkins_rca_console_excerpt():
    console = ("step 1 ok\nstep 2 ok\n"
               "ERROR: compilation failed\njava.lang.NullPointerException\n"
               "Build step 'shell' marked build as failure\nFinished: FAILURE")

    def handler(req):
        if req.url.endswith("/api/json"):
            assert "job/folder/job/api" in req.url
            return {"builds": [
                {"number": 10, "result": "FAILURE", "timestamp": 1},
                {"number": 9, "result": "SUCCESS", "timestamp": 0}]}
        assert req.url.endswith("/folder/job/api/10/consoleText".replace(
            "/folder/job/api", "/job/folder/job/api"))
        return console

    out = run(devops.jenkins_rca(
        transport_of(handler),
        {"base_url": "https://j", "user": "u", "token": "t"}, "folder/api"))
    assert "#10:FAILURE" in out and "#9:SUCCESS" in out
    assert "NullPointerException" in out
    assert "step 1 ok" not in out  # only failure lines excerpted


def test_jenkins_failure_line_extraction():
    lines = devops.extract_failure_lines("a\nERROR x\nb\nfatal: y\nc")
    assert lines == ["ERROR x", "fatal: y"]
    # no matches: fall back to the tail
    assert devops.extract_failure_lines("l1\nl2", max_lines=1) == ["l2"]


# ------------------------------------------------- registry tool wrappers

def test_provider_tools_wired_through_registry():
    """The agent-facing tools run the real clients via the transport seam
    and degrade explicitly when credentials/egress are absent."""
    import aurora_amd.agent.tools.core  # noqa: F401 - registers tools
    from aurora_amd.agent.tools.registry import GLOBAL_REGISTRY, ToolContext

    async def main():
        async def fixture(req):
            assert req.headers["DD-API-KEY"] == "k"
            return {"data": [], "meta": {}}

        ctx = ToolContext(extra={"provider_transport": fixture})
        # no creds bound -> the client raises a credentials error, surfaced
        out = await GLOBAL_REGISTRY.run(ctx, "query_datadog",
                                        {"query": "service:api"})
        assert out.startswith("ERROR: datadog API:")

        class Store:  # minimal org-store shim with a github token
            def get_token_ref(self, uid, provider):
                return None
            def tool_permissions(self):
                return {}
            def connected_providers(self, uid):
                return ["datadog"]
        ctx2 = ToolContext(extra={"provider_transport": fixture})
        ctx2.user_id = "u"
        # patch _creds through direct call instead: call the client module
        from aurora_amd.connectors.clients import datadog as dd
        out = await dd.query_logs(fixture, DD_CREDS, "service:api")
        assert "no matches" in out

    run(main())


# -------------------------------------------------------------- websearch

def test_websearch_compose_build_parse():
    from aurora_amd.connectors.clients import websearch as ws
    q = ws.compose_query(
        "CRITICAL alert: checkout-svc p99 latency is firing on prod-cluster-7")
    assert "checkout-svc" in q and "prod-cluster-7" in q
    assert "alert" not in q.lower().split()
    req = ws.build_search_request("http://searx:8080", q, time_range="week")
    assert req.url == "http://searx:8080/search"
    assert req.params["format"] == "json" and req.params["time_range"] == "week"
    results = ws.parse_search_response({"results": [
        {"title": "Fixing p99 latency", "url": "https://a.io/x",
         "content": "tune the pool", "score": 2.0},
        {"title": "Fixing p99 latency", "url": "https://a.io/y",
         "content": "dup host+title", "score": 1.9},
        {"title": "Other", "url": "https://b.io/z", "content": "c",
         "score": 0.5},
    ]})
    assert len(results) == 2  # deduped by host+title
    assert results[0]["url"] == "https://a.io/x"  # score-ranked
    out = ws.render_results(results, q)
    assert "Fixing p99 latency" in out and "https://b.io/z" in out


def test_websearch_tool_via_fixture_transport():
    import aurora_amd.agent.tools.core  # noqa: F401
    from aurora_amd.agent.tools.registry import GLOBAL_REGISTRY, ToolContext

    async def main():
        async def fixture(req):
            assert req.params["q"]
            return {"results": [{"title": "T", "url": "https://x.io/1",
                                 "content": "hit", "score": 1}]}
        import os
        os.environ["AURORA_SEARXNG_URL"] = "http://searx:8080"
        try:
            out = await GLOBAL_REGISTRY.run(
                ToolContext(extra={"provider_transport": fixture}),
                "web_search", {"query": "redis timeout spike"})
            assert "https://x.io/1" in out
        finally:
            del os.environ["AURORA_SEARXNG_URL"]

    run(main())


# ------------------------------------------------- second observability set

def test_coroot_overview():
    from aurora_amd.connectors.clients import observability2 as o2
    out = run(o2.coroot_query(transport_of(lambda r: [
        {"id": {"name": "checkout"}, "status": "warning",
         "indicators": [{"message": "p99 above SLO"}]},
        {"id": {"name": "db"}, "status": "ok", "indicators": []}]),
        {"base_url": "https://c", "project": "p1"}, ""))
    assert "2 applications, 1 unhealthy" in out
    assert "[warning] checkout: p99 above SLO" in out


def test_thousandeyes_modes():
    from aurora_amd.connectors.clients import observability2 as o2
    creds = {"token": "t"}

    def handler(req):
        if req.url.endswith("/tests"):
            return {"tests": [{"testId": 9, "testName": "dc-ping",
                               "type": "agent-to-server", "interval": 60}]}
        assert "/test-results/9/network" in req.url
        return {"results": [{"agent": {"agentName": "fra"}, "loss": 12.5,
                             "avgLatency": 80, "jitter": 3}]}
    assert "9: dc-ping" in run(o2.thousandeyes_query(transport_of(handler),
                                                     creds, "tests"))
    out = run(o2.thousandeyes_query(transport_of(handler), creds, "9"))
    assert "max loss 12.5%" in out and "agent fra" in out


def test_netdata_series_stats():
    from aurora_amd.connectors.clients import observability2 as o2
    out = run(o2.netdata_query(transport_of(lambda r: {
        "id": "system.cpu", "labels": ["time", "user", "system"],
        "data": [[1, 10.0, 5.0], [2, 30.0, None], [3, 20.0, 7.0]]}),
        {"base_url": "http://n:19999"}, "system.cpu", now=NOW))
    assert "netdata system.cpu" in out
    assert '"max": 30.0' in out


def test_cloudflare_analytics_rollup():
    from aurora_amd.connectors.clients import observability2 as o2

    def handler(req):
        assert "graphql" in req.url
        assert req.json_body["variables"]["zone"] == "ztag"
        return {"data": {"viewer": {"zones": [{"httpRequests1mGroups": [
            {"sum": {"requests": 100, "cachedRequests": 60, "threats": 2,
                     "responseStatusMap": [
                         {"edgeResponseStatus": 200, "requests": 90},
                         {"edgeResponseStatus": 502, "requests": 10}]}},
            {"sum": {"requests": 50, "cachedRequests": 10, "threats": 0,
                     "responseStatusMap": [
                         {"edgeResponseStatus": 200, "requests": 50}]}},
        ]}]}}}
    out = run(o2.cloudflare_query(transport_of(handler), {"token": "t"},
                                  "ztag", now=NOW))
    assert "150 requests" in out and "10 5xx" in out and "200=140" in out


def test_opsgenie_and_incidentio_and_flyio():
    from aurora_amd.connectors.clients import observability2 as o2
    out = run(o2.opsgenie_query(transport_of(lambda r: {"data": [
        {"priority": "P1", "status": "open", "message": "db down",
         "acknowledged": False, "createdAt": "2026"}]}),
        {"api_key": "k"}, ""))
    assert "[P1/open] db down" in out

    def io_handler(req):
        if "incident_updates" in req.url:
            return {"incident_updates": [
                {"created_at": "t1", "new_incident_status": {"name": "fixed"},
                 "message": "rolled back"}]}
        return {"incidents": [{"reference": "INC-7", "name": "API errors",
                               "severity": {"name": "major"},
                               "incident_status": {"name": "open"}}]}
    assert "INC-7 [major/open]" in run(o2.incidentio_query(
        transport_of(io_handler), {"api_key": "k"}, "list"))
    assert "rolled back" in run(o2.incidentio_query(
        transport_of(io_handler), {"api_key": "k"}, "abc123"))

    def fly_handler(req):
        assert "query_range" in req.url
        return {"status": "success", "data": {"result": [
            {"metric": {"status": "500"},
             "values": [[1, "0.5"], [2, "1.5"]]}]}}
    out = run(o2.flyio_query(transport_of(fly_handler),
                             {"token": "t", "app": "web"}, "", now=NOW))
    assert '"status": "500"' in out and '"max": 1.5' in out


def test_notion_postmortem_markdown_to_blocks():
    from aurora_amd.connectors.clients import comms
    md = ("# Postmortem: db outage\n\n## Summary\nPool exhausted.\n"
          "## Timeline\n- 10:00 alert\n- 10:05 picked up\n")
    req = comms.build_notion_postmortem({"token": "t"}, "page-1",
                                        "Postmortem: db outage", md)
    body = req.json_body
    assert body["parent"]["page_id"] == "page-1"
    kinds = [b["type"] for b in body["children"]]
    assert kinds == ["heading_1", "heading_2", "paragraph", "heading_2",
                     "bulleted_list_item", "bulleted_list_item"]
    assert body["children"][0]["heading_1"]["rich_text"][0]["text"][
        "content"] == "Postmortem: db outage"

    async def main():
        out = await comms.notion_publish_postmortem(
            transport_of(lambda r: {"object": "page",
                                    "url": "https://notion.so/x"}),
            {"token": "t"}, "page-1", "T", md)
        assert "https://notion.so/x" in out
        with pytest.raises(ProviderAPIError):
            await comms.notion_publish_postmortem(
                transport_of(lambda r: {"object": "error",
                                        "message": "denied"}),
                {"token": "t"}, "page-1", "T", md)
    run(main())


def test_sharepoint_search_graph():
    from aurora_amd.connectors.clients import comms

    def handler(req):
        assert req.url.endswith("/search/query")
        assert req.json_body["requests"][0]["query"]["queryString"] == "runbook db"
        return {"value": [{"hitsContainers": [{"hits": [
            {"resource": {"name": "db-runbook.docx"},
             "summary": "failover steps"}]}]}]}
    out = run(comms.sharepoint_query(transport_of(handler), {"token": "t"},
                                     "runbook db"))
    assert "db-runbook.docx: failover steps" in out


def test_spinnaker_executions():
    from aurora_amd.connectors.clients import devops

    def handler(req):
        assert "/applications/checkout/pipelines" in req.url
        return [{"name": "deploy-prod", "status": "TERMINAL",
                 "startTime": 1700000000000,
                 "stages": [{"name": "bake", "status": "SUCCEEDED"},
                            {"name": "deploy", "status": "TERMINAL"}]},
                {"name": "deploy-prod", "status": "SUCCEEDED",
                 "startTime": 1699990000000, "stages": []}]
    out = run(devops.spinnaker_rca(transport_of(handler),
                                   {"base_url": "https://gate"}, "checkout"))
    assert "2 recent executions, 1 failed" in out
    assert "failed stages: deploy" in out
