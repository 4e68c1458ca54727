"""Provider API clients: pure request construction + response parsing per
provider, driven through a pluggable async transport (tests inject
recorded fixtures; production uses base.http_transport; air-gapped
deployments degrade with an explicit error at the socket only).

Reference parity: tools/datadog_tool.py, query_newrelic.py,
splunk_tool.py, query_sentry.py, dynatrace, slack/jira/confluence/notion
clusters, github_rca_tool.py, jenkins_rca (SURVEY §2.3 rows
'Observability query tools' / 'Comms/docs tools' / 'RCA tools')."""
from . import comms, datadog, devops, observability, observability2  # noqa: F401
from .base import (NoEgressError, ProviderAPIError, Request,  # noqa: F401
                   http_transport, parse_time, window)

#: tool-facing dispatch: provider -> async query(transport, creds, query, **kw)
QUERY_FNS = {
    "datadog": datadog.query,
    "newrelic": observability.newrelic_query,
    "splunk": observability.splunk_query,
    "sentry": observability.sentry_query,
    "dynatrace": observability.dynatrace_query,
    "grafana": observability.grafana_query,
    "slack": comms.slack_query,
    "jira": comms.jira_query,
    "confluence": comms.confluence_query,
    "notion": comms.notion_query,
    "coroot": observability2.coroot_query,
    "thousandeyes": observability2.thousandeyes_query,
    "netdata": observability2.netdata_query,
    "cloudflare": observability2.cloudflare_query,
    "opsgenie": observability2.opsgenie_query,
    "incidentio": observability2.incidentio_query,
    "flyio": observability2.flyio_query,
    "sharepoint": comms.sharepoint_query,
}
