"""OAuth 2.0 authorization-code flows for the oauth-auth connectors.

Rebuilds the reference's per-provider OAuth plumbing (reference:
server/connectors/<name>_connector/ — authorize-URL construction, CSRF
state cache, code->token exchange, refresh, expiry handling; contract
AGENTS.md:47-82: tokens land in the secrets backend, the DB keeps only a
reference). The HTTP boundary is the same pluggable transport the
provider API clients use (connectors/clients/base.py), so every flow is
fixture-testable without egress.
"""
from __future__ import annotations

import secrets as _secrets
import time
import urllib.parse
from dataclasses import dataclass, field
from typing import Dict, Optional

from .clients.base import ProviderAPIError, Request, Transport

#: provider OAuth endpoints + scopes (public endpoint shapes)
PROVIDERS: Dict[str, dict] = {
    "github": {
        "authorize": "https://github.com/login/oauth/authorize",
        "token": "https://github.com/login/oauth/access_token",
        "scopes": "repo read:org",
    },
    "slack": {
        "authorize": "https://slack.com/oauth/v2/authorize",
        "token": "https://slack.com/api/oauth.v2.access",
        "scopes": "channels:read,channels:history,search:read",
        "scope_param": "scope",
    },
    "gcp": {
        "authorize": "https://accounts.google.com/o/oauth2/v2/auth",
        "token": "https://oauth2.googleapis.com/token",
        "scopes": "https://www.googleapis.com/auth/cloud-platform.read-only",
        "extra_authorize": {"access_type": "offline", "prompt": "consent"},
    },
    "azure": {
        "authorize": "https://login.microsoftonline.com/common/oauth2/v2.0/authorize",
        "token": "https://login.microsoftonline.com/common/oauth2/v2.0/token",
        "scopes": "https://management.azure.com/user_impersonation offline_access",
    },
    "jira": {
        "authorize": "https://auth.atlassian.com/authorize",
        "token": "https://auth.atlassian.com/oauth/token",
        "scopes": "read:jira-work offline_access",
        "extra_authorize": {"audience": "api.atlassian.com"},
    },
    "confluence": {
        "authorize": "https://auth.atlassian.com/authorize",
        "token": "https://auth.atlassian.com/oauth/token",
        "scopes": "read:confluence-content.all offline_access",
        "extra_authorize": {"audience": "api.atlassian.com"},
    },
    "notion": {
        "authorize": "https://api.notion.com/v1/oauth/authorize",
        "token": "https://api.notion.com/v1/oauth/token",
        "scopes": "",
        "extra_authorize": {"owner": "user"},
    },
    "bitbucket": {
        "authorize": "https://bitbucket.org/site/oauth2/authorize",
        "token": "https://bitbucket.org/site/oauth2/access_token",
        "scopes": "repository issue",
    },
    "sharepoint": {
        "authorize": "https://login.microsoftonline.com/common/oauth2/v2.0/authorize",
        "token": "https://login.microsoftonline.com/common/oauth2/v2.0/token",
        "scopes": "Sites.Read.All offline_access",
    },
}


@dataclass
class StateCache:
    """One-shot CSRF state store (reference: OAuth2 state cache in
    utils/auth — states are single-use and expire)."""

    ttl_s: float = 600.0
    _states: Dict[str, dict] = field(default_factory=dict)

    def issue(self, user_id: str, provider: str) -> str:
        state = _secrets.token_urlsafe(24)
        now = time.time()
        self._states = {k: v for k, v in self._states.items()
                        if now - v["ts"] < self.ttl_s}
        self._states[state] = {"user_id": user_id, "provider": provider,
                               "ts": now}
        return state

    def consume(self, state: str) -> Optional[dict]:
        v = self._states.pop(state, None)
        if v is None or time.time() - v["ts"] >= self.ttl_s:
            return None
        return v


def authorize_url(provider: str, client_id: str, redirect_uri: str,
                  state: str) -> str:
    cfg = PROVIDERS.get(provider)
    if cfg is None:
        raise ProviderAPIError(f"no OAuth config for provider {provider!r}")
    params = {
        "client_id": client_id,
        "redirect_uri": redirect_uri,
        "response_type": "code",
        "state": state,
    }
    if cfg.get("scopes"):
        params[cfg.get("scope_param", "scope")] = cfg["scopes"]
    params.update(cfg.get("extra_authorize", {}))
    return cfg["authorize"] + "?" + urllib.parse.urlencode(params)


def build_token_exchange(provider: str, client_id: str, client_secret: str,
                         code: str, redirect_uri: str) -> Request:
    cfg = PROVIDERS.get(provider)
    if cfg is None:
        raise ProviderAPIError(f"no OAuth config for provider {provider!r}")
    return Request("POST", cfg["token"],
                   headers={"Accept": "application/json",
                            "Content-Type": "application/json"},
                   json_body={"grant_type": "authorization_code",
                              "client_id": client_id,
                              "client_secret": client_secret,
                              "code": code,
                              "redirect_uri": redirect_uri})


def build_token_refresh(provider: str, client_id: str, client_secret: str,
                        refresh_token: str) -> Request:
    cfg = PROVIDERS.get(provider)
    if cfg is None:
        raise ProviderAPIError(f"no OAuth config for provider {provider!r}")
    return Request("POST", cfg["token"],
                   headers={"Accept": "application/json",
                            "Content-Type": "application/json"},
                   json_body={"grant_type": "refresh_token",
                              "client_id": client_id,
                              "client_secret": client_secret,
                              "refresh_token": refresh_token})


def parse_token_response(doc: dict) -> dict:
    """-> {access_token, refresh_token?, expires_at?} or raises."""
    if not isinstance(doc, dict) or doc.get("error"):
        raise ProviderAPIError(str((doc or {}).get(
            "error_description", (doc or {}).get("error", "oauth error")))[:300])
    tok = doc.get("access_token") or (doc.get("authed_user") or {}).get(
        "access_token")  # slack nests the user token
    if not tok:
        raise ProviderAPIError("token response carries no access_token")
    out = {"access_token": tok, "token": tok}
    if doc.get("refresh_token"):
        out["refresh_token"] = doc["refresh_token"]
    if doc.get("expires_in"):
        out["expires_at"] = time.time() + float(doc["expires_in"])
    if doc.get("scope"):
        out["scope"] = doc["scope"]
    return out


def needs_refresh(creds: dict, skew_s: float = 120.0) -> bool:
    exp = creds.get("expires_at")
    return bool(exp) and time.time() >= float(exp) - skew_s


async def exchange_code(transport: Transport, provider: str, client_id: str,
                        client_secret: str, code: str,
                        redirect_uri: str) -> dict:
    return parse_token_response(await transport(build_token_exchange(
        provider, client_id, client_secret, code, redirect_uri)))


async def refresh_credentials(transport: Transport, provider: str,
                              client_id: str, client_secret: str,
                              creds: dict) -> dict:
    """Refresh and MERGE (providers often omit a new refresh_token)."""
    rt = creds.get("refresh_token")
    if not rt:
        raise ProviderAPIError("no refresh_token stored")
    fresh = parse_token_response(await transport(build_token_refresh(
        provider, client_id, client_secret, rt)))
    merged = dict(creds)
    merged.update(fresh)
    if "refresh_token" not in fresh:
        merged["refresh_token"] = rt
    return merged
