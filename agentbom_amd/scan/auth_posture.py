"""MCP server + agent→MCP authentication posture governance.

Reference parity: src/agent_bom/mcp_auth_posture.py — reference-only
assessment of the MCP **server auth** surface per the MCP authorization
spec (OAuth 2.1 for remote servers).  Many MCP servers ship with no auth;
a REMOTE server reachable over the network with no token is the single
biggest exposure on this surface.  agent-bom never becomes an auth broker:
it only reads configs and reports posture.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Any

from agentbom_amd.models import Agent, MCPServer

_AUTH_ENV_RE = re.compile(
    r"(?i)(token|api[_-]?key|auth|bearer|secret|password|credential)")
_LOCAL_URL_RE = re.compile(r"^https?://(localhost|127\.0\.0\.1|\[::1\])([:/]|$)")

# locked posture vocabulary
POSTURE_NO_AUTH_REMOTE = "no_auth_remote"          # critical
POSTURE_NO_AUTH_LOCAL = "no_auth_local"            # info
POSTURE_STATIC_TOKEN = "static_token"              # medium
POSTURE_OAUTH = "oauth"                            # hardened
POSTURE_PLAINTEXT_TRANSPORT = "plaintext_transport"  # high
POSTURE_STDIO = "stdio_local"                      # info


@dataclass
class AuthPostureFinding:
    server_name: str
    posture: str
    severity: str
    detail: str
    agent_names: list[str] = field(default_factory=list)

    def to_dict(self) -> dict[str, Any]:
        return {"server_name": self.server_name, "posture": self.posture,
                "severity": self.severity, "detail": self.detail,
                "agent_names": self.agent_names}


def _server_auth_signals(server: MCPServer) -> dict[str, bool]:
    env_keys = " ".join(server.env)
    args_text = " ".join(server.args)
    has_token = bool(_AUTH_ENV_RE.search(env_keys)
                     or _AUTH_ENV_RE.search(args_text))
    has_oauth = any("oauth" in k.lower() or "client_id" in k.lower()
                    for k in server.env) or "oauth" in args_text.lower()
    return {"has_token": has_token, "has_oauth": has_oauth}


def assess_server(server: MCPServer,
                  agent_names: list[str]) -> AuthPostureFinding:
    """Posture of one configured server (reference-only, never spawns it)."""
    signals = _server_auth_signals(server)
    if not server.url:  # stdio regardless of declared transport enum
        return AuthPostureFinding(
            server.name, POSTURE_STDIO, "info",
            "local stdio server: process-boundary auth (OS user)", agent_names)

    url = server.url or ""
    local = bool(_LOCAL_URL_RE.match(url))
    if url.startswith("http://") and not local:
        return AuthPostureFinding(
            server.name, POSTURE_PLAINTEXT_TRANSPORT, "high",
            f"remote MCP over plaintext http: {url}", agent_names)
    if signals["has_oauth"]:
        return AuthPostureFinding(
            server.name, POSTURE_OAUTH, "hardened",
            "OAuth client configuration present (MCP authorization spec)",
            agent_names)
    if signals["has_token"]:
        return AuthPostureFinding(
            server.name, POSTURE_STATIC_TOKEN, "medium",
            "static bearer token/API key: rotate regularly; prefer OAuth 2.1",
            agent_names)
    if local:
        return AuthPostureFinding(
            server.name, POSTURE_NO_AUTH_LOCAL, "info",
            "loopback server without auth (local trust boundary)", agent_names)
    return AuthPostureFinding(
        server.name, POSTURE_NO_AUTH_REMOTE, "critical",
        f"REMOTE MCP server with no authentication configured: {url}",
        agent_names)


def assess_a2a(agents: list[Agent],
               identity_store=None) -> dict[str, Any]:
    """Inter-agent (A2A) auth posture — reference-only, four weakness classes
    (reference: src/agent_bom/a2a_auth_posture.py):

    1. shared static credentials between agents (one token, many callers);
    2. missing mutual auth — no issued identity covers an agent that shares
       infrastructure with others;
    3. over-broad delegation — wildcard ``allowed_tools``/scopes identities;
    4. deep shared-server webs enabling unbounded transitive delegation.
    """
    findings: list[dict[str, Any]] = []

    # credential name -> agents whose servers carry it
    cred_agents: dict[str, set[str]] = {}
    server_agents: dict[str, set[str]] = {}
    for agent in agents:
        for server in agent.mcp_servers:
            server_agents.setdefault(server.name, set()).add(agent.name)
            for cred in server.credential_names:
                cred_agents.setdefault(cred, set()).add(agent.name)

    for cred, names in sorted(cred_agents.items()):
        if len(names) > 1:
            findings.append({
                "weakness": "shared_static_credential", "severity": "high",
                "credential": cred, "agents": sorted(names),
                "detail": f"one static credential reaches {len(names)} agents; "
                          "issue per-agent short-lived identities instead"})

    identified: set[str] = set()
    wildcard: list[str] = []
    if identity_store is not None:
        for ident in identity_store.list(live_only=True):
            identified.add(ident.agent_name)
            if "*" in ident.scopes or not ident.allowed_tools:
                wildcard.append(ident.identity_id)
    for server, names in sorted(server_agents.items()):
        if len(names) > 1:
            unauthenticated = sorted(n for n in names if n not in identified)
            if unauthenticated:
                findings.append({
                    "weakness": "missing_mutual_auth", "severity": "medium",
                    "server": server, "agents": unauthenticated,
                    "detail": "agents share this server with no issued "
                              "identity binding the caller"})
    for ident_id in sorted(wildcard):
        findings.append({
            "weakness": "over_broad_delegation", "severity": "medium",
            "identity": ident_id,
            "detail": "identity carries wildcard scope / unbounded tools"})

    # transitive web: agents connected through >=2 shared servers
    pair_shared: dict[tuple, int] = {}
    for names in server_agents.values():
        ordered = sorted(names)
        for i, a in enumerate(ordered):
            for b in ordered[i + 1:]:
                pair_shared[(a, b)] = pair_shared.get((a, b), 0) + 1
    for (a, b), n in sorted(pair_shared.items()):
        if n >= 2:
            findings.append({
                "weakness": "unbounded_transitive_delegation",
                "severity": "medium", "agents": [a, b],
                "detail": f"{n} shared servers form an unbounded lateral "
                          "delegation web"})

    by_weakness: dict[str, int] = {}
    for f in findings:
        by_weakness[f["weakness"]] = by_weakness.get(f["weakness"], 0) + 1
    return {"schema_version": "1", "findings": findings,
            "by_weakness": by_weakness}


def assess_estate(agents: list[Agent]) -> dict[str, Any]:
    """Auth posture across every discovered agent→server edge."""
    by_server: dict[str, tuple[MCPServer, list[str]]] = {}
    for agent in agents:
        for server in agent.mcp_servers:
            entry = by_server.setdefault(server.name, (server, []))
            if agent.name not in entry[1]:
                entry[1].append(agent.name)

    findings = [assess_server(srv, sorted(names))
                for srv, names in by_server.values()]
    findings.sort(key=lambda f: ({"critical": 0, "high": 1, "medium": 2,
                                  "info": 3, "hardened": 4}.get(f.severity, 5),
                                 f.server_name))
    counts: dict[str, int] = {}
    for f in findings:
        counts[f.posture] = counts.get(f.posture, 0) + 1
    return {
        "schema_version": "1",
        "servers_assessed": len(findings),
        "posture_counts": counts,
        "critical_exposures": [f.to_dict() for f in findings
                               if f.severity == "critical"],
        "findings": [f.to_dict() for f in findings],
    }
