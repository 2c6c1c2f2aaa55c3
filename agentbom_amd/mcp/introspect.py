"""MCP runtime introspection — connect to live stdio MCP servers, read-only.

Spawns each configured server, speaks the MCP stdio transport directly
(line-delimited JSON-RPC 2.0 — no SDK dependency), calls ``initialize`` /
``tools/list`` / ``resources/list`` / ``prompts/list``, and diffs the runtime
capability surface against what the config declared (rug-pull / drift
detection).

Security contract (parity with reference src/agent_bom/mcp_introspect.py:1-17):
- **Read-only** — never calls ``tools/call``.
- **Timeout-guarded** — every phase is bounded; hung servers are killed.
- **Clean shutdown** — subprocesses are terminated (then killed) on exit.
- **Sanitized** — runtime-supplied strings are length-capped and stripped of
  control characters before they reach reports.
"""

from __future__ import annotations

import json
import re
import subprocess
import threading
from dataclasses import dataclass, field
from queue import Empty, Queue
from typing import Any, Optional

from agentbom_amd.models import MCPPrompt, MCPResource, MCPServer, MCPTool

DEFAULT_TIMEOUT = 10.0
HEALTH_CHECK_TIMEOUT = 5.0

_PATH_HINT_RE = re.compile(r"(path|file|dir|cwd|workspace)", re.IGNORECASE)
_URL_HINT_RE = re.compile(r"(url|uri|endpoint|host|domain|webhook)", re.IGNORECASE)
_SHELL_HINT_RE = re.compile(r"(cmd|command|shell|exec|script)", re.IGNORECASE)
_PROMPT_HINT_RE = re.compile(
    r"(?<![A-Za-z0-9])(?:prompt|instruction|system|markdown|html|svg)(?![A-Za-z0-9])",
    re.IGNORECASE)
_CTRL_RE = re.compile(r"[\x00-\x08\x0b-\x1f\x7f]")


def _safe_text(value: Any, max_len: int = 1000) -> str:
    from agentbom_amd.utils.security import fence_untrusted

    return fence_untrusted(value, max_len=max_len)


class IntrospectionError(Exception):
    pass


@dataclass
class ServerIntrospection:
    """Result of introspecting a single MCP server."""

    server_name: str
    success: bool
    protocol_version: Optional[str] = None
    server_info: dict[str, Any] = field(default_factory=dict)
    configured_tool_count: int = 0
    configured_resource_count: int = 0
    configured_prompt_count: int = 0
    runtime_tools: list[MCPTool] = field(default_factory=list)
    runtime_resources: list[MCPResource] = field(default_factory=list)
    runtime_prompts: list[MCPPrompt] = field(default_factory=list)
    error: Optional[str] = None
    tool_schema_findings: list[str] = field(default_factory=list)
    resource_findings: list[str] = field(default_factory=list)
    prompt_findings: list[str] = field(default_factory=list)
    capability_risk_score: float = 0.0
    capability_risk_level: str = "low"
    tools_added: list[str] = field(default_factory=list)
    tools_removed: list[str] = field(default_factory=list)
    resources_added: list[str] = field(default_factory=list)
    resources_removed: list[str] = field(default_factory=list)
    prompts_added: list[str] = field(default_factory=list)
    prompts_removed: list[str] = field(default_factory=list)

    @property
    def has_drift(self) -> bool:
        return bool(self.tools_added or self.tools_removed or self.resources_added
                    or self.resources_removed or self.prompts_added or self.prompts_removed)

    def to_dict(self) -> dict[str, Any]:
        return {
            "server_name": _safe_text(self.server_name, 300),
            "success": self.success,
            "protocol_version": _safe_text(self.protocol_version, 100) or None,
            "server_info": {k: _safe_text(v, 300) for k, v in self.server_info.items()},
            "configured_tool_count": self.configured_tool_count,
            "tool_count": len(self.runtime_tools),
            "resource_count": len(self.runtime_resources),
            "prompt_count": len(self.runtime_prompts),
            "tools_added": self.tools_added,
            "tools_removed": self.tools_removed,
            "resources_added": self.resources_added,
            "resources_removed": self.resources_removed,
            "prompts_added": self.prompts_added,
            "prompts_removed": self.prompts_removed,
            "has_drift": self.has_drift,
            "tool_schema_findings": self.tool_schema_findings,
            "resource_findings": self.resource_findings,
            "prompt_findings": self.prompt_findings,
            "capability_risk_score": self.capability_risk_score,
            "capability_risk_level": self.capability_risk_level,
            "error": _safe_text(self.error, 500) or None,
        }


@dataclass
class IntrospectionReport:
    results: list[ServerIntrospection] = field(default_factory=list)
    warnings: list[str] = field(default_factory=list)

    @property
    def successful(self) -> int:
        return sum(1 for r in self.results if r.success)

    @property
    def drift_count(self) -> int:
        return sum(1 for r in self.results if r.has_drift)

    def to_dict(self) -> dict[str, Any]:
        return {
            "total_servers": len(self.results),
            "successful": self.successful,
            "failed": len(self.results) - self.successful,
            "total_tools": sum(len(r.runtime_tools) for r in self.results),
            "total_resources": sum(len(r.runtime_resources) for r in self.results),
            "total_prompts": sum(len(r.runtime_prompts) for r in self.results),
            "drift_count": self.drift_count,
            "results": [r.to_dict() for r in self.results],
            "warnings": self.warnings,
        }


# ── stdio JSON-RPC client ───────────────────────────────────────────────────


class _StdioClient:
    """Minimal line-delimited JSON-RPC client over a spawned subprocess."""

    def __init__(self, command: str, args: list[str], env: Optional[dict[str, str]],
                 cwd: Optional[str], timeout: float):
        import os

        self.timeout = timeout
        full_env = dict(os.environ)
        full_env.update(env or {})
        self.proc = subprocess.Popen(
            [command, *args], stdin=subprocess.PIPE, stdout=subprocess.PIPE,
            stderr=subprocess.DEVNULL, env=full_env, cwd=cwd, text=True)
        self._queue: Queue = Queue()
        self._reader = threading.Thread(target=self._drain, daemon=True)
        self._reader.start()
        self._next_id = 0

    def _drain(self) -> None:
        try:
            for line in self.proc.stdout:  # type: ignore[union-attr]
                line = line.strip()
                if line:
                    self._queue.put(line)
        except ValueError:
            pass  # stdout closed during shutdown

    def request(self, method: str, params: Optional[dict] = None) -> Any:
        self._next_id += 1
        msg = {"jsonrpc": "2.0", "id": self._next_id, "method": method}
        if params is not None:
            msg["params"] = params
        self._send(msg)
        deadline = self.timeout
        while True:
            try:
                line = self._queue.get(timeout=deadline)
            except Empty:
                raise IntrospectionError(f"timeout waiting for {method} response")
            try:
                resp = json.loads(line)
            except json.JSONDecodeError:
                continue
            if resp.get("id") == self._next_id:
                if "error" in resp:
                    raise IntrospectionError(str(resp["error"].get("message", "error")))
                return resp.get("result")

    def notify(self, method: str) -> None:
        self._send({"jsonrpc": "2.0", "method": method})

    def _send(self, msg: dict) -> None:
        if self.proc.stdin is None or self.proc.poll() is not None:
            raise IntrospectionError("server process exited")
        try:
            self.proc.stdin.write(json.dumps(msg) + "\n")
            self.proc.stdin.flush()
        except (BrokenPipeError, OSError) as exc:
            raise IntrospectionError(f"server pipe closed: {exc}")

    def close(self) -> None:
        try:
            self.proc.terminate()
            self.proc.wait(timeout=2)
        except subprocess.TimeoutExpired:
            self.proc.kill()
        except OSError:
            pass


# ── schema / content linting ────────────────────────────────────────────────


def lint_tool_schema(name: str, schema: Optional[dict]) -> list[str]:
    """Static findings over a tool's declared input schema."""
    findings = []
    props = (schema or {}).get("properties", {})
    prop_text = " ".join(f"{k} {v.get('description', '')}" for k, v in props.items()
                         if isinstance(v, dict))
    blob = f"{name} {prop_text}"
    if _SHELL_HINT_RE.search(blob):
        findings.append("shell-execution-capability")
    if _URL_HINT_RE.search(blob):
        findings.append("network-egress-capability")
    if _PATH_HINT_RE.search(blob):
        findings.append("filesystem-capability")
    if schema and schema.get("additionalProperties") is not False:
        findings.append("open-schema-additional-properties")
    if not props and schema is not None:
        findings.append("unconstrained-input-schema")
    return findings


def lint_resource(uri: str, name: str, description: str) -> list[str]:
    findings = []
    if _PROMPT_HINT_RE.search(f"{uri} {name} {description}"):
        findings.append("prompt-bearing-resource")
    if uri.startswith(("http://", "https://")):
        findings.append("mutable-resource")
    return findings


def lint_prompt(name: str, description: str) -> list[str]:
    findings = []
    if _PROMPT_HINT_RE.search(f"{name} {description}"):
        findings.append("system-prompt-surface")
    return findings


# ── introspection ───────────────────────────────────────────────────────────


def introspect_server(server: MCPServer,
                      timeout: float = DEFAULT_TIMEOUT) -> ServerIntrospection:
    """Spawn + initialize one stdio server, list its capabilities, diff config."""
    result = ServerIntrospection(
        server_name=server.name, success=False,
        configured_tool_count=len(server.tools),
        configured_resource_count=len(server.resources),
        configured_prompt_count=len(server.prompts))
    if not server.command:
        result.error = "no command configured (non-stdio transport?)"
        return result

    client = None
    try:
        client = _StdioClient(server.command, server.args, server.env,
                              server.working_dir, timeout)
        init = client.request("initialize", {
            "protocolVersion": "2024-11-05",
            "capabilities": {},
            "clientInfo": {"name": "agent-bom-introspect", "version": "1.0"},
        }) or {}
        result.protocol_version = init.get("protocolVersion")
        result.server_info = dict(init.get("serverInfo") or {})
        client.notify("notifications/initialized")

        tools = (client.request("tools/list") or {}).get("tools", [])
        for t in tools:
            findings = lint_tool_schema(str(t.get("name", "")), t.get("inputSchema"))
            result.runtime_tools.append(MCPTool(
                name=_safe_text(t.get("name"), 300),
                description=_safe_text(t.get("description"), 1000),
                input_schema=t.get("inputSchema"),
                schema_findings=findings,
                discovery_source="runtime-introspection"))
            for f in findings:
                result.tool_schema_findings.append(f"{t.get('name')}: {f}")

        for method, bucket in (("resources/list", "resources"), ("prompts/list", "prompts")):
            try:
                items = (client.request(method) or {}).get(bucket, [])
            except IntrospectionError:
                items = []  # optional capability
            if bucket == "resources":
                for r in items:
                    findings = lint_resource(str(r.get("uri", "")), str(r.get("name", "")),
                                             str(r.get("description", "")))
                    result.runtime_resources.append(MCPResource(
                        uri=_safe_text(r.get("uri"), 1000),
                        name=_safe_text(r.get("name"), 300),
                        description=_safe_text(r.get("description"), 1000),
                        mime_type=r.get("mimeType"),
                        content_findings=findings))
                    for f in findings:
                        result.resource_findings.append(f"{r.get('uri')}: {f}")
            else:
                for p in items:
                    findings = lint_prompt(str(p.get("name", "")),
                                           str(p.get("description", "")))
                    result.runtime_prompts.append(MCPPrompt(
                        name=_safe_text(p.get("name"), 300),
                        description=_safe_text(p.get("description"), 1000),
                        arguments=list(p.get("arguments") or []),
                        content_findings=findings))
                    for f in findings:
                        result.prompt_findings.append(f"{p.get('name')}: {f}")

        result.success = True
        _apply_drift(result, server)
        _apply_capability_risk(result)
    except (IntrospectionError, OSError, FileNotFoundError) as exc:
        result.error = str(exc)
    finally:
        if client is not None:
            client.close()
    return result


def _apply_drift(result: ServerIntrospection, server: MCPServer) -> None:
    conf_tools = {t.name for t in server.tools}
    run_tools = {t.name for t in result.runtime_tools}
    result.tools_added = sorted(run_tools - conf_tools)
    result.tools_removed = sorted(conf_tools - run_tools)
    conf_res = {r.uri for r in server.resources}
    run_res = {r.uri for r in result.runtime_resources}
    result.resources_added = sorted(run_res - conf_res)
    result.resources_removed = sorted(conf_res - run_res)
    conf_p = {p.name for p in server.prompts}
    run_p = {p.name for p in result.runtime_prompts}
    result.prompts_added = sorted(run_p - conf_p)
    result.prompts_removed = sorted(conf_p - run_p)


def _apply_capability_risk(result: ServerIntrospection) -> None:
    from agentbom_amd.models import MCPServer as _S
    from agentbom_amd.scan.risk import score_server_risk, server_risk_level

    shadow = _S(name=result.server_name, command="x", tools=result.runtime_tools)
    result.capability_risk_score = score_server_risk(shadow)
    result.capability_risk_level = server_risk_level(result.capability_risk_score)


def introspect_servers(servers: list[MCPServer], timeout: float = DEFAULT_TIMEOUT,
                       max_workers: int = 4) -> IntrospectionReport:
    """Introspect servers concurrently (thread-per-server, bounded)."""
    from concurrent.futures import ThreadPoolExecutor

    report = IntrospectionReport()
    stdio = [s for s in servers if s.command]
    skipped = [s.name for s in servers if not s.command]
    for name in skipped:
        report.warnings.append(f"{name}: skipped (no stdio command)")
    with ThreadPoolExecutor(max_workers=max_workers) as pool:
        report.results = list(pool.map(
            lambda s: introspect_server(s, timeout=timeout), stdio))
    return report


# ── health checks ───────────────────────────────────────────────────────────


@dataclass
class HealthStatus:
    server_name: str
    healthy: bool
    latency_ms: Optional[float] = None
    error: Optional[str] = None

    def to_dict(self) -> dict[str, Any]:
        return {"server_name": self.server_name, "healthy": self.healthy,
                "latency_ms": self.latency_ms, "error": _safe_text(self.error, 300) or None}


def health_check_server(server: MCPServer,
                        timeout: float = HEALTH_CHECK_TIMEOUT) -> HealthStatus:
    """Spawn + initialize only — cheapest liveness probe for a stdio server."""
    import time

    if not server.command:
        return HealthStatus(server.name, healthy=False, error="no stdio command")
    client = None
    t0 = time.perf_counter()
    try:
        client = _StdioClient(server.command, server.args, server.env,
                              server.working_dir, timeout)
        client.request("initialize", {
            "protocolVersion": "2024-11-05", "capabilities": {},
            "clientInfo": {"name": "agent-bom-health", "version": "1.0"}})
        return HealthStatus(server.name, healthy=True,
                            latency_ms=round((time.perf_counter() - t0) * 1000, 2))
    except (IntrospectionError, OSError, FileNotFoundError) as exc:
        return HealthStatus(server.name, healthy=False, error=str(exc))
    finally:
        if client is not None:
            client.close()


def health_check_servers(servers: list[MCPServer],
                         timeout: float = HEALTH_CHECK_TIMEOUT,
                         max_workers: int = 4) -> list[HealthStatus]:
    from concurrent.futures import ThreadPoolExecutor

    with ThreadPoolExecutor(max_workers=max_workers) as pool:
        return list(pool.map(lambda s: health_check_server(s, timeout=timeout), servers))


def enrich_servers(servers: list[MCPServer],
                   timeout: float = DEFAULT_TIMEOUT) -> IntrospectionReport:
    """Introspect + write runtime capabilities back onto the server models."""
    report = introspect_servers(servers, timeout=timeout)
    by_name = {r.server_name: r for r in report.results}
    for server in servers:
        r = by_name.get(server.name)
        if r is None or not r.success:
            continue
        server.tools = r.runtime_tools
        server.resources = r.runtime_resources
        server.prompts = r.runtime_prompts
        server.mcp_version = r.protocol_version or server.mcp_version
        server.stamp_child_identities()
        if r.has_drift:
            server.security_warnings.append(
                f"runtime drift: +{len(r.tools_added)} tools, -{len(r.tools_removed)} tools")
    return report
