"""MCP runtime proxy: wrap a target MCP server with inline detectors.

Reference: src/agent_bom/proxy.py (stdio wrapping, per-message detector
pipeline, audited allow/warn/block), proxy_audit.py (JSONL audit trail),
proxy_policy.py (allow/deny tool lists).

``agent-bom proxy -- <command>`` launches the target server as a child
process, relays stdio JSON-RPC both ways, runs every frame through the
detector pipeline, blocks on critical hits (the request is answered with a
JSON-RPC error instead of being forwarded), and appends one audit record
per frame to ``~/.agent-bom/proxy_audit.jsonl``.
"""

from __future__ import annotations

import json
import os
import subprocess
import sys
import threading
import time
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Optional, TextIO

from agentbom_amd.runtime.detectors import DetectorPipeline


@dataclass
class ProxyPolicy:
    """Allow/deny tool lists + default action for detector warnings.

    ``mode``: "enforce" (default — blocks are enforced) or "observe" —
    every decision is still computed and AUDITED (with the action the
    policy WOULD have taken), but nothing is blocked.  Observe-first is
    the standard rollout path for a new detector pipeline (reference:
    observe_enforce.py)."""

    allow_tools: Optional[list[str]] = None  # None = all
    deny_tools: list[str] = field(default_factory=list)
    block_on_warn: bool = False
    mode: str = "enforce"  # enforce | observe

    def tool_allowed(self, tool: str) -> bool:
        if tool in self.deny_tools:
            return False
        if self.allow_tools is not None and tool not in self.allow_tools:
            return False
        return True

    @classmethod
    def load(cls, path: Optional[str]) -> "ProxyPolicy":
        if not path or not Path(path).exists():
            return cls()
        data = json.loads(Path(path).read_text())
        return cls(
            allow_tools=data.get("allow_tools"),
            deny_tools=data.get("deny_tools", []),
            block_on_warn=data.get("block_on_warn", False),
            mode=data.get("mode", "enforce"),
        )


class AuditLog:
    """Append-only JSONL audit sink with chained integrity hashes."""

    def __init__(self, path: Optional[str] = None):
        import hashlib

        self._hashlib = hashlib
        self.path = Path(path or os.path.expanduser("~/.agent-bom/proxy_audit.jsonl"))
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self._prev_hash = "0" * 64
        self._lock = threading.Lock()

    def record(self, entry: dict[str, Any]) -> None:
        from agentbom_amd.utils.security import redact_structure

        with self._lock:
            entry = redact_structure(dict(entry))
            entry["ts"] = time.time()
            entry["prev_hash"] = self._prev_hash
            body = json.dumps(entry, sort_keys=True, default=str)
            entry_hash = self._hashlib.sha256(body.encode()).hexdigest()
            entry["hash"] = entry_hash
            self._prev_hash = entry_hash
            with open(self.path, "a") as f:
                f.write(json.dumps(entry, default=str) + "\n")

    @staticmethod
    def verify(path: str | Path) -> tuple[bool, int]:
        """Replay the HMAC-style hash chain; returns (intact, entries)."""
        import hashlib

        prev = "0" * 64
        n = 0
        for line in Path(path).read_text().splitlines():
            entry = json.loads(line)
            claimed = entry.pop("hash")
            if entry.get("prev_hash") != prev:
                return False, n
            body = json.dumps(entry, sort_keys=True, default=str)
            if hashlib.sha256(body.encode()).hexdigest() != claimed:
                return False, n
            prev = claimed
            n += 1
        return True, n


class McpProxy:
    """Bidirectional stdio relay with inline detection."""

    def __init__(self, command: list[str], policy: Optional[ProxyPolicy] = None,
                 audit: Optional[AuditLog] = None,
                 pipeline: Optional[DetectorPipeline] = None):
        self.command = command
        self.policy = policy or ProxyPolicy()
        self.audit = audit or AuditLog()
        self.pipeline = pipeline or DetectorPipeline()
        self.blocked = 0
        self.relayed = 0

    # frame-level decision (unit-testable without processes)
    def decide(self, frame: dict[str, Any], direction: str) -> tuple[str, list]:
        method = frame.get("method", "")
        if method == "tools/call":
            tool = (frame.get("params") or {}).get("name", "")
            if not self.policy.tool_allowed(tool):
                denial = [{"detector": "policy", "severity": "critical",
                           "message": f"tool {tool!r} denied by proxy policy",
                           "action": "block", "evidence": {}}]
                if self.policy.mode == "observe":
                    denial[0]["would_block"] = True
                    return "warn", denial
                return "block", denial
        action, alerts = self.pipeline.inspect(frame)
        if action == "warn" and self.policy.block_on_warn:
            action = "block"
        out = [a.to_dict() if hasattr(a, "to_dict") else a for a in alerts]
        if self.policy.mode == "observe" and action == "block":
            # observe mode: record what WOULD have happened, enforce nothing
            for a in out:
                a["would_block"] = True
            return "warn", out
        return action, out

    def _relay(self, src: TextIO, dst: TextIO, direction: str) -> None:
        for line in src:
            line = line.strip()
            if not line:
                continue
            try:
                frame = json.loads(line)
            except json.JSONDecodeError:
                dst.write(line + "\n")
                dst.flush()
                continue
            action, alerts = self.decide(frame, direction)
            self.audit.record({
                "direction": direction,
                "method": frame.get("method"),
                "id": frame.get("id"),
                "action": action,
                "alerts": alerts,
            })
            if action == "block" and direction == "client->server":
                self.blocked += 1
                err = {"jsonrpc": "2.0", "id": frame.get("id"),
                       "error": {"code": -32000,
                                 "message": "blocked by agent-bom proxy: "
                                            + "; ".join(a["message"] for a in alerts[:3])}}
                sys.stdout.write(json.dumps(err) + "\n")
                sys.stdout.flush()
                continue
            self.relayed += 1
            dst.write(json.dumps(frame) + "\n")
            dst.flush()

    def run(self) -> int:
        proc = subprocess.Popen(
            self.command, stdin=subprocess.PIPE, stdout=subprocess.PIPE,
            text=True, bufsize=1,
        )
        up = threading.Thread(target=self._relay,
                              args=(sys.stdin, proc.stdin, "client->server"), daemon=True)
        down = threading.Thread(target=self._relay,
                                args=(proc.stdout, sys.stdout, "server->client"), daemon=True)
        up.start()
        down.start()
        try:
            return proc.wait()
        except KeyboardInterrupt:
            proc.terminate()
            return 130


# ── HTTP / SSE transport wrap (reference proxy.py HTTP+SSE frames) ─────────


def parse_sse_stream(text: str) -> list[dict[str, Any]]:
    """Parse a text/event-stream body into JSON-RPC frames.

    Handles multi-line ``data:`` continuation and ignores comments/other
    fields (the subset MCP's SSE transport uses)."""
    frames: list[dict[str, Any]] = []
    data_lines: list[str] = []
    for raw in text.splitlines() + [""]:
        line = raw.rstrip("\r")
        if line.startswith("data:"):
            data_lines.append(line[5:].lstrip())
            continue
        if line == "" and data_lines:
            payload = "\n".join(data_lines)
            data_lines = []
            try:
                frames.append(json.loads(payload))
            except json.JSONDecodeError:
                continue
    return frames


class HttpMcpProxy:
    """stdio-facing proxy for an HTTP(S)/SSE MCP server.

    The client speaks newline JSON-RPC on stdio; each frame passes the
    SAME decide()/audit path as the stdio proxy, then POSTs to the
    upstream URL.  ``application/json`` responses relay directly; a
    ``text/event-stream`` response is parsed into frames and each relayed
    after response-side inspection.  Offline mode refuses construction —
    wrapping a remote server is inherently a network operation."""

    def __init__(self, url: str, policy: Optional[ProxyPolicy] = None,
                 audit: Optional[AuditLog] = None,
                 pipeline: Optional[DetectorPipeline] = None,
                 client=None, headers: Optional[dict[str, str]] = None):
        from agentbom_amd.utils.http_client import check_offline, create_client

        check_offline(url)
        self.url = url
        self.headers = dict(headers or {})
        self._inner = McpProxy([], policy=policy, audit=audit, pipeline=pipeline)
        self.client = client or create_client(timeout=60.0)

    @property
    def blocked(self) -> int:
        return self._inner.blocked

    @property
    def relayed(self) -> int:
        return self._inner.relayed

    def forward(self, frame: dict[str, Any]) -> list[dict[str, Any]]:
        """One request frame -> response frame(s) (SSE may yield several)."""
        action, alerts = self._inner.decide(frame, "client->server")
        self._inner.audit.record({
            "direction": "client->server", "transport": "http",
            "method": frame.get("method"), "id": frame.get("id"),
            "action": action, "alerts": alerts,
        })
        if action == "block":
            self._inner.blocked += 1
            return [{"jsonrpc": "2.0", "id": frame.get("id"),
                     "error": {"code": -32000,
                               "message": "blocked by agent-bom proxy: "
                               + "; ".join(a["message"] for a in alerts[:3])}}]
        resp = self.client.post(self.url, json=frame, headers={
            "Accept": "application/json, text/event-stream", **self.headers})
        ctype = resp.headers.get("content-type", "")
        if ctype.startswith("text/event-stream"):
            replies = parse_sse_stream(resp.text)
        else:
            try:
                replies = [resp.json()]
            except ValueError:
                replies = []
        out: list[dict[str, Any]] = []
        for reply in replies:
            r_action, r_alerts = self._inner.decide(reply, "server->client")
            self._inner.audit.record({
                "direction": "server->client", "transport": "http",
                "id": reply.get("id"), "action": r_action, "alerts": r_alerts,
            })
            if r_action == "block":
                self._inner.blocked += 1
                out.append({"jsonrpc": "2.0", "id": reply.get("id"),
                            "error": {"code": -32000,
                                      "message": "upstream response blocked"}})
                continue
            self._inner.relayed += 1
            out.append(reply)
        return out

    def run(self) -> int:
        """stdio loop: read frames from stdin, emit replies on stdout."""
        for line in sys.stdin:
            line = line.strip()
            if not line:
                continue
            try:
                frame = json.loads(line)
            except json.JSONDecodeError:
                continue
            for reply in self.forward(frame):
                sys.stdout.write(json.dumps(reply) + "\n")
                sys.stdout.flush()
        return 0


# ── opt-in sandbox (reference proxy_sandbox.py) ─────────────────────────────


def sandbox_command(command: list[str], runtime: str = "docker",
                    image: str = "node:20-slim", network: bool = False,
                    mounts: Optional[list[tuple[str, str]]] = None,
                    memory: str = "512m", pids: int = 256) -> list[str]:
    """Wrap a server launch in a container sandbox (docker/podman).

    Deny-by-default posture: no network (unless opted in), read-only
    root, memory/pids limits, all capabilities dropped, no privilege
    escalation.  Returns the argv; the caller execs it when the runtime
    is present (sandbox_available)."""
    if runtime not in ("docker", "podman"):
        raise ValueError(f"unsupported sandbox runtime {runtime!r}")
    argv = [
        runtime, "run", "-i", "--rm",
        "--read-only",
        "--cap-drop=ALL",
        "--security-opt", "no-new-privileges",
        f"--memory={memory}",
        f"--pids-limit={pids}",
    ]
    if not network:
        argv.append("--network=none")
    for host, cont in mounts or []:
        argv += ["-v", f"{host}:{cont}:ro"]
    argv.append(image)
    argv += command
    return argv


def sandbox_available(runtime: str = "docker") -> bool:
    import shutil

    return shutil.which(runtime) is not None


def sandboxed_proxy(command: list[str], policy: Optional[ProxyPolicy] = None,
                    audit: Optional[AuditLog] = None, runtime: str = "docker",
                    **sandbox_kw) -> McpProxy:
    """McpProxy whose child runs inside the container sandbox."""
    if not sandbox_available(runtime):
        raise RuntimeError(
            f"sandbox runtime {runtime!r} not installed; run without "
            "--sandbox or install docker/podman")
    return McpProxy(sandbox_command(command, runtime=runtime, **sandbox_kw),
                    policy=policy, audit=audit)
