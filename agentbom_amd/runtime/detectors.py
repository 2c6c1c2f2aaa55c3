"""Runtime detectors for the MCP proxy/gateway data path.

Reference: src/agent_bom/runtime/detectors.py:168-779 — ToolDriftDetector,
ArgumentAnalyzer, CredentialLeakDetector, prompt-injection patterns,
RateLimitTracker, SequenceAnalyzer, ResponseInspector,
VectorDBInjectionDetector, ReplayDetector, CrossAgentCorrelator,
VisualLeakDetector.  Each detector consumes one JSON-RPC frame (or
response) and yields structured alerts; per-detector counters feed
/metrics.
"""

from __future__ import annotations

import hashlib
import re
import time
from collections import defaultdict, deque
from dataclasses import dataclass, field
from typing import Any, Optional


@dataclass
class DetectorAlert:
    detector: str
    severity: str  # info | warning | critical
    message: str
    evidence: dict[str, Any] = field(default_factory=dict)
    action: str = "warn"  # allow | warn | block

    def to_dict(self) -> dict[str, Any]:
        return {
            "detector": self.detector, "severity": self.severity,
            "message": self.message, "evidence": self.evidence, "action": self.action,
        }


class BaseDetector:
    name = "base"

    def __init__(self) -> None:
        self.hits = 0

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        raise NotImplementedError

    def _alert(self, severity: str, message: str, action: str = "warn", **evidence) -> DetectorAlert:
        self.hits += 1
        return DetectorAlert(self.name, severity, message, dict(evidence), action)


class ToolDriftDetector(BaseDetector):
    """Rug-pull detection: a tool's description/schema changed after first
    sight (hash pinning per tool name)."""

    name = "tool_drift"

    def __init__(self) -> None:
        super().__init__()
        self._pinned: dict[str, str] = {}

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        alerts = []
        result = frame.get("result") or {}
        for tool in result.get("tools", []) if isinstance(result, dict) else []:
            name = tool.get("name", "")
            digest = hashlib.sha256(
                f"{tool.get('description','')}|{tool.get('inputSchema','')}".encode()
            ).hexdigest()
            prior = self._pinned.get(name)
            if prior is None:
                self._pinned[name] = digest
            elif prior != digest:
                alerts.append(self._alert(
                    "critical", f"tool {name!r} definition changed after first sight (rug pull)",
                    action="block", tool=name, previous_hash=prior[:16], new_hash=digest[:16],
                ))
        return alerts


_CRED_PATTERNS = [
    (re.compile(r"AKIA[0-9A-Z]{16}"), "aws_access_key_id"),
    (re.compile(r"(?i)aws_secret_access_key\s*[=:]\s*\S{20,}"), "aws_secret"),
    (re.compile(r"ghp_[A-Za-z0-9]{36}"), "github_pat"),
    (re.compile(r"github_pat_[A-Za-z0-9_]{22,}"), "github_fine_grained_pat"),
    (re.compile(r"sk-[A-Za-z0-9]{20,}"), "openai_api_key"),
    (re.compile(r"sk-ant-[A-Za-z0-9\-_]{20,}"), "anthropic_api_key"),
    (re.compile(r"xox[bpras]-[A-Za-z0-9\-]{10,}"), "slack_token"),
    (re.compile(r"-----BEGIN (?:RSA |EC |OPENSSH )?PRIVATE KEY-----"), "private_key"),
    (re.compile(r"(?i)(password|passwd|secret|token)\s*[=:]\s*['\"][^'\"]{8,}['\"]"), "generic_secret"),
    (re.compile(r"eyJ[A-Za-z0-9_-]{10,}\.eyJ[A-Za-z0-9_-]{10,}\."), "jwt"),
]


class CredentialLeakDetector(BaseDetector):
    """Secret material in tool arguments or responses."""

    name = "credential_leak"

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        text = _frame_text(frame)
        alerts = []
        for pattern, kind in _CRED_PATTERNS:
            if pattern.search(text):
                alerts.append(self._alert(
                    "critical", f"credential material ({kind}) in MCP frame",
                    action="block", kind=kind,
                ))
        return alerts


_INJECTION_PATTERNS = [
    re.compile(r"(?i)ignore (all )?(previous|prior|above) (instructions|prompts)"),
    re.compile(r"(?i)you are now (a|an) "),
    re.compile(r"(?i)system prompt"),
    re.compile(r"(?i)<\s*(system|assistant)\s*>"),
    re.compile(r"(?i)do not (tell|inform|alert) the user"),
    re.compile(r"(?i)exfiltrate|keylog|reverse shell"),
    re.compile(r"(?i)IMPORTANT:?\s+(instructions|you must)"),
]


class ArgumentAnalyzer(BaseDetector):
    """Prompt-injection / suspicious-content patterns in tool arguments."""

    name = "argument_analyzer"

    _SHELL_META = re.compile(r"[;&|`$]\s*(?:rm|curl|wget|nc|bash|sh|python)\b")
    _PATH_TRAVERSAL = re.compile(r"\.\./\.\./|/etc/passwd|/etc/shadow|~/.ssh")

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        params = frame.get("params") or {}
        args_text = str(params.get("arguments", ""))
        alerts = []
        for pattern in _INJECTION_PATTERNS:
            if pattern.search(args_text):
                alerts.append(self._alert(
                    "warning", "prompt-injection pattern in tool arguments",
                    pattern=pattern.pattern[:60],
                ))
                break
        if self._SHELL_META.search(args_text):
            alerts.append(self._alert("critical", "shell metacharacter chain in tool arguments",
                                      action="block"))
        if self._PATH_TRAVERSAL.search(args_text):
            alerts.append(self._alert("warning", "path traversal / sensitive path in arguments"))
        return alerts


class ResponseInspector(BaseDetector):
    """Injection patterns and oversized payloads in tool RESPONSES."""

    name = "response_inspector"
    MAX_BYTES = 1_000_000

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        result = frame.get("result")
        if result is None:
            return []
        text = str(result)
        alerts = []
        if len(text) > self.MAX_BYTES:
            alerts.append(self._alert("warning", f"oversized tool response ({len(text)} bytes)"))
        for pattern in _INJECTION_PATTERNS:
            if pattern.search(text):
                alerts.append(self._alert(
                    "critical", "prompt-injection pattern in tool RESPONSE (tool output attack)",
                    action="block", pattern=pattern.pattern[:60],
                ))
                break
        return alerts


class CloakingDetector(BaseDetector):
    """Invisible Unicode / cloaked instructions in frames."""

    name = "cloaking"
    _INVISIBLE = re.compile(r"[​‌‍⁠﻿­-]")

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        text = _frame_text(frame)
        hits = self._INVISIBLE.findall(text)
        if len(hits) > 3:
            return [self._alert("critical", f"{len(hits)} invisible characters (cloaked content)",
                                action="block", count=len(hits))]
        return []


class RateLimitTracker(BaseDetector):
    """Per-tool call-rate abuse (sliding 60s window)."""

    name = "rate_limit"

    def __init__(self, max_per_minute: int = 120):
        super().__init__()
        self.max_per_minute = max_per_minute
        self._calls: dict[str, deque] = defaultdict(deque)

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        if frame.get("method") != "tools/call":
            return []
        tool = (frame.get("params") or {}).get("name", "?")
        now = time.monotonic()
        window = self._calls[tool]
        window.append(now)
        while window and window[0] < now - 60:
            window.popleft()
        if len(window) > self.max_per_minute:
            return [self._alert("warning", f"tool {tool!r} exceeded {self.max_per_minute}/min",
                                tool=tool, rate=len(window))]
        return []


class SequenceAnalyzer(BaseDetector):
    """Suspicious call sequences: read-sensitive then network-egress."""

    name = "sequence_analyzer"
    _READ = re.compile(r"(?i)read|get|list|cat|fetch_file|download")
    _EGRESS = re.compile(r"(?i)http|send|post|upload|email|webhook|publish")

    def __init__(self, window: int = 5):
        super().__init__()
        self._recent: deque = deque(maxlen=window)

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        if frame.get("method") != "tools/call":
            return []
        tool = (frame.get("params") or {}).get("name", "")
        alerts = []
        if self._EGRESS.search(tool) and any(self._READ.search(t) for t in self._recent):
            alerts.append(self._alert(
                "warning", f"read-then-egress sequence ending in {tool!r} (exfil pattern)",
                recent=list(self._recent),
            ))
        self._recent.append(tool)
        return alerts


class ReplayDetector(BaseDetector):
    """Identical frames replayed within a short window."""

    name = "replay"

    def __init__(self, window_s: float = 10.0):
        super().__init__()
        self.window_s = window_s
        self._seen: dict[str, float] = {}

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        if frame.get("method") != "tools/call":
            return []
        key = hashlib.sha256(str(frame.get("params")).encode()).hexdigest()
        now = time.monotonic()
        prior = self._seen.get(key)
        self._seen[key] = now
        if prior is not None and now - prior < self.window_s:
            return [self._alert("warning", "identical tool call replayed within window",
                                delta_s=round(now - prior, 2))]
        return []


class VectorDBInjectionDetector(BaseDetector):
    """Instruction payloads smuggled into vector upsert/query content."""

    name = "vectordb_injection"
    _VECTOR_TOOLS = re.compile(r"(?i)upsert|embed|vector|index_doc")

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        if frame.get("method") != "tools/call":
            return []
        params = frame.get("params") or {}
        if not self._VECTOR_TOOLS.search(params.get("name", "")):
            return []
        text = str(params.get("arguments", ""))
        for pattern in _INJECTION_PATTERNS:
            if pattern.search(text):
                return [self._alert("critical",
                                    "instruction payload in vector-store content (memory poisoning)",
                                    action="block")]
        return []


_HARM_PATTERNS = [
    (re.compile(r"(?i)\b(kill yourself|kys)\b"), "self-harm-content", "critical"),
    (re.compile(r"(?i)\b(all|those) (women|men|jews|muslims|christians|"
                r"immigrants|blacks|whites|asians) (are|should)\b"),
     "demographic-generalization", "warning"),
    (re.compile(r"(?i)\byou (stupid|worthless|pathetic) (idiot|moron|fool)\b"),
     "toxic-address", "warning"),
]

_FABRICATION_PATTERNS = [
    re.compile(r"(?i)as (officially )?(confirmed|verified) by (the )?"
               r"(CDC|WHO|NASA|FBI|government)"),
    re.compile(r"(?i)100% (guaranteed|certain|proven) (cure|safe|effective)"),
    re.compile(r"(?i)studies (show|prove) that .{0,60}(always|never)"),
]


class ContentSafetyDetector(BaseDetector):
    """Toxicity / bias / fabrication patterns in tool RESPONSES.

    Pattern-class parity with the reference's bias/toxicity/hallucination
    detectors: responses carrying harmful address, sweeping demographic
    claims, or authority-fabrication markers are flagged (warn, not block —
    these are review signals, not exfiltration)."""

    name = "content_safety"

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        result = frame.get("result")
        if result is None:
            return []
        text = str(result)
        alerts = []
        for pattern, kind, severity in _HARM_PATTERNS:
            if pattern.search(text):
                alerts.append(self._alert(severity, f"harmful content ({kind}) "
                                          "in tool response", kind=kind))
        for pattern in _FABRICATION_PATTERNS:
            m = pattern.search(text)
            if m:
                alerts.append(self._alert(
                    "warning", "authority-fabrication marker in tool response",
                    match=m.group(0)[:60]))
                break
        return alerts


class CrossAgentCorrelator(BaseDetector):
    """The same credential fingerprint appearing from DIFFERENT sessions —
    a leaked secret being replayed by another agent."""

    name = "cross_agent_correlator"

    def __init__(self) -> None:
        super().__init__()
        self._cred_sessions: dict[str, set[str]] = defaultdict(set)

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        session = str(frame.get("_session") or frame.get("session") or "default")
        text = _frame_text(frame)
        alerts = []
        for pattern, kind in _CRED_PATTERNS:
            m = pattern.search(text)
            if not m:
                continue
            fp = hashlib.sha256(m.group(0).encode()).hexdigest()[:16]
            seen = self._cred_sessions[fp]
            if seen and session not in seen:
                alerts.append(self._alert(
                    "critical",
                    f"credential ({kind}) observed from multiple sessions "
                    "(cross-agent reuse)",
                    action="block", kind=kind, fingerprint=fp,
                    sessions=len(seen) + 1))
            seen.add(session)
        return alerts


_DATA_URI_RE = re.compile(r"data:image/[a-z+]+;base64,([A-Za-z0-9+/=]{64,})")
_BASE64_BLOB_RE = re.compile(r"\b[A-Za-z0-9+/]{200,}={0,2}\b")


class VisualLeakDetector(BaseDetector):
    """Secret material smuggled inside image payloads / large base64 blobs.

    Decodes embedded base64 (bounded) and re-runs the credential patterns
    over the decoded bytes — an exfil channel plain-text scanners miss."""

    name = "visual_leak"

    MAX_DECODE = 256 * 1024

    def inspect(self, frame: dict[str, Any]) -> list[DetectorAlert]:
        import base64

        text = _frame_text(frame)
        blobs = _DATA_URI_RE.findall(text) or _BASE64_BLOB_RE.findall(text)
        alerts = []
        for blob in blobs[:8]:
            try:
                decoded = base64.b64decode(blob[: self.MAX_DECODE],
                                           validate=False)
            except (ValueError, TypeError):
                continue
            try:
                decoded_text = decoded.decode("utf-8", errors="ignore")
            except Exception:  # noqa: BLE001
                continue
            for pattern, kind in _CRED_PATTERNS:
                if pattern.search(decoded_text):
                    alerts.append(self._alert(
                        "critical",
                        f"credential ({kind}) hidden inside base64/image payload",
                        action="block", kind=kind, blob_bytes=len(decoded)))
                    break
        return alerts


DEFAULT_DETECTORS = (
    ToolDriftDetector, ArgumentAnalyzer, CredentialLeakDetector, ResponseInspector,
    CloakingDetector, RateLimitTracker, SequenceAnalyzer, ReplayDetector,
    VectorDBInjectionDetector, ContentSafetyDetector, CrossAgentCorrelator,
    VisualLeakDetector,
)


class DetectorPipeline:
    """Run every detector over a frame; aggregate the strongest action."""

    def __init__(self, detectors=None):
        self.detectors = [d() for d in (detectors or DEFAULT_DETECTORS)]

    def inspect(self, frame: dict[str, Any]) -> tuple[str, list[DetectorAlert]]:
        alerts: list[DetectorAlert] = []
        for det in self.detectors:
            try:
                alerts.extend(det.inspect(frame))
            except Exception:  # noqa: BLE001 — a detector must never kill the proxy
                continue
        action = "allow"
        if any(a.action == "block" for a in alerts):
            action = "block"
        elif any(a.action == "warn" for a in alerts):
            action = "warn"
        return action, alerts

    def metrics(self) -> dict[str, int]:
        return {d.name: d.hits for d in self.detectors}


def _frame_text(frame: dict[str, Any]) -> str:
    parts = [str(frame.get("params", "")), str(frame.get("result", ""))]
    return " ".join(parts)
