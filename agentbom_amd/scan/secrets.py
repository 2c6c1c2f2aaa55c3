"""Hardcoded secret / credential scanning over files.

Reference: src/agent_bom/secret_scanner.py — pattern-based secret detection
with entropy screening and redacted previews (values never leave the
scanner); findings surface in the unified stream as CREDENTIAL_EXPOSURE.
"""

from __future__ import annotations

import math
import re
from dataclasses import dataclass
from pathlib import Path
from typing import Iterable, Optional

from agentbom_amd.models.finding import Asset, Finding, FindingSource, FindingType, stable_id

# (name, pattern, severity); value group 0 is redacted, never stored.
SECRET_PATTERNS: list[tuple[str, re.Pattern, str]] = [
    ("aws_access_key_id", re.compile(r"\bAKIA[0-9A-Z]{16}\b"), "high"),
    ("aws_secret_access_key",
     re.compile(r"(?i)aws_secret_access_key\s*[=:]\s*['\"]?([A-Za-z0-9/+=]{40})"), "critical"),
    ("github_pat", re.compile(r"\bghp_[A-Za-z0-9]{36}\b"), "critical"),
    ("github_fine_grained", re.compile(r"\bgithub_pat_[A-Za-z0-9_]{22,}\b"), "critical"),
    ("gitlab_pat", re.compile(r"\bglpat-[A-Za-z0-9\-_]{20,}\b"), "critical"),
    ("openai_api_key", re.compile(r"\bsk-(?:proj-)?[A-Za-z0-9_\-]{20,}\b"), "critical"),
    ("anthropic_api_key", re.compile(r"\bsk-ant-[A-Za-z0-9\-_]{20,}\b"), "critical"),
    ("slack_token", re.compile(r"\bxox[bpras]-[A-Za-z0-9\-]{10,}\b"), "high"),
    ("stripe_key", re.compile(r"\b[sr]k_live_[A-Za-z0-9]{20,}\b"), "critical"),
    ("google_api_key", re.compile(r"\bAIza[A-Za-z0-9_\-]{35}\b"), "high"),
    ("private_key_block",
     re.compile(r"-----BEGIN (?:RSA |EC |DSA |OPENSSH |PGP )?PRIVATE KEY(?: BLOCK)?-----"),
     "critical"),
    ("jwt", re.compile(r"\beyJ[A-Za-z0-9_-]{10,}\.eyJ[A-Za-z0-9_-]{10,}\.[A-Za-z0-9_-]{10,}\b"),
     "medium"),
    ("generic_password",
     re.compile(r"(?i)\b(password|passwd|secret|api_key|apikey|token)\b\s*[=:]\s*['\"]([^'\"\s]{8,})['\"]"),
     "medium"),
    ("connection_string",
     re.compile(r"(?i)\b(?:postgres|postgresql|mysql|mongodb(?:\+srv)?|redis|amqp)://[^:\s]+:([^@\s]+)@"),
     "high"),
]

_SKIP_DIRS = {".git", "node_modules", ".venv", "venv", "__pycache__", "dist", "build"}
_TEXT_SUFFIXES = {".py", ".js", ".ts", ".json", ".yaml", ".yml", ".env", ".toml", ".ini",
                  ".cfg", ".sh", ".txt", ".md", ".tf", ".properties", ".xml", ".conf", ""}


def shannon_entropy(s: str) -> float:
    if not s:
        return 0.0
    freq: dict[str, int] = {}
    for ch in s:
        freq[ch] = freq.get(ch, 0) + 1
    return -sum((c / len(s)) * math.log2(c / len(s)) for c in freq.values())


def _redact(value: str) -> str:
    if len(value) <= 8:
        return "***REDACTED***"
    return f"{value[:4]}…{value[-2:]} ({len(value)} chars)"


@dataclass
class SecretHit:
    kind: str
    severity: str
    file: str
    line: int
    preview: str  # redacted, never the raw value
    entropy: float

    def to_dict(self) -> dict:
        return {"kind": self.kind, "severity": self.severity, "file": self.file,
                "line": self.line, "preview": self.preview,
                "entropy": round(self.entropy, 2)}


def scan_text(text: str, path: str = "<memory>") -> list[SecretHit]:
    hits: list[SecretHit] = []
    for lineno, line in enumerate(text.splitlines(), start=1):
        if len(line) > 2000 or "REDACTED" in line:
            continue
        for kind, pattern, severity in SECRET_PATTERNS:
            m = pattern.search(line)
            if not m:
                continue
            value = m.group(m.lastindex or 0)
            ent = shannon_entropy(value)
            # low-entropy generic matches are placeholders, not secrets
            if kind in ("generic_password",) and (ent < 3.0 or value.lower() in
                                                  ("password", "changeme", "example")):
                continue
            hits.append(SecretHit(kind=kind, severity=severity, file=path,
                                  line=lineno, preview=_redact(value), entropy=ent))
    return hits


def scan_paths(root: str | Path, max_file_bytes: int = 1_000_000,
               max_depth: int = 8) -> list[SecretHit]:
    root = Path(root)
    files: Iterable[Path]
    if root.is_file():
        files = [root]
    else:
        files = (
            p for p in root.rglob("*")
            if p.is_file()
            and len(p.relative_to(root).parts) <= max_depth
            and not (_SKIP_DIRS & set(p.relative_to(root).parts[:-1]))
            and p.suffix.lower() in _TEXT_SUFFIXES
        )
    hits: list[SecretHit] = []
    for f in files:
        try:
            if f.stat().st_size > max_file_bytes:
                continue
            hits.extend(scan_text(f.read_text(errors="replace"), str(f)))
        except OSError:
            continue
    return hits


def secret_hit_to_finding(hit: SecretHit) -> Finding:
    return Finding(
        finding_type=FindingType.CREDENTIAL_EXPOSURE,
        source=FindingSource.SECRET_SCAN,
        asset=Asset(name=hit.file, asset_type="source_file", location=hit.file),
        severity=hit.severity,
        title=f"Hardcoded secret ({hit.kind}) in {hit.file}",
        description=f"{hit.kind} at {hit.file}:{hit.line} — value redacted",
        evidence={"kind": hit.kind, "line": hit.line, "preview": hit.preview,
                  "entropy": hit.entropy},
        is_actionable=True,
        id=stable_id("secret", hit.file, str(hit.line), hit.kind, ""),
    )
