"""Security sanitizers: path validation, secret redaction, untrusted fencing.

The one place these policies live (VERDICT r1 partial #6 — reference
src/agent_bom/security.py, 1,035 LoC, plus context_graph.py:47
_untrusted_metadata_text).  Everything that logs, persists or renders
attacker-influenced content routes through here:

- ``validate_path``       directory-traversal / symlink-escape guard for
                          user-supplied file paths
- ``redact_secrets``      canonical credential redaction for logs/audit
- ``sanitize_env``        env maps with secret-named keys masked
- ``fence_untrusted``     untrusted MCP/tool metadata made safe to embed
                          in reports and LLM-bound output (control chars,
                          ANSI, role-tag markers, length bound)
"""

from __future__ import annotations

import re
from pathlib import Path
from typing import Any, Mapping, Optional

# ── path validation ─────────────────────────────────────────────────────────


class PathValidationError(ValueError):
    """User-supplied path escapes the allowed base."""


def validate_path(path: str, base: str, must_exist: bool = False) -> Path:
    """Resolve ``path`` and require it stays under ``base``.

    Symlinks are resolved BEFORE the containment check, so a link pointing
    outside the base is rejected, not followed."""
    basep = Path(base).resolve()
    cand = Path(path)
    if not cand.is_absolute():
        cand = basep / cand
    resolved = cand.resolve()
    try:
        resolved.relative_to(basep)
    except ValueError:
        raise PathValidationError(
            f"path {path!r} escapes the allowed base {base!r}") from None
    if must_exist and not resolved.exists():
        raise PathValidationError(f"path {path!r} does not exist")
    return resolved


# ── secret redaction ────────────────────────────────────────────────────────

_SECRET_PATTERNS = [
    (re.compile(r"\b(AKIA|ASIA)[0-9A-Z]{16}\b"), "[AWS_KEY]"),
    (re.compile(r"\bgh[pousr]_[A-Za-z0-9]{36,}\b"), "[GITHUB_TOKEN]"),
    (re.compile(r"\bsk-[A-Za-z0-9\-_]{20,}\b"), "[API_KEY]"),
    (re.compile(r"\bxox[baprs]-[A-Za-z0-9\-]{10,}\b"), "[SLACK_TOKEN]"),
    (re.compile(r"\beyJ[A-Za-z0-9_\-]{10,}\.[A-Za-z0-9_\-]{5,}\.[A-Za-z0-9_\-]{5,}\b"),
     "[JWT]"),
    (re.compile(r"-----BEGIN [A-Z ]*PRIVATE KEY-----[\s\S]*?-----END [A-Z ]*PRIVATE KEY-----"),
     "[PRIVATE_KEY]"),
    (re.compile(r"(?i)\b(password|passwd|secret|token|api[_-]?key|authorization)"
                r"(\"?\s*[:=]\s*)(\"[^\"]{4,}\"|'[^']{4,}'|[^\s\"',;&]{4,})"),
     r"\1\2[REDACTED]"),
    (re.compile(r"://[^/@\s:]+:[^@\s/]+@"), "://***:***@"),  # creds in URLs
]

_SECRET_ENV_KEY = re.compile(
    r"(?i)(secret|token|password|passwd|credential|api[_-]?key|private[_-]?key"
    r"|access[_-]?key|auth)")


def redact_secrets(text: Optional[str]) -> Optional[str]:
    """Canonical credential redaction for anything headed to logs/audit."""
    if not text:
        return text
    for pattern, repl in _SECRET_PATTERNS:
        text = pattern.sub(repl, text)
    return text


def sanitize_env(env: Mapping[str, Any]) -> dict[str, str]:
    """Env map with secret-named keys masked and values redacted."""
    out = {}
    for k, v in env.items():
        if _SECRET_ENV_KEY.search(str(k)):
            out[str(k)] = "***"
        else:
            out[str(k)] = redact_secrets(str(v)) or ""
    return out


# ── untrusted-metadata fencing ──────────────────────────────────────────────

_ANSI = re.compile(r"\x1b\[[0-9;]*[A-Za-z]")
_CONTROL = re.compile(r"[\x00-\x08\x0b\x0c\x0e-\x1f\x7f]")
_ROLE_TAGS = re.compile(r"</?\s*(system|assistant|user|tool)\s*>", re.IGNORECASE)


def fence_untrusted(text: Any, max_len: int = 500) -> str:
    """Make attacker-influenced metadata (tool descriptions, server names,
    advisory summaries) safe to embed in reports and LLM-bound output:
    strip ANSI/control characters, neutralize chat role tags, redact
    embedded secrets, and bound length (reference context_graph.py:47)."""
    s = str(text or "")
    s = _ANSI.sub("", s)
    s = _CONTROL.sub(" ", s)
    s = _ROLE_TAGS.sub(lambda m: "[" + m.group(0).strip("<>/ ").lower() + "]", s)
    s = redact_secrets(s) or ""
    if len(s) > max_len:
        s = s[: max_len - 1] + "…"
    return s


def redact_structure(obj: Any) -> Any:
    """Recursive redaction over dict/list structures: secret-named keys are
    masked outright, every string value runs through redact_secrets."""
    if isinstance(obj, str):
        return redact_secrets(obj)
    if isinstance(obj, Mapping):
        return {
            str(k): ("***" if _SECRET_ENV_KEY.search(str(k)) and
                     isinstance(v, (str, int, float)) else redact_structure(v))
            for k, v in obj.items()
        }
    if isinstance(obj, (list, tuple)):
        return [redact_structure(v) for v in obj]
    return obj
