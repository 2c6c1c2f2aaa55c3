"""Agent skill-bundle scanning (Claude skills / SKILL.md packages).

Reference parity: src/agent_bom/{skill_bundles,skills_catalog,
skills_policy,skill_intel}.py — skills are instruction bundles an agent
loads verbatim, which makes them a prompt-injection and capability-grant
surface on par with MCP servers.  Scanned signals per bundle:

- **frontmatter contract** — name/description present; ``allowed-tools``
  grants (wildcard = finding);
- **instruction body** — prompt-injection patterns (same detector set the
  runtime proxy uses), hidden-unicode smuggling, credential material;
- **bundled executables** — scripts shipped inside the bundle run with
  the agent's permissions: risky commands are flagged;
- **policy** — an operator allowlist/blocklist evaluated per bundle.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Optional

_FRONTMATTER_RE = re.compile(r"\A---\s*\n(.*?)\n---\s*\n", re.S)
_HIDDEN_UNICODE_RE = re.compile(r"[​-‏‪-‮⁦-⁩]")
_RISKY_SCRIPT_RE = re.compile(
    r"(?i)(?:curl|wget)[^|\n]*\|\s*(?:bash|sh|python)|rm\s+-rf\s+[/~]|"
    r"chmod\s+\+x\s+/|nc\s+-e|base64\s+-d[^|\n]*\|\s*(?:bash|sh)")


@dataclass
class SkillFinding:
    skill: str
    file: str
    rule: str
    severity: str
    detail: str = ""

    def to_dict(self) -> dict[str, Any]:
        return {"skill": self.skill, "file": self.file, "rule": self.rule,
                "severity": self.severity, "detail": self.detail}


@dataclass
class SkillBundle:
    name: str
    path: str
    description: str = ""
    allowed_tools: list[str] = field(default_factory=list)
    findings: list[SkillFinding] = field(default_factory=list)
    files: int = 0

    @property
    def risk_level(self) -> str:
        sev = {f.severity for f in self.findings}
        if "critical" in sev:
            return "critical"
        if "high" in sev:
            return "high"
        if "medium" in sev:
            return "medium"
        return "low"

    def to_dict(self) -> dict[str, Any]:
        return {"name": self.name, "path": self.path,
                "description": self.description,
                "allowed_tools": self.allowed_tools,
                "files": self.files, "risk_level": self.risk_level,
                "findings": [f.to_dict() for f in self.findings]}


def _parse_frontmatter(text: str) -> tuple[dict[str, Any], str]:
    m = _FRONTMATTER_RE.match(text)
    if not m:
        return {}, text
    import yaml

    try:
        data = yaml.safe_load(m.group(1))
    except yaml.YAMLError:
        data = None
    return (data if isinstance(data, dict) else {}), text[m.end():]


def scan_skill_bundle(skill_dir: str | Path) -> Optional[SkillBundle]:
    """Scan one skill directory (SKILL.md + any bundled files)."""
    from agentbom_amd.runtime.detectors import _INJECTION_PATTERNS
    from agentbom_amd.scan.secrets import scan_text

    skill_dir = Path(skill_dir)
    skill_md = skill_dir / "SKILL.md"
    if not skill_md.exists():
        return None
    try:
        text = skill_md.read_text(errors="replace")
    except OSError:
        return None

    meta, body = _parse_frontmatter(text)
    raw_tools = meta.get("allowed-tools") or meta.get("allowed_tools") or []
    if isinstance(raw_tools, str):
        raw_tools = [t.strip() for t in raw_tools.split(",") if t.strip()]
    bundle = SkillBundle(
        name=str(meta.get("name") or skill_dir.name),
        path=str(skill_dir),
        description=str(meta.get("description") or ""),
        allowed_tools=[str(t) for t in raw_tools])

    def add(file: Path, rule: str, severity: str, detail: str = "") -> None:
        bundle.findings.append(SkillFinding(
            skill=bundle.name, file=str(file), rule=rule,
            severity=severity, detail=detail))

    if not meta:
        add(skill_md, "skill-no-frontmatter", "low",
            "SKILL.md has no frontmatter contract")
    if any(t.strip() == "*" for t in bundle.allowed_tools):
        add(skill_md, "skill-wildcard-tools", "high",
            "allowed-tools grants '*' — the skill can invoke anything")
    for t in bundle.allowed_tools:
        if re.search(r"(?i)\b(bash|shell|exec|terminal)\b", t):
            add(skill_md, "skill-shell-grant", "medium",
                f"allowed-tools grants shell-class tool {t!r}")
            break

    # instruction-surface checks over SKILL.md + referenced markdown
    docs = [skill_md] + sorted(p for p in skill_dir.rglob("*.md")
                               if p != skill_md)[:50]
    for doc in docs:
        try:
            doc_text = doc.read_text(errors="replace")
        except OSError:
            continue
        bundle.files += 1
        for pat in _INJECTION_PATTERNS:
            m = pat.search(doc_text)
            if m:
                add(doc, "skill-prompt-injection", "critical",
                    m.group(0)[:80])
                break
        if _HIDDEN_UNICODE_RE.search(doc_text):
            add(doc, "skill-hidden-unicode", "high",
                "zero-width/bidi characters hide instructions from review")
        for hit in scan_text(doc_text, str(doc)):
            add(doc, "skill-embedded-secret", "critical", hit.kind)

    # bundled executables
    for script in sorted(skill_dir.rglob("*"))[:200]:
        if script.suffix not in (".sh", ".py", ".js", ".bash"):
            continue
        try:
            s_text = script.read_text(errors="replace")
        except OSError:
            continue
        bundle.files += 1
        m = _RISKY_SCRIPT_RE.search(s_text)
        if m:
            add(script, "skill-risky-script", "critical", m.group(0)[:80])
    return bundle


def scan_skills_tree(root: str | Path, cap: int = 200) -> list[SkillBundle]:
    """Scan every skill bundle under ``root`` (dirs containing SKILL.md)."""
    root = Path(root)
    out = []
    for skill_md in sorted(root.rglob("SKILL.md"))[:cap]:
        bundle = scan_skill_bundle(skill_md.parent)
        if bundle:
            out.append(bundle)
    return out


def evaluate_skills_policy(bundles: list[SkillBundle],
                           policy: dict[str, Any]) -> dict[str, Any]:
    """Operator policy: {"blocklist": [names], "max_risk_level": "...",
    "require_frontmatter": bool}.  Returns allow/deny per bundle."""
    order = {"low": 0, "medium": 1, "high": 2, "critical": 3}
    max_level = order.get(str(policy.get("max_risk_level", "high")), 2)
    blocklist = {str(b).lower() for b in policy.get("blocklist", [])}
    decisions = []
    for b in bundles:
        reasons = []
        if b.name.lower() in blocklist:
            reasons.append("blocklisted by operator policy")
        if order.get(b.risk_level, 0) > max_level:
            reasons.append(f"risk level {b.risk_level} exceeds policy maximum")
        if policy.get("require_frontmatter") and any(
                f.rule == "skill-no-frontmatter" for f in b.findings):
            reasons.append("frontmatter contract required by policy")
        decisions.append({"skill": b.name, "allowed": not reasons,
                          "reasons": reasons})
    return {"decisions": decisions,
            "denied": sum(1 for d in decisions if not d["allowed"])}
