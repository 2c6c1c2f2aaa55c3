"""Project-level configuration: ``.agent-bom.yaml`` + ``.agent-bom-ignore``.

Reference parity: src/agent_bom/project_config.py (per-repo scan settings)
and src/agent_bom/ignore.py (suppression file with reasons + expiry).

``.agent-bom.yaml`` (any subset):

    offline: true
    include_unfixed: false
    fail_on_severity: high
    fail_on_kev: true
    blast_radius_depth: 2
    exit_zero: false
    policy: .agent-bom-policy.json
    ignore:
      - CVE-2020-14343

``.agent-bom-ignore`` — one rule per line:

    CVE-2023-4863                       # bare id
    CVE-2021-23337 until=2026-12-31 reason=accepted risk, sandboxed
    GHSA-xxxx-yyyy-zzzz reason=not exploitable here

Expired rules are NOT applied (fail-open on expiry: the finding comes
back) and are reported so the report can warn about them.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from datetime import date
from pathlib import Path
from typing import Any, Optional

CONFIG_FILENAME = ".agent-bom.yaml"
IGNORE_FILENAME = ".agent-bom-ignore"

_KNOWN_KEYS = {
    "offline", "include_unfixed", "fail_on_severity", "fail_on_kev",
    "blast_radius_depth", "exit_zero", "policy", "ignore", "use_gpu",
    "profiles",
}

_RULE_RE = re.compile(r"^(?P<id>\S+)(?P<rest>.*)$")


@dataclass
class IgnoreRule:
    vuln_id: str
    reason: str = ""
    until: Optional[str] = None  # ISO date; past date = expired

    @property
    def expired(self) -> bool:
        if not self.until:
            return False
        try:
            return date.fromisoformat(self.until) < date.today()
        except ValueError:
            return True  # unparseable expiry: treat as expired (fail open)

    def to_dict(self) -> dict[str, Any]:
        return {"vuln_id": self.vuln_id, "reason": self.reason,
                "until": self.until, "expired": self.expired}


@dataclass
class ProjectConfig:
    settings: dict[str, Any] = field(default_factory=dict)
    ignore_rules: list[IgnoreRule] = field(default_factory=list)
    warnings: list[str] = field(default_factory=list)
    config_path: Optional[str] = None
    ignore_path: Optional[str] = None

    @property
    def active_ignore_ids(self) -> frozenset:
        return frozenset(r.vuln_id for r in self.ignore_rules if not r.expired)

    @property
    def expired_rules(self) -> list[IgnoreRule]:
        return [r for r in self.ignore_rules if r.expired]


def parse_ignore_line(line: str) -> Optional[IgnoreRule]:
    line = line.split("#", 1)[0].strip()
    if not line:
        return None
    m = _RULE_RE.match(line)
    if not m:
        return None
    vuln_id = m.group("id")
    rest = m.group("rest").strip()
    until = None
    reason = ""
    um = re.search(r"until=(\S+)", rest)
    if um:
        until = um.group(1)
        rest = rest.replace(um.group(0), "").strip()
    rm = re.search(r"reason=(.*)$", rest)
    if rm:
        reason = rm.group(1).strip()
    return IgnoreRule(vuln_id=vuln_id, reason=reason, until=until)


def load_project_config(root: str | Path = ".") -> ProjectConfig:
    """Load both files from ``root``; missing files are simply absent."""
    root = Path(root)
    cfg = ProjectConfig()

    yaml_path = root / CONFIG_FILENAME
    if yaml_path.exists():
        import yaml

        cfg.config_path = str(yaml_path)
        try:
            data = yaml.safe_load(yaml_path.read_text()) or {}
        except yaml.YAMLError as exc:
            cfg.warnings.append(f"{CONFIG_FILENAME}: unparseable ({exc})")
            data = {}
        if not isinstance(data, dict):
            cfg.warnings.append(f"{CONFIG_FILENAME}: not a mapping; ignored")
            data = {}
        for key, value in data.items():
            if key not in _KNOWN_KEYS:
                cfg.warnings.append(f"{CONFIG_FILENAME}: unknown key {key!r}")
                continue
            cfg.settings[key] = value
        for vid in cfg.settings.pop("ignore", []) or []:
            cfg.ignore_rules.append(IgnoreRule(vuln_id=str(vid),
                                               reason="project config"))

    ignore_path = root / IGNORE_FILENAME
    if ignore_path.exists():
        cfg.ignore_path = str(ignore_path)
        for line in ignore_path.read_text().splitlines():
            rule = parse_ignore_line(line)
            if rule:
                cfg.ignore_rules.append(rule)

    for rule in cfg.expired_rules:
        cfg.warnings.append(
            f"ignore rule for {rule.vuln_id} expired ({rule.until}) — "
            "finding will be reported again")
    return cfg


def apply_to_scan_options(cfg: ProjectConfig, options) -> None:
    """Overlay file settings onto a ScanOptions (CLI flags already set win
    only where the file doesn't specify — the file is the project default)."""
    s = cfg.settings
    if "offline" in s:
        options.offline = bool(s["offline"])
    if "include_unfixed" in s:
        options.include_unfixed = bool(s["include_unfixed"])
    if "fail_on_severity" in s:
        options.fail_on_severity = str(s["fail_on_severity"])
    if "fail_on_kev" in s:
        options.fail_on_kev = bool(s["fail_on_kev"])
    if "blast_radius_depth" in s:
        options.blast_radius_depth = int(s["blast_radius_depth"])
    if "exit_zero" in s:
        options.exit_zero = bool(s["exit_zero"])
    if "use_gpu" in s:
        options.use_gpu = bool(s["use_gpu"])
    options.ignore_ids = frozenset(options.ignore_ids) | cfg.active_ignore_ids


def get_profile(cfg: ProjectConfig, name: str) -> dict[str, Any]:
    """Named CLI flag-set from the project file (reference: cli/_profiles.py).

    .agent-bom.yaml::

        profiles:
          ci:       {offline: true, fail_on_severity: high, format: sarif}
          deep:     {blast_radius_depth: 3, include_unfixed: true}

    Unknown profile -> ValueError listing what exists; non-mapping entries
    are rejected the same way (never silently empty)."""
    profiles = cfg.settings.get("profiles") or {}
    if not isinstance(profiles, dict):
        raise ValueError(".agent-bom.yaml: profiles must be a mapping")
    if name not in profiles:
        raise ValueError(
            f"unknown profile {name!r}; defined: {sorted(profiles) or '(none)'}")
    body = profiles[name]
    if not isinstance(body, dict):
        raise ValueError(f"profile {name!r} must be a mapping of flags")
    return dict(body)
