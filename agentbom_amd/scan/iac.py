"""IaC misconfiguration scanning: Terraform / Kubernetes / Dockerfile / compose.

Reference: src/agent_bom/iac/ (terraform_security.py, kubernetes.py,
dockerfile checks, attack_mapping.py) — rule-based misconfiguration
detection with ATT&CK mapping, surfaced as MISCONFIGURATION findings.
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass
from pathlib import Path
from typing import Any, Callable, Optional

from agentbom_amd.models.finding import Asset, Finding, FindingSource, FindingType, stable_id


@dataclass
class IacFinding:
    rule_id: str
    title: str
    severity: str
    file: str
    line: int
    snippet: str
    attack_technique: Optional[str] = None

    def to_dict(self) -> dict[str, Any]:
        return {"rule_id": self.rule_id, "title": self.title, "severity": self.severity,
                "file": self.file, "line": self.line, "snippet": self.snippet,
                "attack_technique": self.attack_technique}


def _line_of(text: str, pos: int) -> int:
    return text.count("\n", 0, pos) + 1


def _grep_rules(text: str, path: str, rules) -> list[IacFinding]:
    out = []
    for rule_id, pattern, title, severity, technique in rules:
        for m in pattern.finditer(text):
            line = _line_of(text, m.start())
            snippet = text.splitlines()[line - 1].strip()[:160]
            out.append(IacFinding(rule_id=rule_id, title=title, severity=severity,
                                  file=path, line=line, snippet=snippet,
                                  attack_technique=technique))
    return out


_TF_RULES = [
    ("TF001", re.compile(r'cidr_blocks\s*=\s*\[\s*"0\.0\.0\.0/0"'),
     "security group open to the internet", "high", "T1190"),
    ("TF002", re.compile(r'acl\s*=\s*"public-read'),
     "S3 bucket with public-read ACL", "high", "T1530"),
    ("TF003", re.compile(r"publicly_accessible\s*=\s*true"),
     "database instance publicly accessible", "critical", "T1190"),
    ("TF004", re.compile(r"(?i)(password|secret|access_key)\s*=\s*\"[^$\"{][^\"]{7,}\""),
     "hardcoded credential in Terraform", "critical", "T1552"),
    ("TF005", re.compile(r'encrypted\s*=\s*false'),
     "encryption disabled on storage resource", "medium", "T1530"),
    ("TF006", re.compile(r'"iam:\*"|Action\s*=\s*"\*"'),
     "wildcard IAM action grant", "high", "T1078"),
    ("TF007", re.compile(r"enable_logging\s*=\s*false|logging\s*{\s*}"),
     "audit logging disabled", "medium", "T1562"),
]

_K8S_RULES = [
    ("K8S001", re.compile(r"privileged:\s*true"),
     "privileged container", "critical", "T1611"),
    ("K8S002", re.compile(r"runAsUser:\s*0\b"),
     "container runs as root (runAsUser 0)", "high", "T1611"),
    ("K8S003", re.compile(r"allowPrivilegeEscalation:\s*true"),
     "privilege escalation allowed", "high", "T1611"),
    ("K8S004", re.compile(r"hostNetwork:\s*true"),
     "pod attached to host network", "high", "T1610"),
    ("K8S005", re.compile(r"hostPath:"),
     "hostPath volume mount", "medium", "T1611"),
    ("K8S006", re.compile(r"readOnlyRootFilesystem:\s*false"),
     "writable root filesystem", "low", "T1105"),
    ("K8S007", re.compile(r"(?i)kind:\s*ClusterRoleBinding[\s\S]{0,400}name:\s*cluster-admin"),
     "cluster-admin role binding", "critical", "T1078"),
    ("K8S008", re.compile(r"imagePullPolicy:\s*Never"),
     "imagePullPolicy Never (stale/unscanned image)", "low", "T1525"),
]

_DOCKER_RULES = [
    ("DKR001", re.compile(r"(?m)^USER\s+root\s*$"),
     "container runs as root user", "medium", "T1611"),
    ("DKR002", re.compile(r"(?m)^\s*(ENV|ARG)\s+\w*(PASSWORD|SECRET|TOKEN|API_KEY)\w*\s*=?\s*\S+"),
     "secret in Dockerfile ENV/ARG", "high", "T1552"),
    ("DKR003", re.compile(r"(?m)curl[^\n]*\|\s*(bash|sh)\b"),
     "curl-pipe-to-shell install", "high", "T1059"),
    ("DKR004", re.compile(r"(?m)^FROM\s+[^\s:@]+\s*$"),
     "unpinned base image (no tag or digest)", "medium", "T1525"),
    ("DKR005", re.compile(r"--no-check-certificate|--insecure\b|-k\s"),
     "TLS verification disabled in build", "medium", "T1557"),
]

_COMPOSE_RULES = [
    ("CMP001", re.compile(r"privileged:\s*true"),
     "privileged compose service", "critical", "T1611"),
    ("CMP002", re.compile(r"/var/run/docker\.sock"),
     "docker socket mounted into container", "critical", "T1611"),
    ("CMP003", re.compile(r"network_mode:\s*[\"']?host"),
     "compose service on host network", "high", "T1610"),
]

# GitHub Actions workflow supply-chain rules (reference: github_actions.py)
_GHA_RULES = [
    ("GHA001", re.compile(r"uses:\s*\S+@(?:main|master|v?\d+(?:\.\d+)?)\s*$",
                          re.M),
     "action pinned to a mutable ref (tag/branch) instead of a commit SHA",
     "medium", "T1195.002"),
    ("GHA002", re.compile(r"(?i)on:\s*\n?\s*-?\s*pull_request_target"),
     "pull_request_target grants secrets to fork PRs (pwn-request surface)",
     "high", "T1195"),
    ("GHA003", re.compile(r"\$\{\{\s*github\.event\.(?:issue|comment|"
                          r"pull_request)\.(?:title|body)[^}]*\}\}"),
     "untrusted event text interpolated into a run step (script injection)",
     "critical", "T1059"),
    ("GHA004", re.compile(r"(?i)permissions:\s*write-all"),
     "workflow requests write-all token permissions", "medium", "T1528"),
    ("GHA005", re.compile(r"(?i)(?:curl|wget)[^|\n]*\|\s*(?:bash|sh)\b"),
     "remote script piped to shell inside a workflow", "critical", "T1059"),
    ("GHA006", re.compile(r"(?i)ACTIONS_ALLOW_UNSECURE_COMMANDS\s*:\s*true"),
     "deprecated unsecure workflow commands enabled", "high", "T1059"),
]


def _is_workflow(p: Path) -> bool:
    return (p.suffix in (".yaml", ".yml")
            and "workflows" in p.parts
            and ".github" in p.parts)


_FILE_RULES: list[tuple[Callable[[Path], bool], list]] = [
    (lambda p: p.suffix == ".tf", _TF_RULES),
    (_is_workflow, _GHA_RULES),
    (lambda p: p.suffix in (".yaml", ".yml") and p.name.startswith(("docker-compose", "compose")),
     _COMPOSE_RULES),
    (lambda p: p.suffix in (".yaml", ".yml"), _K8S_RULES),
    (lambda p: p.name == "Dockerfile" or p.name.startswith("Dockerfile."), _DOCKER_RULES),
]

_SKIP_DIRS = {".git", "node_modules", ".terraform", "__pycache__"}


def scan_iac_text(text: str, path: str) -> list[IacFinding]:
    p = Path(path)
    out: list[IacFinding] = []
    for matcher, rules in _FILE_RULES:
        if matcher(p):
            out.extend(_grep_rules(text, path, rules))
            break
    return out


def scan_iac_tree(root: str | Path, max_depth: int = 8) -> list[IacFinding]:
    root = Path(root)
    files = [root] if root.is_file() else [
        p for p in root.rglob("*")
        if p.is_file()
        and len(p.relative_to(root).parts) <= max_depth
        and not (_SKIP_DIRS & set(p.relative_to(root).parts[:-1]))
        and (p.suffix in (".tf", ".yaml", ".yml") or p.name.startswith("Dockerfile"))
    ]
    out: list[IacFinding] = []
    for f in sorted(files):
        try:
            out.extend(scan_iac_text(f.read_text(errors="replace"), str(f)))
        except OSError:
            continue
    return out


def iac_finding_to_finding(f: IacFinding) -> Finding:
    return Finding(
        finding_type=FindingType.CIS_FAIL,
        source=FindingSource.SAST,
        asset=Asset(name=Path(f.file).name, asset_type="config_file", location=f.file),
        severity=f.severity,
        title=f"IaC misconfiguration {f.rule_id}: {f.title}",
        description=f"{f.title} at {f.file}:{f.line} — {f.snippet}",
        attack_tags=[f.attack_technique] if f.attack_technique else [],
        evidence=f.to_dict(),
        is_actionable=f.severity in ("critical", "high"),
        id=stable_id("iac", f.rule_id, f.file, str(f.line)),
    )
