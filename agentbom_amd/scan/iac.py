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
    ("TF008", re.compile(r'block_public_acls\s*=\s*false|block_public_policy\s*=\s*false'),
     "S3 public access block disabled", "high", "T1530"),
    ("TF009", re.compile(r'skip_final_snapshot\s*=\s*true'),
     "database deleted without a final snapshot", "low", "T1485"),
    ("TF010", re.compile(r'(?m)^\s*versioning\s*{\s*enabled\s*=\s*false'),
     "bucket versioning disabled (ransom/delete resilience)", "medium", "T1485"),
    ("TF011", re.compile(r'force_ssl\s*=\s*false|require_secure_transport\s*=\s*"?OFF'),
     "TLS not enforced on data store", "high", "T1557"),
    ("TF012", re.compile(r'associate_public_ip_address\s*=\s*true'),
     "compute instance with public IP", "medium", "T1190"),
    ("TF013", re.compile(r'(?i)kms_key_id\s*=\s*(""|null)'),
     "customer-managed KMS key absent", "low", "T1530"),
    ("TF014", re.compile(r'min_tls_version\s*=\s*"?(TLS1_0|TLS1_1|1\.0|1\.1)'),
     "legacy TLS version allowed", "medium", "T1557"),
    ("TF015", re.compile(r'(?m)^\s*ingress\s*{[^}]*from_port\s*=\s*(22|3389)\b[\s\S]{0,200}?0\.0\.0\.0/0'),
     "SSH/RDP admin port open to the internet", "critical", "T1021"),
    ("TF016", re.compile(r'(?i)authorized_networks\s*{\s*[^}]*0\.0\.0\.0/0'),
     "cloud SQL authorized network open to the internet", "critical", "T1190"),
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
    ("K8S009", re.compile(r"hostPID:\s*true|hostIPC:\s*true"),
     "pod shares host PID/IPC namespace", "high", "T1611"),
    ("K8S010", re.compile(r"(?m)^\s*capabilities:\s*\n\s*add:[\s\S]{0,80}?(SYS_ADMIN|NET_ADMIN|ALL)\b"),
     "dangerous Linux capability added", "high", "T1611"),
    ("K8S011", re.compile(r"automountServiceAccountToken:\s*true"),
     "service-account token automounted", "medium", "T1528"),
    ("K8S012", re.compile(r"(?i)(?:env|value):[^\n]*(?:PASSWORD|SECRET|TOKEN|API_KEY)[^\n]*value:\s*\S+"),
     "secret literal in pod env", "high", "T1552"),
    ("K8S013", re.compile(r"(?m)^\s*securityContext:\s*{\s*}\s*$"),
     "empty securityContext (no hardening)", "low", "T1611"),
    ("K8S014", re.compile(r"type:\s*NodePort"),
     "NodePort service exposes every node", "medium", "T1190"),
    ("K8S015", re.compile(r"(?i)seccompProfile:\s*\n\s*type:\s*Unconfined"),
     "seccomp unconfined", "high", "T1611"),
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
    ("DKR006", re.compile(r"(?m)^\s*ADD\s+https?://"),
     "remote ADD without checksum verification", "medium", "T1195"),
    ("DKR007", re.compile(r"(?m)^\s*EXPOSE\s+22\b"),
     "SSH port exposed from a container", "medium", "T1021"),
    ("DKR008", re.compile(r"(?m)apt-get[^\n]*install(?![^\n]*--no-install-recommends)"),
     "apt install without --no-install-recommends (bloat/attack surface)",
     "low", "T1105"),
    ("DKR009", re.compile(r"(?m)^\s*COPY\s+(\.|\./|/)\s"),
     "COPY of the whole build context (secrets risk)", "low", "T1552"),
    ("DKR010", re.compile(r"(?m)chmod\s+(-R\s+)?0?777"),
     "world-writable permissions set in image", "medium", "T1222"),
]

_COMPOSE_RULES = [
    ("CMP001", re.compile(r"privileged:\s*true"),
     "privileged compose service", "critical", "T1611"),
    ("CMP002", re.compile(r"/var/run/docker\.sock"),
     "docker socket mounted into container", "critical", "T1611"),
    ("CMP003", re.compile(r"network_mode:\s*[\"']?host"),
     "compose service on host network", "high", "T1610"),
    ("CMP004", re.compile(r"(?m)^\s*-?\s*[\"']?\d+:\d+[\"']?\s*$[\s\S]{0,4}"),
     "service port published to all interfaces", "low", "T1190"),
    ("CMP005", re.compile(r"(?i)^\s*(?:-\s*)?\w*(PASSWORD|SECRET|TOKEN|API_KEY)\w*[=:]\s*\S+", re.M),
     "secret literal in compose environment", "high", "T1552"),
    ("CMP006", re.compile(r"cap_add:[\s\S]{0,60}?(SYS_ADMIN|NET_ADMIN|ALL)\b"),
     "dangerous capability added to compose service", "high", "T1611"),
    ("CMP007", re.compile(r"pid:\s*[\"']?host|ipc:\s*[\"']?host"),
     "compose service shares host PID/IPC", "high", "T1611"),
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
