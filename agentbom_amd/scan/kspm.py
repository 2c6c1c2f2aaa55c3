"""KSPM: Kubernetes cluster security posture (runtime objects, read-only).

Reference parity: src/agent_bom/k8s.py (live cluster posture with the
CIS-K8s 5.2.x pod-security checks + RBAC wildcard analysis) and
mcp_tools/kspm.py (the evidence envelope: per-collector
executed / unevaluable / failed state — a DENIED read is 'unevaluable'
and the run PARTIAL, never a clean pass).

This build evaluates a cluster INVENTORY (exported
``kubectl get pods,roles,clusterroles,rolebindings,clusterrolebindings
-o json`` documents, or any injected reader that returns the same
shapes) — the static-manifest checks stay in scan/iac.py; these are the
runtime-object checks.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Optional

_BUILTIN_ROLE_PREFIXES = ("system:", "cluster-admin", "admin", "edit", "view")


@dataclass
class KspmFinding:
    check_id: str
    title: str
    severity: str
    resource: str          # "ns/name" or role name
    detail: str
    compliance: list[str] = field(default_factory=list)

    def to_dict(self) -> dict[str, Any]:
        return self.__dict__.copy()


@dataclass
class CollectorState:
    name: str
    status: str  # executed | unevaluable | failed
    detail: str = ""


@dataclass
class KspmResult:
    findings: list[KspmFinding] = field(default_factory=list)
    collectors: list[CollectorState] = field(default_factory=list)

    @property
    def status(self) -> str:
        """complete only when EVERY collector executed — a denied or
        failed read makes the posture PARTIAL (never a clean pass)."""
        if any(c.status == "failed" for c in self.collectors):
            return "partial"
        if any(c.status == "unevaluable" for c in self.collectors):
            return "partial"
        return "complete"

    def severity_summary(self) -> dict[str, int]:
        out = {"critical": 0, "high": 0, "medium": 0, "low": 0}
        for f in self.findings:
            out[f.severity] = out.get(f.severity, 0) + 1
        return out

    def to_evidence_dict(self) -> dict[str, Any]:
        return {
            "schema_version": "kspm.cluster.posture.v1",
            "status": self.status,
            "collectors": [c.__dict__ for c in self.collectors],
            "finding_count": len(self.findings),
            "severity_summary": self.severity_summary(),
            "findings": [f.to_dict() for f in self.findings],
        }


def _pod_findings(pod: dict) -> list[KspmFinding]:
    out: list[KspmFinding] = []
    meta = pod.get("metadata") or {}
    ns = meta.get("namespace", "default")
    name = meta.get("name", "?")
    res = f"{ns}/{name}"
    spec = pod.get("spec") or {}

    if spec.get("hostNetwork") is True:
        out.append(KspmFinding(
            "KSPM-POD-001", "Pod uses hostNetwork", "high", res,
            "shares the node's network namespace (sniffing/bypass risk)",
            ["CIS-K8s-5.2.4", "NIST-CM-7"]))
    for key, cid in (("hostPID", "KSPM-POD-002"), ("hostIPC", "KSPM-POD-003")):
        if spec.get(key) is True:
            out.append(KspmFinding(
                cid, f"Pod uses {key}", "high", res,
                f"shares the node's {key[4:]} namespace",
                ["CIS-K8s-5.2.2", "NIST-CM-7"]))
    for vol in spec.get("volumes") or []:
        if isinstance(vol, dict) and vol.get("hostPath"):
            path = (vol["hostPath"] or {}).get("path", "?")
            sev = "critical" if path in ("/", "/etc", "/var/run/docker.sock",
                                         "/proc") else "high"
            out.append(KspmFinding(
                "KSPM-POD-004", "hostPath volume mount", sev, res,
                f"mounts host path {path!r} (container isolation break)",
                ["CIS-K8s-5.2.12", "NIST-SC-7"]))

    sa_automount = spec.get("automountServiceAccountToken")
    if spec.get("serviceAccountName", "default") == "default" \
            and sa_automount is not False:
        out.append(KspmFinding(
            "KSPM-POD-008", "default ServiceAccount token automounted",
            "medium", res,
            "runs as the default SA with its token mounted; scope a "
            "dedicated SA or set automountServiceAccountToken: false",
            ["CIS-K8s-5.1.5", "NIST-AC-6"]))

    for c in (spec.get("containers") or []) + (spec.get("initContainers") or []):
        cname = c.get("name", "?")
        ctx = c.get("securityContext") or {}
        if ctx.get("privileged") is True:
            out.append(KspmFinding(
                "KSPM-POD-005", "Privileged container", "critical",
                f"{res}/{cname}", "privileged: true grants full host access",
                ["CIS-K8s-5.2.1", "NIST-AC-6"]))
        if ctx.get("allowPrivilegeEscalation") is True:
            out.append(KspmFinding(
                "KSPM-POD-006", "allowPrivilegeEscalation enabled", "high",
                f"{res}/{cname}", "setuid-based escalation is not blocked",
                ["CIS-K8s-5.2.5", "NIST-AC-6"]))
        run_as = ctx.get("runAsUser")
        if run_as == 0 or (run_as is None
                           and ctx.get("runAsNonRoot") is not True):
            out.append(KspmFinding(
                "KSPM-POD-007", "Container may run as root",
                "medium" if run_as is None else "high", f"{res}/{cname}",
                "set runAsNonRoot: true or a non-zero runAsUser",
                ["CIS-K8s-5.2.6", "NIST-AC-6"]))
    return out


def _rule_has_wildcard(rule: dict) -> bool:
    return "*" in (rule.get("verbs") or []) \
        or "*" in (rule.get("resources") or []) \
        or "*" in (rule.get("apiGroups") or [])


def _role_findings(role: dict, kind: str) -> list[KspmFinding]:
    name = (role.get("metadata") or {}).get("name", "?")
    if any(name == p or name.startswith(p) for p in _BUILTIN_ROLE_PREFIXES):
        return []  # built-in wildcards are expected and cannot be tightened
    for rule in role.get("rules") or []:
        if isinstance(rule, dict) and _rule_has_wildcard(rule):
            return [KspmFinding(
                "KSPM-RBAC-001", f"Wildcard grant in {kind}", "high", name,
                "grants '*' verbs/resources/apiGroups; scope the role to "
                "the operations it actually needs",
                ["CIS-K8s-5.1.3", "NIST-AC-6"])]
    return []


def _binding_findings(binding: dict, kind: str) -> list[KspmFinding]:
    name = (binding.get("metadata") or {}).get("name", "?")
    out = []
    for subject in binding.get("subjects") or []:
        sname = (subject or {}).get("name", "")
        if sname in ("system:anonymous", "system:unauthenticated"):
            out.append(KspmFinding(
                "KSPM-RBAC-002", f"{kind} grants access to {sname}",
                "critical", name,
                "anonymous/unauthenticated principals bound to a role",
                ["CIS-K8s-5.1.1", "NIST-AC-3"]))
    return out


def evaluate_cluster_inventory(
    inventory: dict[str, Any],
    reader_errors: Optional[dict[str, str]] = None) -> KspmResult:
    """Evaluate exported cluster objects.

    ``inventory`` keys (each a kubectl-style ``{"items": [...]}`` doc or a
    bare list): pods, roles, cluster_roles, role_bindings,
    cluster_role_bindings.  An ABSENT key is recorded as an
    'unevaluable' collector (denied read ≠ clean); ``reader_errors``
    marks collectors 'failed' with the error detail.
    """
    result = KspmResult()
    reader_errors = reader_errors or {}

    def items(key: str) -> Optional[list]:
        if key in reader_errors:
            result.collectors.append(CollectorState(
                key, "failed", reader_errors[key]))
            return None
        raw = inventory.get(key)
        if raw is None:
            result.collectors.append(CollectorState(
                key, "unevaluable", "not collected (absent or read denied)"))
            return None
        result.collectors.append(CollectorState(key, "executed"))
        return raw.get("items", raw) if isinstance(raw, dict) else raw

    pods = items("pods")
    for pod in pods or []:
        if isinstance(pod, dict):
            result.findings.extend(_pod_findings(pod))
    for key, kind in (("roles", "Role"), ("cluster_roles", "ClusterRole")):
        for role in items(key) or []:
            if isinstance(role, dict):
                result.findings.extend(_role_findings(role, kind))
    for key, kind in (("role_bindings", "RoleBinding"),
                      ("cluster_role_bindings", "ClusterRoleBinding")):
        for b in items(key) or []:
            if isinstance(b, dict):
                result.findings.extend(_binding_findings(b, kind))
    return result


def scan_cluster_posture(path: str) -> KspmResult:
    """Evaluate an exported inventory file (one JSON document)."""
    doc = json.loads(Path(path).read_text())
    return evaluate_cluster_inventory(doc if isinstance(doc, dict) else {})


def kspm_finding_to_finding(kf: KspmFinding):
    """Bridge into the unified findings stream."""
    from agentbom_amd.models.finding import Asset, Finding, FindingSource, FindingType

    return Finding(
        finding_type=FindingType.CIS_FAIL,
        source=FindingSource.CLOUD_CIS,
        asset=Asset(name=kf.resource, asset_type="cloud_resource",
                    provider="kubernetes"),
        severity=kf.severity,
        title=f"{kf.check_id}: {kf.title}",
        description=kf.detail,
        compliance_tags=list(kf.compliance),
    )
