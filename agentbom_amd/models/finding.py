"""Unified Finding stream — one model for every issue type across sources.

Reference: src/agent_bom/finding.py:36-600 (FindingType/FindingSource/
ControlTag/Asset/Finding), :1428 (blast_radius_to_finding),
src/agent_bom/exploitability.py:85-161 (fused_triage_priority).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from enum import Enum
from typing import Any, Optional

from agentbom_amd.models.blast import BlastRadius
from agentbom_amd.models.core import Severity
from agentbom_amd.utils.canonical_ids import canonical_finding_id, canonical_id
from agentbom_amd.utils.version_utils import compare_versions


def stable_id(*parts: str) -> str:
    """Deterministic UUIDv5 from content parts."""
    return canonical_id(*parts)


class FindingType(str, Enum):
    CVE = "CVE"
    CIS_FAIL = "CIS_FAIL"
    CIS_ERROR = "CIS_ERROR"
    CLOUD_BEST_PRACTICE_FAIL = "CLOUD_BEST_PRACTICE_FAIL"
    CLOUD_BEST_PRACTICE_ERROR = "CLOUD_BEST_PRACTICE_ERROR"
    CREDENTIAL_EXPOSURE = "CREDENTIAL_EXPOSURE"
    TOOL_DRIFT = "TOOL_DRIFT"
    INJECTION = "INJECTION"
    PROMPT_SECURITY = "PROMPT_SECURITY"
    EXFILTRATION = "EXFILTRATION"
    CLOAKING = "CLOAKING"
    SAST = "SAST"
    SKILL_RISK = "SKILL_RISK"
    BROWSER_EXT = "BROWSER_EXT"
    LICENSE = "LICENSE"
    RATE_LIMIT = "RATE_LIMIT"
    MCP_BLOCKLIST = "MCP_BLOCKLIST"
    COMBINATION = "COMBINATION"
    MALICIOUS_PACKAGE = "MALICIOUS_PACKAGE"
    MALICIOUS_MODEL = "MALICIOUS_MODEL"
    MODEL_INTEGRITY = "MODEL_INTEGRITY"
    CIEM_OVER_PRIVILEGE = "CIEM_OVER_PRIVILEGE"
    SENSITIVE_DATA = "SENSITIVE_DATA"


class FindingSource(str, Enum):
    MCP_SCAN = "MCP_SCAN"
    CONTAINER = "CONTAINER"
    SBOM = "SBOM"
    CLOUD_CIS = "CLOUD_CIS"
    CLOUD_SECURITY = "CLOUD_SECURITY"
    PROXY = "PROXY"
    SAST = "SAST"
    SKILL = "SKILL"
    BROWSER_EXT = "BROWSER_EXT"
    EXTERNAL = "EXTERNAL"
    FILESYSTEM = "FILESYSTEM"
    PROMPT_SCAN = "PROMPT_SCAN"
    SECRET_SCAN = "SECRET_SCAN"
    GRAPH_ANALYSIS = "GRAPH_ANALYSIS"
    DSPM = "DSPM"
    MODEL_SCAN = "MODEL_SCAN"


@dataclass(frozen=True)
class ControlTag:
    """Normalized framework control attached to a finding."""

    framework: str
    control: str
    version: Optional[str] = None
    confidence: Optional[float] = None
    source: Optional[str] = None
    via: Optional[str] = None

    def to_dict(self) -> dict[str, object]:
        return {
            "framework": self.framework,
            "control": self.control,
            "version": self.version,
            "confidence": self.confidence,
            "source": self.source,
            "via": self.via,
        }

    @classmethod
    def from_dict(cls, payload: dict[str, object]) -> "ControlTag":
        raw_conf = payload.get("confidence")
        conf: Optional[float] = None
        if isinstance(raw_conf, (int, float, str)):
            try:
                conf = float(raw_conf)
            except ValueError:
                conf = None
        return cls(
            framework=str(payload.get("framework") or ""),
            control=str(payload.get("control") or ""),
            version=(str(payload["version"]) if payload.get("version") else None),
            confidence=conf,
            source=(str(payload["source"]) if payload.get("source") else None),
            via=(str(payload["via"]) if payload.get("via") else None),
        )


# (field name on BlastRadius/Finding, framework slug) pairs
FRAMEWORK_TAG_FIELDS: tuple[tuple[str, str], ...] = (
    ("owasp_tags", "owasp_llm"),
    ("atlas_tags", "mitre_atlas"),
    ("attack_tags", "mitre_attack"),
    ("nist_ai_rmf_tags", "nist_ai_rmf"),
    ("owasp_mcp_tags", "owasp_mcp"),
    ("owasp_agentic_tags", "owasp_agentic"),
    ("eu_ai_act_tags", "eu_ai_act"),
    ("nist_csf_tags", "nist_csf"),
    ("iso_27001_tags", "iso_27001"),
    ("soc2_tags", "soc2"),
    ("cis_tags", "cis"),
    ("cmmc_tags", "cmmc"),
    ("nist_800_53_tags", "nist_800_53"),
    ("fedramp_tags", "fedramp"),
    ("pci_dss_tags", "pci_dss"),
)


@dataclass
class Asset:
    """What is affected by this finding."""

    name: str
    asset_type: str  # "mcp_server" | "package" | "container" | "cloud_resource" | "agent"
    identifier: Optional[str] = None
    location: Optional[str] = None
    provider: Optional[str] = None
    account_ref: Optional[str] = None
    region: Optional[str] = None
    environment: Optional[str] = None

    @property
    def stable_id(self) -> str:
        identifier = self.identifier or f"{self.name}:{self.location or ''}"
        return stable_id(self.asset_type, identifier)

    @property
    def canonical_id(self) -> str:
        return self.stable_id

    def to_dict(self) -> dict[str, Any]:
        return {
            "name": self.name,
            "asset_type": self.asset_type,
            "identifier": self.identifier,
            "location": self.location,
            "stable_id": self.stable_id,
            "canonical_id": self.canonical_id,
            "provider": self.provider,
            "account_ref": self.account_ref,
            "region": self.region,
            "environment": self.environment,
        }


_SEVERITY_VALUES = {"critical", "high", "medium", "low", "none", "unknown"}


def normalize_severity(value: Any) -> str:
    raw = str(getattr(value, "value", value) or "unknown").strip().lower()
    aliases = {"moderate": "medium", "important": "high", "info": "low", "informational": "low"}
    raw = aliases.get(raw, raw)
    return raw if raw in _SEVERITY_VALUES else "unknown"


@dataclass
class Finding:
    """Unified finding record."""

    finding_type: FindingType
    source: FindingSource
    asset: Asset
    severity: str

    provider: Optional[str] = None
    account_ref: Optional[str] = None
    region: Optional[str] = None
    environment: Optional[str] = None
    vendor_severity: Optional[str] = None
    cvss_severity: Optional[str] = None

    title: str = ""
    description: str = ""
    cve_id: Optional[str] = None
    cwe_ids: list[str] = field(default_factory=list)
    cvss_score: Optional[float] = None
    cvss_vector: Optional[str] = None
    attack_vector: Optional[str] = None
    attack_complexity: Optional[str] = None
    privileges_required: Optional[str] = None
    user_interaction: Optional[str] = None
    network_exploitable: bool = False
    epss_score: Optional[float] = None
    is_kev: bool = False
    is_malicious: bool = False
    malicious_reason: Optional[str] = None

    fixed_version: Optional[str] = None
    remediation_guidance: Optional[str] = None

    compliance_tags: list[str] = field(default_factory=list)
    applicable_frameworks: list[str] = field(default_factory=list)
    controls: list[ControlTag] = field(default_factory=list)
    owasp_tags: list[str] = field(default_factory=list)
    atlas_tags: list[str] = field(default_factory=list)
    attack_tags: list[str] = field(default_factory=list)
    nist_ai_rmf_tags: list[str] = field(default_factory=list)
    owasp_mcp_tags: list[str] = field(default_factory=list)
    owasp_agentic_tags: list[str] = field(default_factory=list)
    eu_ai_act_tags: list[str] = field(default_factory=list)
    nist_csf_tags: list[str] = field(default_factory=list)
    iso_27001_tags: list[str] = field(default_factory=list)
    soc2_tags: list[str] = field(default_factory=list)
    cis_tags: list[str] = field(default_factory=list)
    cmmc_tags: list[str] = field(default_factory=list)
    nist_800_53_tags: list[str] = field(default_factory=list)
    fedramp_tags: list[str] = field(default_factory=list)
    pci_dss_tags: list[str] = field(default_factory=list)

    related_findings: list[str] = field(default_factory=list)
    evidence: dict = field(default_factory=dict)
    node_id: Optional[str] = None
    finding_node_id: Optional[str] = None
    entity_type: Optional[str] = None

    risk_score: float = 0.0
    reachability: Optional[str] = None
    graph_reachable: Optional[bool] = None
    graph_min_hop_distance: Optional[int] = None
    graph_reachable_from_agents: list[str] = field(default_factory=list)
    is_actionable: Optional[bool] = None
    impact_category: Optional[str] = None

    suppressed: bool = False
    suppression_id: Optional[str] = None
    suppression_state: Optional[str] = None
    suppression_reason: Optional[str] = None
    unsuppressed_risk_score: Optional[float] = None

    ai_risk_context: Optional[str] = None
    ai_summary: Optional[str] = None
    attack_vector_summary: Optional[str] = None

    affected_servers: list[str] = field(default_factory=list)
    affected_agents: list[str] = field(default_factory=list)
    exposed_credentials: list[str] = field(default_factory=list)
    exposed_tools: list[str] = field(default_factory=list)

    first_seen: Optional[str] = None
    last_seen: Optional[str] = None

    id: str = field(default="")

    def __post_init__(self) -> None:
        self.severity = normalize_severity(self.severity)
        # Mirror scope between finding and asset.
        for scope_field in ("provider", "account_ref", "region", "environment"):
            fv = getattr(self, scope_field)
            av = getattr(self.asset, scope_field, None)
            if fv is not None and av is None:
                setattr(self.asset, scope_field, fv)
            elif fv is None and av is not None:
                setattr(self, scope_field, av)
        if self.vendor_severity is not None:
            self.vendor_severity = normalize_severity(self.vendor_severity)
        if self.cvss_severity is not None:
            self.cvss_severity = normalize_severity(self.cvss_severity)
        self.controls = [t if isinstance(t, ControlTag) else ControlTag.from_dict(t) for t in self.controls]
        if not self.id:
            # Deterministic: same CVE on same asset+package always same id.
            cve_part = self.vulnerability_id or self.title
            pkg_name = ""
            pkg_version = ""
            if self.asset.asset_type == "package" and self.asset.identifier:
                purl = self.asset.identifier
                pkg_part = purl.split("/")[-1] if "/" in purl else purl
                if "@" in pkg_part:
                    pkg_name, pkg_version = pkg_part.rsplit("@", 1)
            elif isinstance(self.evidence, dict):
                pkg_name = str(self.evidence.get("package_name") or "")
                pkg_version = str(self.evidence.get("package_version") or "")
            self.id = canonical_finding_id(self.asset.stable_id, cve_part, pkg_name, pkg_version)

    @property
    def canonical_id(self) -> str:
        return self.id

    @property
    def vulnerability_id(self) -> Optional[str]:
        if self.cve_id:
            return self.cve_id
        raw = self.evidence.get("vulnerability_id") if isinstance(self.evidence, dict) else None
        return str(raw).strip() or None if raw is not None else None

    @property
    def all_framework_tags(self) -> dict[str, list[str]]:
        return {slug: list(getattr(self, fname)) for fname, slug in FRAMEWORK_TAG_FIELDS if getattr(self, fname)}

    def to_dict(self) -> dict[str, Any]:
        d: dict[str, Any] = {
            "id": self.id,
            "canonical_id": self.canonical_id,
            "finding_type": self.finding_type.value,
            "source": self.source.value,
            "asset": self.asset.to_dict(),
            "severity": self.severity,
            "vendor_severity": self.vendor_severity,
            "cvss_severity": self.cvss_severity,
            "title": self.title,
            "description": self.description,
            "cve_id": self.cve_id,
            "cwe_ids": list(self.cwe_ids),
            "cvss_score": self.cvss_score,
            "cvss_vector": self.cvss_vector,
            "attack_vector": self.attack_vector,
            "attack_complexity": self.attack_complexity,
            "privileges_required": self.privileges_required,
            "user_interaction": self.user_interaction,
            "network_exploitable": self.network_exploitable,
            "epss_score": self.epss_score,
            "is_kev": self.is_kev,
            "is_malicious": self.is_malicious,
            "malicious_reason": self.malicious_reason,
            "fixed_version": self.fixed_version,
            "remediation_guidance": self.remediation_guidance,
            "compliance_tags": list(self.compliance_tags),
            "applicable_frameworks": list(self.applicable_frameworks),
            "controls": [c.to_dict() for c in self.controls],
            "related_findings": list(self.related_findings),
            "evidence": dict(self.evidence),
            "node_id": self.node_id,
            "finding_node_id": self.finding_node_id,
            "entity_type": self.entity_type,
            "risk_score": self.risk_score,
            "reachability": self.reachability,
            "graph_reachable": self.graph_reachable,
            "graph_min_hop_distance": self.graph_min_hop_distance,
            "graph_reachable_from_agents": list(self.graph_reachable_from_agents),
            "is_actionable": self.is_actionable,
            "impact_category": self.impact_category,
            "suppressed": self.suppressed,
            "suppression_id": self.suppression_id,
            "suppression_state": self.suppression_state,
            "suppression_reason": self.suppression_reason,
            "unsuppressed_risk_score": self.unsuppressed_risk_score,
            "ai_risk_context": self.ai_risk_context,
            "ai_summary": self.ai_summary,
            "attack_vector_summary": self.attack_vector_summary,
            "affected_servers": list(self.affected_servers),
            "affected_agents": list(self.affected_agents),
            "exposed_credentials": list(self.exposed_credentials),
            "exposed_tools": list(self.exposed_tools),
            "first_seen": self.first_seen,
            "last_seen": self.last_seen,
        }
        for fname, _slug in FRAMEWORK_TAG_FIELDS:
            d[fname] = list(getattr(self, fname))
        return d


# ── Triage priority fusion (exploitability.py:85-161) ───────────────────────

_SEVERITY_POINTS = {"critical": 40, "high": 30, "medium": 18, "low": 8, "none": 0, "unknown": 0}


def fused_triage_priority(
    *,
    severity: Optional[str],
    is_kev: bool = False,
    epss_score: Optional[float] = None,
    network_exploitable: bool = False,
    impact_category: Optional[str] = None,
    reachable: Optional[bool] = None,
    internet_exposed: bool = False,
    exposed_credential_count: int = 0,
    exposed_tool_count: int = 0,
    symbol_reachability: Optional[str] = None,
) -> dict[str, object]:
    """Bounded, explainable triage priority fused from factual signals."""
    sev = (severity or "unknown").lower()
    score: float = float(_SEVERITY_POINTS.get(sev, 0))
    reasons: list[str] = [f"severity:{sev}"]

    if is_kev:
        score += 25
        reasons.append("cisa_kev")
    if epss_score is not None:
        if epss_score >= 0.7:
            score += 15
            reasons.append("epss>=0.70")
        elif epss_score >= 0.5:
            score += 10
            reasons.append("epss>=0.50")
    if network_exploitable:
        score += 10
        reasons.append("cvss_av_network")
    if impact_category == "code-execution":
        score += 12
        reasons.append("cwe_code_execution")
    elif impact_category == "unknown":
        reasons.append("impact_unknown")
    if reachable is True:
        score += 8
        reasons.append("reachable")
    elif reachable is False:
        score -= 4
        reasons.append("not_reachable")
    if internet_exposed:
        score += 14
        reasons.append("internet_exposed")
    if exposed_credential_count:
        score += min(exposed_credential_count * 4, 12)
        reasons.append("credentials_reachable")
    if exposed_tool_count:
        score += min(exposed_tool_count * 3, 9)
        reasons.append("tools_reachable")
    if symbol_reachability == "function_reachable":
        score += 10
        reasons.append("symbol_function_reachable")
    elif symbol_reachability == "unreachable":
        score -= 8
        reasons.append("symbol_unreachable")

    bounded = max(0, min(int(score), 100))
    if bounded >= 85:
        band = "urgent"
    elif bounded >= 65:
        band = "high"
    elif bounded >= 40:
        band = "medium"
    else:
        band = "low"
    return {"score": bounded, "band": band, "reasons": reasons}


def forward_fixed_version(
    fixed_version: Optional[str], current_version: Optional[str], ecosystem: Optional[str]
) -> Optional[str]:
    """Return fixed_version only when it is a forward upgrade of current."""
    if not isinstance(fixed_version, str) or not fixed_version.strip():
        return None
    fixed = fixed_version.strip()
    current = (current_version or "").strip()
    if not current:
        return fixed
    if compare_versions(current, fixed, (ecosystem or "").strip()):
        return fixed
    return None


def remediation_guidance_for(vuln, pkg) -> str:
    pkg_name = getattr(pkg, "name", "the package")
    if getattr(pkg, "is_malicious", False):
        reason = getattr(pkg, "malicious_reason", None)
        detail = f" ({reason})" if reason else ""
        return f"Remove {pkg_name} from all environments immediately{detail}."
    fixed = forward_fixed_version(
        getattr(vuln, "fixed_version", None), getattr(pkg, "version", None), getattr(pkg, "ecosystem", None)
    )
    if isinstance(fixed, str) and fixed.strip():
        return f"Upgrade {pkg_name} to {fixed.strip()}."
    if getattr(vuln, "is_kev", False):
        vid = getattr(vuln, "id", None)
        cve_context = f" {vid}" if vid else ""
        return f"Prioritize vendor mitigation for{cve_context}; it is listed in CISA KEV and no fixed version is recorded."
    if getattr(vuln, "references", None):
        return "Review the linked advisory references and apply the vendor-recommended mitigation or upgrade path."
    return "Review the advisory and apply the vendor-recommended mitigation, upgrade path, or compensating control."


def _source_for_blast_radius(br: BlastRadius) -> FindingSource:
    surfaces = {getattr(getattr(s, "surface", None), "value", None) for s in br.affected_servers}
    if any(getattr(s, "is_mcp_surface", False) for s in br.affected_servers):
        return FindingSource.MCP_SCAN
    if {"container-image", "oci-tarball"} & surfaces:
        return FindingSource.CONTAINER
    if "filesystem" in surfaces:
        return FindingSource.FILESYSTEM
    if "external-scan" in surfaces:
        return FindingSource.EXTERNAL
    if "sbom" in surfaces:
        return FindingSource.SBOM
    return FindingSource.MCP_SCAN


def blast_radius_to_finding(br: BlastRadius) -> Finding:
    """Convert a BlastRadius row to a unified Finding (dual-write shim)."""
    vuln = br.vulnerability
    pkg = br.package

    if br.affected_servers:
        primary = br.affected_servers[0]
        loc_parts = [primary.command, *primary.args] if primary.command else []
        asset = Asset(
            name=primary.name,
            asset_type="mcp_server",
            identifier=None,
            location=" ".join(loc_parts) or None,
        )
    else:
        asset = Asset(
            name=pkg.name,
            asset_type="package",
            identifier=f"pkg:{pkg.ecosystem}/{pkg.name}@{pkg.version}" if pkg.version else None,
        )

    evidence: dict = {
        "package_name": pkg.name,
        "package_version": pkg.version,
        "ecosystem": pkg.ecosystem,
        "package_is_direct": pkg.is_direct,
        "package_parent": pkg.parent_package,
        "package_dependency_depth": pkg.dependency_depth,
        "package_dependency_scope": pkg.dependency_scope,
        "package_reachability_evidence": pkg.reachability_evidence,
        "affected_server_count": len(br.affected_servers),
        "exposed_credential_count": len(br.exposed_credentials),
        "exposed_tool_count": len(br.exposed_tools),
        "hop_depth": br.hop_depth,
        "delegation_chain": list(br.delegation_chain),
        "transitive_agents": list(br.transitive_agents),
        "transitive_credential_count": len(br.transitive_credentials or []),
        "transitive_risk_score": br.transitive_risk_score,
        "dependency_reachable": br.dependency_reachable,
        "dependency_min_hop_distance": br.dependency_min_hop_distance,
        "dependency_reachable_from_agents": list(br.dependency_reachable_from_agents),
        "graph_reachable": br.graph_reachable,
        "graph_min_hop_distance": br.graph_min_hop_distance,
        "graph_reachable_from_agents": list(br.graph_reachable_from_agents),
        "symbol_reachability": br.symbol_reachability,
        "reachable_affected_symbols": list(br.reachable_affected_symbols),
        "layer_attribution": [o.to_dict() for o in br.layer_attribution],
        "published_at": vuln.published_at,
        "modified_at": vuln.modified_at,
        "severity_source": vuln.severity_source,
        "cvss_vector": vuln.cvss_vector,
        "attack_vector": vuln.attack_vector,
        "attack_complexity": vuln.attack_complexity,
        "privileges_required": vuln.privileges_required,
        "user_interaction": vuln.user_interaction,
        "network_exploitable": vuln.network_exploitable,
        "epss_percentile": vuln.epss_percentile,
        "kev_date_added": vuln.kev_date_added,
        "kev_due_date": vuln.kev_due_date,
        "vulnerability_compliance_tags": dict(vuln.compliance_tags or {}),
    }
    if pkg.is_malicious:
        evidence["package_is_malicious"] = True
        if isinstance(pkg.malicious_reason, str) and pkg.malicious_reason.strip():
            evidence["malicious_reason"] = pkg.malicious_reason.strip()
    if vuln.references:
        evidence["references"] = list(vuln.references[:5])
    if vuln.match_confidence_tier:
        evidence["match_confidence_tier"] = vuln.match_confidence_tier
    if vuln.vex_status:
        evidence["vex_status"] = vuln.vex_status
    if vuln.vex_justification:
        evidence["vex_justification"] = vuln.vex_justification
    if vuln.aliases:
        evidence["advisory_aliases"] = list(vuln.aliases)
    cve_ids = [i for i in [vuln.id, *vuln.aliases] if i and i.upper().startswith("CVE-")]
    if cve_ids:
        evidence["cve_ids"] = sorted(set(cve_ids))

    sev = vuln.severity.value if hasattr(vuln.severity, "value") else str(vuln.severity)
    evidence["triage_priority"] = fused_triage_priority(
        severity=sev,
        is_kev=bool(vuln.is_kev),
        epss_score=vuln.epss_score,
        network_exploitable=bool(vuln.network_exploitable),
        impact_category=br.impact_category,
        reachable=br.graph_reachable,
        exposed_credential_count=len(br.exposed_credentials),
        exposed_tool_count=len(br.exposed_tools),
        symbol_reachability=br.symbol_reachability,
    )

    from agentbom_amd.utils.canonical_ids import canonical_package_key

    package_node_id = f"pkg:{canonical_package_key(pkg.name, pkg.version or '', pkg.ecosystem or '', pkg.purl)}"
    finding_node_id = f"vuln:{vuln.id}" if vuln.id else None

    return Finding(
        finding_type=FindingType.CVE,
        source=_source_for_blast_radius(br),
        asset=asset,
        severity=sev,
        node_id=package_node_id,
        finding_node_id=finding_node_id,
        title=f"{vuln.id}: {pkg.name}@{pkg.version or 'unknown'}",
        description=vuln.summary or "",
        cve_id=vuln.id,
        cwe_ids=list(vuln.cwe_ids or []),
        cvss_score=vuln.cvss_score,
        cvss_vector=vuln.cvss_vector,
        attack_vector=vuln.attack_vector,
        attack_complexity=vuln.attack_complexity,
        privileges_required=vuln.privileges_required,
        user_interaction=vuln.user_interaction,
        network_exploitable=bool(vuln.network_exploitable),
        epss_score=vuln.epss_score,
        is_kev=bool(vuln.is_kev),
        is_malicious=bool(pkg.is_malicious),
        malicious_reason=pkg.malicious_reason,
        fixed_version=forward_fixed_version(vuln.fixed_version, pkg.version, pkg.ecosystem),
        remediation_guidance=remediation_guidance_for(vuln, pkg),
        owasp_tags=list(br.owasp_tags),
        atlas_tags=list(br.atlas_tags),
        attack_tags=list(br.attack_tags),
        nist_ai_rmf_tags=list(br.nist_ai_rmf_tags),
        owasp_mcp_tags=list(br.owasp_mcp_tags),
        owasp_agentic_tags=list(br.owasp_agentic_tags),
        eu_ai_act_tags=list(br.eu_ai_act_tags),
        nist_csf_tags=list(br.nist_csf_tags),
        iso_27001_tags=list(br.iso_27001_tags),
        soc2_tags=list(br.soc2_tags),
        cis_tags=list(br.cis_tags),
        cmmc_tags=list(br.cmmc_tags),
        nist_800_53_tags=list(br.nist_800_53_tags),
        fedramp_tags=list(br.fedramp_tags),
        pci_dss_tags=list(br.pci_dss_tags),
        compliance_tags=sorted(
            {t for fname, _ in FRAMEWORK_TAG_FIELDS for t in getattr(br, fname)}
        ),
        evidence=evidence,
        risk_score=br.risk_score,
        reachability=br.reachability,
        graph_reachable=br.graph_reachable,
        graph_min_hop_distance=br.graph_min_hop_distance,
        graph_reachable_from_agents=list(br.graph_reachable_from_agents),
        is_actionable=br.is_actionable,
        impact_category=br.impact_category,
        suppressed=br.suppressed,
        suppression_id=br.suppression_id,
        suppression_state=br.suppression_state,
        suppression_reason=br.suppression_reason,
        unsuppressed_risk_score=br.unsuppressed_risk_score,
        ai_risk_context=br.ai_risk_context,
        ai_summary=br.ai_summary,
        attack_vector_summary=br.attack_vector_summary,
        affected_servers=[s.name for s in br.affected_servers],
        affected_agents=[a.name for a in br.affected_agents],
        exposed_credentials=list(br.exposed_credentials),
        exposed_tools=[t.name for t in br.exposed_tools],
    )
