"""Core AI-BOM data model.

Feature-parity rebuild of the reference data model
(reference: src/agent_bom/models.py:52-1100, exploitability.py,
advisory_sources.py, constants.py credential heuristics).  Risk-score weights
come from :mod:`agentbom_amd.utils.config` with identical defaults so scores
are numerically reproducible.
"""

from __future__ import annotations

import json
import uuid as _uuid
from dataclasses import dataclass, field
from datetime import datetime, timezone
from enum import Enum
from typing import Any, Optional

from agentbom_amd.utils import config as cfg
from agentbom_amd.utils.canonical_ids import (
    AGENT_BOM_ID_NAMESPACE,
    canonical_agent_id,
    canonical_mcp_prompt_id,
    canonical_mcp_resource_id,
    canonical_mcp_server_id,
    canonical_mcp_tool_id,
    canonical_package_id,
    normalize_package_name,
)


def utc_now_iso() -> str:
    return datetime.now(timezone.utc).isoformat().replace("+00:00", "Z")


# ── Enums ───────────────────────────────────────────────────────────────────


class Severity(str, Enum):
    CRITICAL = "critical"
    HIGH = "high"
    MEDIUM = "medium"
    LOW = "low"
    NONE = "none"
    UNKNOWN = "unknown"  # no severity data — distinct from NONE (no vulnerability)


SEVERITY_ORDER = [Severity.CRITICAL, Severity.HIGH, Severity.MEDIUM, Severity.LOW]

# Fixed numeric codes shared with the GPU engine (ops/csrc): severity is
# carried device-side as u8 with this encoding, highest first for max-reduce.
SEVERITY_CODE = {
    Severity.CRITICAL: 5,
    Severity.HIGH: 4,
    Severity.MEDIUM: 3,
    Severity.LOW: 2,
    Severity.UNKNOWN: 1,
    Severity.NONE: 0,
}
SEVERITY_FROM_CODE = {v: k for k, v in SEVERITY_CODE.items()}


class AgentType(str, Enum):
    CLAUDE_DESKTOP = "claude-desktop"
    CLAUDE_CODE = "claude-code"
    CURSOR = "cursor"
    WINDSURF = "windsurf"
    CLINE = "cline"
    VSCODE_COPILOT = "vscode-copilot"
    CORTEX_CODE = "cortex-code"
    CODEX_CLI = "codex-cli"
    GEMINI_CLI = "gemini-cli"
    GOOSE = "goose"
    SNOWFLAKE_CLI = "snowflake-cli"
    CONTINUE = "continue"
    ZED = "zed"
    OPENCLAW = "openclaw"
    ROO_CODE = "roo-code"
    AMAZON_Q = "amazon-q"
    DOCKER_MCP = "docker-mcp"
    JETBRAINS_AI = "jetbrains-ai"
    JUNIE = "junie"
    COPILOT_CLI = "copilot-cli"
    TABNINE = "tabnine"
    SOURCEGRAPH_CODY = "sourcegraph-cody"
    AIDER = "aider"
    REPLIT_AGENT = "replit-agent"
    VOID_EDITOR = "void"
    AIDE = "aide"
    TRAE = "trae"
    PIECES = "pieces"
    MCP_CLI = "mcp-cli"
    CUSTOM = "custom"


class TransportType(str, Enum):
    STDIO = "stdio"
    SSE = "sse"
    STREAMABLE_HTTP = "streamable-http"
    UNKNOWN = "unknown"


class ServerSurface(str, Enum):
    MCP = "mcp-server"
    CONTAINER_IMAGE = "container-image"
    OCI_TARBALL = "oci-tarball"
    FILESYSTEM = "filesystem"
    SBOM = "sbom"
    EXTERNAL_SCAN = "external-scan"
    OS_PACKAGES = "os-packages"
    SAST = "sast"
    AI_INVENTORY = "ai-inventory"
    OTHER = "other"


class AgentStatus(str, Enum):
    CONFIGURED = "configured"
    INSTALLED_NOT_CONFIGURED = "installed-not-configured"


# ── Advisory source attribution ─────────────────────────────────────────────

PRIMARY_ADVISORY_SOURCES: tuple[str, ...] = ("osv", "ghsa", "nvidia_csaf", "amd_psirt", "intel_psirt")
ENRICHMENT_ADVISORY_SOURCES: tuple[str, ...] = ("nvd", "epss", "cisa_kev")
_PREFERRED_SOURCE_ORDER = PRIMARY_ADVISORY_SOURCES + ENRICHMENT_ADVISORY_SOURCES

_SOURCE_ALIASES = {
    "github": "ghsa",
    "github_advisory": "ghsa",
    "github_security_advisory": "ghsa",
    "kev": "cisa_kev",
    "cisa": "cisa_kev",
    "cisa_kev_catalog": "cisa_kev",
    "nvidia": "nvidia_csaf",
    "nvidia_advisory": "nvidia_csaf",
    "nvidia_csaf_advisory": "nvidia_csaf",
    "amd": "amd_psirt",
    "amd_advisory": "amd_psirt",
    "amd_psirt_advisory": "amd_psirt",
    "intel": "intel_psirt",
    "intel_advisory": "intel_psirt",
    "intel_psirt_advisory": "intel_psirt",
}


def normalize_advisory_source(source: Optional[str]) -> Optional[str]:
    if not source:
        return None
    value = source.strip().lower().replace("-", "_")
    return _SOURCE_ALIASES.get(value, value)


def merge_advisory_sources(*sources: Optional[str]) -> list[str]:
    """Dedup + stable preference ordering (primary, enrichment, then unknown)."""
    seen: set[str] = set()
    merged: list[str] = []
    for s in sources:
        n = normalize_advisory_source(s)
        if not n or n in seen:
            continue
        seen.add(n)
        merged.append(n)
    preferred = [s for s in _PREFERRED_SOURCE_ORDER if s in seen]
    unknown = sorted(s for s in merged if s not in _PREFERRED_SOURCE_ORDER)
    return preferred + unknown


# ── CVSS vector exploitability signals ──────────────────────────────────────

_ATTACK_VECTOR = {"N": "network", "A": "adjacent", "L": "local", "P": "physical"}
_ATTACK_COMPLEXITY = {"L": "low", "H": "high"}
_PRIVILEGES_REQUIRED = {"N": "none", "L": "low", "H": "high"}
_UI_V3 = {"N": "none", "R": "required"}
_UI_V4 = {"N": "none", "P": "passive", "A": "active"}


@dataclass(frozen=True)
class CvssVectorSignals:
    attack_vector: Optional[str] = None
    attack_complexity: Optional[str] = None
    privileges_required: Optional[str] = None
    user_interaction: Optional[str] = None

    @property
    def network_exploitable(self) -> bool:
        return self.attack_vector == "network"

    def to_dict(self) -> dict[str, Any]:
        return {
            "attack_vector": self.attack_vector or "",
            "attack_complexity": self.attack_complexity or "",
            "privileges_required": self.privileges_required or "",
            "user_interaction": self.user_interaction or "",
            "network_exploitable": self.network_exploitable,
        }


def parse_cvss_vector_signals(vector: Optional[str]) -> CvssVectorSignals:
    """Extract exploitability dims from CVSS 3.x/4.0 vectors; never guess."""
    if not vector:
        return CvssVectorSignals()
    raw = vector.strip()
    if not raw.startswith("CVSS:"):
        return CvssVectorSignals()
    try:
        version = raw.split("/", 1)[0]
        metrics = dict(part.split(":", 1) for part in raw.split("/")[1:] if ":" in part)
    except ValueError:
        return CvssVectorSignals()
    ui_map = _UI_V4 if version.startswith("CVSS:4") else _UI_V3
    return CvssVectorSignals(
        attack_vector=_ATTACK_VECTOR.get(metrics.get("AV", "")),
        attack_complexity=_ATTACK_COMPLEXITY.get(metrics.get("AC", "")),
        privileges_required=_PRIVILEGES_REQUIRED.get(metrics.get("PR", "")),
        user_interaction=ui_map.get(metrics.get("UI", "")),
    )


# ── Credential-key heuristic (constants.py:258-340 semantics) ───────────────

_CREDENTIAL_WORDS = frozenset(
    {"apikey", "authorization", "bearer", "credential", "key", "password", "passwd", "secret", "token"}
)
_CREDENTIAL_MATERIAL_WORDS = frozenset({"cert", "certificate", "oauth"})
_LOCATOR_WORDS = frozenset(
    {
        "arn", "dir", "directories", "directory", "endpoint", "file", "filename",
        "host", "hostname", "id", "location", "name", "path", "port", "ref", "url",
    }
)
_SINGULARIZABLE = _CREDENTIAL_WORDS | _CREDENTIAL_MATERIAL_WORDS | _LOCATOR_WORDS
_CREDENTIAL_WORD_PAIRS = frozenset(
    {
        ("ca", "cert"), ("client", "cert"), ("client", "certificate"),
        ("conn", "str"), ("connection", "string"), ("connection", "uri"),
        ("connection", "url"), ("database", "url"), ("db", "url"),
    }
)
_CREDENTIAL_COMPOUND_NAMES = frozenset(
    {"id_dsa", "id_ecdsa", "id_ed25519", "id_rsa", "mysql_pwd", "pgpassword"}
)


def _singularize(token: str) -> str:
    if token.endswith("s") and token[:-1] in _SINGULARIZABLE:
        return token[:-1]
    return token


def is_credential_key(name: str) -> bool:
    """True when an env-var name looks like credential material.

    Locator-qualified material words (``CERTIFICATE_PATH``, ``OAUTH_CLIENT_ID``)
    are NOT credentials — they name files/endpoints, not secrets.
    """
    raw = (name or "").strip().lower()
    if not raw:
        return False
    if raw in _CREDENTIAL_COMPOUND_NAMES:
        return True
    tokens = [_singularize(t) for t in raw.replace("-", "_").split("_") if t]
    token_set = set(tokens)
    has_locator = bool(token_set & _LOCATOR_WORDS)
    if token_set & _CREDENTIAL_WORDS:
        # A locator-qualified pure-material word is a path, but a strong
        # credential word wins even with a locator (e.g. API_KEY_ID is... a
        # borderline; the reference keeps strong words as credentials unless
        # the value is an obvious locator of a weak word).
        return True
    if token_set & _CREDENTIAL_MATERIAL_WORDS and not has_locator:
        return True
    for a, b in zip(tokens, tokens[1:]):
        if (a, b) in _CREDENTIAL_WORD_PAIRS:
            return True
    return False


# ── Vulnerability ───────────────────────────────────────────────────────────


def _strip_reference_host_path(ref: str) -> tuple[str, str]:
    from urllib.parse import urlparse

    try:
        p = urlparse(ref)
    except ValueError:
        return "", ""
    return (p.hostname or "").lower(), (p.path or "").lower()


def _host_matches_domain(host: str, domain: str) -> bool:
    return host == domain or host.endswith(f".{domain}")


EPSS_ACTIVE_EXPLOITATION_THRESHOLD = cfg._float("AGENT_BOM_EPSS_ACTIVE_THRESHOLD", 0.5)


@dataclass
class Vulnerability:
    """A known vulnerability in a package (reference models.py:110-308)."""

    id: str
    summary: str
    severity: Severity
    severity_source: Optional[str] = None
    confidence: Optional[float] = None
    cvss_score: Optional[float] = None
    fixed_version: Optional[str] = None
    references: list[str] = field(default_factory=list)
    epss_score: Optional[float] = None
    epss_percentile: Optional[float] = None
    is_kev: bool = False
    kev_date_added: Optional[str] = None
    kev_due_date: Optional[str] = None
    published_at: Optional[str] = None
    modified_at: Optional[str] = None
    nvd_published: Optional[str] = None
    nvd_modified: Optional[str] = None
    nvd_status: Optional[str] = None
    cwe_ids: list[str] = field(default_factory=list)
    aliases: list[str] = field(default_factory=list)
    exploitability: Optional[str] = None
    vex_status: Optional[str] = None
    vex_justification: Optional[str] = None
    compliance_tags: dict[str, list[str]] = field(default_factory=dict)
    advisory_sources: list[str] = field(default_factory=list)
    match_confidence_tier: Optional[str] = None
    cvss_vector: Optional[str] = None
    attack_vector: Optional[str] = None
    attack_complexity: Optional[str] = None
    privileges_required: Optional[str] = None
    user_interaction: Optional[str] = None
    network_exploitable: bool = False
    affected_symbols: list[str] = field(default_factory=list)
    affected_symbols_by_path: dict[str, list[str]] = field(default_factory=dict)

    def __post_init__(self) -> None:
        self.advisory_sources = merge_advisory_sources(*self.advisory_sources)
        signals = parse_cvss_vector_signals(self.cvss_vector)
        self.attack_vector = self.attack_vector or signals.attack_vector
        self.attack_complexity = self.attack_complexity or signals.attack_complexity
        self.privileges_required = self.privileges_required or signals.privileges_required
        self.user_interaction = self.user_interaction or signals.user_interaction
        self.network_exploitable = bool(self.network_exploitable or signals.network_exploitable)
        # Sanitize fixed_version: git SHAs / digit-free strings are not versions.
        if self.fixed_version:
            v = self.fixed_version.lstrip("v")
            is_hex = all(c in "0123456789abcdef" for c in v)
            if (len(v) == 40 and is_hex) or (7 <= len(v) <= 12 and is_hex) or not any(c.isdigit() for c in v):
                self.fixed_version = None

    @property
    def is_actively_exploited(self) -> bool:
        return self.is_kev or (
            self.epss_score is not None and self.epss_score > EPSS_ACTIVE_EXPLOITATION_THRESHOLD
        )

    @property
    def exploit_likelihood(self) -> str:
        """KEV > EPSS-derived gradation; no fabricated assessment on no signal."""
        if self.is_kev:
            return "actively_exploited"
        if self.epss_score is None and self.epss_percentile is None:
            return "unassessed"
        epss = self.epss_score or 0.0
        pct = self.epss_percentile or 0.0
        if epss >= EPSS_ACTIVE_EXPLOITATION_THRESHOLD or pct >= 95.0:
            return "likely_exploited"
        if pct >= 80.0:
            return "public_exploit"
        return "theoretical"

    @property
    def all_advisory_sources(self) -> list[str]:
        derived: list[Optional[str]] = []
        if self.id.startswith("GHSA-") or any(a.startswith("GHSA-") for a in self.aliases):
            derived.append("ghsa")
        if self.references:
            hp = [_strip_reference_host_path(r) for r in self.references]
            if any(h == "github.com" and (p.startswith("/advisories/") or p.startswith("/advisory/")) for h, p in hp):
                derived.append("ghsa")
            if any(_host_matches_domain(h, "nvidia.com") or _host_matches_domain(h, "nvidia.github.io") for h, _ in hp):
                derived.append("nvidia_csaf")
            if any(h == "nvd.nist.gov" for h, _ in hp):
                derived.append("nvd")
        if self.nvd_status or self.nvd_published or self.nvd_modified:
            derived.append("nvd")
        if self.epss_score is not None:
            derived.append("epss")
        if self.is_kev:
            derived.append("cisa_kev")
        return merge_advisory_sources(*self.advisory_sources, *derived)

    @property
    def advisory_coverage_state(self) -> str:
        sources = self.all_advisory_sources
        has_primary = any(s in {"osv", "ghsa", "nvidia_csaf"} for s in sources)
        has_enrichment = any(s in {"nvd", "epss", "cisa_kev"} for s in sources)
        if has_primary and has_enrichment:
            return "enriched"
        if has_primary:
            return "primary_only"
        if has_enrichment:
            return "enrichment_only"
        return "unknown"

    @property
    def risk_level(self) -> str:
        if self.is_kev:
            return "CRITICAL - Active Exploitation"
        if self.epss_score and self.epss_score > cfg.EPSS_CRITICAL_THRESHOLD:
            return "CRITICAL - High Exploit Probability"
        if self.severity == Severity.CRITICAL:
            return "CRITICAL"
        if self.severity == Severity.HIGH and self.epss_score and self.epss_score > cfg.EPSS_HIGH_LIKELY_THRESHOLD:
            return "HIGH - Likely Exploitable"
        if self.severity == Severity.HIGH:
            return "HIGH"
        if self.severity == Severity.MEDIUM:
            return "MEDIUM"
        return "LOW"


def compute_confidence(vuln: Vulnerability) -> float:
    """0.0-1.0 data-quality confidence (reference models.py:308-330)."""
    score = 0.0
    if vuln.cvss_score is not None:
        score += 0.25
    if vuln.cvss_vector:
        score += 0.05
    if vuln.epss_score is not None:
        score += 0.20
    if vuln.severity_source and vuln.severity_source != "unknown":
        score += 0.15
    if vuln.cwe_ids:
        score += 0.15
    if vuln.fixed_version:
        score += 0.10
    if vuln.cvss_score is not None and vuln.severity_source == "cvss":
        score += 0.15
    return min(score, 1.0)


# ── Package ─────────────────────────────────────────────────────────────────


@dataclass
class PackageOccurrence:
    layer_index: int
    layer_id: str
    package_path: Optional[str] = None
    layer_path: Optional[str] = None
    created_by: Optional[str] = None
    dockerfile_instruction: Optional[str] = None

    def to_dict(self) -> dict[str, object]:
        return {
            "layer_index": self.layer_index,
            "layer_id": self.layer_id,
            "layer_path": self.layer_path,
            "package_path": self.package_path,
            "created_by": self.created_by,
            "dockerfile_instruction": self.dockerfile_instruction,
        }


@dataclass
class Package:
    """A software package dependency (reference models.py:351-494)."""

    name: str
    version: str
    ecosystem: str
    purl: Optional[str] = None
    source_package: Optional[str] = None
    distro_name: Optional[str] = None
    distro_version: Optional[str] = None
    vulnerabilities: list[Vulnerability] = field(default_factory=list)
    is_direct: bool = True
    parent_package: Optional[str] = None
    dependency_depth: int = 0
    dependency_scope: str = "runtime"
    reachability_evidence: str = "runtime_dependency"
    resolved_from_registry: bool = False
    registry_version: Optional[str] = None
    version_source: str = "detected"
    declared_version: Optional[str] = None
    resolved_version: Optional[str] = None
    version_confidence: Optional[str] = None
    version_resolved_at: Optional[str] = None
    version_evidence: list[dict[str, Any]] = field(default_factory=list)
    version_conflicts: list[dict[str, Any]] = field(default_factory=list)
    floating_reference: bool = False
    floating_reference_reason: Optional[str] = None
    is_malicious: bool = False
    malicious_reason: Optional[str] = None
    license: Optional[str] = None
    license_expression: Optional[str] = None
    deps_dev_resolved: bool = False
    supplier: Optional[str] = None
    author: Optional[str] = None
    description: Optional[str] = None
    homepage: Optional[str] = None
    repository_url: Optional[str] = None
    download_url: Optional[str] = None
    copyright_text: Optional[str] = None
    scorecard_score: Optional[float] = None
    scorecard_checks: dict[str, int] = field(default_factory=dict)
    scorecard_repo: Optional[str] = None
    scorecard_lookup_state: Optional[str] = None
    scorecard_lookup_reason: Optional[str] = None
    checksums: dict[str, str] = field(default_factory=dict)
    integrity_verified: Optional[bool] = None
    provenance_attested: Optional[bool] = None
    provenance_source: Optional[str] = None
    provenance_status: Optional[str] = None
    auto_risk_level: Optional[str] = None
    auto_risk_justification: Optional[str] = None
    maintainer_count: Optional[int] = None
    source_repo: Optional[str] = None
    occurrences: list[PackageOccurrence] = field(default_factory=list)
    discovery_provenance: Optional[dict[str, Any]] = None

    @property
    def stable_id(self) -> str:
        return canonical_package_id(self.name, self.version, self.ecosystem, self.purl)

    @property
    def canonical_id(self) -> str:
        return self.stable_id

    @property
    def lookup_names(self) -> list[str]:
        names: list[str] = []

        def add(candidate: Optional[str]) -> None:
            candidate = (candidate or "").strip()
            if not candidate:
                return
            norm = normalize_package_name(candidate, self.ecosystem)
            if all(normalize_package_name(n, self.ecosystem) != norm for n in names):
                names.append(candidate)

        add(self.name)
        if self.source_package:
            src = self.source_package.strip()
            if src and normalize_package_name(src, self.ecosystem) != normalize_package_name(self.name, self.ecosystem):
                add(src)
        return names

    @property
    def has_vulnerabilities(self) -> bool:
        return len(self.vulnerabilities) > 0

    @property
    def primary_occurrence(self) -> Optional[PackageOccurrence]:
        if not self.occurrences:
            return None
        return min(self.occurrences, key=lambda o: (o.layer_index, o.layer_id, o.package_path or ""))

    @property
    def layer_count(self) -> int:
        return len({(o.layer_index, o.layer_id) for o in self.occurrences})

    @property
    def max_severity(self) -> Severity:
        if not self.vulnerabilities:
            return Severity.NONE
        for sev in SEVERITY_ORDER:
            if any(v.severity == sev for v in self.vulnerabilities):
                return sev
        return Severity.NONE


# ── MCP child entities ──────────────────────────────────────────────────────


@dataclass
class MCPTool:
    name: str
    description: str
    discovery_source: Optional[str] = None
    discovery_confidence: Optional[str] = None
    input_schema: Optional[dict[str, Any]] = None
    declared_capabilities: list[str] = field(default_factory=list)
    schema_findings: list[str] = field(default_factory=list)
    schema_rule_findings: list[dict[str, Any]] = field(default_factory=list)
    server_canonical_id: Optional[str] = None

    @property
    def stable_id(self) -> str:
        return canonical_mcp_tool_id(self.name, self.input_schema, server_id=self.server_canonical_id)

    canonical_id = stable_id
    fingerprint = stable_id

    @property
    def risk_score(self) -> int:
        score = 0
        for f in self.schema_findings:
            if "shell-execution-capability" in f:
                score += 4
            elif "network-egress-capability" in f:
                score += 3
            elif "filesystem-capability" in f:
                score += 2
            else:
                score += 1
        return min(score, 10)


@dataclass
class MCPResource:
    uri: str
    name: str
    description: str = ""
    mime_type: Optional[str] = None
    content_findings: list[str] = field(default_factory=list)
    server_canonical_id: Optional[str] = None

    @property
    def stable_id(self) -> str:
        return canonical_mcp_resource_id(self.uri, self.mime_type, server_id=self.server_canonical_id)

    canonical_id = stable_id
    fingerprint = stable_id

    @property
    def risk_score(self) -> int:
        score = 0
        for f in self.content_findings:
            if "hidden-instruction-surface" in f or "prompt-bearing-resource" in f:
                score += 3
            elif "mutable-resource" in f:
                score += 2
            else:
                score += 1
        return min(score, 10)


@dataclass
class MCPPrompt:
    name: str
    description: str = ""
    arguments: list[dict[str, object]] = field(default_factory=list)
    content_findings: list[str] = field(default_factory=list)
    server_canonical_id: Optional[str] = None

    @property
    def stable_id(self) -> str:
        return canonical_mcp_prompt_id(self.name, self.arguments, server_id=self.server_canonical_id)

    canonical_id = stable_id
    fingerprint = stable_id

    @property
    def risk_score(self) -> int:
        score = 0
        for f in self.content_findings:
            if "system-prompt-surface" in f or "hidden-instruction-surface" in f:
                score += 3
            elif "required-freeform-argument" in f:
                score += 2
            else:
                score += 1
        return min(score, 10)


@dataclass
class PermissionProfile:
    runs_as_root: bool = False
    container_privileged: bool = False
    tool_permissions: dict[str, str] = field(default_factory=dict)
    capabilities: list[str] = field(default_factory=list)
    network_access: bool = False
    filesystem_write: bool = False
    shell_access: bool = False
    security_opt: list[str] = field(default_factory=list)

    @property
    def is_elevated(self) -> bool:
        return self.runs_as_root or self.container_privileged or self.shell_access or bool(self.capabilities)

    @property
    def privilege_level(self) -> str:
        if self.container_privileged or "CAP_SYS_ADMIN" in self.capabilities:
            return "critical"
        if self.runs_as_root or self.shell_access:
            return "high"
        if self.filesystem_write or self.network_access or self.capabilities:
            return "medium"
        return "low"


# ── MCPServer ───────────────────────────────────────────────────────────────


@dataclass
class MCPServer:
    """An MCP server with its tools, resources and dependencies."""

    name: str
    command: str = ""
    args: list[str] = field(default_factory=list)
    env: dict[str, str] = field(default_factory=dict)
    transport: TransportType = TransportType.STDIO
    url: Optional[str] = None
    tools: list[MCPTool] = field(default_factory=list)
    resources: list[MCPResource] = field(default_factory=list)
    prompts: list[MCPPrompt] = field(default_factory=list)
    packages: list[Package] = field(default_factory=list)
    config_path: Optional[str] = None
    working_dir: Optional[str] = None
    mcp_version: Optional[str] = None
    registry_verified: bool = False
    registry_id: Optional[str] = None
    permission_profile: Optional[PermissionProfile] = None
    security_blocked: bool = False
    security_warnings: list[str] = field(default_factory=list)
    security_intelligence: list[dict[str, object]] = field(default_factory=list)
    surface: ServerSurface = ServerSurface.MCP
    discovery_sources: list[str] = field(default_factory=list)
    discovery_provenance: Optional[dict[str, Any]] = None

    def __post_init__(self) -> None:
        self.stamp_child_identities()

    def stamp_child_identities(self) -> None:
        scope = self.canonical_id
        for child in (*self.tools, *self.resources, *self.prompts):
            if hasattr(child, "server_canonical_id"):
                child.server_canonical_id = scope

    @property
    def stable_id(self) -> str:
        return canonical_mcp_server_id(
            self.name, self.command, registry_id=self.registry_id, url=self.url, args=self.args
        )

    @property
    def canonical_id(self) -> str:
        return self.stable_id

    @property
    def auth_mode(self) -> str:
        if self.credential_names:
            return "env-credentials"
        if self.url and "@" in self.url:
            return "url-embedded-credentials"
        if self.url:
            return "network-no-auth-observed"
        return "local-stdio"

    @property
    def fingerprint(self) -> str:
        raw = json.dumps(
            {
                "registry_id": self.registry_id,
                "name": self.name,
                "command": self.command,
                "args": self.args,
                "url": self.url,
                "transport": self.transport.value,
                "auth_mode": self.auth_mode,
                "credential_refs": sorted(self.credential_names),
                "tool_ids": sorted(t.stable_id for t in self.tools),
                "resource_ids": sorted(r.stable_id for r in self.resources),
                "prompt_ids": sorted(p.stable_id for p in self.prompts),
            },
            sort_keys=True,
            separators=(",", ":"),
        )
        return str(_uuid.uuid5(AGENT_BOM_ID_NAMESPACE, f"mcp_server_fingerprint:{raw}"))

    @property
    def vulnerable_packages(self) -> list[Package]:
        return [p for p in self.packages if p.has_vulnerabilities]

    @property
    def total_vulnerabilities(self) -> int:
        return sum(len(p.vulnerabilities) for p in self.packages)

    @property
    def has_credentials(self) -> bool:
        return any(is_credential_key(k) for k in self.env)

    @property
    def credential_names(self) -> list[str]:
        return [k for k in self.env if is_credential_key(k)]

    @property
    def is_mcp_surface(self) -> bool:
        return self.surface == ServerSurface.MCP


# ── Agent ───────────────────────────────────────────────────────────────────


@dataclass
class Agent:
    """An AI agent (client) that connects to MCP servers."""

    name: str
    agent_type: AgentType
    config_path: str
    mcp_servers: list[MCPServer] = field(default_factory=list)
    version: Optional[str] = None
    source: Optional[str] = None
    status: AgentStatus = AgentStatus.CONFIGURED
    discovered_at: str = field(default_factory=utc_now_iso)
    last_seen: Optional[str] = None
    parent_agent: Optional[str] = None
    metadata: dict[str, object] = field(default_factory=dict)
    automation_settings: list[Any] = field(default_factory=list)
    discovery_provenance: Optional[dict[str, Any]] = None
    discovery_envelope: Optional[dict[str, Any]] = None
    source_id: Optional[str] = None
    device_fingerprint: Optional[str] = None

    def __post_init__(self) -> None:
        if not self.discovered_at:
            self.discovered_at = utc_now_iso()
        if not self.last_seen:
            self.last_seen = self.discovered_at

    @property
    def stable_id(self) -> str:
        return canonical_agent_id(
            self.agent_type.value,
            self.name,
            source_id=self.source_id or "",
            device_fingerprint=self.device_fingerprint or "",
            config_path=self.config_path,
        )

    @property
    def canonical_id(self) -> str:
        return self.stable_id

    @property
    def total_packages(self) -> int:
        return sum(len(s.packages) for s in self.mcp_servers)

    @property
    def total_vulnerabilities(self) -> int:
        return sum(s.total_vulnerabilities for s in self.mcp_servers)

    @property
    def affected_servers(self) -> list[MCPServer]:
        return [s for s in self.mcp_servers if s.vulnerable_packages]

    @property
    def servers_with_credentials(self) -> list[MCPServer]:
        return [s for s in self.mcp_servers if s.has_credentials]
