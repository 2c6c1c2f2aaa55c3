"""Core data model package — re-exports the public model surface."""

from agentbom_amd.models.blast import (
    BlastRadius,
    HOP_RISK_FACTORS,
    active_blast_radii,
    expand_blast_radius_hops,
    is_vex_suppressed,
    risk_score_from_counts,
)
from agentbom_amd.models.core import (
    Agent,
    AgentStatus,
    AgentType,
    CvssVectorSignals,
    MCPPrompt,
    MCPResource,
    MCPServer,
    MCPTool,
    Package,
    PackageOccurrence,
    PermissionProfile,
    SEVERITY_CODE,
    SEVERITY_FROM_CODE,
    SEVERITY_ORDER,
    ServerSurface,
    Severity,
    TransportType,
    Vulnerability,
    compute_confidence,
    is_credential_key,
    merge_advisory_sources,
    parse_cvss_vector_signals,
    utc_now_iso,
)
from agentbom_amd.models.cwe_impact import (
    build_attack_vector_summary,
    classify_cwe_impact,
    filter_credentials_by_impact,
    filter_tools_by_impact,
)
from agentbom_amd.models.finding import (
    Asset,
    ControlTag,
    FRAMEWORK_TAG_FIELDS,
    Finding,
    FindingSource,
    FindingType,
    blast_radius_to_finding,
    fused_triage_priority,
    stable_id,
)
from agentbom_amd.models.report import (
    AIBOMReport,
    ScanIssue,
    ScanOutcome,
    ScanRun,
    ScanScope,
    ScanScopeStatus,
    classify_agent_kind,
)

__all__ = [
    "Agent", "AgentStatus", "AgentType", "AIBOMReport", "Asset", "BlastRadius",
    "ControlTag", "CvssVectorSignals", "FRAMEWORK_TAG_FIELDS", "Finding",
    "FindingSource", "FindingType", "HOP_RISK_FACTORS", "MCPPrompt",
    "MCPResource", "MCPServer", "MCPTool", "Package", "PackageOccurrence",
    "PermissionProfile", "ScanIssue", "ScanOutcome", "ScanRun", "ScanScope",
    "ScanScopeStatus", "SEVERITY_CODE", "SEVERITY_FROM_CODE", "SEVERITY_ORDER",
    "ServerSurface", "Severity", "TransportType", "Vulnerability",
    "active_blast_radii", "blast_radius_to_finding", "build_attack_vector_summary",
    "classify_agent_kind", "classify_cwe_impact", "compute_confidence",
    "expand_blast_radius_hops", "filter_credentials_by_impact",
    "filter_tools_by_impact", "fused_triage_priority", "is_credential_key",
    "is_vex_suppressed", "merge_advisory_sources", "parse_cvss_vector_signals",
    "risk_score_from_counts", "stable_id", "utc_now_iso",
]
