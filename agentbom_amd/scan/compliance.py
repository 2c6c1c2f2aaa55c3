"""Compliance tagging across 16 frameworks.

Reference: src/agent_bom/{owasp,owasp_mcp,owasp_agentic,atlas,mitre_attack,
nist_ai_rmf,nist_csf,nist_800_53,fedramp,eu_ai_act,iso_27001,soc2,
cis_controls,cmmc,pci_dss}.py + compliance_coverage.py +
package_scan.py:1220 apply_framework_tags.

One tagger per framework, driven by blast-radius signals (severity, KEV,
credentials, tool capabilities, CWE class, package family) so a framework
can never be registered without a live tagger.  ``apply_framework_tags``
stamps all 16 onto a BlastRadius in one place.
"""

from __future__ import annotations

from agentbom_amd.models import BlastRadius, Severity
from agentbom_amd.models.cwe_impact import classify_cwe_impact
from agentbom_amd.scan.risk import ToolCapability, classify_mcp_tool

_HIGH = {Severity.CRITICAL, Severity.HIGH}

_AI_PACKAGES = frozenset({
    "langchain", "langchain-core", "langgraph", "llama-index", "transformers",
    "torch", "tensorflow", "keras", "openai", "anthropic", "autogen", "crewai",
    "semantic-kernel", "haystack", "dspy", "guidance", "litellm", "vllm",
})
_TRAINING_DATA_PACKAGES = frozenset({
    "datasets", "pandas", "numpy", "pyarrow", "dvc", "mlflow", "wandb",
})

# ── Control catalogs (code -> label), used by coverage tables/labels ───────

OWASP_LLM_TOP10 = {
    "LLM01": "Prompt Injection", "LLM02": "Insecure Output Handling",
    "LLM03": "Training Data Poisoning", "LLM04": "Model Denial of Service",
    "LLM05": "Supply Chain Vulnerabilities", "LLM06": "Sensitive Information Disclosure",
    "LLM07": "Insecure Plugin Design", "LLM08": "Excessive Agency",
    "LLM09": "Overreliance", "LLM10": "Model Theft",
}
OWASP_MCP_TOP10 = {
    "MCP01": "Tool Poisoning", "MCP02": "Credential Exposure",
    "MCP03": "Supply Chain Dependencies", "MCP04": "Excessive Tool Permissions",
    "MCP05": "Unvalidated Tool Responses", "MCP06": "Server Impersonation",
    "MCP07": "Cross-Server Escalation", "MCP08": "Transport Security",
    "MCP09": "Insufficient Audit", "MCP10": "Shadow Servers",
}
OWASP_AGENTIC_TOP10 = {
    "ASI01": "Agent Goal Hijacking", "ASI02": "Tool Misuse",
    "ASI03": "Identity Spoofing", "ASI04": "Supply Chain Compromise",
    "ASI05": "Cascading Failures", "ASI06": "Memory Poisoning",
    "ASI07": "Delegation Abuse", "ASI08": "Excessive Autonomy",
    "ASI09": "Insufficient Observability", "ASI10": "Rogue Agents",
}

FRAMEWORK_REGISTRY = [
    "owasp_llm", "owasp_mcp", "owasp_agentic", "mitre_atlas", "mitre_attack",
    "nist_ai_rmf", "nist_csf", "nist_800_53", "fedramp", "eu_ai_act",
    "iso_27001", "soc2", "cis", "cmmc", "pci_dss", "aisvs",
]


def _caps(br: BlastRadius) -> set[ToolCapability]:
    caps: set[ToolCapability] = set()
    for t in br.exposed_tools:
        caps |= classify_mcp_tool(t)
    return caps


def _impact(br: BlastRadius) -> str:
    return br.impact_category or classify_cwe_impact(br.vulnerability.cwe_ids)


# ── per-framework taggers ──────────────────────────────────────────────────


def tag_owasp_llm(br: BlastRadius) -> list[str]:
    tags = {"LLM05"}  # every third-party CVE is supply chain per OWASP
    if br.exposed_credentials:
        tags.add("LLM06")
    caps = _caps(br)
    if ToolCapability.EXECUTE in caps:
        tags.add("LLM02")
    if ToolCapability.READ in caps:
        tags.add("LLM07")
    if len(br.exposed_tools) > 5 and br.vulnerability.severity in _HIGH:
        tags.add("LLM08")
    name = br.package.name.lower()
    if name in _TRAINING_DATA_PACKAGES:
        tags.add("LLM03")
    if name in _AI_PACKAGES and br.vulnerability.severity in _HIGH:
        tags.add("LLM04")
    return sorted(tags)


def tag_owasp_mcp(br: BlastRadius) -> list[str]:
    tags = {"MCP03"}
    if br.exposed_credentials:
        tags.add("MCP02")
    if len(br.exposed_tools) > 5:
        tags.add("MCP04")
    if br.hop_depth > 1 or br.transitive_agents:
        tags.add("MCP07")
    if any(s.transport.value in ("sse", "streamable-http") and not (s.url or "").startswith("https")
           for s in br.affected_servers if s.url):
        tags.add("MCP08")
    if br.package.is_malicious:
        tags.add("MCP01")
    return sorted(tags)


def tag_owasp_agentic(br: BlastRadius) -> list[str]:
    tags = {"ASI04"}
    caps = _caps(br)
    if ToolCapability.EXECUTE in caps and br.vulnerability.severity in _HIGH:
        tags.add("ASI02")
    if br.transitive_agents:
        tags.add("ASI07")
    if len(br.affected_agents) > 2:
        tags.add("ASI05")
    if len(br.exposed_tools) > 5 and ToolCapability.EXECUTE in caps:
        tags.add("ASI08")
    return sorted(tags)


def tag_atlas(br: BlastRadius) -> list[str]:
    tags = {"AML.T0010"}  # ML supply chain compromise
    if br.package.name.lower() in _AI_PACKAGES:
        tags.add("AML.T0010.001")  # ML software
    if br.exposed_credentials:
        tags.add("AML.T0012")  # valid accounts
    if _impact(br) == "code-execution":
        tags.add("AML.T0011")  # user execution
    if br.package.is_malicious:
        tags.add("AML.T0010.002")  # poisoned dependency
    return sorted(tags)


def tag_attack(br: BlastRadius) -> list[str]:
    tags = {"T1195.001"}  # supply chain: dev tools/dependencies
    impact = _impact(br)
    if impact == "code-execution":
        tags.add("T1203")  # exploitation for client execution
    if impact == "credential-access" or br.exposed_credentials:
        tags.add("T1552")  # unsecured credentials
    if impact == "ssrf":
        tags.add("T1090")
    if impact == "availability":
        tags.add("T1499")  # endpoint DoS
    if br.vulnerability.network_exploitable:
        tags.add("T1190")  # exploit public-facing application
    caps = _caps(br)
    if ToolCapability.EXECUTE in caps:
        tags.add("T1059")  # command and scripting interpreter
    if br.transitive_agents:
        tags.add("T1021")  # lateral movement
    return sorted(tags)


def tag_nist_ai_rmf(br: BlastRadius) -> list[str]:
    tags = {"MAP-3.5", "MEASURE-2.6"}  # third-party risk, security eval
    if br.package.name.lower() in _AI_PACKAGES:
        tags.add("GOVERN-6.1")  # third-party AI resources policy
    if br.exposed_credentials:
        tags.add("MANAGE-2.2")
    return sorted(tags)


def tag_nist_csf(br: BlastRadius) -> list[str]:
    tags = {"ID.RA-01"}  # vulnerabilities identified
    if br.vulnerability.is_kev or (br.vulnerability.epss_score or 0) >= 0.5:
        tags.add("ID.RA-02")  # threat intel received
    if br.vulnerability.fixed_version:
        tags.add("ID.RA-06")  # responses prioritized
    if br.exposed_credentials:
        tags.add("PR.AA-01")  # identities/credentials managed
    if br.package.is_malicious:
        tags.add("ID.RA-03")
    return sorted(tags)


_NIST_IMPACT_CONTROLS = {
    "code-execution": ["RA-5", "SI-2", "CM-7"],
    "credential-access": ["RA-5", "SI-2", "IA-5"],
    "file-access": ["RA-5", "SI-2", "AC-6"],
    "injection": ["RA-5", "SI-2", "SI-10"],
    "ssrf": ["RA-5", "SI-2", "SC-7"],
    "data-leak": ["RA-5", "SI-2", "SC-28"],
    "availability": ["RA-5", "SI-2", "SC-5"],
    "client-side": ["RA-5", "SI-2"],
    "unknown": ["RA-5", "SI-2"],
}


def tag_nist_800_53(br: BlastRadius) -> list[str]:
    tags = set(_NIST_IMPACT_CONTROLS.get(_impact(br), ["RA-5", "SI-2"]))
    if br.package.is_malicious:
        tags.add("SR-3")  # supply chain controls
    if br.exposed_credentials:
        tags.add("IA-5")
    return sorted(tags)


def tag_fedramp(br: BlastRadius) -> list[str]:
    # FedRAMP Moderate baseline selects from 800-53; vulnerability-management
    # controls carry over.
    return [t for t in tag_nist_800_53(br) if t in
            {"RA-5", "SI-2", "IA-5", "SC-7", "SC-28", "AC-6", "CM-7", "SR-3"}]


def tag_eu_ai_act(br: BlastRadius) -> list[str]:
    tags = {"ART-15"}  # accuracy, robustness and cybersecurity
    if br.package.name.lower() in _AI_PACKAGES:
        tags.add("ART-9")  # risk management system
    if br.exposed_credentials or _impact(br) == "data-leak":
        tags.add("ART-10")  # data governance
    return sorted(tags)


def tag_iso_27001(br: BlastRadius) -> list[str]:
    tags = {"A.8.8"}  # management of technical vulnerabilities
    if br.exposed_credentials:
        tags.add("A.5.17")  # authentication information
    if br.package.is_malicious:
        tags.add("A.8.7")  # protection against malware
    if _impact(br) == "data-leak":
        tags.add("A.8.12")  # data leakage prevention
    if br.vulnerability.fixed_version:
        tags.add("A.8.19")  # installation of software
    return sorted(tags)


def tag_soc2(br: BlastRadius) -> list[str]:
    tags = {"CC7.1"}  # vulnerability identification
    if br.vulnerability.severity in _HIGH:
        tags.add("CC7.2")  # anomaly monitoring
    if br.exposed_credentials:
        tags.add("CC6.1")  # logical access controls
    if br.package.is_malicious:
        tags.add("CC6.8")  # unauthorized software
    return sorted(tags)


def tag_cis_controls(br: BlastRadius) -> list[str]:
    tags = {"CIS-07.1"}  # vulnerability management process
    if br.vulnerability.fixed_version:
        tags.add("CIS-07.3")  # automated patch management
    if br.package.is_malicious:
        tags.add("CIS-02.3")  # unauthorized software
    if br.exposed_credentials:
        tags.add("CIS-03.3")  # data access control
    return sorted(tags)


def tag_cmmc(br: BlastRadius) -> list[str]:
    tags = {"RA.L2-3.11.2"}  # scan for vulnerabilities
    if br.vulnerability.fixed_version:
        tags.add("SI.L1-3.14.1")  # identify/correct flaws
    if br.exposed_credentials:
        tags.add("IA.L2-3.5.10")  # protect credentials
    return sorted(tags)


def tag_pci_dss(br: BlastRadius) -> list[str]:
    tags = {"Req-6.3"}  # vulnerabilities identified and managed
    if br.vulnerability.severity in _HIGH:
        tags.add("Req-6.3.1")
    if br.vulnerability.fixed_version:
        tags.add("Req-6.3.3")  # patching
    if br.exposed_credentials:
        tags.add("Req-8.6")  # application/system credentials
    return sorted(tags)


def apply_framework_tags(br: BlastRadius) -> None:
    """Stamp every registered framework's tags (one place — a framework can
    not exist in the registry without its tagger firing here)."""
    br.owasp_tags = tag_owasp_llm(br)
    br.atlas_tags = tag_atlas(br)
    br.attack_tags = tag_attack(br)
    br.nist_ai_rmf_tags = tag_nist_ai_rmf(br)
    br.owasp_mcp_tags = tag_owasp_mcp(br)
    br.owasp_agentic_tags = tag_owasp_agentic(br)
    br.eu_ai_act_tags = tag_eu_ai_act(br)
    br.nist_csf_tags = tag_nist_csf(br)
    br.iso_27001_tags = tag_iso_27001(br)
    br.soc2_tags = tag_soc2(br)
    br.cis_tags = tag_cis_controls(br)
    br.cmmc_tags = tag_cmmc(br)
    br.nist_800_53_tags = tag_nist_800_53(br)
    br.fedramp_tags = tag_fedramp(br)
    br.pci_dss_tags = tag_pci_dss(br)


def compliance_coverage_summary() -> dict:
    """Framework registry metadata for the API/compliance surfaces."""
    return {
        "frameworks": FRAMEWORK_REGISTRY,
        "catalogs": {
            "owasp_llm": OWASP_LLM_TOP10,
            "owasp_mcp": OWASP_MCP_TOP10,
            "owasp_agentic": OWASP_AGENTIC_TOP10,
        },
        "count": len(FRAMEWORK_REGISTRY),
    }
