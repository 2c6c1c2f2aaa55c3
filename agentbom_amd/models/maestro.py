"""MAESTRO 7-layer agentic-AI classification of findings.

Reference parity: src/agent_bom/maestro.py — every finding is tagged with
the MAESTRO knowledge-component (KC) layer it affects so outputs
cross-reference cleanly against MAESTRO-based threat modelers (alongside
the ATT&CK / ATLAS tags the compliance layer already stamps).

Classification here keys on this build's own Finding vocabulary
(FindingType + asset_type + category keywords) instead of the
reference's source strings; the KC layer semantics are identical.
"""

from __future__ import annotations

from enum import Enum
from typing import Any


class MaestroLayer(str, Enum):
    KC1_AI_MODELS = "KC1: AI Models"
    KC2_AGENT_ARCHITECTURE = "KC2: Agent Architecture"
    KC3_AGENTIC_PATTERNS = "KC3: Agentic Patterns"
    KC4_MEMORY_CONTEXT = "KC4: Memory & Context"
    KC5_TOOLS_CAPABILITIES = "KC5: Tools & Capabilities"
    KC6_INFRASTRUCTURE = "KC6: Infrastructure"


LAYER_DESCRIPTIONS = {
    MaestroLayer.KC1_AI_MODELS: "LLM, foundation models, embedding models",
    MaestroLayer.KC2_AGENT_ARCHITECTURE: "AI agents, orchestrators, multi-agent patterns",
    MaestroLayer.KC3_AGENTIC_PATTERNS: "RAG pipelines, prompts, guardrails",
    MaestroLayer.KC4_MEMORY_CONTEXT: "Vector databases, memory stores, knowledge bases",
    MaestroLayer.KC5_TOOLS_CAPABILITIES: "MCP servers, tool APIs, capability scope",
    MaestroLayer.KC6_INFRASTRUCTURE: "Cloud infra, networking, compute, deployments",
}

# FindingType value -> layer (the common, unambiguous cases)
_TYPE_TO_LAYER = {
    "MALICIOUS_MODEL": MaestroLayer.KC1_AI_MODELS,
    "MODEL_INTEGRITY": MaestroLayer.KC1_AI_MODELS,
    "TOOL_DRIFT": MaestroLayer.KC3_AGENTIC_PATTERNS,
    "INJECTION": MaestroLayer.KC3_AGENTIC_PATTERNS,
    "PROMPT_SECURITY": MaestroLayer.KC3_AGENTIC_PATTERNS,
    "CLOAKING": MaestroLayer.KC3_AGENTIC_PATTERNS,
    "SKILL_RISK": MaestroLayer.KC3_AGENTIC_PATTERNS,
    "CREDENTIAL_EXPOSURE": MaestroLayer.KC5_TOOLS_CAPABILITIES,
    "MCP_BLOCKLIST": MaestroLayer.KC5_TOOLS_CAPABILITIES,
    "RATE_LIMIT": MaestroLayer.KC5_TOOLS_CAPABILITIES,
    "BROWSER_EXT": MaestroLayer.KC5_TOOLS_CAPABILITIES,
    "CIS_FAIL": MaestroLayer.KC6_INFRASTRUCTURE,
    "CIS_ERROR": MaestroLayer.KC6_INFRASTRUCTURE,
    "CLOUD_BEST_PRACTICE_FAIL": MaestroLayer.KC6_INFRASTRUCTURE,
    "CLOUD_BEST_PRACTICE_ERROR": MaestroLayer.KC6_INFRASTRUCTURE,
    "CIEM_OVER_PRIVILEGE": MaestroLayer.KC6_INFRASTRUCTURE,
    "SENSITIVE_DATA": MaestroLayer.KC4_MEMORY_CONTEXT,
    "EXFILTRATION": MaestroLayer.KC4_MEMORY_CONTEXT,
}

_VECTOR_DB_HINTS = ("qdrant", "weaviate", "chroma", "milvus", "pinecone",
                    "pgvector", "faiss", "vector")
_MODEL_HINTS = ("huggingface", "ollama", "model", "checkpoint", "safetensors",
                "pickle", "gguf")


def classify_finding(finding: Any) -> MaestroLayer:
    """MAESTRO layer for one Finding (object or to_dict form)."""
    ftype = str(getattr(finding, "finding_type", None)
                or (finding.get("finding_type") if isinstance(finding, dict)
                    else "") or "")
    ftype = getattr(getattr(finding, "finding_type", None), "value", ftype) \
        if not isinstance(finding, dict) else ftype
    layer = _TYPE_TO_LAYER.get(ftype)
    if layer is not None:
        return layer

    asset = finding.get("asset", {}) if isinstance(finding, dict) \
        else getattr(finding, "asset", None)
    asset_type = (asset.get("asset_type", "") if isinstance(asset, dict)
                  else str(getattr(asset, "asset_type", "") or "")).lower()
    asset_name = (asset.get("name", "") if isinstance(asset, dict)
                  else str(getattr(asset, "name", "") or "")).lower()

    blob = f"{asset_type} {asset_name}".lower()
    if any(h in blob for h in _VECTOR_DB_HINTS):
        return MaestroLayer.KC4_MEMORY_CONTEXT
    if any(h in blob for h in _MODEL_HINTS):
        return MaestroLayer.KC1_AI_MODELS
    if asset_type == "agent":
        return MaestroLayer.KC2_AGENT_ARCHITECTURE
    if asset_type in ("mcp_server", "tool"):
        return MaestroLayer.KC5_TOOLS_CAPABILITIES
    if asset_type in ("cloud_resource", "container"):
        return MaestroLayer.KC6_INFRASTRUCTURE
    if ftype == "CVE" and asset_type == "package":
        # package CVEs ride the tool/capability surface they are installed on
        return MaestroLayer.KC5_TOOLS_CAPABILITIES
    return MaestroLayer.KC6_INFRASTRUCTURE


def layer_label(layer: MaestroLayer) -> str:
    desc = LAYER_DESCRIPTIONS.get(layer, "")
    return f"{layer.value} ({desc})" if desc else layer.value


def maestro_summary(findings: list[Any]) -> dict[str, Any]:
    """Per-layer histogram + worst severity for the report JSON."""
    order = {"critical": 0, "high": 1, "medium": 2, "low": 3, "info": 4}
    acc: dict[str, dict[str, Any]] = {}
    for f in findings:
        layer = classify_finding(f)
        sev = str(f.get("severity") if isinstance(f, dict)
                  else getattr(f, "severity", "info") or "info").lower()
        row = acc.setdefault(layer.value, {"layer": layer.value,
                                           "description": LAYER_DESCRIPTIONS[layer],
                                           "findings": 0,
                                           "worst_severity": "info"})
        row["findings"] += 1
        if order.get(sev, 9) < order.get(row["worst_severity"], 9):
            row["worst_severity"] = sev
    return {"schema_version": "1",
            "layers": sorted(acc.values(), key=lambda r: r["layer"])}
