"""Toxic-combination detection over the unified graph.

Reference: src/agent_bom/graph/toxic_findings.py (764 LoC) +
toxic_combos.py (561 LoC) — multi-condition findings: conditions that are
individually tolerable but exploitable when chained on one path
(e.g. critical CVE + exposed credential + execute-capable tool on the
same server; internet exposure + data store reach; shared credential +
vulnerable lateral hub).  Emitted as COMBINATION findings so they reach
--fail-on-severity and every machine output.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

from agentbom_amd.graph.container import UnifiedGraph
from agentbom_amd.graph.types import EntityType, RelationshipType
from agentbom_amd.models.finding import Asset, Finding, FindingSource, FindingType, stable_id


@dataclass
class ToxicCombination:
    id: str
    name: str
    severity: str
    nodes: list[str]
    conditions: list[str]
    narrative: str
    score: float = 0.0

    def to_dict(self) -> dict[str, Any]:
        return {
            "id": self.id, "name": self.name, "severity": self.severity,
            "nodes": self.nodes, "conditions": self.conditions,
            "narrative": self.narrative, "score": self.score,
        }


def _vuln_severity(graph: UnifiedGraph, v_id: str) -> str:
    return str(graph.nodes[v_id].properties.get("severity") or "unknown")


def detect_toxic_combinations(graph: UnifiedGraph, max_results: int = 200) -> list[ToxicCombination]:
    out: list[ToxicCombination] = []

    # ── combo 1: critical/high CVE + credential + execute tool on one server
    for s_id, node in sorted(graph.nodes.items()):
        if node.entity_type != EntityType.SERVER:
            continue
        creds = [v for v, e in graph._out(s_id, {RelationshipType.EXPOSES_CRED})]
        tools = [v for v, e in graph._out(s_id, {RelationshipType.PROVIDES_TOOL})]
        exec_tools = [t for t in tools if any(
            k in graph.nodes[t].label.lower()
            for k in ("exec", "shell", "run", "eval", "command", "sql"))]
        vulns: list[str] = []
        for p, e in graph._out(s_id, {RelationshipType.DEPENDS_ON, RelationshipType.CONTAINS}):
            if graph.nodes[p].entity_type != EntityType.PACKAGE:
                continue
            for v, e2 in graph._out(p, {RelationshipType.VULNERABLE_TO}):
                if _vuln_severity(graph, v) in ("critical", "high"):
                    vulns.append(v)
        if vulns and creds and exec_tools:
            worst = "critical" if any(_vuln_severity(graph, v) == "critical" for v in vulns) else "high"
            out.append(ToxicCombination(
                id=stable_id("toxic", "rce-cred-tool", s_id, *sorted(vulns)),
                name="exploitable-server-with-credentials-and-execution",
                severity=worst,
                nodes=[s_id, *sorted(set(vulns))[:5], *creds[:5], *exec_tools[:5]],
                conditions=[
                    f"{len(vulns)} critical/high CVE(s) on server packages",
                    f"{len(creds)} credential(s) in server environment",
                    f"{len(exec_tools)} execution-capable tool(s)",
                ],
                narrative=(
                    f"Server {graph.nodes[s_id].label!r} chains a {worst} CVE, "
                    f"{len(creds)} exposed credential(s), and execution-capable "
                    f"tools — a single exploited request reaches code execution "
                    "with live credentials."
                ),
                score=9.0 if worst == "critical" else 7.5,
            ))

    # ── combo 2: shared credential + any vulnerable server on either side
    cred_edges: dict[str, list[tuple[str, str]]] = {}
    for e in graph.edges:
        if e.relationship == RelationshipType.SHARES_CRED:
            cred_edges.setdefault(e.evidence or "", []).append((e.source, e.target))
    for cred, pairs in sorted(cred_edges.items()):
        involved_agents = sorted({a for pair in pairs for a in pair})
        vulnerable = []
        for a in involved_agents:
            for s, e in graph._out(a, {RelationshipType.USES}):
                for p, e2 in graph._out(s, {RelationshipType.DEPENDS_ON}):
                    for v, e3 in graph._out(p, {RelationshipType.VULNERABLE_TO}):
                        if _vuln_severity(graph, v) in ("critical", "high"):
                            vulnerable.append((a, v))
        if vulnerable:
            out.append(ToxicCombination(
                id=stable_id("toxic", "shared-cred-lateral", cred, *involved_agents),
                name="shared-credential-lateral-blast",
                severity="high",
                nodes=[cred, *involved_agents[:5]],
                conditions=[
                    f"credential shared by {len(involved_agents)} agents",
                    f"{len(set(v for _a, v in vulnerable))} critical/high CVE(s) reachable",
                ],
                narrative=(
                    f"Credential {cred!r} is shared across {len(involved_agents)} "
                    "agents while at least one of them depends on a critical/high "
                    "CVE — compromise of one agent laterally exposes all of them."
                ),
                score=7.5,
            ))

    # ── combo 3: malicious package anywhere near a credential
    for p_id, node in sorted(graph.nodes.items()):
        if node.entity_type != EntityType.PACKAGE or not node.properties.get("is_malicious"):
            continue
        servers = [s for s, e in graph._out(p_id, {RelationshipType.DEPENDS_ON,
                                                   RelationshipType.CONTAINS}, reverse=True)]
        creds = [c for s in servers for c, e in graph._out(s, {RelationshipType.EXPOSES_CRED})]
        if creds:
            out.append(ToxicCombination(
                id=stable_id("toxic", "malicious-cred", p_id),
                name="malicious-package-with-credential-reach",
                severity="critical",
                nodes=[p_id, *servers[:3], *creds[:5]],
                conditions=["known-malicious package installed",
                            f"{len(creds)} credential(s) on its server(s)"],
                narrative=(
                    f"Known-malicious package {node.label!r} runs on a server "
                    f"holding {len(creds)} credential(s) — assume active "
                    "exfiltration."
                ),
                score=9.8,
            ))

    out.sort(key=lambda t: (-t.score, t.id))
    return out[:max_results]


def toxic_combination_to_finding(combo: ToxicCombination) -> Finding:
    return Finding(
        finding_type=FindingType.COMBINATION,
        source=FindingSource.GRAPH_ANALYSIS,
        asset=Asset(name=combo.nodes[0] if combo.nodes else "estate",
                    asset_type="graph_path", identifier=combo.id),
        severity=combo.severity,
        title=f"Toxic combination: {combo.name}",
        description=combo.narrative,
        evidence={"conditions": combo.conditions, "nodes": combo.nodes,
                  "combo_score": combo.score},
        risk_score=combo.score,
        is_actionable=True,
        id=combo.id,
    )
