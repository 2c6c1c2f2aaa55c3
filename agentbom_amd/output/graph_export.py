"""Graph exports: dot / mermaid / graphml / cypher / json.

Reference: src/agent_bom/output/graph.py + graph_export.py (DepGraph,
agent-bom graph -f json|dot|mermaid|graphml|cypher).
"""

from __future__ import annotations

from typing import Any

from agentbom_amd.models import AIBOMReport


def build_graph_dict(report: AIBOMReport) -> dict[str, Any]:
    """Lightweight node/edge projection of the estate for exports."""
    nodes: dict[str, dict] = {}
    edges: list[dict] = []

    def add_node(nid: str, kind: str, label: str, **props) -> None:
        nodes.setdefault(nid, {"id": nid, "kind": kind, "label": label, **props})

    for agent in report.agents:
        a_id = f"agent:{agent.name}"
        add_node(a_id, "agent", agent.name, agent_type=agent.agent_type.value)
        for server in agent.mcp_servers:
            s_id = f"server:{server.name}"
            add_node(s_id, "mcp_server", server.name, transport=server.transport.value)
            edges.append({"source": a_id, "target": s_id, "type": "uses"})
            for cred in server.credential_names:
                c_id = f"credential:{cred}"
                add_node(c_id, "credential", cred)
                edges.append({"source": s_id, "target": c_id, "type": "has_credential"})
            for tool in server.tools:
                t_id = f"tool:{server.name}/{tool.name}"
                add_node(t_id, "tool", tool.name)
                edges.append({"source": s_id, "target": t_id, "type": "provides_tool"})
            for pkg in server.packages:
                p_id = f"pkg:{pkg.ecosystem}:{pkg.name}@{pkg.version}"
                add_node(p_id, "package", f"{pkg.name}@{pkg.version}",
                         ecosystem=pkg.ecosystem, is_malicious=pkg.is_malicious)
                edges.append({"source": s_id, "target": p_id, "type": "contains"})
                for v in pkg.vulnerabilities:
                    v_id = f"vuln:{v.id}"
                    add_node(v_id, "vulnerability", v.id, severity=v.severity.value,
                             is_kev=v.is_kev)
                    edges.append({"source": p_id, "target": v_id, "type": "vulnerable_to"})
    return {"nodes": sorted(nodes.values(), key=lambda n: n["id"]), "edges": edges}


def to_dot(report: AIBOMReport) -> str:
    g = build_graph_dict(report)
    shape = {"agent": "box", "mcp_server": "ellipse", "package": "note",
             "vulnerability": "octagon", "credential": "diamond", "tool": "cds"}
    lines = ["digraph agent_bom {", "  rankdir=LR;"]
    for n in g["nodes"]:
        color = "red" if n.get("severity") in ("critical", "high") or n.get("is_malicious") else "black"
        lines.append(
            f'  "{n["id"]}" [label="{n["label"]}", shape={shape.get(n["kind"], "box")}, color={color}];'
        )
    for e in g["edges"]:
        lines.append(f'  "{e["source"]}" -> "{e["target"]}" [label="{e["type"]}"];')
    lines.append("}")
    return "\n".join(lines) + "\n"


def to_mermaid(report: AIBOMReport) -> str:
    g = build_graph_dict(report)
    ids = {n["id"]: f"n{i}" for i, n in enumerate(g["nodes"])}
    lines = ["graph LR"]
    for n in g["nodes"]:
        label = n["label"].replace('"', "'")
        lines.append(f'  {ids[n["id"]]}["{n["kind"]}: {label}"]')
    for e in g["edges"]:
        lines.append(f'  {ids[e["source"]]} -->|{e["type"]}| {ids[e["target"]]}')
    return "\n".join(lines) + "\n"


def to_graphml(report: AIBOMReport) -> str:
    from xml.sax.saxutils import escape

    g = build_graph_dict(report)
    out = [
        '<?xml version="1.0" encoding="UTF-8"?>',
        '<graphml xmlns="http://graphml.graphdrawing.org/xmlns">',
        '  <key id="kind" for="node" attr.name="kind" attr.type="string"/>',
        '  <key id="label" for="node" attr.name="label" attr.type="string"/>',
        '  <key id="type" for="edge" attr.name="type" attr.type="string"/>',
        '  <graph id="agent-bom" edgedefault="directed">',
    ]
    for n in g["nodes"]:
        out.append(f'    <node id="{escape(n["id"])}">')
        out.append(f'      <data key="kind">{escape(n["kind"])}</data>')
        out.append(f'      <data key="label">{escape(n["label"])}</data>')
        out.append("    </node>")
    for i, e in enumerate(g["edges"]):
        out.append(
            f'    <edge id="e{i}" source="{escape(e["source"])}" target="{escape(e["target"])}">'
            f'<data key="type">{escape(e["type"])}</data></edge>'
        )
    out += ["  </graph>", "</graphml>"]
    return "\n".join(out) + "\n"


def to_cypher(report: AIBOMReport) -> str:
    g = build_graph_dict(report)
    lines = []
    for n in g["nodes"]:
        label = n["kind"].replace("-", "_").title().replace("_", "")
        name = n["label"].replace("'", "\\'")
        lines.append(f"MERGE (:{label} {{id: '{n['id']}', name: '{name}'}});")
    for e in g["edges"]:
        rel = e["type"].upper()
        lines.append(
            f"MATCH (a {{id: '{e['source']}'}}), (b {{id: '{e['target']}'}}) "
            f"MERGE (a)-[:{rel}]->(b);"
        )
    return "\n".join(lines) + "\n"
