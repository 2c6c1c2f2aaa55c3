#!/usr/bin/env python3
"""Regenerate docs/graph_contract.json — the demo-estate graph snapshot gate.

The committed snapshot pins node/edge-type counts (and a risk-rank digest)
for the deterministic demo estate; tests/test_graph_contract.py fails on
drift so graph-shape changes are always an explicit, reviewed rebaseline
(reference analog: scripts/rebaseline_graph_edges.py).
"""

import hashlib
import json
import sys
from collections import Counter
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


def build_contract() -> dict:
    from agentbom_amd.graph.builder import build_unified_graph_from_report
    from agentbom_amd.scan.orchestrator import run_demo_scan

    report = run_demo_scan()
    graph = build_unified_graph_from_report(report)
    nodes = Counter(n.entity_type.value for n in graph.nodes.values())
    edges = Counter(e.relationship.value for e in graph.edges)
    rank = [(br.vulnerability.id, f"{br.package.name}@{br.package.version}")
            for br in report.blast_radii]
    return {
        "_comment": "demo-estate graph contract; regenerate with "
                    "scripts/rebaseline_graph_contract.py",
        "total_nodes": len(graph.nodes),
        "total_edges": len(graph.edges),
        "nodes_by_type": dict(sorted(nodes.items())),
        "edges_by_relationship": dict(sorted(edges.items())),
        "finding_count": len(report.blast_radii),
        "risk_rank_digest": hashlib.sha256(
            json.dumps(rank).encode()).hexdigest(),
    }


if __name__ == "__main__":
    out = Path(__file__).resolve().parents[1] / "docs" / "graph_contract.json"
    out.write_text(json.dumps(build_contract(), indent=2) + "\n")
    print(f"wrote {out}")
