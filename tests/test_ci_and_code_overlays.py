"""CI workflow scanner, repo inventory, and the code/CI/ASPM/CNAPP overlays."""

from __future__ import annotations

import textwrap

import pytest

from agentbom_amd.graph.builder import (
    apply_report_overlays,
    build_unified_graph_from_report_json,
)
from agentbom_amd.graph.container import UnifiedEdge, UnifiedGraph, UnifiedNode
from agentbom_amd.graph.types import EntityType, RelationshipType
from agentbom_amd.scan.ci_workflows import scan_github_actions
from agentbom_amd.scan.repo_inventory import collect_project_inventory


@pytest.fixture
def repo(tmp_path):
    wf = tmp_path / ".github" / "workflows"
    wf.mkdir(parents=True)
    (wf / "triage.yml").write_text(textwrap.dedent("""\
        name: ai-triage
        on: [push]
        env:
          ANTHROPIC_API_KEY: ${{ secrets.ANTHROPIC_API_KEY }}
        jobs:
          triage:
            runs-on: ubuntu-latest
            steps:
              - uses: actions/checkout@v4
              - uses: anthropic/claude-action@v1
              - run: |
                  pip install anthropic langchain requests
                  python triage.py
    """))
    (wf / "build.yml").write_text(textwrap.dedent("""\
        name: build
        on:
          pull_request_target:
        permissions:
          contents: read
        jobs:
          build:
            runs-on: ubuntu-latest
            steps:
              - uses: actions/checkout@8f4b7f84864484a7bf31766abe9204da3cbe65b3
              - run: make test
    """))
    src = tmp_path / "svc"
    src.mkdir()
    (src / "app.py").write_text("import requests\nimport os\nfrom svc import util\n")
    (src / "util.py").write_text("import json\n")
    (src / "requirements.txt").write_text("requests==2.31.0\n")
    (tmp_path / "deploy.yaml").write_text("kind: Deployment\n")
    return tmp_path


class TestGithubActionsScan:
    def test_ai_workflow_becomes_agent(self, repo):
        agents, warnings = scan_github_actions(str(repo))
        assert [a.name for a in agents] == ["gha:triage"]
        agent = agents[0]
        assert agent.source == "github-actions"
        srv = agent.mcp_servers[0]
        assert srv.env == {"ANTHROPIC_API_KEY": "***REDACTED***"}
        pkgs = {(p.name, p.ecosystem) for p in srv.packages}
        assert ("anthropic", "pypi") in pkgs and ("langchain", "pypi") in pkgs
        assert ("requests", "pypi") not in pkgs  # not an AI SDK
        tool_names = {t.name for t in srv.tools}
        assert any("Anthropic action" in t for t in tool_names)
        assert any("Anthropic SDK" == t for t in tool_names)
        # CI surface must not inflate MCP counts
        assert not srv.is_mcp_surface

    def test_hardening_warnings(self, repo):
        _, warnings = scan_github_actions(str(repo))
        text = "\n".join(warnings)
        # triage.yml: no permissions + two unpinned actions
        assert "triage.yml: no top-level permissions" in text
        assert "unpinned action actions/checkout@v4" in text
        assert "unpinned action anthropic/claude-action@v1" in text
        # build.yml: SHA-pinned action must NOT warn; pull_request_target must
        assert "unpinned action actions/checkout@8f4b" not in text
        assert "build.yml: pull_request_target" in text
        assert "AI credentials exposed in triage.yml: ANTHROPIC_API_KEY" in text

    def test_no_workflows_dir(self, tmp_path):
        agents, warnings = scan_github_actions(str(tmp_path))
        assert agents == [] and warnings == []


class TestRepoInventory:
    def test_collects_dirs_files_imports(self, repo):
        inv = collect_project_inventory(str(repo))
        dirs = {d["path"]: d for d in inv["directories"]}
        assert dirs["svc"]["source_files"] == 2
        assert dirs["svc"]["manifests"] == ["requirements.txt"]
        files = {f["path"]: f for f in inv["files"]}
        assert files["svc/app.py"]["kind"] == "source"
        # real-AST import extraction; relative/local handled at overlay stage
        assert set(files["svc/app.py"]["imports"]) == {"requests", "os", "svc"}
        assert files["deploy.yaml"]["kind"] == "config"
        assert not inv["truncated"]

    def test_missing_root(self):
        assert collect_project_inventory("/nonexistent/xyz") is None

    def test_bounded(self, tmp_path):
        for i in range(30):
            (tmp_path / f"f{i:02d}.py").write_text("import os\n")
        inv = collect_project_inventory(str(tmp_path), max_files=10)
        assert len(inv["files"]) <= 10 and inv["truncated"]


def _report_json(repo):
    """Scan the fixture repo into a minimal report JSON for overlay tests."""
    agents, _ = scan_github_actions(str(repo))
    return {
        "agents": [{
            "name": a.name, "source": a.source,
            "config_path": a.config_path,
            "mcp_servers": [{
                "name": s.name,
                "tools": [{"name": t.name} for t in s.tools],
                "packages": [{"name": p.name, "version": p.version,
                              "ecosystem": p.ecosystem}
                             for p in s.packages],
            } for s in a.mcp_servers],
        } for a in agents],
        "project_inventory": collect_project_inventory(str(repo)),
        "findings": [
            {"severity": "high", "cve_id": "CVE-2024-1111",
             "asset": {"name": "requests", "asset_type": "package",
                       "location": "svc/requirements.txt"},
             "reachability": "reachable"},
            {"severity": "high", "cve_id": "CVE-2024-1111",   # duplicate
             "asset": {"name": "requests", "asset_type": "package",
                       "location": "svc/requirements.txt"}},
            {"severity": "low", "title": "world-readable config",
             "asset": {"name": "deploy.yaml", "asset_type": "config",
                       "location": "deploy.yaml"}},
        ],
        "codeowners": {"svc": "@platform-team"},
    }


class TestOverlays:
    def test_repo_structure_and_code_graph(self, repo):
        g = UnifiedGraph()
        res = apply_report_overlays(g, _report_json(repo))
        rs, cg = res["repo_structure"], res["code_graph"]
        assert rs["directories"] >= 2 and rs["files"] >= 5
        assert "dir:svc" in g.nodes and "file:svc/app.py" in g.nodes
        assert g.nodes["file:svc/app.py"].entity_type == EntityType.SOURCE_FILE
        assert g.nodes["file:deploy.yaml"].entity_type == EntityType.CONFIG_FILE
        # module + DEFINES + external imports (svc is local → excluded)
        assert cg["code_modules"] >= 1
        assert "module:svc" in g.nodes
        ext = {n.label for n in g.nodes.values()
               if n.entity_type == EntityType.EXTERNAL_IMPORT}
        assert "requests" in ext and "os" in ext and "svc" not in ext
        rels = {(e.source, e.relationship) for e in g.edges}
        assert ("file:svc/app.py", RelationshipType.DEFINES) in rels

    def test_ci_graph(self, repo):
        g = UnifiedGraph()
        doc = _report_json(repo)
        # place the workflow tool nodes the RUNS edges target
        for a in doc["agents"]:
            for s in a["mcp_servers"]:
                for t in s["tools"]:
                    g.add_node(UnifiedNode(f"tool:x/{t['name']}",
                                           EntityType.TOOL, t["name"]))
        res = apply_report_overlays(g, doc)
        ci = res["ci_graph"]
        assert ci["ci_jobs"] == 1
        assert "ci_job:triage" in g.nodes
        assert ci["configures_edges"] == 1  # workflow file → job
        assert ci["runs_edges"] >= 1

    def test_aspm_apps(self, repo):
        g = UnifiedGraph()
        res = apply_report_overlays(g, _report_json(repo))
        aspm = res["aspm"]
        assert aspm["applications"] == 2          # svc + (repo root)
        assert aspm["deduplicated"] == 1          # duplicate CVE collapsed
        assert aspm["reachable"] == 1
        app = g.nodes["application:svc"]
        assert app.properties["owner"] == "@platform-team"
        assert app.properties["max_severity"] == "high"
        assert app.properties["severity_histogram"] == {"high": 1}

    def test_cnapp_toxic_exposure(self):
        g = UnifiedGraph()
        g.add_node(UnifiedNode("ds:1", EntityType.DATA_STORE,
                               "customer-payments-db",
                               properties={"internet_exposed": True}))
        g.add_node(UnifiedNode("ds:2", EntityType.DATA_STORE, "telemetry-db"))
        res = apply_report_overlays(g, {})
        cn = res["cnapp"]
        assert cn["sensitive_nodes"] == 1 and cn["exposed_sensitive"] == 1
        ds = g.nodes["ds:1"]
        assert ds.properties["data_sensitivity"] == "sensitive"
        assert "pci" in ds.properties["regulatory_frameworks"]
        assert ds.properties["toxic_exposed_sensitive"] is True
        f_id = "misconfig:cnapp-exposed-sensitive:ds:1"
        assert f_id in g.nodes
        assert any(e.source == f_id and e.target == "ds:1"
                   and e.relationship == RelationshipType.AFFECTS
                   for e in g.edges)
        assert "data_sensitivity" not in g.nodes["ds:2"].properties

    def test_idempotent(self, repo):
        g = UnifiedGraph()
        doc = _report_json(repo)
        apply_report_overlays(g, doc)
        n_nodes, n_edges = len(g.nodes), len(g.edges)
        apply_report_overlays(g, doc)
        assert (len(g.nodes), len(g.edges)) == (n_nodes, n_edges)

    def test_empty_noop(self):
        g = UnifiedGraph()
        res = apply_report_overlays(g, {})
        assert len(g.nodes) == 0 and len(g.edges) == 0
        assert all(all(v == 0 for v in c.values()) for c in res.values())

    def test_json_builder_applies_overlays(self, repo):
        doc = _report_json(repo)
        g = build_unified_graph_from_report_json(doc)
        assert "ci_job:triage" in g.nodes
        assert "application:svc" in g.nodes
        assert "module:svc" in g.nodes
