"""Runtime discovery surfaces, fixtures-driven (no Docker/psutil/cluster).

Docker MCP Toolkit registry+catalog, compose label/image detection, the
process sweep patterns, and kubectl-JSON pod/CRD discovery — each via the
injected seam ('fixtures-driven tests for each')."""

from __future__ import annotations

import pytest
import yaml

from agentbom_amd.scan.discovery_runtime import (
    discover_compose_mcp_servers,
    discover_docker_mcp,
    discover_k8s_mcp_servers,
    discover_running_processes,
)


class TestDockerMcp:
    def _write(self, root, registry, catalog=None):
        root.mkdir(parents=True, exist_ok=True)
        (root / "registry.yaml").write_text(yaml.safe_dump(registry))
        if catalog is not None:
            (root / "catalogs").mkdir(exist_ok=True)
            (root / "catalogs" / "docker-mcp.yaml").write_text(yaml.safe_dump(catalog))

    def test_registry_cross_referenced_with_catalog(self, tmp_path):
        self._write(
            tmp_path / "mcp",
            {"registry": {"github": {}, "postgres": {}}},
            {"registry": {
                "github": {"image": "mcp/github:1.2", "secrets": [{"name": "GITHUB_TOKEN"}]},
                "postgres": {"image": "mcp/postgres:9"},
                "disabled-one": {"image": "mcp/never"},
            }},
        )
        agent = discover_docker_mcp(home=tmp_path / "mcp")
        assert agent is not None and agent.source == "docker-mcp"
        by_name = {s.name: s for s in agent.mcp_servers}
        assert set(by_name) == {"github", "postgres"}  # enabled only
        assert "mcp/github:1.2" in by_name["github"].args
        assert "GITHUB_TOKEN" in by_name["github"].env
        assert by_name["github"].env["GITHUB_TOKEN"] == "***"  # redacted

    def test_missing_registry_returns_none(self, tmp_path):
        assert discover_docker_mcp(home=tmp_path / "nope") is None

    def test_malformed_registry_returns_none(self, tmp_path):
        root = tmp_path / "mcp"
        root.mkdir()
        (root / "registry.yaml").write_text(":\n  - not valid yaml: [")
        assert discover_docker_mcp(home=root) is None

    def test_no_catalog_falls_back_to_default_image(self, tmp_path):
        self._write(tmp_path / "mcp", {"registry": {"fetch": {}}})
        agent = discover_docker_mcp(home=tmp_path / "mcp")
        assert "mcp/fetch" in agent.mcp_servers[0].args


class TestCompose:
    def test_labels_and_image_hints(self, tmp_path):
        (tmp_path / "docker-compose.yml").write_text(yaml.safe_dump({
            "services": {
                "github-mcp": {"image": "org/github-mcp:2", "environment":
                               {"GITHUB_TOKEN": "secret"}},
                "db": {"image": "postgres:16"},
                "tagged": {"image": "internal/tool:1",
                           "labels": {"mcp.transport": "sse", "mcp.port": "8080"}},
            }
        }))
        agent = discover_compose_mcp_servers(str(tmp_path))
        names = {s.name for s in agent.mcp_servers}
        assert names == {"github-mcp", "tagged"}  # plain postgres excluded
        tagged = next(s for s in agent.mcp_servers if s.name == "tagged")
        assert tagged.transport.value == "sse"
        gh = next(s for s in agent.mcp_servers if s.name == "github-mcp")
        assert gh.env.get("GITHUB_TOKEN") == "***"  # redacted

    def test_list_style_labels(self, tmp_path):
        (tmp_path / "compose.yaml").write_text(yaml.safe_dump({
            "services": {"svc": {"image": "x/y:1", "labels": ["mcp.enabled=true"]}}
        }))
        agent = discover_compose_mcp_servers(str(tmp_path))
        assert agent and agent.mcp_servers[0].name == "svc"

    def test_no_compose_file(self, tmp_path):
        assert discover_compose_mcp_servers(str(tmp_path)) is None


class TestProcessSweep:
    FIXTURE = [
        {"pid": 100, "name": "node",
         "cmdline": ["npx", "-y", "@modelcontextprotocol/server-filesystem@1.2.3", "/data"]},
        {"pid": 101, "name": "python",
         "cmdline": ["uvx", "mcp-server-git", "--repo", "/src"]},
        {"pid": 102, "name": "python",
         "cmdline": ["python3", "-m", "mcp_weather"]},
        {"pid": 103, "name": "bash", "cmdline": ["bash", "-lc", "sleep 9"]},
        {"pid": 104, "name": "idle", "cmdline": []},
    ]

    def test_patterns_and_package_attribution(self):
        agent = discover_running_processes(process_iter=iter(self.FIXTURE))
        assert agent is not None and agent.source == "running-processes"
        assert len(agent.mcp_servers) == 3  # bash + empty cmdline excluded
        fs = next(s for s in agent.mcp_servers if "100" in s.name)
        assert fs.packages and fs.packages[0].name == "@modelcontextprotocol/server-filesystem"
        assert fs.packages[0].version == "1.2.3"
        assert fs.packages[0].ecosystem == "npm"
        git = next(s for s in agent.mcp_servers if "101" in s.name)
        assert git.packages and git.packages[0].ecosystem == "pypi"

    def test_no_matches(self):
        assert discover_running_processes(process_iter=iter([
            {"pid": 1, "name": "x", "cmdline": ["ls", "-la"]}])) is None


class TestK8s:
    PODS = {"items": [
        {"metadata": {"name": "mcp-gateway-0", "namespace": "ai",
                      "labels": {"app.mcp.io/role": "server"}},
         "spec": {"containers": [{"name": "srv", "image": "ghcr.io/x/mcp-gw:3"}]}},
        {"metadata": {"name": "web-1", "namespace": "ai", "labels": {"app": "web"}},
         "spec": {"containers": [{"name": "web", "image": "nginx:1"}]}},
        {"metadata": {"name": "tools-2", "namespace": "ai", "labels": {}},
         "spec": {"containers": [{"name": "t", "image": "corp/mcp-tools:7"}]}},
    ]}
    CRDS = {"items": [
        {"metadata": {"name": "weather", "namespace": "ai"},
         "spec": {"image": "corp/weather-mcp:2"}},
    ]}

    def _run(self, *args):
        if "pods" in args:
            return self.PODS
        if "mcpservers.mcp.io" in args:
            return self.CRDS
        return None

    def test_pods_and_crds(self):
        agent = discover_k8s_mcp_servers(namespace="ai", run=self._run)
        assert agent is not None and agent.source == "kubernetes"
        names = {s.name for s in agent.mcp_servers}
        assert "ai/mcp-gateway-0/srv" in names       # label signal
        assert "ai/tools-2/t" in names               # image signal
        assert "crd/ai/weather" in names             # CRD
        assert not any("web-1" in n for n in names)  # plain pod excluded

    def test_no_kubectl(self, monkeypatch):
        monkeypatch.setattr("shutil.which", lambda _: None)
        assert discover_k8s_mcp_servers() is None

    def test_kubectl_errors_are_none(self):
        assert discover_k8s_mcp_servers(run=lambda *a: None) is None


def test_discover_all_includes_runtime_surfaces(monkeypatch, tmp_path):
    """discover_all(include_runtime=True) folds runtime agents in and
    swallows collector crashes (best-effort surfaces)."""
    from agentbom_amd.scan import discovery, discovery_runtime

    monkeypatch.setattr(discovery_runtime, "discover_docker_mcp",
                        lambda home=None: None)
    fake = discover_running_processes(process_iter=iter(TestProcessSweep.FIXTURE))
    monkeypatch.setattr(discovery_runtime, "discover_running_processes",
                        lambda process_iter=None: fake)

    def crash(*a, **k):
        raise RuntimeError("collector exploded")

    monkeypatch.setattr(discovery_runtime, "discover_k8s_mcp_servers", crash)
    monkeypatch.setattr(discovery_runtime, "discover_compose_mcp_servers",
                        lambda project_dir=None: None)
    agents = discovery.discover_all(project_root=str(tmp_path))
    assert any(a.source == "running-processes" for a in agents)
