"""Transitive expansion + registry version resolution (scan/transitive.py).

Pinned-fixture corpus: npm/PyPI/Go registry responses served from
httpx.MockTransport; resolution grammars (caret/tilde/x-range, PEP 440
specifiers, go.mod require blocks) checked against known-good picks.
"""

from __future__ import annotations

import json

import httpx
import pytest

from agentbom_amd.models.core import Package
from agentbom_amd.scan import transitive as tr
from agentbom_amd.utils.http_client import set_offline


@pytest.fixture(autouse=True)
def _online():
    set_offline(False)
    yield
    set_offline(False)


class TestNpmRangeResolution:
    VERSIONS = ["0.0.3", "0.1.0", "0.1.9", "1.0.0", "1.2.0", "1.2.5", "1.3.0",
                "2.0.0", "2.1.4", "3.0.0-beta.1"]

    @pytest.mark.parametrize("spec,expect", [
        ("^1.2.0", "1.3.0"),
        ("^1.2.3", "1.3.0"),
        ("~1.2.0", "1.2.5"),
        ("~1.2", "1.2.5"),
        ("^0.1.0", "0.1.9"),
        ("^0.0.3", "0.0.3"),
        ("1.2.x", "1.2.5"),
        ("2", "2.1.4"),
        (">=1.0.0 <2.0.0", "1.3.0"),
        (">1.2.0 <=1.2.5", "1.2.5"),
        ("1.2.5", "1.2.5"),
        ("*", "2.1.4"),          # prereleases excluded from ranges
        ("^9.0.0", None),
        ("^1.0.0 || ^2.0.0", "1.3.0"),  # first matching alternative wins
    ])
    def test_grammar(self, spec, expect):
        assert tr.resolve_npm_range(spec, self.VERSIONS) == expect

    def test_dist_tag(self):
        assert tr.resolve_npm_range("latest", self.VERSIONS,
                                    {"latest": "2.1.4"}) == "2.1.4"


class TestPipResolution:
    RELEASES = ["1.0", "1.5", "2.0", "2.5", "3.0a1"]

    @pytest.mark.parametrize("spec,expect", [
        (">=1.0,<2.0", "1.5"),
        (">=2.0", "2.5"),
        ("==1.0", "1.0"),
        (">=9", None),
        ("", "2.5"),  # unpinned -> highest stable (prerelease excluded)
    ])
    def test_specifiers(self, spec, expect):
        assert tr.resolve_pip_spec(spec, self.RELEASES) == expect

    def test_requires_dist_markers(self):
        deps = tr.parse_requires_dist([
            "urllib3 (>=1.21.1,<3)",
            'PySocks (!=1.5.7,>=1.5.6) ; extra == "socks"',
            "charset-normalizer (>=2,<4)",
            'colorama ; platform_system == "Windows"',
        ])
        names = [n for n, _ in deps]
        assert "urllib3" in names and "charset-normalizer" in names
        assert "PySocks" not in names  # extra-gated: skipped


def test_go_mod_parse():
    text = """module example.com/m

go 1.21

require (
\tgithub.com/pkg/errors v0.9.1
\tgolang.org/x/net v0.17.0 // indirect
)
require github.com/spf13/cobra v1.8.0
"""
    assert tr.parse_go_mod_requires(text) == [
        ("github.com/pkg/errors", "v0.9.1"),
        ("golang.org/x/net", "v0.17.0"),
        ("github.com/spf13/cobra", "v1.8.0"),
    ]


# ── end-to-end expansion against a pinned registry fixture ──────────────────

NPM_DOCS = {
    "express": {
        "dist-tags": {"latest": "4.18.2"},
        "versions": {
            "4.18.2": {"dependencies": {"accepts": "~1.3.8", "body-parser": "1.20.1"}},
        },
    },
    "accepts": {
        "dist-tags": {"latest": "1.3.8"},
        "versions": {"1.3.8": {"dependencies": {"mime-types": "~2.1.34"}},
                     "1.3.7": {"dependencies": {}}},
    },
    "body-parser": {
        "dist-tags": {"latest": "1.20.1"},
        "versions": {"1.20.1": {"dependencies": {}}},
    },
    "mime-types": {
        "dist-tags": {"latest": "2.1.35"},
        "versions": {"2.1.35": {"dependencies": {}}, "2.1.34": {"dependencies": {}}},
    },
}


def _npm_client():
    def handler(request):
        name = request.url.path.lstrip("/")
        doc = NPM_DOCS.get(name)
        if doc is None:
            return httpx.Response(404)
        return httpx.Response(200, json=doc)

    return httpx.Client(transport=httpx.MockTransport(handler))


def test_expand_transitive_npm_tree():
    root = Package(name="express", version="4.18.2", ecosystem="npm")
    out = tr.expand_transitive([root], max_depth=3, client=_npm_client())
    by_name = {p.name: p for p in out}
    assert set(by_name) == {"accepts", "body-parser", "mime-types"}
    acc = by_name["accepts"]
    assert acc.version == "1.3.8"           # ~1.3.8 resolved against registry
    assert acc.is_direct is False
    assert acc.parent_package == "express@4.18.2"
    assert acc.dependency_depth == 1
    assert by_name["mime-types"].dependency_depth == 2
    assert by_name["mime-types"].parent_package == "accepts@1.3.8"
    assert all(p.resolved_from_registry for p in out)


def test_expand_depth_bound():
    root = Package(name="express", version="4.18.2", ecosystem="npm")
    out = tr.expand_transitive([root], max_depth=1, client=_npm_client())
    assert {p.name for p in out} == {"accepts", "body-parser"}


def test_expand_never_duplicates_existing():
    root = Package(name="express", version="4.18.2", ecosystem="npm")
    existing = Package(name="accepts", version="1.3.7", ecosystem="npm")
    out = tr.expand_transitive([root, existing], client=_npm_client())
    assert "accepts" not in {p.name for p in out}


def test_resolve_versionless_package():
    pkgs = [Package(name="express", version="", ecosystem="npm"),
            Package(name="pinned", version="1.0.0", ecosystem="npm")]
    n = tr.resolve_package_versions(pkgs, client=_npm_client())
    assert n == 1
    assert pkgs[0].version == "4.18.2"
    assert pkgs[0].version_source == "registry"
    assert pkgs[0].resolved_from_registry is True
    assert pkgs[1].version == "1.0.0"  # untouched


def test_orchestrator_transitive_option():
    """scan_agents(transitive=True) expands server packages with provenance;
    registry failure degrades to a warning."""
    from unittest.mock import patch

    from agentbom_amd.models.core import Agent, MCPServer
    from agentbom_amd.scan.orchestrator import ScanOptions, scan_agents

    def mk_agents():
        return [Agent(name="a1", agent_type="claude-desktop", config_path="/tmp/cfg.json", mcp_servers=[
            MCPServer(name="s1", command="npx", args=[], packages=[
                Package(name="express", version="4.18.2", ecosystem="npm")])])]

    with patch("agentbom_amd.utils.http_client.create_client", _npm_client), \
         patch("agentbom_amd.scan.transitive.create_client", _npm_client):
        agents = mk_agents()
        report = scan_agents(agents, [], ScanOptions(transitive=True))
    pkg_names = {p.name for a in agents for s in a.mcp_servers for p in s.packages}
    assert {"express", "accepts", "body-parser", "mime-types"} <= pkg_names
    assert any("transitive expansion: +3 packages" in w for w in report.warnings)

    def boom(*a, **k):
        raise RuntimeError("registry down")

    with patch("agentbom_amd.scan.transitive.expand_transitive", side_effect=boom):
        report = scan_agents(mk_agents(), [], ScanOptions(transitive=True))
    assert any("transitive expansion unavailable" in w for w in report.warnings)
