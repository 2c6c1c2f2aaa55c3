"""SBOM ingestion: scan an existing CycloneDX / SPDX document.

Reference: src/agent_bom/sbom.py + scanners package-scan synthetic wrapper —
an ingested SBOM becomes a synthetic agent (``sbom:<name>``) with one
dependency-bearing surface so the matching/blast/graph pipeline runs
unchanged; synthetic agents never inflate MCP counts
(ServerSurface.SBOM, models.core.is_mcp_surface).
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Any

from agentbom_amd.models import Agent, AgentType, MCPServer, Package, ServerSurface

_PURL_ECO = {
    "pypi": "pypi", "npm": "npm", "golang": "go", "cargo": "cargo",
    "maven": "maven", "nuget": "nuget", "gem": "rubygems",
    "composer": "composer", "deb": "deb", "rpm": "rpm", "apk": "apk",
    "hex": "hex", "pub": "pub", "swift": "swifturl", "generic": "generic",
}


def _parse_purl(purl: str) -> tuple[str, str, str]:
    """purl -> (ecosystem, name, version)."""
    body = purl.removeprefix("pkg:")
    ptype, _, rest = body.partition("/")
    rest = rest.split("?")[0]
    name, _, version = rest.rpartition("@")
    if not name:
        name, version = rest, ""
    return _PURL_ECO.get(ptype.lower(), ptype.lower()), name, version


def parse_cyclonedx(data: dict[str, Any]) -> list[Package]:
    out = []
    for comp in data.get("components", []) or []:
        if comp.get("type") not in (None, "library", "framework", "application"):
            continue
        purl = comp.get("purl")
        if purl:
            eco, name, version = _parse_purl(purl)
        else:
            eco, name, version = "generic", comp.get("name", ""), comp.get("version", "")
        if not name:
            continue
        lic = None
        for entry in comp.get("licenses", []) or []:
            lic = (entry.get("license") or {}).get("id") or lic
        out.append(Package(name=name, version=version, ecosystem=eco, purl=purl,
                           license=lic, reachability_evidence="lockfile"))
    return out


def parse_spdx(data: dict[str, Any]) -> list[Package]:
    out = []
    for pkg in data.get("packages", []) or []:
        name = pkg.get("name", "")
        version = pkg.get("versionInfo", "")
        purl = None
        eco = "generic"
        for ref in pkg.get("externalRefs", []) or []:
            if ref.get("referenceType") == "purl":
                purl = ref.get("referenceLocator")
                eco, name, version = _parse_purl(purl)
        if not name or name.startswith("SPDXRef"):
            continue
        lic = pkg.get("licenseConcluded")
        out.append(Package(
            name=name, version=version, ecosystem=eco, purl=purl,
            license=None if lic in (None, "NOASSERTION") else lic,
            reachability_evidence="lockfile",
        ))
    return out


def load_sbom(path: str | Path) -> list[Package]:
    data = json.loads(Path(path).read_text())
    if data.get("bomFormat") == "CycloneDX" or "components" in data:
        return parse_cyclonedx(data)
    if str(data.get("spdxVersion", "")).startswith("SPDX") or "packages" in data:
        return parse_spdx(data)
    raise ValueError(f"unrecognized SBOM format in {path}")


def sbom_to_agent(path: str | Path) -> Agent:
    """Wrap an SBOM as a synthetic agent so the scan pipeline runs as-is."""
    packages = load_sbom(path)
    name = Path(path).stem
    server = MCPServer(
        name=f"sbom:{name}",
        command="",
        packages=packages,
        surface=ServerSurface.SBOM,
        config_path=str(path),
    )
    return Agent(
        name=f"sbom:{name}",
        agent_type=AgentType.CUSTOM,
        config_path=str(path),
        mcp_servers=[server],
        source="sbom_ingest",
    )
