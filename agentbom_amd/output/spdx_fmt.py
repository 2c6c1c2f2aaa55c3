"""SPDX 2.3 SBOM export (reference: src/agent_bom/output/spdx2_fmt.py)."""

from __future__ import annotations

import re
from typing import Any

from agentbom_amd import __version__
from agentbom_amd.models import AIBOMReport


def _spdx_id(text: str) -> str:
    return "SPDXRef-" + re.sub(r"[^A-Za-z0-9.\-]", "-", text)


def to_spdx(report: AIBOMReport) -> dict[str, Any]:
    packages: dict[str, dict] = {}
    relationships: list[dict] = []
    doc_id = "SPDXRef-DOCUMENT"

    for agent in report.agents:
        for server in agent.mcp_servers:
            for pkg in server.packages:
                sid = _spdx_id(f"{pkg.ecosystem}-{pkg.name}-{pkg.version}")
                if sid not in packages:
                    packages[sid] = {
                        "SPDXID": sid,
                        "name": pkg.name,
                        "versionInfo": pkg.version,
                        "downloadLocation": pkg.download_url or "NOASSERTION",
                        "licenseConcluded": pkg.license or "NOASSERTION",
                        "licenseDeclared": pkg.license or "NOASSERTION",
                        "copyrightText": pkg.copyright_text or "NOASSERTION",
                        "supplier": f"Organization: {pkg.supplier}" if pkg.supplier else "NOASSERTION",
                        "externalRefs": [
                            {
                                "referenceCategory": "PACKAGE-MANAGER",
                                "referenceType": "purl",
                                "referenceLocator": pkg.purl
                                or f"pkg:{pkg.ecosystem}/{pkg.name}@{pkg.version}",
                            }
                        ],
                    }
                    relationships.append(
                        {
                            "spdxElementId": doc_id,
                            "relationshipType": "DESCRIBES",
                            "relatedSpdxElement": sid,
                        }
                    )

    return {
        "spdxVersion": "SPDX-2.3",
        "dataLicense": "CC0-1.0",
        "SPDXID": doc_id,
        "name": f"agent-bom-scan-{report.scan_id or 'local'}",
        "documentNamespace": f"https://agent-bom.dev/spdx/{report.scan_id or 'local'}",
        "creationInfo": {
            "created": report.generated_at.isoformat(),
            "creators": [f"Tool: agent-bom-{__version__}"],
        },
        "packages": sorted(packages.values(), key=lambda p: p["SPDXID"]),
        "relationships": relationships,
    }


def to_spdx3(report: AIBOMReport) -> dict[str, Any]:
    """SPDX 3.0 JSON-LD export (reference ships SPDX 2 and 3 variants).

    Element graph: one SpdxDocument + CreationInfo, one software_Package
    element per distinct package (purl as ExternalIdentifier), and
    ``describes`` relationships from the document."""
    from datetime import datetime, timezone

    created = datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")
    creation_info = {
        "type": "CreationInfo",
        "@id": "_:creationinfo",
        "specVersion": "3.0.1",
        "created": created,
        "createdBy": ["urn:agent-bom:agent"],
    }
    elements: list[dict] = [
        {
            "type": "Tool",
            "spdxId": "urn:agent-bom:agent",
            "name": f"agent-bom {__version__}",
            "creationInfo": "_:creationinfo",
        }
    ]
    described: list[str] = []
    seen: set[str] = set()
    for agent in report.agents:
        for server in agent.mcp_servers:
            for pkg in server.packages:
                purl = pkg.purl or f"pkg:{pkg.ecosystem}/{pkg.name}@{pkg.version}"
                spdx_id = f"urn:agent-bom:pkg:{_spdx_id(f'{pkg.ecosystem}-{pkg.name}-{pkg.version}')}"
                if spdx_id in seen:
                    continue
                seen.add(spdx_id)
                elements.append({
                    "type": "software_Package",
                    "spdxId": spdx_id,
                    "creationInfo": "_:creationinfo",
                    "name": pkg.name,
                    "software_packageVersion": pkg.version,
                    "software_downloadLocation": pkg.download_url or "NOASSERTION",
                    "externalIdentifier": [{
                        "type": "ExternalIdentifier",
                        "externalIdentifierType": "packageUrl",
                        "identifier": purl,
                    }],
                })
                described.append(spdx_id)
    doc_id = "urn:agent-bom:document"
    elements.append({
        "type": "SpdxDocument",
        "spdxId": doc_id,
        "creationInfo": "_:creationinfo",
        "name": f"agent-bom scan {report.scan_id or 'report'}",
        "rootElement": described,
        "profileConformance": ["core", "software"],
    })
    for i, sid in enumerate(described):
        elements.append({
            "type": "Relationship",
            "spdxId": f"urn:agent-bom:rel:{i}",
            "creationInfo": "_:creationinfo",
            "relationshipType": "describes",
            "from": doc_id,
            "to": [sid],
        })
    return {
        "@context": "https://spdx.org/rdf/3.0.1/spdx-context.jsonld",
        "@graph": [creation_info] + elements,
    }
