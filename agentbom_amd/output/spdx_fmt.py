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
