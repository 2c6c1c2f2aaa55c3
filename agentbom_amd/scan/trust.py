"""Supply-chain trust scoring for packages and MCP servers.

Reference parity: src/agent_bom/trust_score.py — one 0-100 trust score
per package with reasons, combining only signals available offline:
malicious/typosquat checks, advisory density against the local arena,
KEV presence, name heuristics (homoglyph/entropy), registry verification
for MCP server packages, and scorecard fields when the package carries
them.  Grades: A >= 80, B >= 60, C >= 40, D >= 20, F below.
"""

from __future__ import annotations

import math
from collections import Counter
from typing import Any, Optional

from agentbom_amd.models import Package


def _grade(score: float) -> str:
    for cutoff, grade in ((80, "A"), (60, "B"), (40, "C"), (20, "D")):
        if score >= cutoff:
            return grade
    return "F"


def _name_entropy(name: str) -> float:
    counts = Counter(name)
    n = len(name)
    return -sum(c / n * math.log2(c / n) for c in counts.values()) if n else 0.0


_HOMOGLYPH_SETS = ({"0", "o"}, {"1", "l", "i"}, {"rn", "m"})


def trust_score(package: Package,
                advisory_windows: Optional[list] = None) -> dict[str, Any]:
    """0-100 trust score with explicit reasons (higher = more trustworthy)."""
    from agentbom_amd.mcp.registry import check_blocklist, lookup_package
    from agentbom_amd.scan.malicious import check_typosquat, flag_malicious_packages

    score = 70.0  # neutral prior
    reasons: list[dict[str, Any]] = []

    def adjust(delta: float, reason: str) -> None:
        nonlocal score
        score += delta
        reasons.append({"delta": delta, "reason": reason})

    flag_malicious_packages([package])
    if package.is_malicious:
        adjust(-70, f"known-malicious: {package.malicious_reason}")
    typo = check_typosquat(package.name, package.ecosystem)
    if typo:
        adjust(-40, f"typosquat candidate of {typo!r}")
    if check_blocklist(package.name):
        adjust(-50, "registry blocklist pattern match")

    reg = lookup_package(package.name)
    if reg is not None:
        if reg.get("verified"):
            adjust(+15, "verified entry in the known-server registry")
        risk = reg.get("risk_level")
        if risk == "critical":
            adjust(-10, "registry risk level critical (capability surface)")
        elif risk == "high":
            adjust(-5, "registry risk level high")

    # advisory density against the supplied arena windows
    if advisory_windows:
        from agentbom_amd.utils.canonical_ids import normalize_package_name

        norm = normalize_package_name(package.name, package.ecosystem)
        mine = [w for w in advisory_windows
                if w.ecosystem.lower() == package.ecosystem.lower()
                and normalize_package_name(w.package_name, w.ecosystem) == norm]
        vulns = {w.vuln_id for w in mine}
        if len(vulns) >= 10:
            adjust(-15, f"{len(vulns)} known advisories (high historical density)")
        elif len(vulns) >= 3:
            adjust(-8, f"{len(vulns)} known advisories")
        if any(w.is_kev for w in mine):
            adjust(-15, "appears in the CISA KEV catalog")
        unfixed = [w for w in mine if w.unfixed]
        if unfixed:
            adjust(-10, f"{len(unfixed)} advisories with no fixed version")

    # name heuristics
    name = package.name.lower()
    if len(name) <= 2:
        adjust(-8, "very short name (confusion surface)")
    if _name_entropy(name) > 4.0 and len(name) > 16:
        adjust(-10, "high-entropy name (generated-looking)")
    for a in _HOMOGLYPH_SETS:
        present = [g for g in a if g in name]
        if len(present) >= 2:
            adjust(-5, f"mixed homoglyph characters {sorted(present)}")
            break

    # scorecard fields when enrichment populated them
    sc = getattr(package, "scorecard_score", None)
    if isinstance(sc, (int, float)) and sc >= 0:
        adjust((sc - 5.0) * 3, f"OpenSSF scorecard {sc:.1f}/10")

    score = max(0.0, min(100.0, score))
    return {
        "package": f"{package.name}@{package.version}",
        "ecosystem": package.ecosystem,
        "score": round(score, 1),
        "grade": _grade(score),
        "reasons": reasons,
    }
