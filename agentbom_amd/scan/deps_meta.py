"""deps.dev / OpenSSF Scorecard supply-chain metadata enrichment.

Reference parity: src/agent_bom/deps_dev.py + scorecard.py (SURVEY.md §2.3
— supply-chain metadata + Scorecard enrichment feeding the risk boost at
models.py:1041-1048).  Air-gapped design: metadata comes from a LOCAL
bundle file (the shape a `deps.dev` bulk export produces) plus an
injectable fetcher for online deployments; lookups that find nothing set
``scorecard_lookup_state`` so an absent score is never mistaken for a
good one (same fail-visible rule as scan coverage).

Bundle shape (``AGENT_BOM_DEPS_BUNDLE`` or operator-passed path)::

    {"packages": {"pypi:requests": {
        "scorecard_score": 9.1,
        "scorecard_checks": {"Maintained": 10, "Code-Review": 8},
        "scorecard_repo": "github.com/psf/requests",
        "maintainer_count": 5, "license": "Apache-2.0",
        "deprecated": false, "source_repo": "github.com/psf/requests"}}}
"""

from __future__ import annotations

import json
import os
from pathlib import Path
from typing import Any, Callable, Optional

from agentbom_amd.utils.canonical_ids import (
    normalize_package_ecosystem,
    normalize_package_name,
)

Fetcher = Callable[[str, str], Optional[dict]]  # (ecosystem, name) -> row


def _bundle_key(ecosystem: str, name: str) -> str:
    eco = normalize_package_ecosystem(ecosystem).lower()
    return f"{eco}:{normalize_package_name(name, ecosystem)}"


def load_deps_bundle(path: Optional[str] = None) -> dict[str, dict]:
    """Load the offline metadata bundle; {} when absent/unreadable."""
    p = path or os.environ.get("AGENT_BOM_DEPS_BUNDLE")
    if not p:
        return {}
    try:
        doc = json.loads(Path(p).read_text())
    except (OSError, ValueError):
        return {}
    pkgs = doc.get("packages") if isinstance(doc, dict) else None
    if not isinstance(pkgs, dict):
        return {}
    return {str(k).lower(): v for k, v in pkgs.items() if isinstance(v, dict)}


def enrich_packages_with_deps_meta(
    packages,
    bundle: Optional[dict[str, dict]] = None,
    fetcher: Optional[Fetcher] = None,
) -> int:
    """Stamp supply-chain metadata onto Package objects.

    Source order: explicit bundle > $AGENT_BOM_DEPS_BUNDLE > injectable
    fetcher (online deployments only — never called unless provided).
    Returns the number of packages enriched.  Lookup state is always
    recorded: found / not_in_bundle / error."""
    if bundle is None:
        bundle = load_deps_bundle()
    enriched = 0
    for pkg in packages:
        key = _bundle_key(pkg.ecosystem, pkg.name)
        row = bundle.get(key)
        if row is None and fetcher is not None:
            try:
                row = fetcher(pkg.ecosystem, pkg.name)
            except Exception:
                pkg.scorecard_lookup_state = "error"
                pkg.scorecard_lookup_reason = "fetcher raised"
                continue
        if row is None:
            if pkg.scorecard_lookup_state is None:
                pkg.scorecard_lookup_state = "not_in_bundle"
            continue
        sc = row.get("scorecard_score")
        if isinstance(sc, (int, float)):
            pkg.scorecard_score = float(sc)
        checks = row.get("scorecard_checks")
        if isinstance(checks, dict):
            pkg.scorecard_checks = {str(k): int(v) for k, v in checks.items()
                                    if isinstance(v, (int, float))}
        for src_key, attr in (("scorecard_repo", "scorecard_repo"),
                              ("source_repo", "source_repo"),
                              ("license", "license")):
            v = row.get(src_key)
            if isinstance(v, str) and v and getattr(pkg, attr, None) in (None, ""):
                setattr(pkg, attr, v)
        mc = row.get("maintainer_count")
        if isinstance(mc, int):
            pkg.maintainer_count = mc
        if row.get("deprecated") is True:
            pkg.auto_risk_level = pkg.auto_risk_level or "elevated"
            pkg.auto_risk_justification = (pkg.auto_risk_justification
                                           or "deps.dev marks package deprecated")
        pkg.deps_dev_resolved = True
        pkg.scorecard_lookup_state = "found"
        enriched += 1
    return enriched


def supply_chain_risk_notes(pkg) -> list[str]:
    """Human-readable supply-chain concerns for one package (used by the
    trust report and console rendering)."""
    notes: list[str] = []
    if pkg.scorecard_score is not None and pkg.scorecard_score < 4.0:
        notes.append(f"low OpenSSF scorecard ({pkg.scorecard_score:.1f}/10)")
    if pkg.maintainer_count is not None and pkg.maintainer_count <= 1:
        notes.append("single-maintainer package")
    if (pkg.auto_risk_justification or "").startswith("deps.dev marks"):
        notes.append("deprecated upstream")
    checks = pkg.scorecard_checks or {}
    if checks.get("Dangerous-Workflow", 10) <= 2:
        notes.append("dangerous CI workflow patterns flagged by scorecard")
    if checks.get("Signed-Releases", 10) <= 2:
        notes.append("releases are unsigned")
    return notes
