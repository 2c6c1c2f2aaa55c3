"""Local threat-intel lookup + report enrichment.

Reference parity: src/agent_bom/intel_lookup.py / intel_fetch.py — a local
threat-intelligence store (indicators keyed by package, CVE, domain or
file hash) consulted during enrichment, plus an OPTIONAL live-search hook
(the reference's only third-party egress tool).  This environment is
air-gapped, so the live hook is an injectable callable that defaults to
None and the store is a local JSON/JSONL file:

    {"indicators": [
       {"type": "package", "key": "pypi:evil-pkg", "source": "...",
        "note": "...", "confidence": 0.9, "exploited": false},
       {"type": "cve", "key": "CVE-2024-0001", "note": "PoC public", ...}
    ]}

Enrichment effects (mirrors the reference's risk wiring):
- a ``cve`` indicator with ``exploited: true`` sets
  ``Vulnerability.exploitability = "active_exploitation"`` (feeds the
  existing exploit-likelihood risk boost) and appends ``intel`` to
  ``advisory_sources``;
- a ``package`` indicator marks the package malicious when
  ``malicious: true`` (fail-closed, same path as the typosquat screen);
- every hit is recorded on ``report.intel_matches`` for evidence.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Callable, Optional

_BUNDLED = Path(__file__).resolve().parent.parent / "data" / "threat_intel.json"

_TYPES = ("package", "cve", "domain", "hash", "actor")


@dataclass
class IntelIndicator:
    type: str
    key: str
    source: str = "local"
    note: str = ""
    confidence: float = 0.5
    exploited: bool = False
    malicious: bool = False
    last_seen: str = ""

    def to_dict(self) -> dict:
        return {
            "type": self.type, "key": self.key, "source": self.source,
            "note": self.note, "confidence": self.confidence,
            "exploited": self.exploited, "malicious": self.malicious,
            "last_seen": self.last_seen,
        }


def _indicator_from(row: Any) -> Optional[IntelIndicator]:
    if not isinstance(row, dict):
        return None
    t, k = row.get("type"), row.get("key")
    if not isinstance(t, str) or t not in _TYPES or not isinstance(k, str) or not k:
        return None
    try:
        conf = float(row.get("confidence", 0.5))
    except (TypeError, ValueError):
        conf = 0.5
    return IntelIndicator(
        type=t, key=k.lower(),
        source=str(row.get("source") or "local"),
        note=str(row.get("note") or ""),
        confidence=max(0.0, min(1.0, conf)),
        exploited=bool(row.get("exploited")),
        malicious=bool(row.get("malicious")),
        last_seen=str(row.get("last_seen") or ""),
    )


class IntelStore:
    """In-memory index over local threat-intel indicators."""

    def __init__(self, indicators: Optional[list[IntelIndicator]] = None):
        self._by_key: dict[tuple[str, str], IntelIndicator] = {}
        for ind in indicators or []:
            self.add(ind)

    def add(self, ind: IntelIndicator) -> None:
        self._by_key[(ind.type, ind.key)] = ind

    def __len__(self) -> int:
        return len(self._by_key)

    @classmethod
    def load(cls, path: Optional[str] = None) -> "IntelStore":
        """Bundled feed, overlaid with $AGENT_BOM_INTEL_DB when set.

        Fail-soft: unreadable files yield an empty (or partial) store.
        """
        store = cls()
        paths = [_BUNDLED]
        env = path or os.environ.get("AGENT_BOM_INTEL_DB")
        if env:
            paths.append(Path(env))
        for p in paths:
            try:
                doc = json.loads(Path(p).read_text())
            except Exception:
                continue
            rows = doc.get("indicators") if isinstance(doc, dict) else doc
            if not isinstance(rows, list):
                continue
            for row in rows:
                ind = _indicator_from(row)
                if ind is not None:
                    store.add(ind)
        return store

    # ── lookups ────────────────────────────────────────────────────────
    def lookup_package(self, ecosystem: str, name: str) -> Optional[IntelIndicator]:
        key = f"{(ecosystem or '').lower()}:{(name or '').lower()}"
        hit = self._by_key.get(("package", key))
        if hit is None:
            hit = self._by_key.get(("package", (name or "").lower()))
        return hit

    def lookup_cve(self, vuln_id: str) -> Optional[IntelIndicator]:
        return self._by_key.get(("cve", (vuln_id or "").lower()))

    def lookup(self, kind: str, key: str) -> Optional[IntelIndicator]:
        return self._by_key.get((kind, (key or "").lower()))

    def search(self, query: str, limit: int = 20) -> list[IntelIndicator]:
        q = (query or "").lower()
        out = [i for (t, k), i in sorted(self._by_key.items())
               if q in k or q in i.note.lower()]
        return out[:limit]


def enrich_report_with_intel(
    report,
    store: Optional[IntelStore] = None,
    live_search: Optional[Callable[[str], list[dict]]] = None,
) -> list[dict]:
    """Stamp intel hits onto a finished report.  Returns the match list.

    ``live_search`` is the injectable egress hook (reference: You.com tool);
    it is NEVER called unless explicitly provided — air-gapped default.
    """
    store = store or IntelStore.load()
    matches: list[dict] = []
    seen_cve: set[str] = set()
    for br in getattr(report, "blast_radii", []) or []:
        vuln = br.vulnerability
        pkg = br.package
        ind = store.lookup_cve(vuln.id)
        if ind is None:
            for alias in vuln.aliases or []:
                ind = store.lookup_cve(alias)
                if ind is not None:
                    break
        if ind is not None and vuln.id not in seen_cve:
            seen_cve.add(vuln.id)
            if ind.exploited:
                vuln.exploitability = "active_exploitation"
            if "intel" not in (vuln.advisory_sources or []):
                vuln.advisory_sources = list(vuln.advisory_sources or []) + ["intel"]
            matches.append({"entity": "vulnerability", "id": vuln.id,
                            **ind.to_dict()})
            br.calculate_risk_score()
        pind = store.lookup_package(pkg.ecosystem, pkg.name)
        if pind is not None:
            if pind.malicious and not pkg.is_malicious:
                pkg.is_malicious = True
                pkg.malicious_reason = pind.note or f"threat-intel: {pind.source}"
            matches.append({"entity": "package",
                            "id": f"{pkg.ecosystem}:{pkg.name}",
                            **pind.to_dict()})
    if live_search is not None:
        for br in (getattr(report, "blast_radii", []) or [])[:5]:
            try:
                rows = live_search(br.vulnerability.id) or []
            except Exception:
                continue
            for row in rows[:3]:
                if isinstance(row, dict):
                    matches.append({"entity": "live", "id": br.vulnerability.id,
                                    **{k: row[k] for k in ("title", "url", "snippet")
                                       if isinstance(row.get(k), str)}})
    # dedupe (entity, id, key) so repeated blast radii of one package
    # don't multiply rows
    uniq: dict[tuple, dict] = {}
    for m in matches:
        uniq[(m.get("entity"), m.get("id"), m.get("key"))] = m
    out = list(uniq.values())
    if hasattr(report, "intel_matches"):
        report.intel_matches = out
    else:
        setattr(report, "intel_matches", out)
    return out
