"""Vendor security-bulletin supplemental matching (AMD PSIRT / NVIDIA CSAF /
Intel / firmware).

Reference parity: src/agent_bom/scanners/{ghsa_advisory,nvidia_advisory,
amd_advisory}.py and the ROCm prefix list mirrored at
src/agent_bom/scanners/package_scan.py:1674-1699 — vendor bulletins that
never land in OSV (driver stacks, runtime libraries, firmware) are matched
AFTER the OSV arena pass as supplemental advisory sources.

MI355X-first design notes: this deployment runs on ROCm, so the AMD PSIRT
prefix family (rocm/hip/rccl/miopen/…) is first-class, not an
afterthought.  There is no network egress in the target environment:
matching runs against a LOCAL feed — the bundled
``agentbom_amd/data/vendor_advisories.json`` (machinery examples, gated),
an operator-supplied feed via ``AGENT_BOM_VENDOR_FEED`` (a JSON file or a
directory of CSAF 2.0 documents), or feeds handed in by the caller.

Matching semantics stay fail-closed and tiered:

- an advisory row with version bounds matches only when
  :func:`version_in_range` says so → ``match_confidence_tier="vendor_range"``;
- a row with NO bounds matches by product/prefix only →
  ``match_confidence_tier="vendor_prefix"`` (lower confidence, never
  silently dropped);
- unparseable bounds never establish a match (version_utils fail-closed).
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Optional

from agentbom_amd.models.core import Severity, Vulnerability, compute_confidence
from agentbom_amd.utils.version_utils import version_in_range

# ── vendor product-prefix maps ──────────────────────────────────────────────
# Package-name prefixes that identify a vendor's runtime stack.  Mirrors the
# reference's ROCm prefix list (package_scan.py:1674-1699) in spirit; the
# names here are the real distribution names on PyPI / system ecosystems.

AMD_PACKAGE_PREFIXES: tuple[str, ...] = (
    "rocm", "hip", "hsa-", "rocblas", "rocsparse", "rocsolver", "rocfft",
    "rocrand", "rocprim", "rocthrust", "rccl", "miopen", "migraphx",
    "roctracer", "rocprofiler", "amdsmi", "amd-smi", "hipblas", "hipblaslt",
    "hipsparse", "hipfft", "hipsolver", "hipcub", "hiprand", "rocwmma",
    "composable-kernel", "ck-tile", "amdgpu", "rocminfo", "half-rocm",
)

NVIDIA_PACKAGE_PREFIXES: tuple[str, ...] = (
    "cuda", "nvidia-", "cudnn", "nccl", "cublas", "cufft", "curand",
    "cusparse", "cusolver", "cutensor", "tensorrt", "nvjpeg", "nvrtc",
    "dali", "nvml", "pynvml", "triton-inference",
)

INTEL_PACKAGE_PREFIXES: tuple[str, ...] = (
    "intel-", "openvino", "mkl", "oneapi", "oneccl", "onednn", "onemkl",
    "ipex", "dpctl", "dpnp",
)

_VENDOR_PREFIXES: dict[str, tuple[str, ...]] = {
    "amd": AMD_PACKAGE_PREFIXES,
    "nvidia": NVIDIA_PACKAGE_PREFIXES,
    "intel": INTEL_PACKAGE_PREFIXES,
}


def vendor_for_package(name: str) -> Optional[str]:
    """Which vendor runtime stack (if any) a package name belongs to."""
    n = (name or "").lower()
    for vendor, prefixes in _VENDOR_PREFIXES.items():
        for p in prefixes:
            if n == p.rstrip("-") or n.startswith(p):
                return vendor
    return None


# ── advisory model ──────────────────────────────────────────────────────────

@dataclass
class VendorAffected:
    """One affected product row inside a vendor bulletin."""

    ecosystem: str = "*"            # "*" matches any ecosystem
    name: Optional[str] = None      # exact normalized name …
    name_prefix: Optional[str] = None  # … or a prefix
    introduced: Optional[str] = None
    fixed: Optional[str] = None
    last_affected: Optional[str] = None

    def matches_name(self, ecosystem: str, name: str) -> bool:
        if self.ecosystem not in ("*", "", None) and self.ecosystem.lower() != ecosystem.lower():
            return False
        n = name.lower()
        if self.name:
            return n == self.name.lower()
        if self.name_prefix:
            return n.startswith(self.name_prefix.lower())
        return False

    @property
    def has_bounds(self) -> bool:
        return bool(self.introduced or self.fixed or self.last_affected)


@dataclass
class VendorAdvisory:
    """A vendor security bulletin (PSIRT / CSAF) row."""

    vendor: str
    advisory_id: str
    title: str = ""
    severity: str = "unknown"
    cve_ids: list[str] = field(default_factory=list)
    affected: list[VendorAffected] = field(default_factory=list)
    url: str = ""
    fixed_version: Optional[str] = None
    example: bool = False           # machinery demo rows, skipped by default


# ── feed loading ────────────────────────────────────────────────────────────

_BUNDLED_FEED = Path(__file__).resolve().parent.parent / "data" / "vendor_advisories.json"


def _advisory_from_dict(row: dict) -> Optional[VendorAdvisory]:
    if not isinstance(row, dict):
        return None
    aid = row.get("advisory_id") or row.get("id")
    if not aid or not isinstance(aid, str):
        return None
    affected = []
    for a in row.get("affected") or []:
        if not isinstance(a, dict):
            continue
        affected.append(VendorAffected(
            ecosystem=str(a.get("ecosystem") or "*"),
            name=a.get("name") if isinstance(a.get("name"), str) else None,
            name_prefix=a.get("name_prefix") if isinstance(a.get("name_prefix"), str) else None,
            introduced=a.get("introduced") if isinstance(a.get("introduced"), str) else None,
            fixed=a.get("fixed") if isinstance(a.get("fixed"), str) else None,
            last_affected=a.get("last_affected") if isinstance(a.get("last_affected"), str) else None,
        ))
    return VendorAdvisory(
        vendor=str(row.get("vendor") or "unknown").lower(),
        advisory_id=aid,
        title=str(row.get("title") or ""),
        severity=str(row.get("severity") or "unknown").lower(),
        cve_ids=[c for c in (row.get("cve_ids") or []) if isinstance(c, str)],
        affected=affected,
        url=str(row.get("url") or ""),
        fixed_version=row.get("fixed_version") if isinstance(row.get("fixed_version"), str) else None,
        example=bool(row.get("example")),
    )


def parse_csaf_document(doc: Any) -> list[VendorAdvisory]:
    """Tolerant CSAF 2.0 ingestion (the format AMD PSIRT and NVIDIA publish).

    Walks document/tracking/id + title, vulnerabilities[].cve and the
    product_tree's full_product_names; never raises on malformed shapes —
    unparseable documents yield [].
    """
    if not isinstance(doc, dict):
        return []
    d = doc.get("document")
    if not isinstance(d, dict):
        return []
    tracking = d.get("tracking") if isinstance(d.get("tracking"), dict) else {}
    aid = tracking.get("id")
    if not isinstance(aid, str) or not aid:
        return []
    title = d.get("title") if isinstance(d.get("title"), str) else ""
    publisher = d.get("publisher") if isinstance(d.get("publisher"), dict) else {}
    vendor = str(publisher.get("name") or "unknown").split()[0].lower()
    severity = "unknown"
    agg = d.get("aggregate_severity")
    if isinstance(agg, dict) and isinstance(agg.get("text"), str):
        severity = agg["text"].lower()

    # product names → prefix rows (CSAF product trees name binaries/branches;
    # we match them as name prefixes, unbounded → vendor_prefix tier)
    names: list[str] = []

    def _walk_products(node: Any) -> None:
        if isinstance(node, dict):
            fpn = node.get("name")
            if isinstance(fpn, str) and node.get("product_id") and fpn:
                names.append(fpn)
            for v in node.values():
                _walk_products(v)
        elif isinstance(node, list):
            for v in node:
                _walk_products(v)

    _walk_products(doc.get("product_tree"))

    cves: list[str] = []
    vulns = doc.get("vulnerabilities")
    if isinstance(vulns, list):
        for v in vulns:
            if isinstance(v, dict) and isinstance(v.get("cve"), str):
                cves.append(v["cve"])

    affected = [
        VendorAffected(ecosystem="*", name_prefix=n.split()[0].lower())
        for n in names[:64]
        if n.split()
    ]
    return [VendorAdvisory(
        vendor=vendor, advisory_id=aid, title=title, severity=severity,
        cve_ids=cves, affected=affected,
    )]


def load_vendor_feed(path: Optional[str] = None,
                     include_examples: Optional[bool] = None) -> list[VendorAdvisory]:
    """Load the vendor feed: explicit path > $AGENT_BOM_VENDOR_FEED > bundled.

    A path may be a JSON file (``{"advisories": [...]}`` or a bare list) or a
    directory of CSAF 2.0 ``*.json`` documents.  Fail-soft: unreadable or
    malformed entries are skipped, never fatal.
    """
    if include_examples is None:
        include_examples = os.environ.get("AGENT_BOM_VENDOR_EXAMPLES", "") == "1"
    p = Path(path or os.environ.get("AGENT_BOM_VENDOR_FEED") or _BUNDLED_FEED)
    out: list[VendorAdvisory] = []
    try:
        if p.is_dir():
            for f in sorted(p.glob("*.json")):
                try:
                    out.extend(parse_csaf_document(json.loads(f.read_text())))
                except Exception:
                    continue
        elif p.exists():
            doc = json.loads(p.read_text())
            rows = doc.get("advisories") if isinstance(doc, dict) else doc
            if isinstance(doc, dict) and "document" in doc:
                out.extend(parse_csaf_document(doc))
            elif isinstance(rows, list):
                for row in rows:
                    adv = _advisory_from_dict(row)
                    if adv is not None:
                        out.append(adv)
    except Exception:
        return []
    return [a for a in out if include_examples or not a.example]


# ── matching ────────────────────────────────────────────────────────────────

_SEV = {s.value: s for s in Severity}


def check_vendor_advisories(
    unique_pkgs: list[tuple[str, str, str]],
    feed: Optional[list[VendorAdvisory]] = None,
) -> dict[tuple[str, str, str], list[Vulnerability]]:
    """Match (eco, name, version) tuples against vendor bulletins.

    Returns supplemental :class:`Vulnerability` rows keyed by package tuple;
    the orchestrator appends them after the OSV pass (dedup by vuln id).
    """
    if feed is None:
        feed = load_vendor_feed()
    if not feed:
        return {}
    out: dict[tuple[str, str, str], list[Vulnerability]] = {}
    for key in unique_pkgs:
        eco, name, version = key
        for adv in feed:
            row_hit: Optional[VendorAffected] = None
            tier = None
            for a in adv.affected:
                if not a.matches_name(eco, name):
                    continue
                if a.has_bounds:
                    if version_in_range(version, a.introduced, a.fixed,
                                        a.last_affected, eco):
                        row_hit, tier = a, "vendor_range"
                        break
                else:
                    row_hit, tier = a, "vendor_prefix"
                    # keep scanning: a bounded row wins over a prefix row
            if row_hit is None:
                continue
            vuln = Vulnerability(
                id=adv.advisory_id,
                summary=adv.title or f"{adv.vendor} security bulletin",
                severity=_SEV.get(adv.severity, Severity.UNKNOWN),
                severity_source="vendor",
                fixed_version=adv.fixed_version or row_hit.fixed,
                references=[adv.url] if adv.url else [],
                aliases=list(adv.cve_ids),
                advisory_sources=[f"vendor:{adv.vendor}"],
                match_confidence_tier=tier,
            )
            vuln.confidence = compute_confidence(vuln)
            if tier == "vendor_prefix" and vuln.confidence is not None:
                vuln.confidence = round(vuln.confidence * 0.6, 3)
            out.setdefault(key, []).append(vuln)
    return out
