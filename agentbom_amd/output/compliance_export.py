"""Signed compliance evidence bundles per framework.

Reference parity: src/agent_bom/output/compliance_export.py — per-control
evidence rows collected from blast radii + unified findings, a bundle
manifest with content digests, completeness grading, and an HMAC-SHA256
signature over the canonical manifest when ``AGENT_BOM_AUDIT_HMAC_KEY``
is set (unsigned local bundles say so honestly).

Baseline comparator: the reference signs a 10k-control bundle in 19.4 ms
(BASELINE.md); `tests/test_compliance_export.py` pins our time under that.
"""

from __future__ import annotations

import hashlib
import hmac
import json
import os
import time
from collections import defaultdict
from typing import Any, Optional

from agentbom_amd.models import FRAMEWORK_TAG_FIELDS, AIBOMReport

_COMPLETENESS_LEVELS = ("evidence_backed", "partial_evidence", "no_evidence")


def _digest(data: bytes) -> str:
    return hashlib.sha256(data).hexdigest()


def _json_bytes(value: Any) -> bytes:
    return json.dumps(value, sort_keys=True, separators=(",", ":"),
                      default=str).encode()


def _framework_field(framework: str) -> Optional[str]:
    for field_name, slug in FRAMEWORK_TAG_FIELDS:
        if slug == framework:
            return field_name
    return None


def export_compliance_bundle(report: AIBOMReport, framework: str,
                             hmac_key: Optional[bytes] = None) -> dict[str, Any]:
    """One framework's control→evidence bundle with manifest + signature."""
    field_name = _framework_field(framework)
    if field_name is None:
        raise ValueError(f"unknown framework {framework!r}; one of "
                         f"{[slug for _f, slug in FRAMEWORK_TAG_FIELDS]}")

    controls: dict[str, list[dict[str, Any]]] = defaultdict(list)
    for br in report.blast_radii:
        tags = getattr(br, field_name) or []
        for tag in tags:
            controls[str(tag)].append({
                "kind": "blast_radius",
                "vulnerability_id": br.vulnerability.id,
                "package": f"{br.package.name}@{br.package.version}",
                "severity": br.vulnerability.severity.value,
                "risk_score": round(float(br.risk_score), 4),
                "is_kev": bool(br.vulnerability.is_kev),
            })
    for finding in report.findings or []:
        for tag in getattr(finding, "controls", []) or []:
            if getattr(tag, "framework", None) == framework:
                controls[str(tag.control)].append({
                    "kind": "finding",
                    "finding_id": finding.id,
                    "severity": finding.severity,
                    "title": finding.title,
                })

    sev_rank = {"critical": 4, "high": 3, "medium": 2, "low": 1}
    control_rows = []
    chain = hashlib.sha256()
    for control in sorted(controls):
        rows = controls[control]
        evidence_bytes = _json_bytes(rows)  # serialized exactly once
        evidence_digest = _digest(evidence_bytes)
        chain.update(control.encode())
        chain.update(evidence_digest.encode())
        control_rows.append({
            "control": control,
            "evidence_count": len(rows),
            "worst_severity": max(
                rows, key=lambda r: sev_rank.get(str(r.get("severity")), 0)
            )["severity"],
            "evidence": rows,
            "evidence_digest": evidence_digest,
        })

    if control_rows and report.blast_radii:
        completeness = "evidence_backed"
    elif control_rows:
        completeness = "partial_evidence"
    else:
        completeness = "no_evidence"

    manifest = {
        "schema_version": "1",
        "framework": framework,
        "scan_id": report.scan_id,
        "control_count": len(control_rows),
        "total_evidence_rows": sum(c["evidence_count"] for c in control_rows),
        "completeness": completeness,
        # Merkle-style: hash of (control, evidence_digest) pairs in order —
        # verifiable per control without re-serializing the whole bundle
        "controls_digest": chain.hexdigest(),
    }

    if hmac_key is None:
        env_key = os.environ.get("AGENT_BOM_AUDIT_HMAC_KEY", "")
        hmac_key = env_key.encode() if env_key else None
    if hmac_key:
        signature = {
            "status": "signed",
            "algorithm": "HMAC-SHA256",
            "signed_payload": "manifest",
            "value": hmac.new(hmac_key, _json_bytes(manifest),
                              hashlib.sha256).hexdigest(),
        }
    else:
        signature = {
            "status": "unsigned_local_bundle",
            "algorithm": None,
            "note": "set AGENT_BOM_AUDIT_HMAC_KEY to sign bundles",
        }

    return {"manifest": manifest, "controls": control_rows,
            "signature": signature,
            "generated_in_ms": None}  # stamped by timed wrapper below


def export_compliance_bundle_timed(report: AIBOMReport, framework: str,
                                   hmac_key: Optional[bytes] = None) -> dict[str, Any]:
    t0 = time.perf_counter()
    bundle = export_compliance_bundle(report, framework, hmac_key=hmac_key)
    bundle["generated_in_ms"] = round((time.perf_counter() - t0) * 1000, 3)
    return bundle


def verify_compliance_bundle(bundle: dict[str, Any],
                             hmac_key: bytes) -> dict[str, Any]:
    """Re-verify a bundle: signature over manifest + content digests."""
    sig = bundle.get("signature") or {}
    if sig.get("status") != "signed":
        return {"valid": False, "reason": "bundle is unsigned"}
    expect = hmac.new(hmac_key, _json_bytes(bundle["manifest"]),
                      hashlib.sha256).hexdigest()
    if not hmac.compare_digest(expect, str(sig.get("value", ""))):
        return {"valid": False, "reason": "manifest signature mismatch"}
    chain = hashlib.sha256()
    for c in bundle["controls"]:
        if c["evidence_digest"] != _digest(_json_bytes(c["evidence"])):
            return {"valid": False,
                    "reason": f"evidence digest mismatch for {c['control']}"}
        chain.update(str(c["control"]).encode())
        chain.update(str(c["evidence_digest"]).encode())
    if bundle["manifest"]["controls_digest"] != chain.hexdigest():
        return {"valid": False, "reason": "controls digest mismatch"}
    return {"valid": True, "reason": "ok"}
