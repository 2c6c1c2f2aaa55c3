"""Signed, portable per-instance MCP scan attestations (DSSE envelopes).

Reference parity: src/agent_bom/mcp_scan_attestation.py — each scanned MCP
server instance is bound into a signed evidence object (instance digest,
scanner version, observation time, verdict, capability fingerprint,
normalized evidence digest) that an independent verifier validates ONLY
against an operator-configured trust policy (pinned signers, expected
tenant, freshness).  Trust is never derived from key material embedded in
the artifact being verified.

Signature scheme: this offline image ships no asymmetric-crypto library,
so envelopes are signed with **HMAC-SHA256** over the DSSE PAE encoding
(`DSSEv1 <type> <payload>`); the envelope records ``alg: "hmac-sha256"``
and a key id, and the verifier resolves the key id through its own trust
policy — the same pinned-signer model as the reference's Ed25519 path,
with the signing primitive swappable.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import time
from typing import Any, Optional

from agentbom_amd import __version__

PAYLOAD_TYPE = "application/vnd.agent-bom.mcp-scan-attestation+json"
SIG_ALG = "hmac-sha256"
DEFAULT_MAX_AGE_S = 7 * 24 * 3600.0


def _pae(payload_type: str, payload: bytes) -> bytes:
    """DSSE pre-authentication encoding."""
    return b" ".join([
        b"DSSEv1",
        str(len(payload_type)).encode(), payload_type.encode(),
        str(len(payload)).encode(), payload,
    ])


def _b64(data: bytes) -> str:
    return base64.standard_b64encode(data).decode()


def _unb64(text: str) -> bytes:
    return base64.standard_b64decode(text)


def capability_fingerprint(tool_names: list[str]) -> str:
    return hashlib.sha256("\n".join(sorted(tool_names)).encode()).hexdigest()


def build_attestation_statement(*, server_name: str, instance_digest: str,
                                verdict: str, tool_names: list[str],
                                tenant_id: str = "default",
                                evidence: Optional[dict[str, Any]] = None,
                                observed_at: Optional[float] = None) -> dict[str, Any]:
    """The in-toto-style statement bound into the envelope."""
    evidence_doc = json.dumps(evidence or {}, sort_keys=True, default=str)
    return {
        "_type": "https://in-toto.io/Statement/v1",
        "predicateType": PAYLOAD_TYPE,
        "subject": [{"name": server_name,
                     "digest": {"sha256": instance_digest}}],
        "predicate": {
            "scanner": {"name": "agent-bom", "version": __version__},
            "tenant_id": tenant_id,
            "observed_at": observed_at if observed_at is not None else time.time(),
            "verdict": verdict,  # pass | warn | block
            "capability_fingerprint": capability_fingerprint(tool_names),
            "tool_count": len(tool_names),
            "evidence_digest": hashlib.sha256(evidence_doc.encode()).hexdigest(),
        },
    }


def sign_attestation(statement: dict[str, Any], key: bytes,
                     key_id: str) -> dict[str, Any]:
    """Wrap a statement in a signed DSSE envelope."""
    payload = json.dumps(statement, sort_keys=True, default=str).encode()
    sig = hmac.new(key, _pae(PAYLOAD_TYPE, payload), hashlib.sha256).digest()
    return {
        "payloadType": PAYLOAD_TYPE,
        "payload": _b64(payload),
        "signatures": [{"keyid": key_id, "alg": SIG_ALG, "sig": _b64(sig)}],
    }


def verify_attestation(envelope: dict[str, Any],
                       trust_policy: dict[str, Any]) -> dict[str, Any]:
    """Validate an envelope against an OPERATOR trust policy.

    ``trust_policy``: {"keys": {key_id: hex_or_bytes}, "expected_tenant":
    str | None, "max_age_s": float}.  Returns {"valid": bool, "reason": ...,
    "statement": ... (only when valid)}.  Never trusts anything inside the
    envelope beyond the payload it authenticates.
    """
    keys = trust_policy.get("keys") or {}
    if envelope.get("payloadType") != PAYLOAD_TYPE:
        return {"valid": False, "reason": "unexpected payload type"}
    try:
        payload = _unb64(envelope["payload"])
        sigs = envelope["signatures"]
    except (KeyError, ValueError, TypeError):
        return {"valid": False, "reason": "malformed envelope"}

    verified = False
    for sig in sigs:
        key_id = str(sig.get("keyid", ""))
        raw = keys.get(key_id)
        if raw is None:
            continue  # unpinned signer: ignored, never trusted
        key = bytes.fromhex(raw) if isinstance(raw, str) else raw
        expect = hmac.new(key, _pae(PAYLOAD_TYPE, payload),
                          hashlib.sha256).digest()
        try:
            if hmac.compare_digest(expect, _unb64(str(sig.get("sig", "")))):
                verified = True
                break
        except (ValueError, TypeError):
            continue
    if not verified:
        return {"valid": False, "reason": "no signature from a pinned signer"}

    try:
        statement = json.loads(payload)
    except json.JSONDecodeError:
        return {"valid": False, "reason": "payload is not JSON"}
    predicate = statement.get("predicate") or {}

    expected_tenant = trust_policy.get("expected_tenant")
    if expected_tenant and predicate.get("tenant_id") != expected_tenant:
        return {"valid": False,
                "reason": f"tenant mismatch: {predicate.get('tenant_id')!r}"}

    max_age = float(trust_policy.get("max_age_s", DEFAULT_MAX_AGE_S))
    observed = float(predicate.get("observed_at") or 0.0)
    if observed <= 0 or time.time() - observed > max_age:
        return {"valid": False, "reason": "attestation stale or undated"}

    return {"valid": True, "reason": "ok", "statement": statement}


def attest_scanned_server(server, verdict: str, key: bytes, key_id: str,
                          tenant_id: str = "default") -> dict[str, Any]:
    """Convenience: statement + envelope for one scanned MCPServer."""
    instance_digest = hashlib.sha256(
        f"{server.name}|{server.command}|{' '.join(server.args)}".encode()
    ).hexdigest()
    statement = build_attestation_statement(
        server_name=server.name,
        instance_digest=instance_digest,
        verdict=verdict,
        tool_names=[t.name for t in server.tools],
        tenant_id=tenant_id,
        evidence={"config_path": server.config_path,
                  "registry_id": server.registry_id,
                  "security_warnings": server.security_warnings})
    return sign_attestation(statement, key, key_id)
