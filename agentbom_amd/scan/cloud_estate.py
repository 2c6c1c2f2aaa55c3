"""Multi-cloud estate posture: Azure/GCP/Snowflake/Databricks CIS checks,
IAM over-privilege evaluation, audit-trail anomaly ingest, DSPM data
classification.

Reference parity: the src/agent_bom/cloud/ package (SURVEY.md §2.2 —
aws/azure/gcp/snowflake/databricks inventories, per-provider CIS
benchmarks, IAM collectors/evaluators, audit_trail ingest, DSPM
classifiers).  There is no cloud egress in this environment, so every
evaluator consumes an EXPORTED inventory document (the same
collector-or-operator-assembled JSON shape `scan/cloud.py` uses for AWS);
live collectors plug in where the file loads happen today.

All evaluators return :class:`CisCheckResult` rows and reuse
``cis_result_to_finding`` so findings land in the unified stream with the
same CLOUD_CIS provenance the AWS path has.
"""

from __future__ import annotations

import json
import re
from pathlib import Path
from typing import Any, Optional

from agentbom_amd.scan.cloud import CisCheckResult, _check


def _rows(inv: Any, key: str) -> list[dict]:
    """Tolerant list-of-dicts extraction: hostile/hand-edited inventory
    documents may put anything under these keys (fuzz-found)."""
    v = inv.get(key) if isinstance(inv, dict) else None
    if not isinstance(v, list):
        return []
    return [r for r in v if isinstance(r, dict)]


def _section(inv: Any, key: str) -> Optional[dict]:
    v = inv.get(key) if isinstance(inv, dict) else None
    return v if isinstance(v, dict) else None

# ── Azure CIS subset ────────────────────────────────────────────────────────


def evaluate_azure_inventory(inv: dict[str, Any]) -> list[CisCheckResult]:
    """Azure CIS Foundations subset over an exported inventory document.

    {"storage_accounts": [...], "nsgs": [...], "sql_servers": [...],
     "aad_users": [...], "ml_workspaces": [...], "activity_log": {...}}
    """
    results: list[CisCheckResult] = []
    for sa in _rows(inv, "storage_accounts"):
        name = sa.get("name", "?")
        _check(results, "AZ-3.1", "Storage account denies public blob access", "high",
               f"storage:{name}", not sa.get("allowBlobPublicAccess", False))
        _check(results, "AZ-3.2", "Storage requires secure transfer (HTTPS)", "medium",
               f"storage:{name}", bool(sa.get("supportsHttpsTrafficOnly", True)))
        _check(results, "AZ-3.8", "Storage soft delete enabled", "low",
               f"storage:{name}", bool(sa.get("softDelete", True)))
    for nsg in _rows(inv, "nsgs"):
        name = nsg.get("name", "?")
        open_world = any(
            r.get("sourceAddressPrefix") in ("*", "0.0.0.0/0", "Internet")
            and str(r.get("destinationPortRange", "")) in ("22", "3389", "*")
            and r.get("access", "Allow") == "Allow"
            for r in nsg.get("rules", []) or [] if isinstance(r, dict))
        _check(results, "AZ-6.1", "No admin ports open to the internet", "critical",
               f"nsg:{name}", not open_world,
               "SSH/RDP open to Internet" if open_world else "")
    for sql in _rows(inv, "sql_servers"):
        name = sql.get("name", "?")
        _check(results, "AZ-4.1", "SQL server auditing enabled", "medium",
               f"sql:{name}", bool(sql.get("auditingEnabled", True)))
        _check(results, "AZ-4.2", "SQL public network access disabled", "critical",
               f"sql:{name}", not sql.get("publicNetworkAccess", False))
    for user in _rows(inv, "aad_users"):
        name = user.get("userPrincipalName", "?")
        _check(results, "AZ-1.1", "MFA enabled for privileged users", "high",
               f"aad:{name}",
               bool(user.get("mfaEnabled", True)) or not user.get("privileged", False))
    for ws in _rows(inv, "ml_workspaces"):
        name = ws.get("name", "?")
        _check(results, "AIINF-1", "ML workspace endpoint requires auth", "critical",
               f"mlworkspace:{name}", bool(ws.get("authRequired", True)))
        _check(results, "AIINF-2", "ML workspace not internet-exposed", "high",
               f"mlworkspace:{name}", not ws.get("publicEndpoint", False))
    log = _section(inv, "activity_log")
    if log is not None:
        _check(results, "AZ-5.1", "Activity log export configured", "high",
               "activity-log", bool(log.get("exportEnabled", False)))
    return results


# ── GCP CIS subset ──────────────────────────────────────────────────────────


def evaluate_gcp_inventory(inv: dict[str, Any]) -> list[CisCheckResult]:
    """GCP CIS Foundations subset.

    {"buckets": [...], "firewalls": [...], "service_accounts": [...],
     "cloudsql": [...], "vertex_endpoints": [...], "audit_config": {...}}
    """
    results: list[CisCheckResult] = []
    for b in _rows(inv, "buckets"):
        name = b.get("name", "?")
        public = any(m in ("allUsers", "allAuthenticatedUsers")
                     for m in b.get("iamMembers", []) or [])
        _check(results, "GCP-5.1", "Bucket not publicly accessible", "high",
               f"gcs:{name}", not public,
               "allUsers/allAuthenticatedUsers granted" if public else "")
        _check(results, "GCP-5.2", "Bucket uniform access enabled", "medium",
               f"gcs:{name}", bool(b.get("uniformBucketLevelAccess", True)))
    for fw in _rows(inv, "firewalls"):
        name = fw.get("name", "?")
        open_world = ("0.0.0.0/0" in (fw.get("sourceRanges") or []) and any(
            str(p) in ("22", "3389", "all")
            for a in (fw.get("allowed") or []) if isinstance(a, dict)
            for p in (a.get("ports") or ["all"])))
        _check(results, "GCP-3.6", "No admin ports open to 0.0.0.0/0", "critical",
               f"firewall:{name}", not open_world)
    for sa in _rows(inv, "service_accounts"):
        email = sa.get("email", "?")
        _check(results, "GCP-1.5", "Service account has no Owner/Editor role", "high",
               f"sa:{email}",
               not any(r in ("roles/owner", "roles/editor")
                       for r in sa.get("roles", []) or []))
        _check(results, "GCP-1.4", "Service account keys rotated <= 90 days", "medium",
               f"sa:{email}", not sa.get("keyStale", False))
    for db in _rows(inv, "cloudsql"):
        name = db.get("name", "?")
        _check(results, "GCP-6.5", "Cloud SQL not open to the world", "critical",
               f"cloudsql:{name}",
               "0.0.0.0/0" not in (db.get("authorizedNetworks") or []))
    for ep in _rows(inv, "vertex_endpoints"):
        name = ep.get("name", "?")
        _check(results, "AIINF-1", "Vertex endpoint requires auth", "critical",
               f"vertex:{name}", bool(ep.get("authRequired", True)))
    audit = _section(inv, "audit_config")
    if audit is not None:
        _check(results, "GCP-2.1", "Cloud audit logging configured", "high",
               "audit-config", bool(audit.get("enabled", False)))
    return results


# ── Snowflake / Databricks subsets ─────────────────────────────────────────


def evaluate_snowflake_inventory(inv: dict[str, Any]) -> list[CisCheckResult]:
    results: list[CisCheckResult] = []
    for u in _rows(inv, "users"):
        name = u.get("name", "?")
        _check(results, "SF-1.1", "Snowflake user has MFA or key-pair auth", "high",
               f"sf-user:{name}",
               bool(u.get("mfaEnabled") or u.get("hasRsaKey")))
        _check(results, "SF-1.3", "ACCOUNTADMIN not used as default role", "high",
               f"sf-user:{name}", u.get("defaultRole") != "ACCOUNTADMIN")
    for sh in _rows(inv, "shares"):
        name = sh.get("name", "?")
        _check(results, "SF-2.1", "Share restricted to named accounts", "medium",
               f"sf-share:{name}", not sh.get("public", False))
    params = _section(inv, "account_parameters") or {}
    if params:
        _check(results, "SF-3.1", "Network policy attached to account", "medium",
               "sf-account", bool(params.get("networkPolicy")))
    return results


def evaluate_databricks_inventory(inv: dict[str, Any]) -> list[CisCheckResult]:
    results: list[CisCheckResult] = []
    for c in _rows(inv, "clusters"):
        name = c.get("name", "?")
        _check(results, "DBX-1.1", "Cluster not publicly reachable", "critical",
               f"dbx-cluster:{name}", not c.get("publicIp", False))
        _check(results, "DBX-1.2", "Credential passthrough / UC governance on", "medium",
               f"dbx-cluster:{name}",
               bool(c.get("unityCatalog") or c.get("credentialPassthrough")))
    for t in _rows(inv, "tokens"):
        _check(results, "DBX-2.1", "PAT lifetime bounded", "medium",
               f"dbx-token:{t.get('comment', '?')}",
               t.get("lifetimeDays") is not None and t.get("lifetimeDays", 999) <= 90)
    for sv in _rows(inv, "model_serving"):
        name = sv.get("name", "?")
        _check(results, "AIINF-1", "Model serving endpoint requires auth", "critical",
               f"dbx-serving:{name}", bool(sv.get("authRequired", True)))
    return results


# ── IAM over-privilege evaluator ───────────────────────────────────────────


def evaluate_iam_policies(docs: list[dict[str, Any]]) -> list[CisCheckResult]:
    """Over-privilege analysis of IAM policy documents (reference:
    cloud/iam_* collectors/evaluators).

    Accepts AWS-style policy documents: {"name": ..., "attached_to": ...,
    "Statement": [{"Effect": "Allow", "Action": [...], "Resource": [...]}]}.
    """
    results: list[CisCheckResult] = []
    for doc in docs or []:
        if not isinstance(doc, dict):
            continue
        name = str(doc.get("name") or doc.get("PolicyName") or "?")
        attached = str(doc.get("attached_to") or "")
        stmts = doc.get("Statement")
        if isinstance(stmts, dict):
            stmts = [stmts]
        if not isinstance(stmts, list):
            continue
        star_action = star_resource = passrole = False
        for st in stmts:
            if not isinstance(st, dict) or st.get("Effect") != "Allow":
                continue
            actions = st.get("Action") or []
            actions = [actions] if isinstance(actions, str) else actions
            resources = st.get("Resource") or []
            resources = [resources] if isinstance(resources, str) else resources
            if any(a == "*" for a in actions):
                star_action = True
            if any(r == "*" for r in resources):
                star_resource = True
            if any(isinstance(a, str) and a.lower() == "iam:passrole" for a in actions) \
                    and any(r == "*" for r in resources):
                passrole = True
        _check(results, "IAM-1", "Policy avoids Action:* with Resource:*", "critical",
               f"policy:{name}", not (star_action and star_resource),
               f"full-admin policy attached to {attached}" if attached else "")
        if star_resource and not star_action:
            _check(results, "IAM-2", "Policy scopes resources (no Resource:*)",
                   "medium", f"policy:{name}", False, "wildcard resource")
        if passrole:
            _check(results, "IAM-3", "iam:PassRole is resource-scoped", "high",
                   f"policy:{name}", False,
                   "unscoped PassRole enables privilege escalation")
        svc = doc.get("service_identity") or doc.get("is_service_account")
        if svc and (star_action or "AdministratorAccess" in name):
            _check(results, "IAM-4", "Service identities are least-privilege", "high",
                   f"policy:{name}", False, "admin rights on a non-human identity")
    return results


# ── audit-trail anomaly ingest ─────────────────────────────────────────────

_SENSITIVE_EVENTS = {
    "DeleteTrail": ("critical", "audit logging deleted"),
    "StopLogging": ("critical", "audit logging stopped"),
    "PutBucketPolicy": ("medium", "bucket policy changed"),
    "CreateAccessKey": ("medium", "new access key minted"),
    "AttachUserPolicy": ("medium", "policy attached to user"),
    "ConsoleLogin": (None, None),  # handled specially (MFA check)
    "DeactivateMFADevice": ("high", "MFA device deactivated"),
    "AuthorizeSecurityGroupIngress": ("medium", "ingress rule opened"),
}


def ingest_audit_trail(rows: list[dict[str, Any]],
                       enumeration_threshold: int = 50) -> list[CisCheckResult]:
    """CloudTrail-shaped event rows → anomaly results (reference:
    cloud/audit_trail.py).  Detections: root account usage, console login
    without MFA, logging tamper, privilege grants, mass enumeration by one
    principal, and access-denied bursts (recon)."""
    results: list[CisCheckResult] = []
    by_principal_reads: dict[str, int] = {}
    by_principal_denied: dict[str, int] = {}
    for ev in rows or []:
        if not isinstance(ev, dict):
            continue
        name = str(ev.get("eventName") or "")
        ident = ev.get("userIdentity") or {}
        principal = str(ident.get("arn") or ident.get("userName") or "?")
        if ident.get("type") == "Root":
            _check(results, "TRAIL-1", "No root account API usage", "high",
                   principal, False, f"root performed {name}")
        if name == "ConsoleLogin":
            mfa = str((ev.get("additionalEventData") or {}).get("MFAUsed", "Yes"))
            if mfa.lower() == "no":
                _check(results, "TRAIL-2", "Console logins use MFA", "high",
                       principal, False, "console login without MFA")
        sev_note = _SENSITIVE_EVENTS.get(name)
        if sev_note and sev_note[0]:
            _check(results, "TRAIL-3", f"Sensitive event: {name}", sev_note[0],
                   principal, False, sev_note[1])
        if re.match(r"^(List|Describe|Get)", name):
            by_principal_reads[principal] = by_principal_reads.get(principal, 0) + 1
        if ev.get("errorCode") in ("AccessDenied", "Client.UnauthorizedOperation"):
            by_principal_denied[principal] = by_principal_denied.get(principal, 0) + 1
    for principal, n in by_principal_reads.items():
        if n >= enumeration_threshold:
            _check(results, "TRAIL-4", "No mass enumeration bursts", "medium",
                   principal, False, f"{n} List/Describe/Get calls in window")
    for principal, n in by_principal_denied.items():
        if n >= 10:
            _check(results, "TRAIL-5", "No access-denied bursts (recon)", "high",
                   principal, False, f"{n} AccessDenied errors in window")
    return results


# ── DSPM data classification ───────────────────────────────────────────────

_DSPM_PATTERNS: list[tuple[str, str, re.Pattern]] = [
    ("pii-email", "email addresses", re.compile(r"[\w.+-]+@[\w-]+\.[\w.]+")),
    ("pii-ssn", "US SSNs", re.compile(r"\b\d{3}-\d{2}-\d{4}\b")),
    ("pii-phone", "phone numbers", re.compile(r"\b\+?\d{1,2}[ .-]?\(?\d{3}\)?[ .-]?\d{3}[ .-]?\d{4}\b")),
    ("cred-aws-key", "AWS access keys", re.compile(r"\bAKIA[0-9A-Z]{16}\b")),
    ("cred-private-key", "private key material", re.compile(r"-----BEGIN (?:RSA |EC )?PRIVATE KEY-----")),
    ("cred-token", "bearer tokens", re.compile(r"\b(?:ghp|gho|xoxb|sk)[-_][A-Za-z0-9_-]{16,}\b")),
    ("phi-mrn", "medical record numbers", re.compile(r"\bMRN[:# ]\s*\d{6,10}\b", re.I)),
]

_NAME_HINTS: list[tuple[str, str]] = [
    ("pii", r"(?i)(customer|user|person|contact|email|ssn|dob)"),
    ("credentials", r"(?i)(secret|credential|token|apikey|api_key|password)"),
    ("financial", r"(?i)(invoice|payment|card|billing|payroll)"),
    ("phi", r"(?i)(patient|medical|clinical|health)"),
]


def classify_data_asset(name: str, sample_text: str = "",
                        public: bool = False) -> dict[str, Any]:
    """DSPM classification of one data asset (bucket/table/file).

    Classification uses the asset NAME (schema hints) plus an optional
    bounded content sample; returns categories + matched detectors and a
    severity that escalates when a sensitive asset is also public."""
    categories: set[str] = set()
    detectors: list[str] = []
    for cat, pat in _NAME_HINTS:
        if re.search(pat, name or ""):
            categories.add(cat)
    sample = (sample_text or "")[:200_000]
    for det_id, _desc, pat in _DSPM_PATTERNS:
        if pat.search(sample):
            detectors.append(det_id)
            categories.add(det_id.split("-")[0].replace("cred", "credentials"))
    sensitive = bool(categories)
    severity = "info"
    if sensitive:
        severity = "medium"
        if "credentials" in categories:
            severity = "high"
        if public:
            severity = "critical"
    return {"asset": name, "categories": sorted(categories),
            "detectors": detectors, "public": bool(public),
            "sensitive": sensitive, "severity": severity}


def dspm_scan_inventory(assets: list[dict[str, Any]]) -> list[CisCheckResult]:
    """Classify a data-asset inventory: [{"name": ..., "public": bool,
    "sample": "..."}] → failing rows for public-and-sensitive combos."""
    results: list[CisCheckResult] = []
    for a in assets or []:
        if not isinstance(a, dict):
            continue
        cls = classify_data_asset(str(a.get("name") or "?"),
                                  str(a.get("sample") or ""),
                                  bool(a.get("public")))
        if cls["sensitive"]:
            _check(results, "DSPM-1",
                   f"Sensitive data ({', '.join(cls['categories'])}) governed",
                   cls["severity"], f"data:{cls['asset']}",
                   not cls["public"],
                   "PUBLIC asset contains sensitive data" if cls["public"]
                   else "")
    return results


# ── dispatch ────────────────────────────────────────────────────────────────

_EVALUATORS = {
    "azure": evaluate_azure_inventory,
    "gcp": evaluate_gcp_inventory,
    "snowflake": evaluate_snowflake_inventory,
    "databricks": evaluate_databricks_inventory,
}


def scan_cloud_estate(path: str | Path, provider: str) -> list[CisCheckResult]:
    """File-based inventory scan for any supported provider (aws handled by
    scan/cloud.py; this adds azure/gcp/snowflake/databricks plus the
    iam/audit_trail/dspm sections when present in the same document)."""
    inv = json.loads(Path(path).read_text())
    if provider == "aws":
        from agentbom_amd.scan.cloud import evaluate_aws_inventory

        results = evaluate_aws_inventory(inv)
    elif provider in _EVALUATORS:
        results = _EVALUATORS[provider](inv)
    else:
        raise ValueError(f"unsupported provider {provider!r}; expected one of "
                         f"aws, {', '.join(sorted(_EVALUATORS))}")
    results.extend(evaluate_iam_policies(inv.get("iam_policies") or []))
    results.extend(ingest_audit_trail(inv.get("audit_events") or []))
    results.extend(dspm_scan_inventory(inv.get("data_assets") or []))
    return results
