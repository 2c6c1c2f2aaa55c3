"""Cloud estate posture: CIS-style checks over exported inventory files.

Reference: src/agent_bom/cloud/ (51,920 LoC: live AWS/Azure/GCP/Snowflake
collectors + CIS benchmarks + GPU-infra posture).  This build evaluates
the same CHECK classes over exported inventory JSON (``aws ... --output
json`` dumps, or this tool's own collector output) — the live-API
collector layer plugs in where ``load_inventory`` reads files, keeping
scans runnable in air-gapped environments.

Check packs: AWS CIS Foundations subset (IAM/S3/EC2/RDS/CloudTrail),
GPU-infrastructure posture (exposed inference endpoints, unauthenticated
model servers).
"""

from __future__ import annotations

import json
from dataclasses import dataclass
from pathlib import Path
from typing import Any, Callable, Optional

from agentbom_amd.models.finding import Asset, Finding, FindingSource, FindingType, stable_id


@dataclass
class CisCheckResult:
    check_id: str
    title: str
    severity: str
    status: str  # pass | fail | error
    resource: str
    detail: str = ""

    def to_dict(self) -> dict[str, Any]:
        return {"check_id": self.check_id, "title": self.title,
                "severity": self.severity, "status": self.status,
                "resource": self.resource, "detail": self.detail}


def _check(results, check_id, title, severity, resource, ok: bool, detail=""):
    results.append(CisCheckResult(
        check_id=check_id, title=title, severity=severity,
        status="pass" if ok else "fail", resource=resource, detail=detail))


def evaluate_aws_inventory(inv: dict[str, Any]) -> list[CisCheckResult]:
    """AWS CIS Foundations subset over an exported inventory document.

    Expected shape (produced by collectors or assembled by operators):
    {"iam_users": [...], "s3_buckets": [...], "security_groups": [...],
     "rds_instances": [...], "cloudtrail": {...}, "inference_endpoints": [...]}
    """
    results: list[CisCheckResult] = []

    for user in inv.get("iam_users", []) or []:
        name = user.get("UserName", "?")
        _check(results, "CIS-1.4", "IAM user access keys rotated <= 90 days", "medium",
               f"iam:{name}", not user.get("AccessKeyStale", False),
               "stale access key" if user.get("AccessKeyStale") else "")
        _check(results, "CIS-1.10", "MFA enabled for console users", "high",
               f"iam:{name}",
               bool(user.get("MFAEnabled", True)) or not user.get("ConsoleAccess", False))
        if user.get("AttachedPolicies") and any(
            p.get("PolicyName") == "AdministratorAccess" for p in user["AttachedPolicies"]
        ):
            _check(results, "CIS-1.16", "No directly-attached AdministratorAccess", "high",
                   f"iam:{name}", False, "AdministratorAccess attached directly")

    for bucket in inv.get("s3_buckets", []) or []:
        name = bucket.get("Name", "?")
        _check(results, "CIS-2.1.1", "S3 bucket blocks public access", "high",
               f"s3:{name}", bool(bucket.get("PublicAccessBlock", True)))
        _check(results, "CIS-2.1.2", "S3 bucket encryption enabled", "medium",
               f"s3:{name}", bool(bucket.get("Encryption", True)))

    for sg in inv.get("security_groups", []) or []:
        gid = sg.get("GroupId", "?")
        open_world = any(
            r.get("CidrIp") == "0.0.0.0/0" and r.get("FromPort") in (22, 3389, None)
            for r in sg.get("IngressRules", []) or []
        )
        _check(results, "CIS-5.2", "No admin ports open to 0.0.0.0/0", "critical",
               f"sg:{gid}", not open_world,
               "SSH/RDP (or all ports) open to the internet" if open_world else "")

    for rds in inv.get("rds_instances", []) or []:
        rid = rds.get("DBInstanceIdentifier", "?")
        _check(results, "CIS-2.3.1", "RDS not publicly accessible", "critical",
               f"rds:{rid}", not rds.get("PubliclyAccessible", False))
        _check(results, "CIS-2.3.2", "RDS storage encrypted", "medium",
               f"rds:{rid}", bool(rds.get("StorageEncrypted", True)))

    trail = inv.get("cloudtrail")
    if trail is not None:
        _check(results, "CIS-3.1", "CloudTrail enabled in all regions", "high",
               "cloudtrail", bool(trail.get("MultiRegion", False)))
        _check(results, "CIS-3.2", "CloudTrail log file validation enabled", "medium",
               "cloudtrail", bool(trail.get("LogFileValidation", True)))
        _check(results, "CIS-3.7", "CloudTrail logs encrypted with KMS", "medium",
               "cloudtrail", bool(trail.get("KmsKeyId")))

    root = inv.get("root_account")
    if root is not None:
        _check(results, "CIS-1.5", "Root account MFA enabled", "critical",
               "iam:root", bool(root.get("MFAEnabled", False)))
        _check(results, "CIS-1.7", "Root access keys do not exist", "critical",
               "iam:root", not root.get("AccessKeysPresent", False))
        _check(results, "CIS-1.1", "Root account not used in 90 days", "high",
               "iam:root", not root.get("RecentlyUsed", False))

    pw = inv.get("password_policy")
    if pw is not None:
        _check(results, "CIS-1.8", "Password policy minimum length >= 14", "medium",
               "iam:password-policy", int(pw.get("MinimumLength", 0)) >= 14)
        _check(results, "CIS-1.9", "Password reuse prevented (>= 24)", "low",
               "iam:password-policy", int(pw.get("ReusePrevention", 0)) >= 24)

    for vol in inv.get("ebs_volumes", []) or []:
        vid = vol.get("VolumeId", "?")
        _check(results, "CIS-2.2.1", "EBS volume encrypted", "medium",
               f"ebs:{vid}", bool(vol.get("Encrypted", True)))

    for fn in inv.get("lambda_functions", []) or []:
        name = fn.get("FunctionName", "?")
        env = (fn.get("Environment") or {}).get("Variables") or {}
        leaked = [k for k in env
                  if any(m in k.upper() for m in ("SECRET", "PASSWORD", "TOKEN",
                                                  "API_KEY", "ACCESS_KEY"))]
        _check(results, "LMB-1", "Lambda env free of secret-named variables",
               "high", f"lambda:{name}", not leaked,
               f"secret-named env: {', '.join(leaked[:3])}" if leaked else "")
        _check(results, "LMB-2", "Lambda not exposed via public URL without auth",
               "high", f"lambda:{name}",
               not (fn.get("FunctionUrlAuthType") == "NONE"))

    for kms in inv.get("kms_keys", []) or []:
        kid = kms.get("KeyId", "?")
        _check(results, "CIS-3.8", "KMS key rotation enabled", "medium",
               f"kms:{kid}", bool(kms.get("RotationEnabled", True)))

    for eks in inv.get("eks_clusters", []) or []:
        name = eks.get("Name", "?")
        _check(results, "EKS-1", "EKS API endpoint not public", "high",
               f"eks:{name}", not eks.get("EndpointPublicAccess", False))
        _check(results, "EKS-2", "EKS control-plane logging enabled", "medium",
               f"eks:{name}", bool(eks.get("LoggingEnabled", True)))

    # GPU/AI infrastructure posture (cloud/gpu_infra.py analog)
    for ep in inv.get("inference_endpoints", []) or []:
        name = ep.get("Name", "?")
        _check(results, "AIINF-1", "Inference endpoint requires authentication", "critical",
               f"inference:{name}", bool(ep.get("AuthRequired", True)),
               "unauthenticated model endpoint" if not ep.get("AuthRequired", True) else "")
        _check(results, "AIINF-2", "Inference endpoint not internet-exposed", "high",
               f"inference:{name}", not ep.get("PublicEndpoint", False))

    return results


def load_inventory(path: str | Path) -> dict[str, Any]:
    return json.loads(Path(path).read_text())


def scan_cloud_inventory(path: str | Path, provider: str = "aws") -> list[CisCheckResult]:
    inv = load_inventory(path)
    if provider == "aws":
        return evaluate_aws_inventory(inv)
    raise ValueError(f"provider {provider!r}: only file-based aws inventories "
                     "are supported in this build")


def cis_result_to_finding(r: CisCheckResult, provider: str = "aws") -> Optional[Finding]:
    if r.status == "pass":
        return None
    return Finding(
        finding_type=FindingType.CIS_FAIL if r.status == "fail" else FindingType.CIS_ERROR,
        source=FindingSource.CLOUD_CIS,
        asset=Asset(name=r.resource, asset_type="cloud_resource",
                    identifier=r.resource, provider=provider),
        severity=r.severity,
        title=f"{r.check_id}: {r.title}",
        description=r.detail or r.title,
        evidence=r.to_dict(),
        is_actionable=r.severity in ("critical", "high"),
        id=stable_id("cloud-cis", provider, r.check_id, r.resource),
    )
