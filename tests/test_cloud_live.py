"""Live cloud collectors: SigV4 vectors + collectors over MockTransport."""

from __future__ import annotations

import datetime
import json

import httpx
import pytest

from agentbom_amd.scan.cloud_live import (
    AwsCollector,
    AwsCredentials,
    collect_azure_inventory,
    collect_gcp_inventory,
    sigv4_headers,
)
from agentbom_amd.utils.http_client import OfflineError, set_offline

_T = datetime.datetime(2015, 8, 30, 12, 36, 0, tzinfo=datetime.timezone.utc)
_AK = "AKIDEXAMPLE"
_SK = "wJalrXUtnFEMI/K7MDENG+bPxRfiCYEXAMPLEKEY"


@pytest.fixture(autouse=True)
def _online():
    set_offline(False)
    yield
    set_offline(False)


class TestSigV4Vectors:
    """AWS's published SigV4 test-suite vectors (exact signatures)."""

    def _sig(self, headers):
        return headers["Authorization"].rsplit("Signature=", 1)[1]

    def test_get_vanilla(self):
        h = sigv4_headers("GET", "https://example.amazonaws.com/",
                          "us-east-1", "service", _AK, _SK, now=_T)
        assert self._sig(h) == ("5fa00fa31553b73ebf1942676e86291e"
                                "8372ff2a2260956d9b8aae1d763fbf31")
        assert "SignedHeaders=host;x-amz-date" in h["Authorization"]

    def test_get_vanilla_query_order_key(self):
        h = sigv4_headers(
            "GET", "https://example.amazonaws.com/?Param2=value2&Param1=value1",
            "us-east-1", "service", _AK, _SK, now=_T)
        assert self._sig(h) == ("b97d918cfa904a5beff61c982a1b6f45"
                                "8b799221646efd99d3219ec94cdf2500")

    def test_session_token_signed(self):
        h = sigv4_headers("GET", "https://example.amazonaws.com/",
                          "us-east-1", "service", _AK, _SK, now=_T,
                          session_token="FQoGZXIvYXdzEXAMPLE")
        assert "x-amz-security-token" in h["Authorization"]
        assert h["X-Amz-Security-Token"] == "FQoGZXIvYXdzEXAMPLE"

    def test_body_changes_signature(self):
        a = sigv4_headers("POST", "https://iam.amazonaws.com/", "us-east-1",
                          "iam", _AK, _SK, body=b"Action=A", now=_T)
        b = sigv4_headers("POST", "https://iam.amazonaws.com/", "us-east-1",
                          "iam", _AK, _SK, body=b"Action=B", now=_T)
        assert self._sig(a) != self._sig(b)


@pytest.fixture
def aws():
    """Synthetic AWS Query/S3 endpoints, recording signed requests."""
    seen = {"hosts": [], "auth_ok": True}

    def handler(request: httpx.Request) -> httpx.Response:
        seen["hosts"].append(request.url.host)
        auth = request.headers.get("Authorization", "")
        if not auth.startswith("AWS4-HMAC-SHA256 Credential=AKIA"):
            seen["auth_ok"] = False
        body = request.content.decode()
        host = request.url.host
        if host.startswith("iam.") and "Action=ListUsers" in body:
            return httpx.Response(200, text=(
                "<ListUsersResponse><Users>"
                "<member><UserName>alice</UserName></member>"
                "<member><UserName>bob</UserName></member>"
                "</Users></ListUsersResponse>"))
        if host.startswith("iam.") and "Action=ListMFADevices" in body:
            if "alice" in body:
                return httpx.Response(200, text=(
                    "<r><member><SerialNumber>arn:mfa</SerialNumber></member></r>"))
            return httpx.Response(200, text="<r></r>")
        if host.startswith("s3."):
            return httpx.Response(200, text=(
                "<ListAllMyBucketsResult><Buckets>"
                "<Bucket><Name>data-lake</Name></Bucket>"
                "</Buckets></ListAllMyBucketsResult>"))
        if host.startswith("ec2."):
            return httpx.Response(200, text=(
                "<DescribeSecurityGroupsResponse><securityGroupInfo><item>"
                "<groupId>sg-123</groupId>"
                "<ipRanges><item><cidrIp>0.0.0.0/0</cidrIp></item></ipRanges>"
                "</item></securityGroupInfo></DescribeSecurityGroupsResponse>"))
        if host.startswith("rds."):
            return httpx.Response(200, text=(
                "<R><DBInstance>"
                "<DBInstanceIdentifier>prod-db</DBInstanceIdentifier>"
                "<PubliclyAccessible>true</PubliclyAccessible>"
                "<StorageEncrypted>false</StorageEncrypted>"
                "</DBInstance></R>"))
        if host.startswith("cloudtrail."):
            return httpx.Response(200, text=(
                "<R><IsMultiRegionTrail>true</IsMultiRegionTrail>"
                "<LogFileValidationEnabled>true</LogFileValidationEnabled></R>"))
        return httpx.Response(404)

    return handler, seen


def _collector(handler):
    client = httpx.Client(transport=httpx.MockTransport(handler))
    creds = AwsCredentials(access_key="AKIAXXXXEXAMPLE",
                           secret_key="secretsecretsecret", region="us-east-1")
    return AwsCollector(creds=creds, client=client)


def test_aws_collect_inventory_feeds_cis(aws):
    handler, seen = aws
    inv = _collector(handler).collect_inventory()
    assert seen["auth_ok"], "requests were not SigV4-signed"
    users = {u["UserName"]: u for u in inv["iam_users"]}
    assert users["alice"]["MFAEnabled"] and not users["bob"]["MFAEnabled"]
    assert inv["s3_buckets"][0]["Name"] == "data-lake"
    assert inv["security_groups"][0]["IngressRules"][0]["CidrIp"] == "0.0.0.0/0"
    assert inv["rds_instances"][0]["PubliclyAccessible"] is True
    assert inv["cloudtrail"]["MultiRegion"] is True

    # the live inventory drives the SAME check pack as the file drop
    from agentbom_amd.scan.cloud import evaluate_aws_inventory

    fails = {r.check_id for r in evaluate_aws_inventory(inv)
             if r.status == "fail"}
    assert "CIS-1.10" in fails   # bob without MFA
    assert "CIS-5.2" in fails    # open security group
    assert "CIS-2.3.1" in fails  # public RDS


def test_aws_missing_creds_points_to_file_path(monkeypatch):
    for var in ("AWS_ACCESS_KEY_ID", "AWS_SECRET_ACCESS_KEY"):
        monkeypatch.delenv(var, raising=False)
    with pytest.raises(RuntimeError, match="aws-inventory"):
        AwsCollector(creds=AwsCredentials(access_key="", secret_key=""))


def test_aws_offline_refused(aws):
    handler, _ = aws
    coll = _collector(handler)
    set_offline(True)
    with pytest.raises(OfflineError):
        coll.collect_iam_users()


def test_azure_inventory(monkeypatch):
    def handler(request):
        assert request.headers["Authorization"] == "Bearer az-tok"
        if "storageAccounts" in request.url.path:
            return httpx.Response(200, json={"value": [{
                "name": "sa1", "properties": {"allowBlobPublicAccess": True,
                                              "supportsHttpsTrafficOnly": False}}]})
        return httpx.Response(200, json={"value": [{"name": "vm1",
                                                    "properties": {}}]})

    client = httpx.Client(transport=httpx.MockTransport(handler))
    inv = collect_azure_inventory("sub-1", "az-tok", client=client)
    assert inv["storage_accounts"][0]["allowBlobPublicAccess"] is True
    assert inv["storage_accounts"][0]["supportsHttpsTrafficOnly"] is False
    assert inv["virtual_machines"][0]["name"] == "vm1"


def test_gcp_inventory():
    def handler(request):
        assert request.headers["Authorization"] == "Bearer gcp-tok"
        if request.url.host == "storage.googleapis.com":
            return httpx.Response(200, json={"items": [{
                "name": "b1", "iamConfiguration": {
                    "publicAccessPrevention": "inherited",
                    "uniformBucketLevelAccess": {"enabled": False}}}]})
        return httpx.Response(200, json={"items": {"zones/z1": {
            "instances": [{"name": "i1", "networkInterfaces": [
                {"accessConfigs": [{"natIP": "1.2.3.4"}]}]}]}}})

    client = httpx.Client(transport=httpx.MockTransport(handler))
    inv = collect_gcp_inventory("proj", "gcp-tok", client=client)
    assert inv["buckets"][0]["publicAccessPrevention"] == "inherited"
    assert inv["instances"][0]["publicIp"] is True
