"""Endpoint inventory, cloud CIS checks, Shield, tracing."""

import json

import pytest

from agentbom_amd.runtime.shield import PermissionError403, Shield
from agentbom_amd.scan.cloud import (
    cis_result_to_finding,
    evaluate_aws_inventory,
    scan_cloud_inventory,
)
from agentbom_amd.utils.tracing import Tracer, make_traceparent, parse_traceparent


class TestCloudCis:
    INV = {
        "iam_users": [
            {"UserName": "alice", "MFAEnabled": False, "ConsoleAccess": True},
            {"UserName": "ci-bot", "AccessKeyStale": True,
             "AttachedPolicies": [{"PolicyName": "AdministratorAccess"}]},
        ],
        "s3_buckets": [{"Name": "public-data", "PublicAccessBlock": False,
                        "Encryption": False}],
        "security_groups": [{"GroupId": "sg-1", "IngressRules": [
            {"CidrIp": "0.0.0.0/0", "FromPort": 22}]}],
        "rds_instances": [{"DBInstanceIdentifier": "prod-db",
                           "PubliclyAccessible": True}],
        "cloudtrail": {"MultiRegion": False},
        "inference_endpoints": [{"Name": "llm-prod", "AuthRequired": False,
                                 "PublicEndpoint": True}],
    }

    def test_failures_detected(self):
        results = evaluate_aws_inventory(self.INV)
        fails = {r.check_id for r in results if r.status == "fail"}
        assert {"CIS-1.10", "CIS-1.4", "CIS-1.16", "CIS-2.1.1", "CIS-5.2",
                "CIS-2.3.1", "CIS-3.1", "AIINF-1", "AIINF-2"} <= fails

    def test_pass_on_clean(self):
        results = evaluate_aws_inventory({
            "s3_buckets": [{"Name": "safe", "PublicAccessBlock": True,
                            "Encryption": True}]})
        assert all(r.status == "pass" for r in results)

    def test_to_finding(self, tmp_path):
        (tmp_path / "inv.json").write_text(json.dumps(self.INV))
        results = scan_cloud_inventory(tmp_path / "inv.json")
        findings = [f for f in (cis_result_to_finding(r) for r in results) if f]
        assert findings
        f = next(x for x in findings if "AIINF-1" in x.title)
        assert f.severity == "critical"
        assert f.source.value == "CLOUD_CIS"


class TestEndpointInventory:
    def test_collect_runs(self):
        from agentbom_amd.scan.endpoint import collect_endpoint_inventory

        inv = collect_endpoint_inventory()
        doc = inv.to_dict()
        assert doc["schema_version"] == "1"
        assert isinstance(doc["processes"], list)
        # cmdline redaction contract
        for p in doc["processes"]:
            assert "--api-key " not in (p.get("cmdline") or "")

    def test_redaction(self):
        from agentbom_amd.scan.endpoint import _redact_cmdline

        out = _redact_cmdline(["server", "--api-key", "supersecret", "--token=abc"])
        assert "supersecret" not in out and "abc" not in out


class TestShield:
    def test_inline_block_paths(self):
        s = Shield()
        s.apply_write_action("block_tool", "run_shell", admin=True, reason="risk")
        d = s.decide({"method": "tools/call", "params": {"name": "run_shell"}})
        assert d.action == "block"
        s.apply_write_action("quarantine_upstream", "up1", admin=True, reason="drift")
        assert s.decide({}, target="up1").action == "block"
        s.apply_write_action("revoke_credential", "AKIA123", admin=True, reason="leak")
        d = s.decide({"method": "tools/call",
                      "params": {"name": "x", "arguments": {"k": "AKIA123"}}})
        assert d.action == "block"

    def test_write_actions_fail_closed(self):
        s = Shield()
        with pytest.raises(PermissionError403):
            s.apply_write_action("block_tool", "x", admin=False, reason="r")
        with pytest.raises(ValueError):
            s.apply_write_action("block_tool", "x", admin=True, reason="  ")
        with pytest.raises(ValueError):
            s.apply_write_action("nuke", "x", admin=True, reason="r")
        assert s.audit == []

    def test_audit_trail(self):
        s = Shield()
        entry = s.apply_write_action("block_tool", "t", admin=True, reason="why")
        assert entry["reason"] == "why" and s.audit == [entry]


class TestTracing:
    def test_traceparent_roundtrip(self):
        tp = make_traceparent("a" * 32, "b" * 16)
        assert parse_traceparent(tp) == ("a" * 32, "b" * 16)
        assert parse_traceparent("junk") is None
        assert parse_traceparent("00-" + "0" * 32 + "-" + "b" * 16 + "-01") is None

    def test_span_nesting(self):
        t = Tracer()
        t.start_trace()
        with t.span("scan", demo=True) as outer:
            with t.span("match") as inner:
                assert inner.parent_span_id == outer.span_id
        spans = t.export()
        assert [s["name"] for s in spans] == ["match", "scan"]
        assert all(s["duration_ms"] >= 0 for s in spans)

    def test_propagated_parent(self):
        t = Tracer()
        tid = t.start_trace(make_traceparent("c" * 32, "d" * 16))
        assert tid == "c" * 32
        with t.span("child") as s:
            assert s.parent_span_id == "d" * 16

    def test_error_status(self):
        t = Tracer()
        with pytest.raises(RuntimeError):
            with t.span("boom"):
                raise RuntimeError("x")
        assert t.export()[0]["status"] == "error"
