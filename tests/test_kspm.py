"""KSPM cluster-posture checks: pods, RBAC, collector evidence states."""

from __future__ import annotations

import json

from agentbom_amd.scan.kspm import (
    evaluate_cluster_inventory,
    kspm_finding_to_finding,
    scan_cluster_posture,
)


def _pod(name="p1", ns="default", **spec):
    base = {"containers": [{"name": "c1", "securityContext":
                            {"runAsNonRoot": True}}]}
    base.update(spec)
    return {"metadata": {"name": name, "namespace": ns}, "spec": base}


def _inventory(**kw):
    inv = {"pods": {"items": []}, "roles": {"items": []},
           "cluster_roles": {"items": []}, "role_bindings": {"items": []},
           "cluster_role_bindings": {"items": []}}
    inv.update(kw)
    return inv


class TestPodChecks:
    def test_privileged_and_hostpath(self):
        pod = _pod(spec_extra=None, hostNetwork=True, volumes=[
            {"name": "sock", "hostPath": {"path": "/var/run/docker.sock"}}],
            containers=[{"name": "c1", "securityContext": {
                "privileged": True, "allowPrivilegeEscalation": True}}])
        res = evaluate_cluster_inventory(_inventory(pods={"items": [pod]}))
        ids = {f.check_id for f in res.findings}
        assert {"KSPM-POD-001", "KSPM-POD-004", "KSPM-POD-005",
                "KSPM-POD-006"} <= ids
        crit = [f for f in res.findings if f.check_id == "KSPM-POD-004"]
        assert crit[0].severity == "critical"  # docker.sock mount
        assert res.status == "complete"

    def test_root_and_default_sa(self):
        pod = _pod(containers=[{"name": "c1", "securityContext":
                                {"runAsUser": 0}}])
        res = evaluate_cluster_inventory(_inventory(pods={"items": [pod]}))
        ids = {f.check_id: f.severity for f in res.findings}
        assert ids.get("KSPM-POD-007") == "high"   # explicit root
        assert "KSPM-POD-008" in ids               # default SA automount
        # compliant pod -> only the SA finding disappears with a named SA
        good = _pod(serviceAccountName="scanner-sa")
        res2 = evaluate_cluster_inventory(_inventory(pods={"items": [good]}))
        assert not {f.check_id for f in res2.findings} & {"KSPM-POD-008",
                                                          "KSPM-POD-007"}


class TestRbac:
    def test_wildcards_skip_builtin(self):
        custom = {"metadata": {"name": "ops-role"},
                  "rules": [{"verbs": ["*"], "resources": ["pods"]}]}
        builtin = {"metadata": {"name": "system:controller:x"},
                   "rules": [{"verbs": ["*"], "resources": ["*"]}]}
        res = evaluate_cluster_inventory(_inventory(
            cluster_roles={"items": [custom, builtin]}))
        hits = [f for f in res.findings if f.check_id == "KSPM-RBAC-001"]
        assert len(hits) == 1 and hits[0].resource == "ops-role"

    def test_anonymous_binding_critical(self):
        b = {"metadata": {"name": "bad-bind"},
             "subjects": [{"kind": "User", "name": "system:anonymous"}]}
        res = evaluate_cluster_inventory(_inventory(
            cluster_role_bindings={"items": [b]}))
        f = [x for x in res.findings if x.check_id == "KSPM-RBAC-002"][0]
        assert f.severity == "critical"


class TestEvidence:
    def test_denied_read_is_partial_never_clean(self):
        res = evaluate_cluster_inventory({"pods": {"items": []}})
        assert res.status == "partial"  # roles etc. absent -> unevaluable
        states = {c.name: c.status for c in res.collectors}
        assert states["pods"] == "executed"
        assert states["roles"] == "unevaluable"

    def test_failed_reader(self):
        res = evaluate_cluster_inventory(
            _inventory(), reader_errors={"pods": "403 forbidden"})
        states = {c.name: c.status for c in res.collectors}
        assert states["pods"] == "failed" and res.status == "partial"

    def test_evidence_dict_and_file_scan(self, tmp_path):
        pod = _pod(hostPID=True)
        p = tmp_path / "cluster.json"
        p.write_text(json.dumps(_inventory(pods={"items": [pod]})))
        res = scan_cluster_posture(str(p))
        ev = res.to_evidence_dict()
        assert ev["schema_version"] == "kspm.cluster.posture.v1"
        assert ev["finding_count"] == len(ev["findings"]) >= 1
        assert ev["severity_summary"]["high"] >= 1

    def test_bridge_to_unified_finding(self):
        pod = _pod(hostNetwork=True)
        res = evaluate_cluster_inventory(_inventory(pods={"items": [pod]}))
        f = kspm_finding_to_finding(res.findings[0])
        assert f.severity == "high" and "KSPM-POD-001" in f.title
        assert "CIS-K8s-5.2.4" in f.compliance_tags
