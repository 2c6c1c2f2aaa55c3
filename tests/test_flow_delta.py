"""React-Flow exports + finding delta stream tests."""

import json

import pytest

from agentbom_amd.output.delta_stream import DeltaStreamer
from agentbom_amd.output.flow_fmt import build_agent_mesh, build_attack_flow
from agentbom_amd.scan.orchestrator import run_demo_scan


@pytest.fixture(scope="module")
def report():
    return run_demo_scan()


class TestAttackFlow:
    def test_chain_shape(self, report):
        flow = build_attack_flow(report)
        kinds = {n["data"]["kind"] for n in flow["nodes"]}
        assert {"cve", "package", "server"} <= kinds
        ids = {n["id"] for n in flow["nodes"]}
        for e in flow["edges"]:
            assert e["source"] in ids and e["target"] in ids
        assert flow["stats"]["findings_rendered"] == len(report.blast_radii)

    def test_cve_filter(self, report):
        flow = build_attack_flow(report, cve="CVE-2020-14343")
        cves = [n for n in flow["nodes"] if n["data"]["kind"] == "cve"]
        assert len(cves) == 1 and cves[0]["data"]["label"] == "CVE-2020-14343"
        assert cves[0]["data"]["severity"] == "critical"

    def test_severity_filter(self, report):
        flow = build_attack_flow(report, min_severity="critical")
        for n in flow["nodes"]:
            if n["data"]["kind"] == "cve":
                assert n["data"]["severity"] == "critical"

    def test_kev_edges_animated(self, report):
        flow = build_attack_flow(report, cve="CVE-2023-4863")  # KEV in demo
        affected = [e for e in flow["edges"] if e["label"] == "affects"]
        assert affected and affected[0]["animated"]


class TestAgentMesh:
    def test_mesh_stats_and_sharing(self, report):
        mesh = build_agent_mesh(report)
        assert mesh["stats"]["total_agents"] == len(report.agents)
        server_nodes = [n for n in mesh["nodes"]
                        if n["data"]["kind"] == "server"]
        # shared servers are merged into one node
        assert len(server_nodes) == mesh["stats"]["total_servers"]
        assert any(n["data"]["vuln_count"] > 0 for n in server_nodes)

    def test_serializable(self, report):
        assert json.loads(json.dumps(build_agent_mesh(report)))


class TestDeltaStream:
    def test_first_emit_all_new_then_quiet(self, report):
        ds = DeltaStreamer()
        events = ds.emit(report)
        assert events and all(e["kind"] == "new" for e in events)
        assert [e["seq"] for e in events] == list(range(1, len(events) + 1))
        assert ds.watermark["seq"] == len(events)
        # unchanged re-emit -> no events, watermark stable
        assert ds.emit(report) == []
        assert ds.watermark["seq"] == len(events)

    def test_changed_and_resolved(self, report):
        ds = DeltaStreamer()
        ds.emit(report)
        # mutate one finding + drop another
        report2 = run_demo_scan()
        report2.blast_radii[0].risk_score = 1.23
        dropped = report2.blast_radii.pop()
        events = ds.emit(report2)
        kinds = {e["kind"] for e in events}
        assert kinds == {"changed", "resolved"}
        changed = next(e for e in events if e["kind"] == "changed")
        assert "risk_score" in changed["changes"]
        resolved = next(e for e in events if e["kind"] == "resolved")
        assert resolved["finding"]["vulnerability_id"] == dropped.vulnerability.id
        # sequence continues across emits
        assert min(e["seq"] for e in events) > 0

    def test_ndjson_sink_and_ocsf(self, report):
        lines: list[str] = []
        ds = DeltaStreamer(fmt="ocsf")
        ds.emit(report, sink=lines.append)
        assert lines
        doc = json.loads(lines[0])
        assert doc["class_uid"] == 2002
        assert doc["activity_id"] == 1  # new
        assert doc["metadata"]["sequence"] == 1

    def test_persistent_state(self, tmp_path, report):
        path = str(tmp_path / "delta.db")
        ds1 = DeltaStreamer(state_path=path)
        n = len(ds1.emit(report))
        ds1.close()
        ds2 = DeltaStreamer(state_path=path)
        assert ds2.emit(report) == []  # state survived restart
        assert ds2.watermark["seq"] == n


class TestFlowRoutes:
    def test_routes(self):
        from fastapi.testclient import TestClient

        from agentbom_amd.api.server import create_app

        c = TestClient(create_app())
        job = c.post("/v1/scan", json={"demo": True}).json()
        import time

        for _ in range(100):
            if c.get(f"/v1/scan/{job['job_id']}").json()["status"] == "done":
                break
            time.sleep(0.1)
        flow = c.get("/v1/graph/attack-flow?min_severity=critical").json()
        assert flow["nodes"]
        mesh = c.get("/v1/mesh").json()
        assert mesh["stats"]["total_agents"] > 0
        d1 = c.get("/v1/findings/delta").json()
        assert d1["events"] and d1["events"][0]["kind"] == "new"
        d2 = c.get("/v1/findings/delta").json()
        assert d2["events"] == []
        assert d2["watermark"]["seq"] == d1["watermark"]["seq"]
