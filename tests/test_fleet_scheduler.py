"""Fleet registry, scan scheduler, backpressure, audit + compliance routes."""

import time

import pytest
from fastapi.testclient import TestClient

from agentbom_amd.api.fleet import BackpressureController, FleetRegistry, ScanScheduler
from agentbom_amd.api.server import create_app


class TestFleetRegistry:
    def test_heartbeat_and_reconcile(self):
        fleet = FleetRegistry()
        fleet.heartbeat({"member_id": "h1", "hostname": "host1", "agents": 3,
                         "servers": 7, "findings": 12})
        fleet.heartbeat({"member_id": "h2", "hostname": "host2", "agents": 2})
        fleet.heartbeat({"member_id": "h1", "hostname": "host1", "agents": 4})
        rec = fleet.reconcile()
        assert rec["members"] == 2 and rec["healthy"] == 2
        assert rec["total_agents"] == 6  # h1 updated to 4
        assert rec["observations"] == 3

    def test_stale_member(self):
        fleet = FleetRegistry()
        m = fleet.heartbeat({"member_id": "old", "hostname": "old"})
        m.last_heartbeat = time.time() - 3600
        assert fleet.reconcile()["stale"] == 1

    def test_reconcile_throughput(self):
        """10k observations well above the reference's 64.6k obs/s budget."""
        fleet = FleetRegistry()
        t0 = time.perf_counter()
        for i in range(10_000):
            fleet.heartbeat({"member_id": f"m{i % 500}", "hostname": f"m{i % 500}",
                             "agents": 1})
        elapsed = time.perf_counter() - t0
        assert fleet.reconcile()["members"] == 500
        assert elapsed < 2.0, f"10k heartbeats took {elapsed:.2f}s"


class TestScheduler:
    def test_tick_fires_due(self):
        runs = []
        sched = ScanScheduler(run_scan=runs.append)
        sched.add("s1", interval_s=10, demo=True)
        now = time.time()
        assert sched.tick(now) == 0  # not due yet
        assert sched.tick(now + 11) == 1
        assert runs[0]["schedule_id"] == "s1"
        s = sched.schedules["s1"]
        assert s.runs == 1 and s.next_run > now + 11

    def test_remove(self):
        sched = ScanScheduler(run_scan=lambda p: None)
        sched.add("x", 5)
        assert sched.remove("x") and not sched.remove("x")


class TestBackpressure:
    def test_admission_and_adaptation(self):
        bp = BackpressureController(max_concurrent=2)
        ok1, _ = bp.try_acquire()
        ok2, _ = bp.try_acquire()
        ok3, retry = bp.try_acquire()
        assert ok1 and ok2 and not ok3
        assert retry > 0 and bp.rejections == 1
        bp.release()
        ok4, _ = bp.try_acquire()
        assert ok4


class TestApiRoutes:
    @pytest.fixture(scope="class")
    def client(self):
        c = TestClient(create_app())
        assert c.post("/v1/scan", json={"demo": True}).json()["status"] == "done"
        return c

    def test_fleet_routes(self, client):
        r = client.post("/v1/fleet/heartbeat",
                        json={"member_id": "w1", "hostname": "w1", "agents": 5,
                              "findings": 9})
        assert r.json()["status"] == "healthy"
        fleet = client.get("/v1/fleet").json()
        assert fleet["reconciliation"]["total_findings"] == 9

    def test_schedule_routes(self, client):
        r = client.post("/v1/schedules", json={"schedule_id": "nightly",
                                               "interval_s": 3600})
        assert r.status_code == 201
        assert any(s["schedule_id"] == "nightly"
                   for s in client.get("/v1/schedules").json()["schedules"])
        assert client.delete("/v1/schedules/nightly").status_code == 204
        assert client.delete("/v1/schedules/nightly").status_code == 404

    def test_audit_routes(self, client):
        r = client.post("/v1/proxy/audit",
                        json={"entries": [{"action": "allow", "method": "tools/call"}]})
        assert r.json()["ingested"] == 1
        assert client.get("/v1/proxy/audit").json()["total"] >= 1

    def test_compliance_report(self, client):
        r = client.get("/v1/compliance/nist_800_53/report").json()
        assert r["tagged_findings"] > 0
        assert "RA-5" in r["controls"]
        assert client.get("/v1/compliance/bogus/report").status_code == 404


class TestOpenApiContract:
    def test_spec_matches_committed_snapshot(self):
        """The committed docs/openapi/v1.json is the REST contract; route
        changes must re-export it (reference: PROJECT_STRUCTURE.md:31)."""
        import json
        from pathlib import Path

        from agentbom_amd.api.server import create_app

        committed = json.loads(
            (Path(__file__).resolve().parents[1] / "docs/openapi/v1.json").read_text())
        live = create_app().openapi()
        assert sorted(live["paths"]) == sorted(committed["paths"]), (
            "REST surface changed — regenerate docs/openapi/v1.json")
