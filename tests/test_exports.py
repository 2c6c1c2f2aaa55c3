"""Scheduled exports: destinations, renderers, delivery, schedule ticks."""

from __future__ import annotations

import json

import pytest

from agentbom_amd.api.exports import (
    EXPORT_FORMATS,
    ExportDestination,
    ExportManager,
)


@pytest.fixture(scope="module")
def report():
    from agentbom_amd.scan.orchestrator import run_demo_scan

    return run_demo_scan()


@pytest.fixture
def mgr(report, tmp_path):
    posts = []

    def post(url, body):
        posts.append((url, body))
        return "fail" not in url

    m = ExportManager(get_report=lambda: report, file_root=str(tmp_path),
                      http_post=post)
    m._posts = posts
    return m


class TestDelivery:
    def test_file_export_all_formats(self, mgr, tmp_path):
        for fmt in EXPORT_FORMATS:
            dest = mgr.add_destination(ExportDestination(
                target=f"out/report.{fmt}", format=fmt))
            out = mgr.run_export(dest.destination_id)
            assert out["ok"], (fmt, out)
            assert (tmp_path / "out" / f"report.{fmt}").exists()
        # json file parses
        doc = json.loads((tmp_path / "out" / "report.json").read_text())
        assert doc["schema_version"] == "1.0"

    def test_webhook_export_and_failure(self, mgr):
        ok_dest = mgr.add_destination(ExportDestination(
            target="https://siem.example/ingest", format="sarif"))
        bad_dest = mgr.add_destination(ExportDestination(
            target="https://fail.example/x", format="json"))
        assert mgr.run_export(ok_dest.destination_id)["ok"]
        assert not mgr.run_export(bad_dest.destination_id)["ok"]
        url, body = mgr._posts[0]
        assert "siem.example" in url and "sarif" in body[:200].lower() or body

    def test_path_escape_refused(self, mgr):
        dest = mgr.add_destination(ExportDestination(
            target="../../etc/pwned", format="json"))
        out = mgr.run_export(dest.destination_id)
        assert not out["ok"] and "escapes" in out["error"]

    def test_unknown_format_rejected(self):
        with pytest.raises(ValueError, match="unknown export format"):
            ExportDestination(target="x", format="xml")


class TestSchedule:
    def test_tick_fires_due_only(self, mgr):
        dest = mgr.add_destination(ExportDestination(
            target="https://siem.example/a", format="json"))
        sched = mgr.add_schedule(dest.destination_id, interval_s=100)
        assert mgr.tick(now=sched.next_run - 1) == 0
        assert mgr.tick(now=sched.next_run + 1) == 1
        assert sched.runs == 1 and sched.last_status == "ok"
        # rescheduled into the future
        assert mgr.tick(now=sched.next_run - 1) == 0
        assert mgr.add_schedule("dest-missing", 10) is None
        assert mgr.remove_schedule(sched.schedule_id)


class TestApi:
    def test_export_endpoints(self, tmp_path, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_EXPORT_ROOT", str(tmp_path))
        from starlette.testclient import TestClient

        from agentbom_amd.api.server import create_app

        client = TestClient(create_app())
        client.post("/v1/scan", json={"demo": True})
        r = client.post("/v1/exports/destinations",
                        json={"target": "reports/latest.sarif",
                              "format": "sarif", "name": "siem-file"})
        assert r.status_code == 201
        did = r.json()["destination_id"]
        assert client.get("/v1/exports/destinations").json()["destinations"]
        out = client.post(f"/v1/exports/run/{did}").json()
        assert out["ok"] and (tmp_path / "reports" / "latest.sarif").exists()
        s = client.post("/v1/exports/schedules",
                        json={"destination_id": did, "interval_s": 3600})
        assert s.status_code == 201
        listed = client.get("/v1/exports/schedules").json()
        assert listed["schedules"] and listed["recent_deliveries"]
        assert client.post("/v1/exports/run/dest-nope").status_code == 404
        assert client.post("/v1/exports/destinations",
                           json={"target": "x", "format": "xml"}).status_code == 400
