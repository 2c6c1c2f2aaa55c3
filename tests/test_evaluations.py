"""Evaluation runs: lineage-checked recording + regression compare."""

from __future__ import annotations

import pytest

from agentbom_amd.api.dataset_versions import (
    DatasetVersionRecord,
    DatasetVersionStore,
)
from agentbom_amd.api.evaluations import EvaluationRun, EvaluationStore


class TestStore:
    def test_lineage_enforced(self):
        ds = DatasetVersionStore()
        store = EvaluationStore(dataset_store=ds)
        with pytest.raises(ValueError, match="lineage"):
            store.put(EvaluationRun(tenant_id="t1", name="safety",
                                    dataset_id="d1",
                                    dataset_version_id="v1"))
        ds.put(DatasetVersionRecord(tenant_id="t1", dataset_id="d1",
                                    version_id="v1", digest="x"))
        run = store.put(EvaluationRun(tenant_id="t1", name="safety",
                                      dataset_id="d1",
                                      dataset_version_id="v1",
                                      scores={"accuracy": 0.9}))
        assert store.get("t1", run.evaluation_id).scores == {"accuracy": 0.9}
        assert store.get("t2", run.evaluation_id) is None

    def test_compare_flags_regressions(self):
        store = EvaluationStore()
        store.put(EvaluationRun(tenant_id="t1", name="safety",
                                scores={"accuracy": 0.90, "refusal": 0.99},
                                created_at="2026-09-01T00:00:00+00:00"))
        store.put(EvaluationRun(tenant_id="t1", name="safety",
                                scores={"accuracy": 0.95, "refusal": 0.97},
                                created_at="2026-09-02T00:00:00+00:00"))
        out = store.compare("t1", "safety")
        assert out["status"] == "ok"
        assert out["deltas"]["accuracy"]["delta"] == pytest.approx(0.05)
        assert out["regressions"] == ["refusal"]
        assert store.compare("t1", "other")["status"] == "insufficient_history"


def test_endpoints():
    from starlette.testclient import TestClient

    from agentbom_amd.api.server import create_app

    client = TestClient(create_app())
    client.post("/v1/datasets/d1/versions", json={"version_id": "v1",
                                                  "digest": "abc"})
    r = client.post("/v1/evaluations", json={
        "name": "safety", "dataset_id": "d1", "dataset_version_id": "v1",
        "scores": {"accuracy": 0.9}})
    assert r.status_code == 201
    client.post("/v1/evaluations", json={"name": "safety",
                                         "scores": {"accuracy": 0.8}})
    assert client.get("/v1/evaluations?name=safety").json()["total"] == 2
    cmp = client.get("/v1/evaluations/compare/safety").json()
    assert cmp["regressions"] == ["accuracy"]
    bad = client.post("/v1/evaluations", json={
        "name": "x", "dataset_id": "d9", "dataset_version_id": "v9"})
    assert bad.status_code == 409
