"""Dataset version registry: write-once digests, verify, API."""

from __future__ import annotations

import pytest

from agentbom_amd.api.dataset_versions import (
    DatasetVersionConflict,
    DatasetVersionRecord,
    DatasetVersionStore,
)


def _rec(v="v1", digest="abc123"):
    return DatasetVersionRecord(tenant_id="t1", dataset_id="train-set",
                                version_id=v, digest=digest,
                                source="s3://bucket/train")


class TestStore:
    def test_put_get_list_ordering(self):
        store = DatasetVersionStore()
        store.put(_rec("v1"))
        store.put(_rec("v2", digest="def456"))
        assert store.get("t1", "train-set", "v1").digest == "abc123"
        assert [r.version_id for r in store.list("t1", "train-set")] \
            == ["v2", "v1"]  # newest first
        assert store.list("t2") == []

    def test_write_once_conflict(self):
        store = DatasetVersionStore()
        store.put(_rec("v1", digest="abc123"))
        store.put(_rec("v1", digest="abc123"))  # idempotent same content
        with pytest.raises(DatasetVersionConflict, match="immutable"):
            store.put(_rec("v1", digest="TAMPERED"))

    def test_verify(self):
        store = DatasetVersionStore()
        store.put(_rec("v1", digest="abc123"))
        assert store.verify("t1", "train-set", "v1", "abc123")["status"] == "match"
        assert store.verify("t1", "train-set", "v1", "zzz")["status"] == "MISMATCH"
        assert store.verify("t1", "train-set", "v9", "x")["status"] == \
            "unknown_version"
        store.put(DatasetVersionRecord(tenant_id="t1", dataset_id="d2",
                                       version_id="v1"))
        assert store.verify("t1", "d2", "v1", "x")["status"] == \
            "no_pinned_digest"


def test_endpoints():
    from starlette.testclient import TestClient

    from agentbom_amd.api.server import create_app

    client = TestClient(create_app())
    r = client.post("/v1/datasets/train-set/versions",
                    json={"version_id": "v1", "digest": "abc123"})
    assert r.status_code == 201
    assert client.post("/v1/datasets/train-set/versions",
                       json={"version_id": "v1",
                             "digest": "OTHER"}).status_code == 409
    listed = client.get("/v1/datasets/train-set/versions").json()
    assert listed["total"] == 1
    v = client.post("/v1/datasets/train-set/versions/v1/verify",
                    json={"digest": "abc123"}).json()
    assert v["status"] == "match"
