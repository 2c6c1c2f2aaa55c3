"""Tenant-scoped dataset version registry (AI-BOM dataset lineage).

Reference parity: src/agent_bom/api/dataset_version_store.py — datasets
are first-class AI-BOM entities; this registry records immutable,
digest-pinned versions per (tenant, dataset) so the graph's DATASET
nodes and training-pipeline evidence can reference an exact artifact:

- a version is WRITE-ONCE: re-putting an existing (dataset, version)
  with a DIFFERENT digest is refused (provenance must not be rewritten);
- digests are algorithm-tagged (sha256 default); drift between a
  declared digest and a re-registration is surfaced, never merged.
"""

from __future__ import annotations

import json
import sqlite3
import threading
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Optional


class DatasetVersionConflict(ValueError):
    """Same (dataset, version) re-registered with different content."""


@dataclass(frozen=True)
class DatasetVersionRecord:
    tenant_id: str
    dataset_id: str
    version_id: str
    source: str = ""
    artifact_uri: Optional[str] = None
    digest: Optional[str] = None
    digest_algorithm: str = "sha256"
    created_at: str = ""
    metadata: dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> dict[str, Any]:
        return {"tenant_id": self.tenant_id, "dataset_id": self.dataset_id,
                "version_id": self.version_id, "source": self.source,
                "artifact_uri": self.artifact_uri, "digest": self.digest,
                "digest_algorithm": self.digest_algorithm,
                "created_at": self.created_at,
                "metadata": dict(self.metadata)}


_SCHEMA = """
CREATE TABLE IF NOT EXISTS dataset_versions (
    tenant_id TEXT NOT NULL,
    dataset_id TEXT NOT NULL,
    version_id TEXT NOT NULL,
    doc TEXT NOT NULL,
    PRIMARY KEY (tenant_id, dataset_id, version_id)
);
"""


class DatasetVersionStore:
    def __init__(self, path: str = ":memory:"):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._lock = threading.Lock()

    def put(self, record: DatasetVersionRecord) -> DatasetVersionRecord:
        if not record.created_at:
            record = DatasetVersionRecord(
                **{**record.to_dict(),
                   "created_at": datetime.now(timezone.utc).isoformat()})
        with self._lock:
            existing = self.get(record.tenant_id, record.dataset_id,
                                record.version_id)
            if existing is not None:
                if (existing.digest, existing.digest_algorithm) != \
                        (record.digest, record.digest_algorithm):
                    raise DatasetVersionConflict(
                        f"{record.dataset_id}@{record.version_id} already "
                        f"registered with digest {existing.digest!r}; a "
                        "version's content is immutable")
                return existing  # idempotent re-put of the same content
            self._db.execute(
                "INSERT INTO dataset_versions (tenant_id, dataset_id,"
                " version_id, doc) VALUES (?,?,?,?)",
                (record.tenant_id, record.dataset_id, record.version_id,
                 json.dumps(record.to_dict())))
            self._db.commit()
        return record

    def get(self, tenant_id: str, dataset_id: str,
            version_id: str) -> Optional[DatasetVersionRecord]:
        row = self._db.execute(
            "SELECT doc FROM dataset_versions WHERE tenant_id=? AND"
            " dataset_id=? AND version_id=?",
            (tenant_id, dataset_id, version_id)).fetchone()
        return DatasetVersionRecord(**json.loads(row[0])) if row else None

    def list(self, tenant_id: str,
             dataset_id: Optional[str] = None) -> list[DatasetVersionRecord]:
        q = "SELECT doc FROM dataset_versions WHERE tenant_id=?"
        args: list[Any] = [tenant_id]
        if dataset_id:
            q += " AND dataset_id=?"
            args.append(dataset_id)
        rows = [DatasetVersionRecord(**json.loads(doc))
                for (doc,) in self._db.execute(q, args)]
        return sorted(rows, key=lambda r: r.created_at, reverse=True)

    def verify(self, tenant_id: str, dataset_id: str, version_id: str,
               digest: str) -> dict[str, Any]:
        """Does an observed artifact digest match the registered version?"""
        rec = self.get(tenant_id, dataset_id, version_id)
        if rec is None:
            return {"status": "unknown_version"}
        if not rec.digest:
            return {"status": "no_pinned_digest"}
        return {"status": "match" if digest == rec.digest else "MISMATCH",
                "registered": rec.digest, "observed": digest}
