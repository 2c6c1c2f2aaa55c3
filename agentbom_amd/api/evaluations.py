"""Tenant-scoped evaluation-run registry (model/agent eval evidence).

Reference parity: src/agent_bom/api/evaluation_store.py — headless
clients record LLM/agent evaluation runs (scores, per-case results,
the dataset VERSION they ran against, the model and prompt hash) so
governance can answer "what evidence backs this deployment?":

- runs link to the digest-pinned dataset registry
  (api/dataset_versions.py) when a dataset_version_id is supplied —
  an unknown version is refused (evidence must reference real lineage);
- score regressions between consecutive runs of the same evaluation
  are surfaced by ``compare``.
"""

from __future__ import annotations

import json
import sqlite3
import threading
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Optional
from uuid import uuid4


@dataclass
class EvaluationRun:
    tenant_id: str
    name: str
    evaluation_id: str = ""
    status: str = "completed"
    dataset_id: Optional[str] = None
    dataset_version_id: Optional[str] = None
    model: Optional[str] = None
    prompt_hash: Optional[str] = None
    source: str = "api"
    scores: dict[str, float] = field(default_factory=dict)
    cases: list[dict[str, Any]] = field(default_factory=list)
    metadata: dict[str, Any] = field(default_factory=dict)
    created_at: str = ""

    def __post_init__(self) -> None:
        if not self.evaluation_id:
            self.evaluation_id = f"eval-{uuid4().hex[:10]}"
        if not self.created_at:
            self.created_at = datetime.now(timezone.utc).isoformat()

    def to_dict(self) -> dict[str, Any]:
        return {k: getattr(self, k) for k in (
            "tenant_id", "name", "evaluation_id", "status", "dataset_id",
            "dataset_version_id", "model", "prompt_hash", "source",
            "scores", "cases", "metadata", "created_at")}


_SCHEMA = """
CREATE TABLE IF NOT EXISTS evaluation_runs (
    evaluation_id TEXT PRIMARY KEY,
    tenant_id TEXT NOT NULL,
    name TEXT NOT NULL,
    created_at TEXT NOT NULL,
    doc TEXT NOT NULL
);
"""


class EvaluationStore:
    def __init__(self, path: str = ":memory:", dataset_store=None):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._lock = threading.Lock()
        self.dataset_store = dataset_store

    def put(self, run: EvaluationRun) -> EvaluationRun:
        if run.dataset_id and run.dataset_version_id \
                and self.dataset_store is not None:
            if self.dataset_store.get(run.tenant_id, run.dataset_id,
                                      run.dataset_version_id) is None:
                raise ValueError(
                    f"dataset {run.dataset_id}@{run.dataset_version_id} is "
                    "not registered — evaluation evidence must reference "
                    "real dataset lineage")
        with self._lock:
            self._db.execute(
                "INSERT OR REPLACE INTO evaluation_runs (evaluation_id,"
                " tenant_id, name, created_at, doc) VALUES (?,?,?,?,?)",
                (run.evaluation_id, run.tenant_id, run.name, run.created_at,
                 json.dumps(run.to_dict())))
            self._db.commit()
        return run

    def get(self, tenant_id: str,
            evaluation_id: str) -> Optional[EvaluationRun]:
        row = self._db.execute(
            "SELECT doc FROM evaluation_runs WHERE evaluation_id=? AND"
            " tenant_id=?", (evaluation_id, tenant_id)).fetchone()
        return EvaluationRun(**json.loads(row[0])) if row else None

    def list(self, tenant_id: str, name: Optional[str] = None,
             limit: int = 100) -> list[EvaluationRun]:
        q = "SELECT doc FROM evaluation_runs WHERE tenant_id=?"
        args: list[Any] = [tenant_id]
        if name:
            q += " AND name=?"
            args.append(name)
        q += " ORDER BY created_at DESC LIMIT ?"
        args.append(limit)
        return [EvaluationRun(**json.loads(doc))
                for (doc,) in self._db.execute(q, args)]

    def compare(self, tenant_id: str, name: str) -> dict[str, Any]:
        """Latest vs previous run of one evaluation: per-metric deltas,
        regressions flagged (metric dropped)."""
        runs = self.list(tenant_id, name=name, limit=2)
        if len(runs) < 2:
            return {"status": "insufficient_history", "runs": len(runs)}
        latest, prev = runs[0], runs[1]
        deltas = {}
        regressions = []
        for metric in sorted(set(latest.scores) | set(prev.scores)):
            a, b = prev.scores.get(metric), latest.scores.get(metric)
            if a is None or b is None:
                deltas[metric] = {"previous": a, "latest": b,
                                  "delta": None}
                continue
            deltas[metric] = {"previous": a, "latest": b,
                              "delta": round(b - a, 6)}
            if b < a:
                regressions.append(metric)
        return {"status": "ok", "latest": latest.evaluation_id,
                "previous": prev.evaluation_id, "deltas": deltas,
                "regressions": sorted(regressions)}
