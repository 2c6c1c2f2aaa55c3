"""Versioned AI-system blueprints + drift incidents (governance tier).

Reference parity: src/agent_bom/api/{blueprint_store,drift_incident_store}.py
— a blueprint is the durable, queryable description of an APPROVED AI
system (agents, models, tools, datasets, identities, owners, guardrails).

Lifecycle: a blueprint owns immutable numbered versions; each version walks
``draft → pending → approved | rejected``.  Approval requires an
accountable approver DIFFERENT from the submitter (four-eyes); an approved
version is immutable — edits always open a fresh draft.  Version diffs use
the added/removed/persistent vocabulary per composition axis.

Drift: ``evaluate_drift`` compares the in-effect (approved) composition
against a scan report's observed agents/tools and opens incidents for
unexpected/missing entities; incidents live in a store with an
open → resolved lifecycle.
"""

from __future__ import annotations

import json
import sqlite3
import threading
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Optional
from uuid import uuid4

STATUS_DRAFT = "draft"
STATUS_PENDING = "pending"
STATUS_APPROVED = "approved"
STATUS_REJECTED = "rejected"

_AXES = ("agents", "models", "tools", "datasets", "identities", "owners",
         "guardrails")


def _now_iso() -> str:
    return datetime.now(timezone.utc).isoformat()


class BlueprintApprovalError(ValueError):
    pass


class SelfApprovalError(BlueprintApprovalError):
    """Four-eyes violation: the submitter may not approve their own version."""


@dataclass
class BlueprintComposition:
    agents: list[str] = field(default_factory=list)
    models: list[str] = field(default_factory=list)
    tools: list[str] = field(default_factory=list)
    datasets: list[str] = field(default_factory=list)
    identities: list[str] = field(default_factory=list)
    owners: list[str] = field(default_factory=list)
    guardrails: list[str] = field(default_factory=list)

    def to_dict(self) -> dict[str, list[str]]:
        return {axis: sorted(set(getattr(self, axis))) for axis in _AXES}

    @classmethod
    def from_dict(cls, data: dict) -> "BlueprintComposition":
        return cls(**{axis: [str(x) for x in data.get(axis, []) or []]
                      for axis in _AXES})


@dataclass
class BlueprintVersion:
    blueprint_id: str
    version: int
    composition: BlueprintComposition
    status: str = STATUS_DRAFT
    submitted_by: str = ""
    approved_by: str = ""
    note: str = ""
    created_at: str = field(default_factory=_now_iso)
    decided_at: str = ""

    def to_dict(self) -> dict[str, Any]:
        return {"blueprint_id": self.blueprint_id, "version": self.version,
                "composition": self.composition.to_dict(),
                "status": self.status, "submitted_by": self.submitted_by,
                "approved_by": self.approved_by, "note": self.note,
                "created_at": self.created_at, "decided_at": self.decided_at}


@dataclass
class Blueprint:
    name: str
    tenant_id: str = "default"
    blueprint_id: str = ""
    owner: str = ""
    seeded_from: str = ""
    current_approved_version: int = 0
    created_at: str = field(default_factory=_now_iso)

    def __post_init__(self) -> None:
        if not self.blueprint_id:
            self.blueprint_id = f"bp-{uuid4().hex[:12]}"

    def to_dict(self) -> dict[str, Any]:
        return {"blueprint_id": self.blueprint_id, "name": self.name,
                "tenant_id": self.tenant_id, "owner": self.owner,
                "seeded_from": self.seeded_from,
                "current_approved_version": self.current_approved_version,
                "created_at": self.created_at}


_SCHEMA = """
CREATE TABLE IF NOT EXISTS blueprints (
    blueprint_id TEXT PRIMARY KEY,
    tenant_id TEXT NOT NULL,
    doc TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS blueprint_versions (
    blueprint_id TEXT NOT NULL,
    version INTEGER NOT NULL,
    status TEXT NOT NULL,
    doc TEXT NOT NULL,
    PRIMARY KEY (blueprint_id, version)
);
CREATE TABLE IF NOT EXISTS drift_incidents (
    incident_id TEXT PRIMARY KEY,
    tenant_id TEXT NOT NULL,
    blueprint_id TEXT NOT NULL,
    status TEXT NOT NULL,
    doc TEXT NOT NULL
);
"""


class BlueprintStore:
    """SQLite store for blueprints, versions and drift incidents."""

    def __init__(self, path: str = ":memory:"):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._lock = threading.Lock()

    # ── blueprints + versions ─────────────────────────────────────────────

    def _save_bp(self, bp: Blueprint) -> None:
        self._db.execute(
            "INSERT OR REPLACE INTO blueprints (blueprint_id, tenant_id, doc)"
            " VALUES (?,?,?)",
            (bp.blueprint_id, bp.tenant_id, json.dumps(bp.to_dict())))

    def _save_ver(self, v: BlueprintVersion) -> None:
        self._db.execute(
            "INSERT OR REPLACE INTO blueprint_versions (blueprint_id, version,"
            " status, doc) VALUES (?,?,?,?)",
            (v.blueprint_id, v.version, v.status, json.dumps(v.to_dict())))

    def get(self, tenant_id: str, blueprint_id: str) -> Optional[Blueprint]:
        row = self._db.execute(
            "SELECT doc FROM blueprints WHERE blueprint_id=? AND tenant_id=?",
            (blueprint_id, tenant_id)).fetchone()
        return Blueprint(**json.loads(row[0])) if row else None

    def list(self, tenant_id: str) -> list[Blueprint]:
        return [Blueprint(**json.loads(doc)) for (doc,) in self._db.execute(
            "SELECT doc FROM blueprints WHERE tenant_id=? ORDER BY blueprint_id",
            (tenant_id,))]

    def get_version(self, blueprint_id: str,
                    version: int) -> Optional[BlueprintVersion]:
        row = self._db.execute(
            "SELECT doc FROM blueprint_versions WHERE blueprint_id=? AND version=?",
            (blueprint_id, version)).fetchone()
        if row is None:
            return None
        d = json.loads(row[0])
        d["composition"] = BlueprintComposition.from_dict(d["composition"])
        return BlueprintVersion(**d)

    def list_versions(self, blueprint_id: str) -> list[BlueprintVersion]:
        out = []
        for (doc,) in self._db.execute(
                "SELECT doc FROM blueprint_versions WHERE blueprint_id=?"
                " ORDER BY version", (blueprint_id,)):
            d = json.loads(doc)
            d["composition"] = BlueprintComposition.from_dict(d["composition"])
            out.append(BlueprintVersion(**d))
        return out

    def create(self, tenant_id: str, name: str, composition: BlueprintComposition,
               author: str, owner: str = "", seeded_from: str = "") -> Blueprint:
        """New blueprint with version 1 as a draft."""
        bp = Blueprint(name=name, tenant_id=tenant_id, owner=owner,
                       seeded_from=seeded_from)
        v1 = BlueprintVersion(blueprint_id=bp.blueprint_id, version=1,
                              composition=composition, submitted_by=author)
        with self._lock:
            self._save_bp(bp)
            self._save_ver(v1)
            self._db.commit()
        return bp

    def create_draft(self, tenant_id: str, blueprint_id: str,
                     composition: BlueprintComposition,
                     author: str) -> Optional[BlueprintVersion]:
        """Approved versions are immutable — edits open the next draft."""
        with self._lock:
            if self.get(tenant_id, blueprint_id) is None:
                return None
            versions = self.list_versions(blueprint_id)
            nxt = (versions[-1].version + 1) if versions else 1
            v = BlueprintVersion(blueprint_id=blueprint_id, version=nxt,
                                 composition=composition, submitted_by=author)
            self._save_ver(v)
            self._db.commit()
            return v

    def submit(self, blueprint_id: str, version: int) -> Optional[BlueprintVersion]:
        with self._lock:
            v = self.get_version(blueprint_id, version)
            if v is None or v.status != STATUS_DRAFT:
                return None
            v.status = STATUS_PENDING
            self._save_ver(v)
            self._db.commit()
            return v

    def approve(self, tenant_id: str, blueprint_id: str, version: int,
                approver: str, note: str = "") -> BlueprintVersion:
        """Approve a pending version.  Enforces an accountable approver and
        four-eyes (approver != submitter)."""
        if not approver.strip():
            raise BlueprintApprovalError(
                "an accountable approver is required to approve a version")
        with self._lock:
            v = self.get_version(blueprint_id, version)
            if v is None or v.status != STATUS_PENDING:
                raise BlueprintApprovalError(
                    f"version {version} is not pending approval")
            if approver == v.submitted_by:
                raise SelfApprovalError(
                    "submitter may not approve their own version (four-eyes)")
            v.status = STATUS_APPROVED
            v.approved_by = approver
            v.note = note
            v.decided_at = _now_iso()
            self._save_ver(v)
            bp = self.get(tenant_id, blueprint_id)
            if bp is not None:
                bp.current_approved_version = version
                self._save_bp(bp)
            self._db.commit()
            return v

    def reject(self, blueprint_id: str, version: int, approver: str,
               note: str = "") -> Optional[BlueprintVersion]:
        with self._lock:
            v = self.get_version(blueprint_id, version)
            if v is None or v.status != STATUS_PENDING:
                return None
            v.status = STATUS_REJECTED
            v.approved_by = approver
            v.note = note
            v.decided_at = _now_iso()
            self._save_ver(v)
            self._db.commit()
            return v

    # ── drift incidents ───────────────────────────────────────────────────

    def record_incident(self, incident: dict[str, Any]) -> dict[str, Any]:
        incident = dict(incident)
        incident.setdefault("incident_id", f"drift-{uuid4().hex[:12]}")
        incident.setdefault("status", "open")
        incident.setdefault("opened_at", _now_iso())
        with self._lock:
            self._db.execute(
                "INSERT OR REPLACE INTO drift_incidents (incident_id, tenant_id,"
                " blueprint_id, status, doc) VALUES (?,?,?,?,?)",
                (incident["incident_id"], incident.get("tenant_id", "default"),
                 incident.get("blueprint_id", ""), incident["status"],
                 json.dumps(incident)))
            self._db.commit()
        return incident

    def list_incidents(self, tenant_id: str,
                       status: Optional[str] = None) -> list[dict[str, Any]]:
        q = "SELECT doc FROM drift_incidents WHERE tenant_id=?"
        args: list[Any] = [tenant_id]
        if status:
            q += " AND status=?"
            args.append(status)
        return [json.loads(doc)
                for (doc,) in self._db.execute(q + " ORDER BY incident_id", args)]

    def resolve_incident(self, incident_id: str, actor: str,
                         note: str = "") -> Optional[dict[str, Any]]:
        with self._lock:
            row = self._db.execute(
                "SELECT doc FROM drift_incidents WHERE incident_id=?",
                (incident_id,)).fetchone()
            if row is None:
                return None
            doc = json.loads(row[0])
            doc["status"] = "resolved"
            doc["resolved_by"] = actor
            doc["resolved_at"] = _now_iso()
            doc["resolution_note"] = note
            self._db.execute(
                "UPDATE drift_incidents SET status=?, doc=? WHERE incident_id=?",
                ("resolved", json.dumps(doc), incident_id))
            self._db.commit()
            return doc


def diff_versions(store: BlueprintStore, blueprint_id: str,
                  from_version: int, to_version: int) -> Optional[dict[str, Any]]:
    """added / removed / persistent per composition axis + net change."""
    a = store.get_version(blueprint_id, from_version)
    b = store.get_version(blueprint_id, to_version)
    if a is None or b is None:
        return None
    axes: dict[str, dict[str, list[str]]] = {}
    added_n = removed_n = persistent_n = 0
    for axis in _AXES:
        prev = set(getattr(a.composition, axis))
        curr = set(getattr(b.composition, axis))
        added, removed = sorted(curr - prev), sorted(prev - curr)
        persistent = sorted(prev & curr)
        added_n += len(added)
        removed_n += len(removed)
        persistent_n += len(persistent)
        axes[axis] = {"added": added, "removed": removed,
                      "persistent": persistent}
    return {"blueprint_id": blueprint_id, "from_version": from_version,
            "to_version": to_version, "axes": axes, "added_count": added_n,
            "removed_count": removed_n, "persistent_count": persistent_n,
            "net_change": added_n - removed_n}


def evaluate_drift(blueprint: Blueprint, version: BlueprintVersion,
                   report) -> list[dict[str, Any]]:
    """Compare the approved composition with a scan report's observation.

    Incident kinds: ``unexpected_agent`` (observed but not governed),
    ``missing_agent`` (governed but absent from the scan), and
    ``unexpected_tool`` (a tool surfaced on a governed agent's servers that
    the blueprint does not list) — only when the blueprint constrains
    tools at all (an empty tools axis means unconstrained).
    """
    if version.status != STATUS_APPROVED:
        return []
    want = version.composition
    observed_agents = {a.name for a in report.agents}
    governed = set(want.agents)
    incidents = []

    def mk(kind: str, entity: str, detail: str) -> dict[str, Any]:
        return {"tenant_id": blueprint.tenant_id,
                "blueprint_id": blueprint.blueprint_id,
                "blueprint_version": version.version,
                "kind": kind, "entity": entity, "detail": detail}

    for name in sorted(observed_agents - governed):
        incidents.append(mk("unexpected_agent", name,
                            "agent observed in scan but not in the approved "
                            "blueprint composition"))
    for name in sorted(governed - observed_agents):
        incidents.append(mk("missing_agent", name,
                            "agent in the approved blueprint was not observed"))
    if want.tools:
        allowed = set(want.tools)
        for agent in report.agents:
            if agent.name not in governed:
                continue
            for srv in agent.mcp_servers:
                for tool in srv.tools:
                    if tool.name not in allowed:
                        incidents.append(mk(
                            "unexpected_tool", f"{agent.name}/{tool.name}",
                            f"tool {tool.name!r} on governed agent "
                            f"{agent.name!r} is not in the approved tools"))
    return incidents
