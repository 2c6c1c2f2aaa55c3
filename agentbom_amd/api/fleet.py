"""Fleet reconciliation, scan scheduling, and backpressure.

Reference: src/agent_bom/api/scheduler.py + fleet/sync_client.py (scan
schedules, fleet heartbeat sync), backpressure.py (adaptive concurrency ->
HTTP 429 + Retry-After), api/audit_log.py + audit_integrity.py
(hash-chained audit ingest + verify).
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Optional

STALE_AFTER_S = 300.0


@dataclass
class FleetMember:
    member_id: str
    hostname: str
    version: str = ""
    last_heartbeat: float = 0.0
    agents: int = 0
    servers: int = 0
    findings: int = 0
    labels: dict[str, str] = field(default_factory=dict)

    @property
    def status(self) -> str:
        return "healthy" if time.time() - self.last_heartbeat < STALE_AFTER_S else "stale"

    def to_dict(self) -> dict[str, Any]:
        return {
            "member_id": self.member_id, "hostname": self.hostname,
            "version": self.version, "last_heartbeat": self.last_heartbeat,
            "status": self.status, "agents": self.agents, "servers": self.servers,
            "findings": self.findings, "labels": self.labels,
        }


class FleetRegistry:
    """Heartbeat-driven fleet membership with reconciliation summaries."""

    def __init__(self) -> None:
        self._members: dict[str, FleetMember] = {}
        self._lock = threading.Lock()
        self.observations = 0

    def heartbeat(self, payload: dict[str, Any]) -> FleetMember:
        member_id = str(payload.get("member_id") or payload.get("hostname") or "unknown")
        with self._lock:
            m = self._members.get(member_id) or FleetMember(
                member_id=member_id, hostname=str(payload.get("hostname", member_id))
            )
            m.version = str(payload.get("version", m.version))
            m.last_heartbeat = time.time()
            m.agents = int(payload.get("agents", m.agents))
            m.servers = int(payload.get("servers", m.servers))
            m.findings = int(payload.get("findings", m.findings))
            if isinstance(payload.get("labels"), dict):
                m.labels.update(payload["labels"])
            self._members[member_id] = m
            self.observations += 1
            return m

    def reconcile(self) -> dict[str, Any]:
        with self._lock:
            members = list(self._members.values())
        healthy = [m for m in members if m.status == "healthy"]
        return {
            "members": len(members),
            "healthy": len(healthy),
            "stale": len(members) - len(healthy),
            "total_agents": sum(m.agents for m in healthy),
            "total_servers": sum(m.servers for m in healthy),
            "total_findings": sum(m.findings for m in healthy),
            "observations": self.observations,
        }

    def list_members(self) -> list[dict[str, Any]]:
        with self._lock:
            return [m.to_dict() for m in sorted(self._members.values(),
                                                key=lambda m: m.member_id)]


@dataclass
class ScanSchedule:
    schedule_id: str
    interval_s: float
    demo: bool = True
    enabled: bool = True
    last_run: Optional[float] = None
    next_run: Optional[float] = None
    runs: int = 0

    def to_dict(self) -> dict[str, Any]:
        return {
            "schedule_id": self.schedule_id, "interval_s": self.interval_s,
            "demo": self.demo, "enabled": self.enabled, "last_run": self.last_run,
            "next_run": self.next_run, "runs": self.runs,
        }


class ScanScheduler:
    """Interval scheduler driving the scan pipeline on a daemon thread."""

    def __init__(self, run_scan: Callable[[dict], Any], tick_s: float = 1.0):
        self.run_scan = run_scan
        self.tick_s = tick_s
        self.schedules: dict[str, ScanSchedule] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def add(self, schedule_id: str, interval_s: float, demo: bool = True) -> ScanSchedule:
        sched = ScanSchedule(schedule_id=schedule_id, interval_s=max(interval_s, 1.0),
                             demo=demo, next_run=time.time() + interval_s)
        with self._lock:
            self.schedules[schedule_id] = sched
        return sched

    def remove(self, schedule_id: str) -> bool:
        with self._lock:
            return self.schedules.pop(schedule_id, None) is not None

    def tick(self, now: Optional[float] = None) -> int:
        """Run due schedules once; returns number fired (unit-testable)."""
        now = now or time.time()
        fired = 0
        with self._lock:
            due = [s for s in self.schedules.values()
                   if s.enabled and s.next_run is not None and s.next_run <= now]
        for sched in due:
            try:
                self.run_scan({"demo": sched.demo, "schedule_id": sched.schedule_id})
            finally:
                sched.last_run = now
                sched.next_run = now + sched.interval_s
                sched.runs += 1
                fired += 1
        return fired

    def start(self) -> None:
        if self._thread is not None:
            return

        def loop() -> None:
            while not self._stop.wait(self.tick_s):
                self.tick()

        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()


class BackpressureController:
    """Process-local adaptive concurrency gate (reference backpressure.py:35).

    Admission above the limit is rejected with a retry-after hint; the limit
    adapts down on rejections and recovers slowly on success.
    """

    def __init__(self, max_concurrent: int = 8, min_limit: int = 2):
        self.hard_max = max_concurrent
        self.limit = max_concurrent
        self.min_limit = min_limit
        self.active = 0
        self.rejections = 0
        self._lock = threading.Lock()

    def try_acquire(self) -> tuple[bool, float]:
        """(admitted, retry_after_s)."""
        with self._lock:
            if self.active >= self.limit:
                self.rejections += 1
                self.limit = max(self.min_limit, int(self.limit * 0.9))
                return False, max(0.5, self.active * 0.25)
            self.active += 1
            return True, 0.0

    def release(self) -> None:
        with self._lock:
            self.active = max(0, self.active - 1)
            if self.limit < self.hard_max:
                self.limit = min(self.hard_max, self.limit + 1)
