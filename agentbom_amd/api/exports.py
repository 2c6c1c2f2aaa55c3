"""Scheduled report exports: destinations + schedules + delivery.

Reference parity: src/agent_bom/api/{export_destination_store,
export_schedule_store,export_scheduler}.py — operators register WHERE
reports go (webhook URL or file path, with a format) and WHEN (interval
schedules); the exporter renders the latest report in the destination's
format and delivers it, with per-delivery outcomes recorded (webhook
failures land in the registry's dead-letter queue pattern).

Formats: json, sarif, csv, markdown, cyclonedx, spdx, ocsf, prometheus.
Delivery transports: ``https://...`` → POST (offline-guarded,
transport-injected in tests); anything else → file write (path
validated to stay under the destination directory root).
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass
from pathlib import Path
from typing import Any, Callable, Optional
from uuid import uuid4

_RENDERERS: dict[str, Callable[[Any], str]] = {}


def _renderer(fmt: str):
    """Lazy renderer lookup (import-on-use keeps API startup light)."""
    import json as _json

    def render_json(report):
        from agentbom_amd.output.json_fmt import to_json

        return _json.dumps(to_json(report), default=str)

    def render_sarif(report):
        from agentbom_amd.output.sarif import to_sarif

        return _json.dumps(to_sarif(report), default=str)

    def render_cdx(report):
        from agentbom_amd.output.cyclonedx_fmt import to_cyclonedx

        return _json.dumps(to_cyclonedx(report), default=str)

    def render_spdx(report):
        from agentbom_amd.output.spdx_fmt import to_spdx

        return _json.dumps(to_spdx(report), default=str)

    def render_ocsf(report):
        from agentbom_amd.output.ocsf import to_ocsf_events

        return _json.dumps(to_ocsf_events(report), default=str)

    def render_csv(report):
        from agentbom_amd.output.misc_fmt import to_csv

        return to_csv(report)

    def render_md(report):
        from agentbom_amd.output.misc_fmt import to_markdown

        return to_markdown(report)

    def render_prom(report):
        from agentbom_amd.output.misc_fmt import to_prometheus

        return to_prometheus(report)

    table = {"json": render_json, "sarif": render_sarif,
             "cyclonedx": render_cdx, "spdx": render_spdx,
             "ocsf": render_ocsf, "csv": render_csv,
             "markdown": render_md, "prometheus": render_prom}
    fn = table.get(fmt)
    if fn is None:
        raise ValueError(f"unknown export format {fmt!r}; "
                         f"one of {sorted(table)}")
    return fn


EXPORT_FORMATS = ("json", "sarif", "cyclonedx", "spdx", "ocsf", "csv",
                  "markdown", "prometheus")


@dataclass
class ExportDestination:
    target: str              # https URL -> POST; else file path under root
    format: str = "json"
    destination_id: str = ""
    tenant_id: str = "default"
    name: str = ""

    def __post_init__(self) -> None:
        if not self.destination_id:
            self.destination_id = f"dest-{uuid4().hex[:10]}"
        if self.format not in EXPORT_FORMATS:
            raise ValueError(f"unknown export format {self.format!r}")

    def to_dict(self) -> dict[str, Any]:
        return {"destination_id": self.destination_id, "target": self.target,
                "format": self.format, "tenant_id": self.tenant_id,
                "name": self.name}


@dataclass
class ExportSchedule:
    destination_id: str
    interval_s: float
    schedule_id: str = ""
    next_run: float = 0.0
    runs: int = 0
    last_status: str = ""

    def __post_init__(self) -> None:
        if not self.schedule_id:
            self.schedule_id = f"expsched-{uuid4().hex[:10]}"
        self.interval_s = max(self.interval_s, 1.0)
        if not self.next_run:
            self.next_run = time.time() + self.interval_s

    def to_dict(self) -> dict[str, Any]:
        return {"schedule_id": self.schedule_id,
                "destination_id": self.destination_id,
                "interval_s": self.interval_s, "runs": self.runs,
                "last_status": self.last_status}


class ExportManager:
    """Destinations + schedules + render/deliver, file-root confined."""

    def __init__(self, get_report: Callable[[], Any],
                 file_root: Optional[str] = None,
                 http_post: Optional[Callable[[str, str], bool]] = None):
        self.get_report = get_report
        self.file_root = Path(file_root) if file_root else None
        self.http_post = http_post or self._default_post
        self.destinations: dict[str, ExportDestination] = {}
        self.schedules: dict[str, ExportSchedule] = {}
        self.deliveries: list[dict[str, Any]] = []
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    @staticmethod
    def _default_post(url: str, body: str) -> bool:
        from agentbom_amd.utils.http_client import (
            check_offline,
            create_client,
            request_with_retry,
        )

        check_offline(url)
        resp = request_with_retry(
            create_client(timeout=30.0), "POST", url, content=body,
            headers={"Content-Type": "application/json"})
        return resp is not None and 200 <= resp.status_code < 300

    # ── registry ──────────────────────────────────────────────────────────

    def add_destination(self, dest: ExportDestination) -> ExportDestination:
        with self._lock:
            self.destinations[dest.destination_id] = dest
        return dest

    def add_schedule(self, destination_id: str,
                     interval_s: float) -> Optional[ExportSchedule]:
        if destination_id not in self.destinations:
            return None
        sched = ExportSchedule(destination_id=destination_id,
                               interval_s=interval_s)
        with self._lock:
            self.schedules[sched.schedule_id] = sched
        return sched

    def remove_schedule(self, schedule_id: str) -> bool:
        with self._lock:
            return self.schedules.pop(schedule_id, None) is not None

    # ── delivery ──────────────────────────────────────────────────────────

    def run_export(self, destination_id: str) -> dict[str, Any]:
        dest = self.destinations.get(destination_id)
        if dest is None:
            return {"ok": False, "error": "unknown destination"}
        outcome: dict[str, Any] = {"destination_id": destination_id,
                                   "format": dest.format, "at": time.time()}
        try:
            body = _renderer(dest.format)(self.get_report())
            if dest.target.startswith(("http://", "https://")):
                ok = self.http_post(dest.target, body)
                outcome |= {"ok": bool(ok), "transport": "http",
                            "bytes": len(body)}
            else:
                root = self.file_root or Path(".")
                path = (root / dest.target).resolve()
                if not str(path).startswith(str(root.resolve())):
                    raise ValueError("export path escapes the file root")
                path.parent.mkdir(parents=True, exist_ok=True)
                path.write_text(body)
                outcome |= {"ok": True, "transport": "file",
                            "path": str(path), "bytes": len(body)}
        except Exception as exc:  # noqa: BLE001 — delivery boundary
            outcome |= {"ok": False, "error": f"{type(exc).__name__}: {exc}"}
        with self._lock:
            self.deliveries.append(outcome)
            del self.deliveries[:-500]
        return outcome

    # ── scheduler loop ────────────────────────────────────────────────────

    def tick(self, now: Optional[float] = None) -> int:
        """Run every due schedule once; returns the number fired."""
        now = now or time.time()
        fired = 0
        for sched in list(self.schedules.values()):
            if now < sched.next_run:
                continue
            out = self.run_export(sched.destination_id)
            sched.runs += 1
            sched.last_status = "ok" if out.get("ok") else "failed"
            sched.next_run = now + sched.interval_s
            fired += 1
        return fired

    def start(self, tick_s: float = 1.0) -> None:
        if self._thread is not None:
            return

        def loop() -> None:
            while not self._stop.wait(tick_s):
                self.tick()

        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
