"""Webhook subscriptions + HMAC-signed delivery with bounded retries.

Reference parity: the api/ webhook store + delivery client (retries,
DLQ, idempotency).  Subscriptions filter by event kind; every delivery
carries an ``X-AgentBom-Signature`` (HMAC-SHA256 over the body with the
subscription secret) and an idempotency key.  Failed deliveries retry
with backoff up to ``max_attempts`` and then land in the dead-letter
queue — never silently dropped.

Transport is injectable (tests use a recorder); the default transport
uses urllib with a short timeout and treats any non-2xx as failure.
"""

from __future__ import annotations

import hashlib
import hmac
import json
import secrets
import threading
import time
from typing import Any, Callable, Optional

EVENT_KINDS = ("scan.completed", "scan.failed", "finding.new",
               "identity.issued", "shield.action")


def default_transport(url: str, body: bytes, headers: dict[str, str]) -> bool:
    import urllib.error
    import urllib.request

    req = urllib.request.Request(url, data=body, headers=headers, method="POST")
    try:
        with urllib.request.urlopen(req, timeout=5) as resp:  # noqa: S310
            return 200 <= resp.status < 300
    except (urllib.error.URLError, OSError, ValueError):
        return False


class WebhookRegistry:
    """In-memory subscriptions + synchronous best-effort delivery."""

    def __init__(self, transport: Optional[Callable[..., bool]] = None,
                 max_attempts: int = 3, backoff_s: float = 0.05):
        self._lock = threading.Lock()
        self._subs: dict[str, dict[str, Any]] = {}
        self.deliveries: list[dict[str, Any]] = []
        self.dead_letters: list[dict[str, Any]] = []
        self.transport = transport or default_transport
        self.max_attempts = max_attempts
        self.backoff_s = backoff_s

    # ── subscriptions ─────────────────────────────────────────────────────

    def subscribe(self, url: str, events: list[str],
                  secret: Optional[str] = None) -> dict[str, Any]:
        bad = [e for e in events if e not in EVENT_KINDS]
        if bad:
            raise ValueError(f"unknown event kinds {bad}; one of {EVENT_KINDS}")
        sub = {
            "webhook_id": f"wh-{secrets.token_hex(5)}",
            "url": url,
            "events": sorted(set(events)),
            "secret": secret or secrets.token_hex(16),
            "created_at": time.time(),
            "delivered": 0,
            "failed": 0,
        }
        with self._lock:
            self._subs[sub["webhook_id"]] = sub
        return {k: v for k, v in sub.items() if k != "secret"} | {
            "secret": sub["secret"]}  # secret shown once at creation

    def unsubscribe(self, webhook_id: str) -> bool:
        with self._lock:
            return self._subs.pop(webhook_id, None) is not None

    def list(self) -> list[dict[str, Any]]:
        with self._lock:
            return [{k: v for k, v in s.items() if k != "secret"}
                    for s in self._subs.values()]

    # ── delivery ──────────────────────────────────────────────────────────

    def emit(self, kind: str, payload: dict[str, Any]) -> int:
        """Deliver one event to every matching subscription; returns count."""
        if kind not in EVENT_KINDS:
            raise ValueError(f"unknown event kind {kind!r}")
        with self._lock:
            targets = [s for s in self._subs.values() if kind in s["events"]]
        event = {
            "kind": kind,
            "id": f"evt-{secrets.token_hex(6)}",
            "ts": time.time(),
            "payload": payload,
        }
        body = json.dumps(event, default=str).encode()
        delivered = 0
        for sub in targets:
            sig = hmac.new(sub["secret"].encode(), body,
                           hashlib.sha256).hexdigest()
            headers = {
                "Content-Type": "application/json",
                "X-AgentBom-Signature": f"sha256={sig}",
                "X-AgentBom-Event": kind,
                "X-AgentBom-Delivery": event["id"],  # idempotency key
            }
            ok = False
            attempts = 0
            for attempt in range(self.max_attempts):
                attempts = attempt + 1
                ok = bool(self.transport(sub["url"], body, headers))
                if ok:
                    break
                time.sleep(self.backoff_s * (2 ** attempt))
            record = {"webhook_id": sub["webhook_id"], "event": event["id"],
                      "kind": kind, "ok": ok, "attempts": attempts,
                      "ts": event["ts"]}
            with self._lock:
                self.deliveries.append(record)
                if ok:
                    sub["delivered"] += 1
                    delivered += 1
                else:
                    sub["failed"] += 1
                    self.dead_letters.append({**record, "body": body.decode()})
        return delivered
