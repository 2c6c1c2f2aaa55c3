#!/usr/bin/env python3
"""Control-plane load test (reference parity: deploy/loadtest/k6-*.js).

k6 is not available in the air-gapped image, so this is a self-contained
thread-pool driver with the same scenario shapes:

- ``read-heavy``  — findings/graph/rollup reads (the k6 smoke profile)
- ``scan-burst``  — concurrent demo-scan submissions exercising the
  backpressure gate (expects some 429 + Retry-After)
- ``mixed``       — 80/20 read/write

Run against a live server::

    python deploy/loadtest/loadtest.py --base-url http://127.0.0.1:8000 \
        --scenario read-heavy --vus 8 --duration 10

or in-process (no server, CI-friendly)::

    python deploy/loadtest/loadtest.py --in-process --vus 4 --duration 3

Prints one JSON summary line: requests, errors, 429s, p50/p95/p99 ms.
"""

from __future__ import annotations

import argparse
import json
import random
import threading
import time
from typing import Callable


def _percentile(xs: list[float], q: float) -> float:
    if not xs:
        return 0.0
    xs = sorted(xs)
    i = min(len(xs) - 1, int(q * (len(xs) - 1)))
    return xs[i]


READ_PATHS = [
    ("GET", "/v1/findings?limit=50"),
    ("GET", "/v1/graph?limit=100"),
    ("GET", "/v1/graph/rollup"),
    ("GET", "/v1/graph/paths?limit=10"),
    ("GET", "/v1/posture"),
    ("GET", "/healthz"),
]


def run_load(request: Callable[[str, str], tuple[int, float]],
             scenario: str, vus: int, duration_s: float,
             seed: int = 7) -> dict:
    """Drive ``request(method, path) -> (status, ms)`` from ``vus`` threads."""
    rng = random.Random(seed)
    lat: list[float] = []
    codes: dict[int, int] = {}
    lock = threading.Lock()
    stop = time.monotonic() + duration_s

    def pick() -> tuple[str, str]:
        if scenario == "scan-burst":
            return ("POST", "/v1/scan")
        if scenario == "mixed" and rng.random() < 0.2:
            return ("POST", "/v1/scan")
        return rng.choice(READ_PATHS)

    def worker() -> None:
        while time.monotonic() < stop:
            method, path = pick()
            try:
                status, ms = request(method, path)
            except Exception:
                status, ms = -1, 0.0
            with lock:
                codes[status] = codes.get(status, 0) + 1
                if status == 200:
                    lat.append(ms)

    threads = [threading.Thread(target=worker, daemon=True) for _ in range(vus)]
    t0 = time.monotonic()
    for t in threads:
        t.start()
    for t in threads:
        t.join(duration_s + 30)
    wall = time.monotonic() - t0
    total = sum(codes.values())
    return {
        "scenario": scenario, "vus": vus, "duration_s": round(wall, 2),
        "requests": total,
        "rps": round(total / wall, 1) if wall else 0.0,
        "status_codes": {str(k): v for k, v in sorted(codes.items())},
        "throttled_429": codes.get(429, 0),
        "errors": sum(v for k, v in codes.items() if k < 0 or k >= 500),
        "latency_ms": {
            "p50": round(_percentile(lat, 0.50), 2),
            "p95": round(_percentile(lat, 0.95), 2),
            "p99": round(_percentile(lat, 0.99), 2),
        },
    }


def make_http_requester(base_url: str, api_key: str = ""):
    import httpx

    client = httpx.Client(base_url=base_url, timeout=30,
                          headers={"X-API-Key": api_key} if api_key else {})

    def request(method: str, path: str) -> tuple[int, float]:
        t0 = time.perf_counter()
        r = client.request(method, path,
                           json={"demo": True} if method == "POST" else None)
        return r.status_code, (time.perf_counter() - t0) * 1000

    return request


def make_inprocess_requester():
    from starlette.testclient import TestClient

    from agentbom_amd.api.server import create_app

    client = TestClient(create_app())
    # seed one demo scan so the read paths have a report to serve
    client.post("/v1/scan", json={"demo": True})

    def request(method: str, path: str) -> tuple[int, float]:
        t0 = time.perf_counter()
        r = client.request(method, path,
                           json={"demo": True} if method == "POST" else None)
        return r.status_code, (time.perf_counter() - t0) * 1000

    return request


def main() -> int:
    ap = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    ap.add_argument("--base-url", default="http://127.0.0.1:8000")
    ap.add_argument("--api-key", default="")
    ap.add_argument("--in-process", action="store_true",
                    help="Spin the app inside this process (no server needed).")
    ap.add_argument("--scenario", default="read-heavy",
                    choices=["read-heavy", "scan-burst", "mixed"])
    ap.add_argument("--vus", type=int, default=8)
    ap.add_argument("--duration", type=float, default=10.0)
    args = ap.parse_args()

    req = (make_inprocess_requester() if args.in_process
           else make_http_requester(args.base_url, args.api_key))
    summary = run_load(req, args.scenario, args.vus, args.duration)
    print(json.dumps(summary))
    return 1 if summary["errors"] else 0


if __name__ == "__main__":
    raise SystemExit(main())
