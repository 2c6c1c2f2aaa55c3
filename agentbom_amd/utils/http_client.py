"""HTTP client: retries + jittered backoff, per-host circuit breaker,
offline-mode guard, URL sanitizing.

Reference: src/agent_bom/http_client.py (create_client :209,
registry_breaker_tripped :57, set_offline/check_offline :111,122) with the
reference's config defaults (HTTP_MAX_RETRIES=3, backoff 1s..30s,
breaker threshold 3x 429 per host per scan).
"""

from __future__ import annotations

import random
import re
import threading
import time
from typing import Any, Callable, Optional

import httpx

from agentbom_amd.utils import config as cfg

_offline = threading.Event()
if cfg.OFFLINE:
    _offline.set()


class OfflineError(RuntimeError):
    """Raised when a network call is attempted in offline mode."""


def set_offline(value: bool = True) -> None:
    if value:
        _offline.set()
    else:
        _offline.clear()


def is_offline() -> bool:
    return _offline.is_set()


def check_offline(url: str) -> None:
    if _offline.is_set():
        raise OfflineError(f"offline mode: refusing network call to {sanitize_url(url)}")


_CRED_IN_URL = re.compile(r"//[^/@]+@")
_TOKEN_PARAM = re.compile(r"([?&](?:token|key|api_key|apikey|secret|signature)=)[^&]+",
                          re.IGNORECASE)


def sanitize_url(url: Optional[str]) -> Optional[str]:
    """Strip embedded credentials and token query params for logs/output."""
    if not url:
        return url
    url = _CRED_IN_URL.sub("//***@", url)
    return _TOKEN_PARAM.sub(r"\1***", url)


class HostBreaker:
    """Per-host 429 circuit breaker, reset per scan.

    After N rate-limit responses from one host, further calls to that host
    short-circuit to the cached/bundled fallback for the rest of the run.
    """

    def __init__(self, threshold: Optional[int] = None):
        self.threshold = threshold or cfg.HTTP_RATE_LIMIT_BREAKER_THRESHOLD
        self._hits: dict[str, int] = {}
        self._lock = threading.Lock()

    def record_rate_limit(self, host: str) -> None:
        with self._lock:
            self._hits[host] = self._hits.get(host, 0) + 1

    def tripped(self, host: str) -> bool:
        with self._lock:
            return self._hits.get(host, 0) >= self.threshold

    def reset(self) -> None:
        with self._lock:
            self._hits.clear()


_breaker = HostBreaker()


def registry_breaker_tripped(host: str) -> bool:
    return _breaker.tripped(host)


def reset_rate_limit_breaker() -> None:
    _breaker.reset()


def create_client(timeout: Optional[float] = None, **kw) -> httpx.Client:
    return httpx.Client(timeout=timeout or cfg.HTTP_DEFAULT_TIMEOUT,
                        follow_redirects=True, **kw)


def request_with_retry(
    client: httpx.Client,
    method: str,
    url: str,
    max_retries: Optional[int] = None,
    sleep: Callable[[float], None] = time.sleep,
    **kw: Any,
) -> Optional[httpx.Response]:
    """Retry with jittered exponential backoff; honors offline mode and the
    per-host breaker.  Returns None after the retry budget is exhausted or
    when the breaker is open (callers fall back to cache/bundles)."""
    check_offline(url)
    host = httpx.URL(url).host or ""
    if _breaker.tripped(host):
        return None
    retries = cfg.HTTP_MAX_RETRIES if max_retries is None else max_retries
    backoff = cfg.HTTP_INITIAL_BACKOFF
    last: Optional[httpx.Response] = None
    for attempt in range(retries + 1):
        try:
            resp = client.request(method, url, **kw)
        except (httpx.TransportError, httpx.TimeoutException):
            resp = None
        if resp is not None:
            last = resp
            if resp.status_code == 429:
                _breaker.record_rate_limit(host)
                if _breaker.tripped(host):
                    return None
            elif resp.status_code < 500:
                return resp
        if attempt < retries:
            retry_after = None
            if resp is not None:
                try:
                    retry_after = float(resp.headers.get("Retry-After", ""))
                except (TypeError, ValueError):
                    retry_after = None
            delay = retry_after if retry_after is not None else backoff * (0.5 + random.random())
            sleep(min(delay, cfg.HTTP_MAX_BACKOFF))
            backoff = min(backoff * 2, cfg.HTTP_MAX_BACKOFF)
    return last
