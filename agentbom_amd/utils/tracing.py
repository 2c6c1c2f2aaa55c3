"""W3C traceparent propagation + in-process span recording.

Reference: src/agent_bom/api/tracing.py:41-221 — traceparent parsing/
generation and spans around scan/graph phases; the OTLP exporter plugs in
where ``Tracer.export`` reads finished spans.  On the MI355X build the
GPU-side profiling story is rocprofv3 (profiles/); this module carries
the request-level trace contract.
"""

from __future__ import annotations

import contextlib
import random
import re
import time
from dataclasses import dataclass, field
from typing import Any, Iterator, Optional

_TRACEPARENT = re.compile(r"^([0-9a-f]{2})-([0-9a-f]{32})-([0-9a-f]{16})-([0-9a-f]{2})$")


def parse_traceparent(header: Optional[str]) -> Optional[tuple[str, str]]:
    """Returns (trace_id, parent_span_id) or None on malformed input."""
    if not header:
        return None
    m = _TRACEPARENT.match(header.strip())
    if not m or m.group(2) == "0" * 32 or m.group(3) == "0" * 16:
        return None
    return m.group(2), m.group(3)


def make_traceparent(trace_id: str, span_id: str, sampled: bool = True) -> str:
    return f"00-{trace_id}-{span_id}-{'01' if sampled else '00'}"


def _rand_hex(n: int) -> str:
    return "".join(random.choices("0123456789abcdef", k=n))


@dataclass
class Span:
    name: str
    trace_id: str
    span_id: str
    parent_span_id: Optional[str]
    start: float
    end: Optional[float] = None
    attributes: dict[str, Any] = field(default_factory=dict)
    status: str = "ok"

    @property
    def duration_ms(self) -> Optional[float]:
        return (self.end - self.start) * 1000 if self.end else None

    def to_dict(self) -> dict[str, Any]:
        return {
            "name": self.name, "trace_id": self.trace_id, "span_id": self.span_id,
            "parent_span_id": self.parent_span_id, "start": self.start,
            "duration_ms": self.duration_ms, "attributes": self.attributes,
            "status": self.status,
        }


class Tracer:
    """Minimal tracer: span stack + finished-span buffer (OTLP-shaped)."""

    def __init__(self, max_spans: int = 10_000):
        self.finished: list[Span] = []
        self.max_spans = max_spans
        self._stack: list[Span] = []
        self.trace_id: Optional[str] = None

    def start_trace(self, traceparent: Optional[str] = None) -> str:
        parsed = parse_traceparent(traceparent)
        self.trace_id = parsed[0] if parsed else _rand_hex(32)
        self._parent = parsed[1] if parsed else None
        return self.trace_id

    @contextlib.contextmanager
    def span(self, name: str, **attributes: Any) -> Iterator[Span]:
        if self.trace_id is None:
            self.start_trace()
        parent = self._stack[-1].span_id if self._stack else getattr(self, "_parent", None)
        s = Span(name=name, trace_id=self.trace_id, span_id=_rand_hex(16),
                 parent_span_id=parent, start=time.time(), attributes=dict(attributes))
        self._stack.append(s)
        try:
            yield s
        except Exception:
            s.status = "error"
            raise
        finally:
            s.end = time.time()
            self._stack.pop()
            if len(self.finished) < self.max_spans:
                self.finished.append(s)

    def current_traceparent(self) -> Optional[str]:
        if self.trace_id is None:
            return None
        span_id = self._stack[-1].span_id if self._stack else _rand_hex(16)
        return make_traceparent(self.trace_id, span_id)

    def export(self) -> list[dict[str, Any]]:
        out = [s.to_dict() for s in self.finished]
        self.finished.clear()
        return out
