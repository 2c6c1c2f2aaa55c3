"""Cost & behavior anomaly detection — explainable statistics, no ML.

Reference parity: src/agent_bom/api/anomaly.py — a runaway agent becomes
visible PROACTIVELY via a modified z-score (median + MAD) over per-agent
spend and per-session tool-call volume, plus an EWMA spike check over
hourly spend buckets.  Median/MAD (not mean/std) on purpose: a single
outlier inflates its own mean-based baseline enough to cap its plain
z-score below threshold in small samples — the robust form flags exactly
the runaway we care about.  Thresholds and math stay operator-readable.
"""

from __future__ import annotations

from datetime import datetime
from typing import Any, Optional

DEFAULT_Z_THRESHOLD = 3.5
_MIN_SAMPLES = 4
_MAD_SCALE = 0.6745  # Φ⁻¹(0.75): scales MAD to a std-dev equivalent
_EWMA_ALPHA = 0.3
_EWMA_SPIKE_FACTOR = 3.0


def _median(values: list[float]) -> float:
    s = sorted(values)
    n = len(s)
    mid = n // 2
    return s[mid] if n % 2 else (s[mid - 1] + s[mid]) / 2.0


def _baseline(values: list[float]) -> tuple[float, float]:
    med = _median(values)
    mad = _median([abs(v - med) for v in values])
    return med, mad


def modified_z(value: float, med: float, mad: float) -> float:
    if mad == 0.0:
        return 0.0 if value == med else float("inf")
    return _MAD_SCALE * (value - med) / mad


def _detect(values_by_key: dict[str, float], kind: str,
            z_threshold: float) -> list[dict[str, Any]]:
    if len(values_by_key) < _MIN_SAMPLES:
        return []  # too few peers for a meaningful baseline
    med, mad = _baseline(list(values_by_key.values()))
    out = []
    for key, value in sorted(values_by_key.items()):
        z = modified_z(value, med, mad)
        if z >= z_threshold:
            out.append({"kind": kind, "subject": key,
                        "value": round(value, 6),
                        "baseline_median": round(med, 6),
                        "baseline_mad": round(mad, 6),
                        "z": round(z, 2) if z != float("inf") else "inf",
                        "threshold": z_threshold})
    out.sort(key=lambda r: (-(float("inf") if r["z"] == "inf" else r["z"]),
                            r["subject"]))
    return out


def detect_cost_anomalies(spend_by_agent: dict[str, float],
                          z_threshold: float = DEFAULT_Z_THRESHOLD
                          ) -> list[dict[str, Any]]:
    """Agents whose spend is a robust outlier vs their peers."""
    return _detect(spend_by_agent, "cost", z_threshold)


def detect_behavior_anomalies(calls_by_session: dict[str, int],
                              z_threshold: float = DEFAULT_Z_THRESHOLD
                              ) -> list[dict[str, Any]]:
    """Sessions whose tool-call volume is a robust outlier."""
    return _detect({k: float(v) for k, v in calls_by_session.items()},
                   "behavior", z_threshold)


def _hour_bucket(observed_at: str) -> Optional[str]:
    try:
        ts = datetime.fromisoformat(str(observed_at).replace("Z", "+00:00"))
    except (TypeError, ValueError):
        return None
    return ts.strftime("%Y-%m-%dT%H")


def detect_temporal_cost_anomalies(records: list[Any],
                                   spike_factor: float = _EWMA_SPIKE_FACTOR
                                   ) -> list[dict[str, Any]]:
    """Hourly spend buckets whose value spikes past the EWMA of the
    PRECEDING buckets (the spike never inflates its own baseline)."""
    buckets: dict[str, float] = {}
    for rec in records:
        at = getattr(rec, "observed_at", None) or (
            rec.get("observed_at") if isinstance(rec, dict) else None)
        cost = getattr(rec, "cost_usd", None) or (
            rec.get("cost_usd", 0.0) if isinstance(rec, dict) else 0.0)
        b = _hour_bucket(at)
        if b is not None:
            buckets[b] = buckets.get(b, 0.0) + float(cost)
    ordered = sorted(buckets.items())
    if len(ordered) < _MIN_SAMPLES:
        return []
    out = []
    ewma: Optional[float] = None
    for bucket, value in ordered:
        if ewma is not None and ewma > 0 and value >= spike_factor * ewma:
            out.append({"kind": "temporal_cost", "bucket": bucket,
                        "value": round(value, 6),
                        "ewma_baseline": round(ewma, 6),
                        "factor": round(value / ewma, 2),
                        "threshold_factor": spike_factor})
        ewma = value if ewma is None else \
            _EWMA_ALPHA * value + (1 - _EWMA_ALPHA) * ewma
    return out
