"""OTLP/JSON trace ingestion + Langfuse bridge + span anomaly analysis.

Reference parity: src/agent_bom/otel_ingest.py (OTLP ingest of external
agent traces) and langfuse_otel.py (Langfuse bridge) — agent runtimes
export their LLM/tool spans, agent-bom ingests them offline and mines the
attribute payloads for security signal: injection text reaching prompts,
credential material leaking into span attributes, error storms, and a
per-service activity summary (which tools were actually called) that
feeds runtime-vs-blueprint drift checks.

Both formats normalize to one span shape::

    {"trace_id", "span_id", "parent_span_id", "name", "service",
     "start_ns", "end_ns", "duration_ms", "status", "attributes": {...}}
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Any, Optional

# ── OTLP AnyValue decoding ──────────────────────────────────────────────────


def _any_value(v: Any) -> Any:
    if not isinstance(v, dict):
        return v
    for key in ("stringValue", "boolValue", "doubleValue"):
        if key in v:
            return v[key]
    if "intValue" in v:
        try:
            return int(v["intValue"])
        except (TypeError, ValueError):
            return v["intValue"]
    if "arrayValue" in v:
        vals = v["arrayValue"].get("values") if isinstance(v["arrayValue"], dict) else None
        return [_any_value(x) for x in (vals or [])]
    if "kvlistValue" in v:
        kvs = v["kvlistValue"].get("values") if isinstance(v["kvlistValue"], dict) else None
        return _attrs(kvs or [])
    return v


def _attrs(pairs: Any) -> dict[str, Any]:
    out: dict[str, Any] = {}
    if not isinstance(pairs, list):
        return out
    for p in pairs:
        if isinstance(p, dict) and isinstance(p.get("key"), str):
            out[p["key"]] = _any_value(p.get("value"))
    return out


def parse_otlp_json(doc: Any) -> list[dict[str, Any]]:
    """OTLP/JSON trace export → normalized span rows.  Fail-soft."""
    if isinstance(doc, (str, bytes)):
        try:
            doc = json.loads(doc)
        except (ValueError, TypeError):
            return []
    if not isinstance(doc, dict):
        return []
    spans: list[dict[str, Any]] = []
    for rs in doc.get("resourceSpans") or []:
        if not isinstance(rs, dict):
            continue
        res_attrs = _attrs((rs.get("resource") or {}).get("attributes"))
        service = str(res_attrs.get("service.name") or "unknown")
        for ss in rs.get("scopeSpans") or rs.get("instrumentationLibrarySpans") or []:
            if not isinstance(ss, dict):
                continue
            for sp in ss.get("spans") or []:
                if not isinstance(sp, dict) or not isinstance(sp.get("name"), str):
                    continue
                try:
                    start = int(sp.get("startTimeUnixNano") or 0)
                    end = int(sp.get("endTimeUnixNano") or 0)
                except (TypeError, ValueError):
                    start = end = 0
                status = sp.get("status") or {}
                code = status.get("code") if isinstance(status, dict) else None
                spans.append({
                    "trace_id": str(sp.get("traceId") or ""),
                    "span_id": str(sp.get("spanId") or ""),
                    "parent_span_id": str(sp.get("parentSpanId") or ""),
                    "name": sp["name"],
                    "service": service,
                    "start_ns": start, "end_ns": end,
                    "duration_ms": max(0.0, (end - start) / 1e6),
                    "status": ("error" if code in (2, "STATUS_CODE_ERROR")
                               else "ok"),
                    "attributes": _attrs(sp.get("attributes")),
                })
    return spans


def looks_like_otlp(doc: Any) -> bool:
    if isinstance(doc, (str, bytes)):
        try:
            doc = json.loads(doc)
        except (ValueError, TypeError):
            return False
    return isinstance(doc, dict) and "resourceSpans" in doc


def parse_langfuse_export(doc: Any) -> list[dict[str, Any]]:
    """Langfuse trace/observation export → the same normalized span shape.

    Accepts {"observations": [...]} or a bare list of observation objects
    (type GENERATION/SPAN/EVENT with traceId/id/name/input/output)."""
    if isinstance(doc, (str, bytes)):
        try:
            doc = json.loads(doc)
        except (ValueError, TypeError):
            return []
    rows = doc.get("observations") if isinstance(doc, dict) else doc
    if not isinstance(rows, list):
        return []
    spans: list[dict[str, Any]] = []
    for ob in rows:
        if not isinstance(ob, dict) or not isinstance(ob.get("name"), str):
            continue
        attrs: dict[str, Any] = {
            "langfuse.type": str(ob.get("type") or "SPAN"),
        }
        if ob.get("model"):
            attrs["gen_ai.request.model"] = str(ob["model"])
        if ob.get("input") is not None:
            attrs["gen_ai.prompt"] = json.dumps(ob["input"], default=str)[:4000]
        if ob.get("output") is not None:
            attrs["gen_ai.completion"] = json.dumps(ob["output"], default=str)[:4000]
        spans.append({
            "trace_id": str(ob.get("traceId") or ""),
            "span_id": str(ob.get("id") or ""),
            "parent_span_id": str(ob.get("parentObservationId") or ""),
            "name": ob["name"],
            "service": str(ob.get("projectId") or "langfuse"),
            "start_ns": 0, "end_ns": 0, "duration_ms": 0.0,
            "status": "error" if ob.get("level") == "ERROR" else "ok",
            "attributes": attrs,
        })
    return spans


# ── analysis ────────────────────────────────────────────────────────────────

_TOOL_SPAN_HINTS = ("tool", "mcp", "function_call")


def summarize_agent_activity(spans: list[dict[str, Any]]) -> dict[str, Any]:
    """Per-service activity rollup: tools called, models used, error rates —
    the runtime truth that drift checks compare against the blueprint."""
    services: dict[str, dict[str, Any]] = {}
    for sp in spans:
        svc = services.setdefault(sp["service"], {
            "spans": 0, "errors": 0, "tools_called": set(), "models": set()})
        svc["spans"] += 1
        if sp["status"] == "error":
            svc["errors"] += 1
        attrs = sp.get("attributes") or {}
        tool = attrs.get("mcp.tool.name") or attrs.get("tool.name")
        if tool:
            svc["tools_called"].add(str(tool))
        elif any(h in sp["name"].lower() for h in _TOOL_SPAN_HINTS):
            svc["tools_called"].add(sp["name"])
        model = attrs.get("gen_ai.request.model") or attrs.get("llm.model")
        if model:
            svc["models"].add(str(model))
    return {svc: {"spans": d["spans"], "errors": d["errors"],
                  "error_rate": round(d["errors"] / d["spans"], 3) if d["spans"] else 0.0,
                  "tools_called": sorted(d["tools_called"]),
                  "models": sorted(d["models"])}
            for svc, d in sorted(services.items())}


def detect_span_anomalies(spans: list[dict[str, Any]],
                          error_burst_threshold: int = 5) -> list[dict[str, Any]]:
    """Security signal mined from span payloads:

    - prompt-injection text in gen_ai.* attributes (runtime detectors' rules)
    - credential material in ANY attribute value (secret scanner)
    - per-service error bursts (>= threshold error spans)
    """
    from agentbom_amd.runtime.detectors import _INJECTION_PATTERNS
    from agentbom_amd.scan.secrets import scan_text

    findings: list[dict[str, Any]] = []
    errors_by_service: dict[str, int] = {}
    for sp in spans:
        attrs = sp.get("attributes") or {}
        if sp["status"] == "error":
            errors_by_service[sp["service"]] = errors_by_service.get(sp["service"], 0) + 1
        for key, val in attrs.items():
            if not isinstance(val, str) or not val:
                continue
            if key.startswith(("gen_ai.", "llm.")) or "prompt" in key:
                for pat in _INJECTION_PATTERNS:
                    m = pat.search(val)
                    if m:
                        findings.append({
                            "rule": "otel-prompt-injection", "severity": "high",
                            "service": sp["service"], "span": sp["name"],
                            "attribute": key, "matched": m.group(0)[:80]})
                        break
            for hit in scan_text(val, path=f"span:{sp['span_id']}")[:2]:
                findings.append({
                    "rule": "otel-credential-in-span", "severity": "critical",
                    "service": sp["service"], "span": sp["name"],
                    "attribute": key, "secret_kind": hit.kind})
    for svc, n in sorted(errors_by_service.items()):
        if n >= error_burst_threshold:
            findings.append({"rule": "otel-error-burst", "severity": "medium",
                             "service": svc, "span": "*",
                             "attribute": "status", "count": n})
    return findings


def ingest_trace_file(path: str | Path) -> dict[str, Any]:
    """One-call ingest: OTLP JSON, Langfuse export, or span JSONL →
    {spans, activity, anomalies}."""
    try:
        text = Path(path).read_text()
    except OSError as exc:
        return {"error": str(exc), "spans": 0, "activity": {}, "anomalies": []}
    spans: list[dict[str, Any]] = []
    try:
        doc = json.loads(text)
    except ValueError:
        doc = None
    if doc is not None:
        if looks_like_otlp(doc):
            spans = parse_otlp_json(doc)
        else:
            spans = parse_langfuse_export(doc)
    if not spans:  # JSONL of already-normalized spans (proxy audit style)
        for line in text.splitlines():
            line = line.strip()
            if not line:
                continue
            try:
                row = json.loads(line)
            except ValueError:
                continue
            if isinstance(row, dict) and row.get("name"):
                row.setdefault("service", "unknown")
                row.setdefault("status", "ok")
                row.setdefault("span_id", "")
                row.setdefault("attributes", {})
                spans.append(row)
    return {"spans": len(spans),
            "activity": summarize_agent_activity(spans),
            "anomalies": detect_span_anomalies(spans)}
