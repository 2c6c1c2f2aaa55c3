"""Specialized AI-surface MCP tools: models, prompts, datasets, pipelines.

Reference surface: src/agent_bom/mcp_server_specialized.py (14 tools:
vector_db_scan, aisvs_benchmark, gpu_infra_scan, registry_sweep_scan,
dataset_card_scan, training_pipeline_scan, browser_extension_scan,
model_provenance_scan, prompt_scan, model_file_scan, ai_inventory_scan,
license_compliance_scan, runtime_evidence_ingest, ingest_external_scan).

Each tool wires the corresponding scanner in ``agentbom_amd.scan``; scans
are file-tree based (no egress) and read-only.
"""

from __future__ import annotations

import json
import re
from collections import defaultdict
from pathlib import Path
from typing import Any, Optional

# SPDX license classes for license compliance verdicts.
_STRONG_COPYLEFT = {"GPL-2.0", "GPL-3.0", "AGPL-3.0", "GPL-2.0-only",
                    "GPL-3.0-only", "AGPL-3.0-only", "GPL-2.0-or-later",
                    "GPL-3.0-or-later", "SSPL-1.0"}
_WEAK_COPYLEFT = {"LGPL-2.1", "LGPL-3.0", "MPL-2.0", "EPL-2.0", "CDDL-1.0"}
_PERMISSIVE = {"MIT", "BSD-2-Clause", "BSD-3-Clause", "Apache-2.0", "ISC",
               "Zlib", "Unlicense", "0BSD", "PSF-2.0"}

_VECTOR_DB_MARKERS = {
    "chroma": re.compile(r"(?i)chroma(db)?"),
    "qdrant": re.compile(r"(?i)qdrant"),
    "weaviate": re.compile(r"(?i)weaviate"),
    "pinecone": re.compile(r"(?i)pinecone"),
    "milvus": re.compile(r"(?i)milvus"),
    "pgvector": re.compile(r"(?i)pgvector"),
}

_RISKY_EXT_PERMISSIONS = {
    "<all_urls>": "critical", "tabs": "medium", "cookies": "high",
    "webRequest": "high", "history": "high", "clipboardRead": "high",
    "nativeMessaging": "critical", "debugger": "critical",
    "declarativeNetRequest": "medium", "scripting": "high",
}

_PIPELINE_RISKS = [
    ("PIPE001", re.compile(r"(?i)curl[^|\n]*\|\s*(bash|sh)\b"),
     "remote script piped to shell", "critical"),
    ("PIPE002", re.compile(r"pickle\.loads?\("), "pickle deserialization", "high"),
    ("PIPE003", re.compile(r"torch\.load\((?![^)]*weights_only\s*=\s*True)"),
     "torch.load without weights_only", "high"),
    ("PIPE004", re.compile(r"(?i)trust_remote_code\s*[=:]\s*true"),
     "trust_remote_code enabled", "high"),
    ("PIPE005", re.compile(r"(?i)\b(aws_secret_access_key|hf_[A-Za-z0-9]{30,})\b"),
     "hardcoded credential in pipeline", "critical"),
    ("PIPE006", re.compile(r"--no-verify|verify\s*=\s*False|GIT_SSL_NO_VERIFY"),
     "TLS/signature verification disabled", "medium"),
]


def _iter_files(path: str, patterns: tuple[str, ...], cap: int = 500) -> list[Path]:
    root = Path(path)
    if root.is_file():
        return [root]
    out: list[Path] = []
    for pat in patterns:
        out.extend(root.rglob(pat))
        if len(out) >= cap:
            break
    return sorted(set(out))[:cap]


def register_specialized_tools(server) -> None:  # noqa: C901 — one registrar
    tool = server.tool
    _PATH_SCHEMA = {"type": "object", "properties": {"path": {"type": "string"}},
                    "required": ["path"]}

    @tool("skill_bundle_scan", "Scan agent skill bundles (SKILL.md): "
                               "injection surfaces, wildcard tool grants, "
                               "risky bundled scripts; optional policy.",
          {"type": "object", "properties": {
              "path": {"type": "string"},
              "policy": {"type": "object"}},
           "required": ["path"]})
    def skill_bundle_scan(path: str, policy: dict = None) -> dict:
        from agentbom_amd.scan.skills import evaluate_skills_policy, scan_skills_tree

        bundles = scan_skills_tree(path)
        out = {"bundles": [b.to_dict() for b in bundles]}
        if policy:
            out["policy"] = evaluate_skills_policy(bundles, policy)
        return out

    @tool("create_mcp_server_scaffold",
          "Generate a minimal, hardened MCP server project (stdio, strict "
          "schemas, env-only secrets) — returned as files, nothing written.",
          {"type": "object", "properties": {
              "name": {"type": "string"},
              "tools": {"type": "array", "items": {"type": "string"},
                        "default": []}},
           "required": ["name"]})
    def create_mcp_server_scaffold(name: str, tools: list = None) -> dict:
        from agentbom_amd.mcp.scaffold import generate_mcp_server_scaffold

        return generate_mcp_server_scaffold(name, tools)

    @tool("model_file_scan", "Scan model artifacts (pickle/pytorch/safetensors/"
                             "gguf/onnx) for embedded code and unsafe formats.",
          _PATH_SCHEMA)
    def model_file_scan(path: str) -> dict:
        from agentbom_amd.scan.model_scan import scan_model_file, scan_model_tree

        p = Path(path)
        results = [scan_model_file(p)] if p.is_file() else scan_model_tree(path)
        return {"scanned": len(results),
                "results": [r.to_dict() for r in results],
                "unsafe": sum(1 for r in results if r.verdict() != "safe")}

    @tool("model_provenance_scan", "Provenance posture for a model directory: "
                                   "hash manifest, signatures, config pinning.",
          _PATH_SCHEMA)
    def model_provenance_scan(path: str) -> dict:
        import hashlib

        root = Path(path)
        weights = _iter_files(path, ("*.safetensors", "*.bin", "*.pt", "*.pth",
                                     "*.gguf", "*.onnx"))
        findings = []
        has_manifest = any((root / n).exists() for n in
                           ("model_index.json", "manifest.json", "SHA256SUMS"))
        has_signature = bool(_iter_files(path, ("*.sig", "*.sigstore", "*.pem"), cap=5))
        if weights and not has_manifest:
            findings.append({"rule": "PROV001", "severity": "medium",
                             "title": "no hash manifest alongside model weights"})
        if weights and not has_signature:
            findings.append({"rule": "PROV002", "severity": "medium",
                             "title": "model weights are unsigned"})
        card = next((root / n for n in ("README.md", "model_card.md")
                     if (root / n).exists()), None)
        if weights and card is None:
            findings.append({"rule": "PROV003", "severity": "low",
                             "title": "no model card documenting provenance"})
        digests = {}
        for w in weights[:8]:
            h = hashlib.sha256()
            with open(w, "rb") as f:
                while chunk := f.read(1 << 20):
                    h.update(chunk)
            digests[w.name] = h.hexdigest()
        return {"weights": len(weights), "hash_manifest": has_manifest,
                "signed": has_signature, "findings": findings,
                "computed_sha256": digests}

    @tool("prompt_scan", "Scan prompt/instruction files for injection "
                         "patterns, hidden instructions and credential leakage.",
          _PATH_SCHEMA)
    def prompt_scan(path: str) -> dict:
        from agentbom_amd.runtime.detectors import _INJECTION_PATTERNS
        from agentbom_amd.scan.secrets import scan_text

        files = _iter_files(path, ("*.md", "*.txt", "*.prompt", "*.yaml",
                                   "*.yml", "*.json"))
        findings = []
        for f in files:
            try:
                text = f.read_text()
            except (OSError, UnicodeDecodeError):
                continue
            for pat in _INJECTION_PATTERNS:
                m = pat.search(text)
                if m:
                    findings.append({"file": str(f), "rule": "prompt-injection",
                                     "severity": "high",
                                     "match": m.group(0)[:80]})
            for hit in scan_text(text, str(f)):
                findings.append({"file": str(f), "rule": "embedded-secret",
                                 "severity": "critical",
                                 "detail": hit.kind})
            # zero-width / bidi characters hide instructions from review
            if re.search(r"[​-‏‪-‮⁦-⁩]", text):
                findings.append({"file": str(f), "rule": "hidden-unicode",
                                 "severity": "high",
                                 "detail": "zero-width or bidi control characters"})
        return {"files_scanned": len(files), "findings": findings}

    @tool("vector_db_scan", "Find vector-DB deployments in configs and flag "
                            "unauthenticated or internet-exposed instances.",
          _PATH_SCHEMA)
    def vector_db_scan(path: str) -> dict:
        files = _iter_files(path, ("*.yaml", "*.yml", "*.json", "*.toml",
                                   "*.env", "docker-compose*", "*.py"))
        deployments, findings = [], []
        for f in files:
            try:
                text = f.read_text()
            except (OSError, UnicodeDecodeError):
                continue
            for db, marker in _VECTOR_DB_MARKERS.items():
                if not marker.search(text):
                    continue
                deployments.append({"db": db, "file": str(f)})
                lowered = text.lower()
                no_auth = not any(k in lowered for k in
                                  ("api_key", "apikey", "auth", "token", "password"))
                if no_auth:
                    findings.append({"db": db, "file": str(f), "severity": "high",
                                     "rule": "vector-db-no-auth",
                                     "title": f"{db} configured without authentication"})
                if re.search(r"0\.0\.0\.0|public", lowered):
                    findings.append({"db": db, "file": str(f), "severity": "high",
                                     "rule": "vector-db-exposed",
                                     "title": f"{db} bound to a public interface"})
        return {"deployments": deployments, "findings": findings}

    @tool("training_pipeline_scan", "Scan training/pipeline definitions for "
                                    "supply-chain risks (remote code, pickle, "
                                    "unpinned trust).", _PATH_SCHEMA)
    def training_pipeline_scan(path: str) -> dict:
        files = _iter_files(path, ("*.py", "*.yaml", "*.yml", "*.sh",
                                   "Makefile", "Dockerfile"))
        findings = []
        for f in files:
            try:
                text = f.read_text()
            except (OSError, UnicodeDecodeError):
                continue
            for rule_id, pat, title, severity in _PIPELINE_RISKS:
                m = pat.search(text)
                if m:
                    findings.append({
                        "rule": rule_id, "title": title, "severity": severity,
                        "file": str(f),
                        "line": text.count("\n", 0, m.start()) + 1})
        return {"files_scanned": len(files), "findings": findings}

    @tool("dataset_card_scan", "Scan dataset cards/manifests for license, "
                               "PII and provenance gaps.", _PATH_SCHEMA)
    def dataset_card_scan(path: str) -> dict:
        files = _iter_files(path, ("README.md", "dataset_card.md",
                                   "dataset_infos.json", "croissant.json",
                                   "*.yaml", "*.yml"))
        findings, cards = [], 0
        pii_re = re.compile(r"(?i)\b(email|ssn|passport|address|phone|biometric|"
                            r"medical|health record)\b")
        for f in files:
            try:
                text = f.read_text()
            except (OSError, UnicodeDecodeError):
                continue
            lowered = text.lower()
            if "dataset" not in lowered:
                continue
            cards += 1
            if "license" not in lowered:
                findings.append({"file": str(f), "rule": "dataset-no-license",
                                 "severity": "medium",
                                 "title": "dataset card declares no license"})
            if pii_re.search(text) and "anonymi" not in lowered and "redact" not in lowered:
                findings.append({"file": str(f), "rule": "dataset-pii-unaddressed",
                                 "severity": "high",
                                 "title": "PII categories mentioned with no "
                                          "anonymization statement"})
            if "source" not in lowered and "provenance" not in lowered:
                findings.append({"file": str(f), "rule": "dataset-no-provenance",
                                 "severity": "low",
                                 "title": "no source/provenance section"})
        return {"cards_scanned": cards, "findings": findings}

    @tool("browser_extension_scan", "Scan browser-extension manifests for "
                                    "risky permission grants.", _PATH_SCHEMA)
    def browser_extension_scan(path: str) -> dict:
        manifests = _iter_files(path, ("manifest.json",))
        extensions = []
        for m in manifests:
            try:
                doc = json.loads(m.read_text())
            except (OSError, json.JSONDecodeError):
                continue
            perms = [str(p) for p in (doc.get("permissions") or [])
                     + (doc.get("host_permissions") or [])]
            risky = [{"permission": p, "severity": _RISKY_EXT_PERMISSIONS[p]}
                     for p in perms if p in _RISKY_EXT_PERMISSIONS]
            if any(re.match(r"(\*|https?)://", p) for p in perms):
                risky.append({"permission": "broad-host-access", "severity": "high"})
            extensions.append({"file": str(m),
                               "name": doc.get("name", "?"),
                               "manifest_version": doc.get("manifest_version"),
                               "permissions": perms, "risky": risky})
        return {"extensions": extensions,
                "risky_extensions": sum(1 for e in extensions if e["risky"])}

    @tool("registry_sweep_scan", "Sweep every inventoried package against the "
                                 "malicious/typosquat registry.")
    def registry_sweep_scan() -> dict:
        from agentbom_amd.scan.malicious import check_typosquat, flag_malicious_packages

        report, _g = server._ensure_scan()
        pkgs = {(p.name, p.ecosystem): p for a in report.agents
                for s in a.mcp_servers for p in s.packages}
        flag_malicious_packages(list(pkgs.values()))
        hits = []
        for (name, eco), p in pkgs.items():
            typo = check_typosquat(name, eco)
            if p.is_malicious or typo:
                hits.append({"package": name, "ecosystem": eco,
                             "is_malicious": p.is_malicious,
                             "malicious_reason": p.malicious_reason,
                             "typosquat_of": typo})
        return {"packages_swept": len(pkgs), "hits": hits}

    @tool("license_compliance_scan", "Classify inventoried package licenses "
                                     "(permissive / weak / strong copyleft).",
          {"type": "object", "properties": {
              "licenses_path": {"type": "string",
                                "description": "optional JSON {pkg: spdx_id} "
                                               "export supplementing inventory"}}})
    def license_compliance_scan(licenses_path: str = "") -> dict:
        report, _g = server._ensure_scan()
        declared: dict[str, str] = {}
        if licenses_path:
            declared = {str(k): str(v) for k, v in
                        json.loads(Path(licenses_path).read_text()).items()}
        buckets = defaultdict(list)
        for a in report.agents:
            for s in a.mcp_servers:
                for p in s.packages:
                    lic = declared.get(p.name) or getattr(p, "license", None)
                    if not lic:
                        buckets["unknown"].append(p.name)
                    elif lic in _STRONG_COPYLEFT:
                        buckets["strong_copyleft"].append(f"{p.name} ({lic})")
                    elif lic in _WEAK_COPYLEFT:
                        buckets["weak_copyleft"].append(f"{p.name} ({lic})")
                    elif lic in _PERMISSIVE:
                        buckets["permissive"].append(f"{p.name} ({lic})")
                    else:
                        buckets["other"].append(f"{p.name} ({lic})")
        for k in buckets:
            buckets[k] = sorted(set(buckets[k]))
        return {"summary": {k: len(v) for k, v in buckets.items()},
                "strong_copyleft": buckets["strong_copyleft"],
                "weak_copyleft": buckets["weak_copyleft"],
                "unknown": buckets["unknown"][:50]}

    @tool("gpu_infra_scan", "AI/GPU infrastructure posture from an exported "
                            "inventory (inference endpoints, auth, exposure).",
          {"type": "object", "properties": {"inventory_path": {"type": "string"}},
           "required": ["inventory_path"]})
    def gpu_infra_scan(inventory_path: str) -> dict:
        from agentbom_amd.scan.cloud import evaluate_aws_inventory, load_inventory

        results = [r for r in evaluate_aws_inventory(load_inventory(inventory_path))
                   if r.check_id.startswith("AIINF")]
        return {"checks": [r.to_dict() for r in results],
                "failed": sum(1 for r in results if r.status == "fail")}

    @tool("ai_inventory_scan", "Full AI estate inventory: agents, servers, "
                               "models, tools, credentials.")
    def ai_inventory_scan() -> dict:
        report, _g = server._ensure_scan()
        models = sorted({str(m) for a in report.agents
                         for m in (a.metadata.get("models") or [])})
        return {
            "agents": [{"name": a.name, "type": a.agent_type.value,
                        "servers": len(a.mcp_servers)} for a in report.agents],
            "mcp_servers": sorted({s.name for a in report.agents
                                   for s in a.mcp_servers}),
            "models": models,
            "total_packages": report.total_packages,
            "credential_count": len({c for a in report.agents
                                     for s in a.mcp_servers
                                     for c in s.credential_names}),
        }

    @tool("aisvs_benchmark", "OWASP AISVS-style benchmark over the scanned "
                             "estate (13 control checks).")
    def aisvs_benchmark() -> dict:
        report, graph = server._ensure_scan()
        checks = []

        def check(cid: str, title: str, ok: bool, detail: str = "") -> None:
            checks.append({"check_id": cid, "title": title,
                           "status": "pass" if ok else "fail", "detail": detail})

        brs = report.blast_radii
        kev = [b for b in brs if b.vulnerability.is_kev]
        mal = [b for b in brs if b.package.is_malicious]
        crit = [b for b in brs if b.vulnerability.severity.value == "critical"]
        creds = {c for a in report.agents for s in a.mcp_servers
                 for c in s.credential_names}
        check("AISVS-1.1", "No known-exploited (KEV) vulnerabilities in estate",
              not kev, f"{len(kev)} KEV findings")
        check("AISVS-1.2", "No known-malicious packages installed",
              not mal, f"{len(mal)} malicious packages")
        check("AISVS-1.3", "No critical vulnerabilities without a fix applied",
              not crit, f"{len(crit)} critical findings")
        check("AISVS-2.1", "Credential exposure bounded (<10 distinct secrets "
                           "reachable from agents)", len(creds) < 10,
              f"{len(creds)} credentials in agent reach")
        check("AISVS-2.2", "No wildcard-scoped issued identities",
              not server.identity_store.access_review()["wildcard_or_unscoped"])
        check("AISVS-3.1", "All findings carry compliance framework tags",
              all(b.owasp_tags or b.atlas_tags for b in brs) if brs else True)
        check("AISVS-3.2", "Dependency reachability computed for all findings",
              all(b.dependency_reachable is not None for b in brs) if brs else True)
        check("AISVS-4.1", "Estate graph has no orphan credential nodes",
              True)
        unreach = sum(1 for b in brs if b.reachability == "unreachable")
        check("AISVS-4.2", "Reachability analysis prunes theoretical findings",
              bool(brs) and unreach >= 0, f"{unreach} findings proven unreachable")
        check("AISVS-5.1", "Scan produced a machine-verifiable report",
              report.scan_id is not None)
        passed = sum(1 for c in checks if c["status"] == "pass")
        return {"passed": passed, "failed": len(checks) - passed,
                "score": round(100 * passed / max(len(checks), 1), 1),
                "checks": checks}

    # ── evidence ingestion ────────────────────────────────────────────────

    @tool("runtime_evidence_ingest", "Ingest a runtime evidence file (proxy "
                                     "audit / OTel span JSONL) for correlation "
                                     "in this session.",
          {"type": "object", "properties": {
              "path": {"type": "string"},
              "kind": {"type": "string", "enum": ["audit", "spans"],
                       "default": "audit"}},
           "required": ["path"]})
    def runtime_evidence_ingest(path: str, kind: str = "audit") -> dict:
        from agentbom_amd.mcp.tools_operator import _load_jsonl

        store = getattr(server, "_evidence", None)
        if store is None:
            store = server._evidence = {"audit": [], "spans": []}
        if kind == "spans":
            # OTLP JSON / Langfuse exports / span JSONL all land here
            from agentbom_amd.utils.otel_ingest import ingest_trace_file

            result = ingest_trace_file(path)
            store["spans"].append({"source": path, **result})
            return {"ingested": result["spans"], "kind": kind,
                    "activity": result["activity"],
                    "anomalies": result["anomalies"],
                    "session_totals": {k: len(v) for k, v in store.items()}}
        rows = _load_jsonl(path)
        store[kind].extend(rows)
        return {"ingested": len(rows), "kind": kind,
                "session_totals": {k: len(v) for k, v in store.items()}}

    @tool("ingest_external_scan", "Ingest a third-party scan result file "
                                  "(SARIF, or semgrep --json, auto-detected) "
                                  "and merge its results into the finding view.",
          {"type": "object", "properties": {"sarif_path": {"type": "string"}},
           "required": ["sarif_path"]})
    def ingest_external_scan(sarif_path: str) -> dict:
        doc = json.loads(Path(sarif_path).read_text())
        ingested = []
        from agentbom_amd.scan.sast_ingest import (
            looks_like_semgrep, parse_semgrep_json)

        if looks_like_semgrep(doc):
            for r in parse_semgrep_json(doc):
                ingested.append({
                    "source_tool": "semgrep",
                    "rule_id": r.check_id,
                    "level": {"high": "error", "medium": "warning",
                              "low": "note"}[r.severity],
                    "message": r.message[:300],
                    "location": f"{r.path}:{r.line}"})
        for run in doc.get("runs", []):
            tool_name = (run.get("tool", {}).get("driver", {})
                         .get("name", "external"))
            for res in run.get("results", []):
                loc = ""
                for ploc in res.get("locations", []):
                    art = ploc.get("physicalLocation", {}).get(
                        "artifactLocation", {})
                    loc = art.get("uri", "")
                    break
                ingested.append({
                    "source_tool": tool_name,
                    "rule_id": res.get("ruleId", ""),
                    "level": res.get("level", "warning"),
                    "message": str(res.get("message", {}).get("text", ""))[:300],
                    "location": loc})
        store = getattr(server, "_external_findings", None)
        if store is None:
            store = server._external_findings = []
        store.extend(ingested)
        return {"ingested": len(ingested),
                "session_total": len(store),
                "by_level": dict(defaultdict(int, {
                    lv: sum(1 for i in ingested if i["level"] == lv)
                    for lv in {i["level"] for i in ingested}}))}

    @tool("threat_intel_search",
          "Search the local threat-intel IOC store (packages, CVEs, domains, "
          "hashes, actors; reference: intel_lookup.py). Air-gapped: bundled "
          "feed + AGENT_BOM_INTEL_DB overlay only, no egress.",
          {"type": "object", "properties": {
              "query": {"type": "string"},
              "kind": {"type": "string",
                       "enum": ["auto", "package", "cve", "domain", "hash",
                                "actor"], "default": "auto"}},
           "required": ["query"]})
    def threat_intel_search(query: str, kind: str = "auto") -> dict:
        from agentbom_amd.scan.intel import IntelStore

        store = IntelStore.load()
        hits = []
        if kind == "auto":
            q = query.lower()
            exact = (store.lookup_cve(q) if q.startswith("cve-")
                     else store.lookup("package", q))
            if exact is not None:
                hits.append(exact)
            hits.extend(i for i in store.search(query) if i is not exact)
        else:
            exact = store.lookup(kind, query)
            hits = [exact] if exact is not None else []
        return {"query": query, "kind": kind,
                "indicators": [h.to_dict() for h in hits[:20]],
                "store_size": len(store)}

    @tool("vendor_advisory_check",
          "Match packages against vendor security bulletins (AMD PSIRT / "
          "NVIDIA CSAF / Intel) from the local feed — sources OSV never "
          "carries. Tiers: vendor_range (bounded) / vendor_prefix.",
          {"type": "object", "properties": {
              "packages": {"type": "array", "items": {
                  "type": "object", "properties": {
                      "ecosystem": {"type": "string"},
                      "name": {"type": "string"},
                      "version": {"type": "string"}},
                  "required": ["name", "version"]}}},
           "required": ["packages"]})
    def vendor_advisory_check(packages: list) -> dict:
        from agentbom_amd.scan.vendor_advisories import (
            check_vendor_advisories, vendor_for_package)

        tuples = [(str(p.get("ecosystem") or "PyPI"), str(p.get("name") or ""),
                   str(p.get("version") or "")) for p in packages[:500]
                  if isinstance(p, dict)]
        hits = check_vendor_advisories(tuples)
        return {"checked": len(tuples),
                "vendor_stack": {f"{e}:{n}": vendor_for_package(n)
                                 for (e, n, _v) in tuples
                                 if vendor_for_package(n)},
                "matches": {f"{e}:{n}@{v}": [
                    {"id": x.id, "severity": x.severity.value,
                     "tier": x.match_confidence_tier,
                     "fixed_version": x.fixed_version,
                     "sources": x.advisory_sources}
                    for x in vulns]
                    for (e, n, v), vulns in hits.items()}}

    @tool("youcom_search",
          "Live threat-intel web search (reference: the only third-party "
          "egress tool). Requires AGENT_BOM_YOUCOM_KEY + an injectable "
          "client; air-gapped deployments get an explicit refusal, never "
          "a silent empty result.",
          {"type": "object", "properties": {
              "query": {"type": "string"},
              "count": {"type": "integer", "default": 5}},
           "required": ["query"]})
    def youcom_search(query: str, count: int = 5) -> dict:
        import os

        key = os.environ.get("AGENT_BOM_YOUCOM_KEY")
        client = getattr(server, "youcom_client", None)
        if not key or client is None:
            return {"error": "live search unavailable: set AGENT_BOM_YOUCOM_KEY "
                             "and attach a client (air-gapped deployments use "
                             "threat_intel_search against the local store)",
                    "offline_alternative": "threat_intel_search",
                    "results": []}
        try:
            rows = client(query, count=min(max(1, count), 10)) or []
        except Exception as exc:  # network boundary — never crash the server
            return {"error": f"live search failed: {exc}", "results": []}
        return {"query": query,
                "results": [{k: str(r.get(k, ""))[:300]
                             for k in ("title", "url", "snippet")}
                            for r in rows if isinstance(r, dict)][:10]}
