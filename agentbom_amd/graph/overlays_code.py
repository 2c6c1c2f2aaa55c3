"""Code/CI/posture graph overlays: repo structure, code graph, CI graph,
ASPM application correlation, CNAPP exposure classification.

Reference parity: src/agent_bom/graph/{repo_structure_overlay,
code_graph_overlay,ci_graph_overlay,aspm_overlay,cnapp_overlay}.py.
Shared contract (same as overlays.py): pure in-place mutation, idempotent,
deterministic (sorted iteration), all-zero no-op on empty input.

Inputs are report-side blocks this package produces itself:
``project_inventory`` (scan/repo_inventory.py), github-actions agents
(scan/ci_workflows.py), the unified ``findings`` list, and cloud nodes the
builder already materialised.  The code-graph overlay goes further than the
reference: per-file Python import lists become IMPORTS edges to
EXTERNAL_IMPORT nodes (the reference stops at directory-module grouping).
"""

from __future__ import annotations

from collections import defaultdict
from typing import Any, Optional

from agentbom_amd.graph.container import UnifiedEdge, UnifiedGraph, UnifiedNode
from agentbom_amd.graph.types import (
    EntityType,
    GraphSemanticLayer,
    NodeStatus,
    RelationshipType,
)

_SEV_RANK = {"critical": 4, "high": 3, "medium": 2, "low": 1, "info": 0}


def _norm(path: Any) -> str:
    s = str(path or "").replace("\\", "/").strip()
    while s.startswith("./"):
        s = s[2:]
    return s.rstrip("/")


def _dir_id(path: str) -> str:
    return f"dir:{path or '.'}"


def _file_id(path: str) -> str:
    return f"file:{path}"


def _ancestors(dir_path: str) -> list[str]:
    """'' plus every prefix of dir_path ('a/b' → ['', 'a', 'a/b'])."""
    out = [""]
    if not dir_path:
        return out
    parts = dir_path.split("/")
    for i in range(1, len(parts) + 1):
        out.append("/".join(parts[:i]))
    return out


def _file_dir(path: str) -> str:
    return path.rsplit("/", 1)[0] if "/" in path else ""


# ── repo structure ──────────────────────────────────────────────────────────


def apply_repo_structure_overlay(graph: UnifiedGraph,
                                 report_json: dict[str, Any]) -> dict[str, int]:
    """DIRECTORY / SOURCE_FILE / CONFIG_FILE nodes + CONTAINS tree.

    Reads ``project_inventory`` and the file paths carried by existing
    MISCONFIGURATION nodes; links finding → file with AFFECTS.
    """
    counts = {"directories": 0, "files": 0, "contains_edges": 0,
              "affects_edges": 0}
    inv = report_json.get("project_inventory") \
        if isinstance(report_json, dict) else None
    dirs = inv.get("directories") if isinstance(inv, dict) else None
    file_recs = inv.get("files") if isinstance(inv, dict) else None

    needed_dirs: set[str] = set()
    dir_records: dict[str, dict] = {}
    if isinstance(dirs, list):
        for rec in dirs:
            if isinstance(rec, dict) and isinstance(rec.get("path"), str):
                p = _norm(rec["path"])
                dir_records[p] = rec
                needed_dirs.update(_ancestors(p))

    # misconfiguration nodes that name a file path
    finding_files: dict[str, str] = {}
    for node in sorted(graph.nodes.values(), key=lambda n: n.id):
        if node.entity_type != EntityType.MISCONFIGURATION:
            continue
        raw = next((node.properties.get(k) for k in
                    ("file_path", "path", "location", "config_path")
                    if isinstance(node.properties.get(k), str)
                    and "/" in str(node.properties.get(k))), None)
        if raw:
            fp = _norm(raw)
            finding_files[node.id] = fp
            needed_dirs.update(_ancestors(_file_dir(fp)))

    if not needed_dirs and not file_recs:
        return counts

    for d in sorted(needed_dirs):
        if graph.add_node(UnifiedNode(
                _dir_id(d), EntityType.DIRECTORY,
                d.rsplit("/", 1)[-1] if d else "(repo root)",
                GraphSemanticLayer.CODE,
                properties={"path": d or ".",
                            **({k: dir_records[d][k] for k in
                                ("source_files", "config_files", "manifests")
                                if k in dir_records.get(d, {})}
                               if d in dir_records else {})})):
            counts["directories"] += 1
        parent = _file_dir(d)
        if d:
            if graph.add_edge(UnifiedEdge(_dir_id(parent), _dir_id(d),
                                          RelationshipType.CONTAINS)):
                counts["contains_edges"] += 1

    def add_file(path: str, kind: str, extra: Optional[dict] = None) -> None:
        et = EntityType.CONFIG_FILE if kind in ("config", "manifest") \
            else EntityType.SOURCE_FILE
        props = {"path": path, "kind": kind, **(extra or {})}
        if graph.add_node(UnifiedNode(_file_id(path), et,
                                      path.rsplit("/", 1)[-1],
                                      GraphSemanticLayer.CODE,
                                      properties=props)):
            counts["files"] += 1
        d = _file_dir(path)
        for anc in _ancestors(d):
            graph.add_node(UnifiedNode(_dir_id(anc), EntityType.DIRECTORY,
                                       anc.rsplit("/", 1)[-1] or "(repo root)",
                                       GraphSemanticLayer.CODE,
                                       properties={"path": anc or "."}))
        if graph.add_edge(UnifiedEdge(_dir_id(d), _file_id(path),
                                      RelationshipType.CONTAINS)):
            counts["contains_edges"] += 1

    if isinstance(file_recs, list):
        for rec in sorted((r for r in file_recs
                           if isinstance(r, dict) and r.get("path")),
                          key=lambda r: str(r["path"])):
            extra = {}
            if rec.get("language"):
                extra["language"] = rec["language"]
            if rec.get("imports"):
                extra["imports"] = list(rec["imports"])
            add_file(_norm(rec["path"]), str(rec.get("kind", "source")), extra)

    for node_id, fp in sorted(finding_files.items()):
        if _file_id(fp) not in graph.nodes:
            add_file(fp, "config" if fp.rsplit(".", 1)[-1] in
                     ("yml", "yaml", "json", "toml", "ini", "cfg", "env", "tf")
                     else "source")
        if graph.add_edge(UnifiedEdge(node_id, _file_id(fp),
                                      RelationshipType.AFFECTS)):
            counts["affects_edges"] += 1
    return counts


# ── code graph ──────────────────────────────────────────────────────────────


def apply_code_graph_overlay(graph: UnifiedGraph,
                             report_json: dict[str, Any]) -> dict[str, int]:
    """CODE_MODULE per source directory (+DEFINES/CONTAINS), plus IMPORTS
    edges to EXTERNAL_IMPORT nodes from per-file Python import evidence."""
    del report_json  # evidence is read from SOURCE_FILE nodes already placed
    counts = {"code_modules": 0, "defines_edges": 0, "contains_edges": 0,
              "external_imports": 0, "imports_edges": 0}
    files_by_dir: dict[str, list[UnifiedNode]] = defaultdict(list)
    for node in graph.nodes.values():
        if node.entity_type == EntityType.SOURCE_FILE:
            p = _norm(node.properties.get("path") or node.label)
            if p:
                files_by_dir[_file_dir(p)].append(node)
    if not files_by_dir:
        return counts

    # local top-level module names (their dirs) don't become EXTERNAL_IMPORT
    local_tops = {d.split("/", 1)[0] for d in files_by_dir if d}
    local_tops.update(
        _norm(n.properties.get("path", "")).rsplit("/", 1)[-1][:-3]
        for ns in files_by_dir.values() for n in ns
        if str(n.properties.get("path", "")).endswith(".py"))

    for d in sorted(files_by_dir):
        mod_id = f"module:{d or '.'}"
        label = d.rsplit("/", 1)[-1] if d else "(repo root)"
        if graph.add_node(UnifiedNode(
                mod_id, EntityType.CODE_MODULE, label,
                GraphSemanticLayer.CODE,
                properties={"path": d or ".",
                            "source_file_count": len(files_by_dir[d])})):
            counts["code_modules"] += 1
        if _dir_id(d) in graph.nodes and graph.add_edge(
                UnifiedEdge(_dir_id(d), mod_id, RelationshipType.CONTAINS)):
            counts["contains_edges"] += 1
        ext_here: set[str] = set()
        for fnode in sorted(files_by_dir[d], key=lambda n: n.id):
            if graph.add_edge(UnifiedEdge(fnode.id, mod_id,
                                          RelationshipType.DEFINES)):
                counts["defines_edges"] += 1
            for imp in fnode.properties.get("imports", []) or []:
                if isinstance(imp, str) and imp and imp not in local_tops:
                    ext_here.add(imp)
        for imp in sorted(ext_here):
            ext_id = f"extimport:{imp}"
            if graph.add_node(UnifiedNode(ext_id, EntityType.EXTERNAL_IMPORT,
                                          imp, GraphSemanticLayer.CODE)):
                counts["external_imports"] += 1
            if graph.add_edge(UnifiedEdge(mod_id, ext_id,
                                          RelationshipType.IMPORTS)):
                counts["imports_edges"] += 1
    return counts


# ── CI graph ────────────────────────────────────────────────────────────────


def apply_ci_graph_overlay(graph: UnifiedGraph,
                           report_json: dict[str, Any]) -> dict[str, int]:
    """CI_JOB per github-actions agent; CONFIGURES from its workflow file;
    RUNS to the tools the workflow exercises."""
    counts = {"ci_jobs": 0, "configures_edges": 0, "runs_edges": 0}
    agents = report_json.get("agents") if isinstance(report_json, dict) else None
    if not isinstance(agents, list):
        return counts
    gha = [a for a in agents if isinstance(a, dict)
           and a.get("source") == "github-actions"]
    if not gha:
        return counts

    files_by_path = {_norm(n.properties.get("path") or n.label): n.id
                     for n in graph.nodes.values()
                     if n.entity_type in (EntityType.CONFIG_FILE,
                                          EntityType.SOURCE_FILE)}
    tools_by_label = {n.label: n.id for n in graph.nodes.values()
                      if n.entity_type == EntityType.TOOL}

    for agent in sorted(gha, key=lambda a: str(a.get("name", ""))):
        name = str(agent.get("name") or "")
        stem = name[4:] if name.startswith("gha:") else name
        if not stem:
            continue
        job_id = f"ci_job:{stem}"
        cfg = _norm(agent.get("config_path") or "")
        # config paths may be absolute on disk; match by workflow-relative tail
        cfg_node = files_by_path.get(cfg)
        if cfg_node is None and cfg:
            tail = ".github/workflows/" + cfg.rsplit("/", 1)[-1]
            cfg_node = files_by_path.get(tail)
        if graph.add_node(UnifiedNode(
                job_id, EntityType.CI_JOB, stem, GraphSemanticLayer.CI,
                properties={"workflow": stem, "config_path": cfg,
                            "agent_name": name, "source": "github-actions"})):
            counts["ci_jobs"] += 1
        if cfg_node and graph.add_edge(UnifiedEdge(
                cfg_node, job_id, RelationshipType.CONFIGURES)):
            counts["configures_edges"] += 1
        for server in agent.get("mcp_servers") or []:
            if not isinstance(server, dict):
                continue
            for tool in server.get("tools") or []:
                tname = tool.get("name") if isinstance(tool, dict) else tool
                tid = tools_by_label.get(str(tname or ""))
                if tid and graph.add_edge(UnifiedEdge(
                        job_id, tid, RelationshipType.RUNS)):
                    counts["runs_edges"] += 1
    return counts


# ── ASPM application correlation ────────────────────────────────────────────


def apply_aspm_overlay(graph: UnifiedGraph,
                       report_json: dict[str, Any]) -> dict[str, int]:
    """Correlate findings around APPLICATION nodes (manifest-rooted apps).

    App identity: the deepest project_inventory manifest root that owns the
    finding's file location, else the asset's top-level path segment, else
    ``(repo root)``.  Within an app, (component, rule) pairs dedupe; each app
    node carries a severity histogram, max severity, a reachable count (from
    the finding's own reachability/actionability signals plus node exposure
    flags), and an optional CODEOWNERS owner from ``report_json["codeowners"]``.
    """
    zero = {"applications": 0, "correlated_findings": 0, "deduplicated": 0,
            "reachable": 0, "belongs_to_edges": 0}
    findings = report_json.get("findings") if isinstance(report_json, dict) \
        else None
    if not isinstance(findings, list) or not findings:
        return zero
    findings = [f for f in findings if isinstance(f, dict)]
    if not findings:
        return zero

    inv = report_json.get("project_inventory") or {}
    manifest_roots = sorted(
        {_norm(d.get("path")) for d in inv.get("directories", [])
         if isinstance(d, dict) and d.get("manifests")},
        key=len, reverse=True)
    owners = report_json.get("codeowners") or {}
    if not isinstance(owners, dict):
        owners = {}

    def app_for(f: dict) -> str:
        asset = f.get("asset") if isinstance(f.get("asset"), dict) else {}
        loc = _norm(asset.get("location") or asset.get("identifier") or "")
        for root in manifest_roots:
            if loc == root or loc.startswith(root + "/"):
                return root or "(repo root)"
        if "/" in loc:
            return loc.split("/", 1)[0]
        return "(repo root)"

    def owner_for(app: str) -> str:
        best, owner = -1, ""
        for prefix, who in owners.items():
            p = _norm(prefix)
            if (app == p or app.startswith(p + "/") or p == "") \
                    and len(p) > best:
                best, owner = len(p), str(who)
        return owner

    apps: dict[str, dict[str, Any]] = {}
    dedup: dict[str, set[tuple[str, str]]] = defaultdict(set)
    stats = dict(zero)
    for f in findings:
        app = app_for(f)
        asset = f.get("asset") if isinstance(f.get("asset"), dict) else {}
        comp = str(asset.get("name") or asset.get("identifier") or "?")
        rule = str(f.get("cve_id") or f.get("title") or f.get("finding_type")
                   or "?")
        sev = str(f.get("severity") or "info").lower()
        rec = apps.setdefault(app, {"histogram": defaultdict(int),
                                    "reachable": 0, "findings": 0})
        if (comp, rule) in dedup[app]:
            stats["deduplicated"] += 1
            continue
        dedup[app].add((comp, rule))
        rec["findings"] += 1
        rec["histogram"][sev] += 1
        stats["correlated_findings"] += 1
        reachable = (f.get("is_actionable") is True
                     or str(f.get("reachability") or "").lower() in
                     ("reachable", "confirmed", "exploitable")
                     or f.get("graph_reachable") is True)
        if not reachable and f.get("node_id") in graph.nodes:
            p = graph.nodes[f["node_id"]].properties
            reachable = bool(p.get("internet_exposed")
                             or p.get("toxic_combination")
                             or p.get("on_attack_path"))
        if reachable:
            rec["reachable"] += 1
            stats["reachable"] += 1

    for app in sorted(apps):
        rec = apps[app]
        hist = dict(sorted(rec["histogram"].items()))
        worst = max(hist, key=lambda s: _SEV_RANK.get(s, 0), default="info")
        app_id = f"application:{app}"
        props = {"findings": rec["findings"], "severity_histogram": hist,
                 "max_severity": worst, "reachable_findings": rec["reachable"]}
        who = owner_for(app)
        if who:
            props["owner"] = who
        if graph.add_node(UnifiedNode(
                app_id, EntityType.APPLICATION, app, GraphSemanticLayer.APP,
                status=(NodeStatus.VULNERABLE
                        if _SEV_RANK.get(worst, 0) >= 3 else NodeStatus.ACTIVE),
                properties=props)):
            stats["applications"] += 1
        # stitch the app onto the repo-structure tree when present
        droot = "" if app == "(repo root)" else app
        if _dir_id(droot) in graph.nodes and graph.add_edge(UnifiedEdge(
                _dir_id(droot), app_id, RelationshipType.BELONGS_TO)):
            stats["belongs_to_edges"] += 1
    return stats


# ── CNAPP exposure classification ───────────────────────────────────────────

_SENSITIVE_KEYWORDS = ("customer", "payment", "billing", "medical", "health",
                       "patient", "ssn", "salary", "payroll", "pii", "phi",
                       "credit", "card", "passport", "secret", "credential",
                       "financial", "invoice", "tax")
_FRAMEWORKS = {"pci": ("payment", "billing", "credit", "card", "invoice"),
               "hipaa": ("medical", "health", "patient", "phi"),
               "gdpr": ("customer", "pii", "passport", "ssn")}


def apply_cnapp_overlay(graph: UnifiedGraph) -> dict[str, int]:
    """Classify data stores for sensitivity and flag internet-exposed
    sensitive data as toxic MISCONFIGURATION findings."""
    counts = {"sensitive_nodes": 0, "exposed_sensitive": 0,
              "findings_created": 0}
    store_types = {EntityType.DATA_STORE, EntityType.CLOUD_RESOURCE,
                   EntityType.RESOURCE}
    for node in sorted(graph.nodes.values(), key=lambda n: n.id):
        if node.entity_type not in store_types:
            continue
        text = " ".join([node.label,
                         str(node.properties.get("description", "")),
                         " ".join(map(str, node.tags))]).lower()
        hits = sorted({kw for kw in _SENSITIVE_KEYWORDS if kw in text})
        if not hits:
            continue
        frameworks = sorted({fw for fw, kws in _FRAMEWORKS.items()
                             if any(k in hits for k in kws)})
        node.properties["data_sensitivity"] = "sensitive"
        node.properties["sensitivity_evidence"] = hits
        if frameworks:
            node.properties["regulatory_frameworks"] = frameworks
        counts["sensitive_nodes"] += 1

        exposed = bool(node.properties.get("internet_exposed")
                       or node.properties.get("public")
                       or node.properties.get("publicly_accessible"))
        if not exposed:
            continue
        node.properties["toxic_exposed_sensitive"] = True
        counts["exposed_sensitive"] += 1
        f_id = f"misconfig:cnapp-exposed-sensitive:{node.id}"
        if graph.add_node(UnifiedNode(
                f_id, EntityType.MISCONFIGURATION,
                f"Internet-exposed sensitive data: {node.label}",
                GraphSemanticLayer.FINDING, status=NodeStatus.VULNERABLE,
                properties={"pattern": "internet_exposed_sensitive_data",
                            "severity": "critical",
                            "frameworks": frameworks})):
            counts["findings_created"] += 1
        graph.add_edge(UnifiedEdge(f_id, node.id, RelationshipType.AFFECTS,
                                   weight=9.0))
    return counts
