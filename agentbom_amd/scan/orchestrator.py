"""Scan orchestration: inventory -> match -> blast radius -> report.

Feature-parity rebuild of the reference's package-scan hot path
(reference: src/agent_bom/scanners/package_scan.py:1307-2028 scan_packages/
scan_agents; :1851-2007 BlastRadius construction with CWE filtering).

The matching engine is the same advisory-arena machinery the GPU uses:
windows are encoded to u128 keys (db/arena.build_arena) and matched via the
HIP kernel on a GPU box or the bit-identical CPU oracle elsewhere; windows
or versions the encoder cannot represent resolve through the exact
comparator (fail-closed).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional, Sequence

import numpy as np

from agentbom_amd.db.arena import (
    AdvisoryArena,
    AdvisoryWindow,
    build_arena,
    hash_name,
    match_cpu_fallback,
)
from agentbom_amd.models import (
    Agent,
    AgentType,
    AgentStatus,
    AIBOMReport,
    BlastRadius,
    MCPServer,
    MCPTool,
    Package,
    ScanIssue,
    ScanRun,
    Severity,
    TransportType,
    Vulnerability,
    build_attack_vector_summary,
    classify_cwe_impact,
    compute_confidence,
    expand_blast_radius_hops,
    filter_credentials_by_impact,
    filter_tools_by_impact,
)
from agentbom_amd.models.cwe_impact import IMPACT_UNKNOWN
from agentbom_amd.ops import cpu_ref
from agentbom_amd.scan.malicious import flag_malicious_packages
from agentbom_amd.utils.canonical_ids import normalize_package_ecosystem, normalize_package_name
from agentbom_amd.utils.version_keys import encode_version
from agentbom_amd.utils.version_utils import set_scan_warning_sink


def inventory_to_agents(inventory: dict[str, Any]) -> list[Agent]:
    """Hydrate a JSON inventory (demo/fleet shape) into model objects."""
    agents: list[Agent] = []
    for a in inventory.get("agents", []):
        servers: list[MCPServer] = []
        for s in a.get("mcp_servers", []):
            tools = [MCPTool(name=t["name"], description=t.get("description", ""))
                     for t in s.get("tools", [])]
            packages = [
                Package(
                    name=p["name"], version=p.get("version", ""),
                    ecosystem=normalize_package_ecosystem(p.get("ecosystem", "")),
                    is_direct=p.get("is_direct", True),
                )
                for p in s.get("packages", [])
            ]
            servers.append(
                MCPServer(
                    name=s["name"],
                    command=s.get("command", ""),
                    args=list(s.get("args", [])),
                    env=dict(s.get("env", {})),
                    transport=TransportType(s.get("transport", "stdio")),
                    url=s.get("url"),
                    tools=tools,
                    packages=packages,
                )
            )
        agents.append(
            Agent(
                name=a["name"],
                agent_type=AgentType(a.get("agent_type", "custom")),
                config_path=a.get("config_path", ""),
                source=a.get("source"),
                mcp_servers=servers,
                status=AgentStatus(a.get("status", "configured")),
            )
        )
    return agents


@dataclass
class ScanOptions:
    offline: bool = False
    demo: bool = False
    include_unfixed: bool = False
    blast_radius_depth: int = 1
    fail_on_severity: str = "critical"
    exit_zero: bool = False
    fail_on_kev: bool = False
    use_gpu: Optional[bool] = None  # None = auto
    ignore_ids: frozenset = frozenset()
    vendor_advisories: bool = True  # AMD PSIRT / NVIDIA CSAF / Intel feeds
    threat_intel: bool = True       # local IOC store enrichment
    live_osv: bool = False          # query api.osv.dev for the package batch
    transitive: bool = False        # expand the dependency tree via registries
    max_depth: int = 3              # transitive expansion depth bound
    fail_if_ai_risk: bool = False   # gate on findings carrying AI risk context
    warn_on: Optional[str] = None   # severity that warns (stderr) w/o failing


def _match_packages(
    unique_pkgs: list[tuple[str, str, str]],
    arena: AdvisoryArena,
    use_gpu: bool,
    include_unfixed: bool = False,
) -> list[tuple[int, int]]:
    """Match unique (eco, name, version) tuples; returns (pkg_i, window_i)."""
    P = len(unique_pkgs)
    gk = np.empty(P, dtype=np.uint64)
    khi = np.zeros(P, dtype=np.uint64)
    klo = np.zeros(P, dtype=np.uint64)
    fl = np.zeros(P, dtype=np.uint8)
    unencodable: set[int] = set()
    for i, (eco, name, version) in enumerate(unique_pkgs):
        gk[i] = hash_name(eco, name)
        hi, lo, ok = encode_version(version, eco)
        khi[i], klo[i] = hi, lo
        if ok:
            fl[i] = cpu_ref.PF_ENCODABLE
        else:
            unencodable.add(i)

    windows = {
        "intro_hi": arena.intro_hi, "intro_lo": arena.intro_lo,
        "fixed_hi": arena.fixed_hi, "fixed_lo": arena.fixed_lo,
        "last_hi": arena.last_hi, "last_lo": arena.last_lo,
        "flags": arena.flags,
    }
    if use_gpu:
        import torch

        from agentbom_amd.ops import native

        dev = torch.device("cuda")
        at = arena.to_torch(dev)
        t = lambda a, dt: torch.from_numpy(np.ascontiguousarray(a)).to(device=dev, dtype=dt)
        gp, gw = native.match(
            t(gk.view(np.int64), torch.int64), t(khi.view(np.int64), torch.int64),
            t(klo.view(np.int64), torch.int64), t(fl, torch.uint8),
            at["group_keys"], at["group_off"], at["windows"],
        )
        pairs = list(zip(gp.cpu().tolist(), gw.cpu().tolist()))
    else:
        cp, cw = cpu_ref.match(gk, khi, klo, fl, arena.group_keys, arena.group_off, windows)
        pairs = list(zip(cp.tolist(), cw.tolist()))

    fallback = match_cpu_fallback(
        arena,
        [(i, e, n, v) for i, (e, n, v) in enumerate(unique_pkgs)],
        unencodable,
        include_unfixed=include_unfixed,
    )
    return sorted(set(pairs) | set(fallback))


def scan_agents(
    agents: list[Agent],
    advisory_windows: Sequence[AdvisoryWindow],
    options: Optional[ScanOptions] = None,
) -> AIBOMReport:
    """Match every package across the estate and build blast radii."""
    import time as _time

    options = options or ScanOptions()
    warnings: list[str] = []
    set_scan_warning_sink(warnings.append)
    perf: dict[str, float] = {}
    _t0 = _time.perf_counter()

    def _mark(stage: str) -> None:
        nonlocal _t0
        now = _time.perf_counter()
        perf[f"{stage}_ms"] = round((now - _t0) * 1000, 3)
        _t0 = now

    # ── transitive expansion + registry version resolution ─────────────────
    if options.transitive and not options.offline:
        try:
            from agentbom_amd.scan.transitive import (
                _MetaCache,
                expand_transitive,
                resolve_package_versions,
            )
            from agentbom_amd.utils.http_client import create_client

            client = create_client()
            cache = _MetaCache()
            n_resolved = n_added = 0
            for agent in agents:
                for server in agent.mcp_servers:
                    n_resolved += resolve_package_versions(server.packages, client, cache)
                    children = expand_transitive(server.packages,
                                                 max_depth=options.max_depth,
                                                 client=client, cache=cache)
                    server.packages.extend(children)
                    n_added += len(children)
            if n_resolved or n_added:
                warnings.append(
                    f"transitive expansion: +{n_added} packages, "
                    f"{n_resolved} versions resolved from registries")
        except Exception as exc:
            warnings.append(f"transitive expansion unavailable: {exc}; "
                            "scanned declared dependencies only")
    _mark("transitive")

    # ── collect + dedup packages across the estate ─────────────────────────
    pkg_refs: dict[tuple[str, str, str], list[Package]] = {}
    for agent in agents:
        for server in agent.mcp_servers:
            for pkg in server.packages:
                key = (
                    normalize_package_ecosystem(pkg.ecosystem),
                    normalize_package_name(pkg.name, pkg.ecosystem),
                    pkg.version,
                )
                pkg_refs.setdefault(key, []).append(pkg)

    unique = list(pkg_refs.keys())

    _mark("dedup")

    # malicious screening (fails closed)
    all_pkgs = [p for plist in pkg_refs.values() for p in plist]
    flag_malicious_packages(all_pkgs)
    _mark("malicious_screen")

    # ── matching ───────────────────────────────────────────────────────────
    advisory_windows = list(advisory_windows)
    if options.live_osv and not options.offline:
        # live OSV batch supplement (reference package_scan.py:573): merges
        # into the same arena; failures warn and fall back to local windows
        # (fail-open with a recorded coverage gap, never silently clean)
        try:
            from agentbom_amd.db.live import osv_windows_for_packages

            advisory_windows.extend(osv_windows_for_packages(unique))
        except Exception as exc:  # OfflineError, transport failure, ...
            warnings.append(f"live OSV query unavailable: {exc}; "
                            "matched against the local advisory DB only")
    arena = build_arena(advisory_windows, include_unfixed=options.include_unfixed)
    use_gpu = options.use_gpu
    if use_gpu is None:
        try:
            import torch

            use_gpu = torch.cuda.is_available() and len(unique) >= 10_000
        except Exception:
            use_gpu = False
    _mark("arena_build")
    pairs = _match_packages(unique, arena, use_gpu,
                            include_unfixed=options.include_unfixed)
    _mark("match")

    # ── attach Vulnerability objects to packages ───────────────────────────
    # group matched windows by (package, vuln_id): several windows of one
    # advisory collapse onto one Vulnerability row
    for pkg_i, win_i in pairs:
        w = arena.windows[win_i]
        if w.vuln_id in options.ignore_ids:
            continue
        key = unique[pkg_i]
        vuln = Vulnerability(
            id=w.vuln_id,
            summary=w.summary,
            severity=w.severity,
            severity_source="cvss" if w.cvss_score else None,
            cvss_score=w.cvss_score,
            fixed_version=w.fixed_version or w.fixed,
            epss_score=w.epss_score,
            is_kev=w.is_kev,
            cwe_ids=list(w.cwe_ids),
            aliases=list(w.aliases),
            advisory_sources=["osv"],
            match_confidence_tier="osv_range",
        )
        vuln.confidence = compute_confidence(vuln)
        for pkg in pkg_refs[key]:
            if all(v.id != vuln.id for v in pkg.vulnerabilities):
                pkg.vulnerabilities.append(vuln)

    # vendor-bulletin supplemental pass (AMD PSIRT / NVIDIA CSAF / Intel):
    # bulletins that never reach OSV, matched fail-closed and tiered
    # (reference: scanners/{amd,nvidia,ghsa}_advisory.py supplemental joins)
    if options.vendor_advisories:
        from agentbom_amd.scan.vendor_advisories import check_vendor_advisories

        for key, extra in check_vendor_advisories(unique).items():
            for vuln in extra:
                if vuln.id in options.ignore_ids:
                    continue
                for pkg in pkg_refs[key]:
                    if all(v.id != vuln.id for v in pkg.vulnerabilities):
                        pkg.vulnerabilities.append(vuln)
    _mark("vendor_advisories")

    # malicious packages surface as synthetic advisory rows (fail closed)
    for pkg in all_pkgs:
        if pkg.is_malicious and not any(v.id.startswith("MAL-") for v in pkg.vulnerabilities):
            pkg.vulnerabilities.append(
                Vulnerability(
                    id=f"MAL-TYPOSQUAT-{pkg.name}",
                    summary=pkg.malicious_reason or f"known-malicious package {pkg.name}",
                    severity=Severity.CRITICAL,
                    severity_source="heuristic",
                    advisory_sources=["osv"],
                )
            )

    # ── blast radius per (package identity, vulnerability) ────────────────
    # index servers/agents containing each package identity
    servers_of: dict[tuple[str, str, str], list[MCPServer]] = {}
    agents_of: dict[tuple[str, str, str], dict[str, Agent]] = {}
    for agent in agents:
        for server in agent.mcp_servers:
            for pkg in server.packages:
                key = (
                    normalize_package_ecosystem(pkg.ecosystem),
                    normalize_package_name(pkg.name, pkg.ecosystem),
                    pkg.version,
                )
                lst = servers_of.setdefault(key, [])
                if server not in lst:
                    lst.append(server)
                agents_of.setdefault(key, {})[agent.name] = agent

    blast_radii: list[BlastRadius] = []
    seen_pair: set[tuple] = set()
    for key, plist in pkg_refs.items():
        rep = plist[0]
        vulns = {}
        for p in plist:
            for v in p.vulnerabilities:
                vulns[v.id] = v
        for vid, vuln in vulns.items():
            if (key, vid) in seen_pair:
                continue
            seen_pair.add((key, vid))
            srvs = servers_of.get(key, [])
            ags = list(agents_of.get(key, {}).values())
            all_creds: list[str] = []
            all_tools: list[MCPTool] = []
            for s in srvs:
                for c in s.credential_names:
                    if c not in all_creds:
                        all_creds.append(c)
                all_tools.extend(s.tools)
            impact = classify_cwe_impact(vuln.cwe_ids)
            if rep.is_malicious:
                impact = "code-execution"  # malicious package = full compromise
            creds = filter_credentials_by_impact(impact, all_creds)
            tools = filter_tools_by_impact(impact, all_tools)
            br = BlastRadius(
                vulnerability=vuln,
                package=rep,
                affected_servers=srvs,
                affected_agents=ags,
                exposed_credentials=creds,
                exposed_tools=tools,
                impact_category=impact,
                all_server_credentials=all_creds,
                all_server_tools=all_tools,
            )
            br.attack_vector_summary = build_attack_vector_summary(
                vuln.cwe_ids, impact, creds, tools,
                severity=vuln.severity.value, is_kev=vuln.is_kev,
            )
            from agentbom_amd.scan.compliance import apply_framework_tags

            apply_framework_tags(br)
            br.calculate_risk_score()
            blast_radii.append(br)

    _mark("blast_radius")
    expand_blast_radius_hops(blast_radii, agents, max_depth=options.blast_radius_depth)
    blast_radii.sort(key=lambda b: -b.risk_score)
    _mark("hop_expand_rank")

    report = AIBOMReport(agents=agents, blast_radii=blast_radii)
    report.warnings = warnings
    report.scan_sources = ["agent_discovery"] if not options.demo else ["demo_inventory"]
    report.scan_run = ScanRun()
    # local threat-intel pass: active-exploitation rescore + malicious
    # indicators (reference: intel_lookup during enrichment) — runs BEFORE
    # finding fusion so findings carry the boosted scores
    if options.threat_intel:
        from agentbom_amd.scan.intel import enrich_report_with_intel

        enrich_report_with_intel(report)
        _mark("threat_intel")
    report.findings = report.to_findings()
    _mark("finding_fusion")
    # counter-in-report contract (reference: scan_performance counters,
    # package_scan.py:1324 / models.py:1243): per-stage ms + volume counters
    report.scan_performance_data = {
        "stages": perf,
        "total_ms": round(sum(perf.values()), 3),
        "unique_packages": len(unique),
        "advisory_windows": len(arena.windows),
        "match_pairs": len(pairs),
        "blast_radii": len(blast_radii),
        "match_path": "gpu" if use_gpu else "cpu",
    }
    set_scan_warning_sink(None)
    return report


def compute_exit_code(report: AIBOMReport, options: ScanOptions) -> int:
    """Exit-code contract (site-docs/reference/exit-codes.md):

    0 clean/under threshold · 1 gate matched (severity threshold, KEV gate,
    malicious fails closed, incomplete scan fails closed) · 2 usage/empty.
    --exit-zero suppresses only the severity gate, never the fail-closed
    gates."""
    from agentbom_amd.models import ScanOutcome, active_blast_radii

    # fail-closed: malicious package
    if any(br.package.is_malicious for br in report.blast_radii):
        return 1
    # fail-closed: incomplete evidence
    if report.scan_run.outcome != ScanOutcome.COMPLETE:
        return 1
    if options.exit_zero:
        return 0
    threshold_order = ["low", "medium", "high", "critical"]
    try:
        t = threshold_order.index(options.fail_on_severity)
    except ValueError:
        t = threshold_order.index("critical")
    gate = set(threshold_order[t:])
    active = active_blast_radii(report.blast_radii)
    if any(br.vulnerability.severity.value in gate for br in active):
        return 1
    if options.fail_on_kev and any(br.vulnerability.is_kev for br in active):
        return 1
    if options.fail_if_ai_risk and any(br.ai_risk_context for br in active):
        return 1
    return 0


def run_demo_scan(options: Optional[ScanOptions] = None) -> AIBOMReport:
    """BASELINE config 1: the deterministic demo estate, fully offline."""
    from agentbom_amd.scan.demo import DEMO_INVENTORY, demo_advisory_windows

    options = options or ScanOptions(demo=True, offline=True)
    agents = inventory_to_agents(DEMO_INVENTORY)
    return scan_agents(agents, demo_advisory_windows(), options)
