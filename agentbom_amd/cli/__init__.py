"""agent-bom CLI (click) — reference: src/agent_bom/cli/__init__.py.

Commands: ``agents``/``scan`` (aliases), ``serve``, ``mcp server``,
``graph``, ``db sync/status``, ``bench``, ``version``.
Exit codes follow site-docs/reference/exit-codes.md (0 clean, 1 gate
matched / fail-closed, 2 usage/empty).
"""

from __future__ import annotations

import json
import sys
from pathlib import Path
from typing import Optional

import click

from agentbom_amd import __version__

FORMATS = [
    "console", "json", "html", "sarif", "cyclonedx", "spdx", "spdx2", "spdx3", "ocsf",
    "csv", "markdown", "plain", "junit", "prometheus", "parquet", "pdf",
    "svg", "badge", "graph", "mermaid", "dot", "graphml", "cypher",
]


@click.group(name="agent-bom")
@click.version_option(__version__, prog_name="agent-bom")
def main() -> None:
    """AI-BOM security scanner and blast-radius graph engine (MI355X-native)."""


def _render(report, fmt: str, output: Optional[str], verbose: bool) -> None:
    from agentbom_amd.output import json_fmt, misc_fmt

    text: Optional[str] = None
    if fmt == "console":
        from rich.console import Console

        from agentbom_amd.output.console_render import render_report

        console = Console(file=open(output, "w") if output else None)
        render_report(report, console=console, verbose=verbose)
        return
    if fmt == "json":
        text = json.dumps(json_fmt.to_json(report), indent=2, default=str)
    elif fmt == "sarif":
        from agentbom_amd.output.sarif import to_sarif

        text = json.dumps(to_sarif(report), indent=2)
    elif fmt == "cyclonedx":
        from agentbom_amd.output.cyclonedx_fmt import to_cyclonedx

        text = json.dumps(to_cyclonedx(report), indent=2)
    elif fmt in ("spdx", "spdx2"):
        from agentbom_amd.output.spdx_fmt import to_spdx

        text = json.dumps(to_spdx(report), indent=2)
    elif fmt == "spdx3":
        from agentbom_amd.output.spdx_fmt import to_spdx3

        text = json.dumps(to_spdx3(report), indent=2)
    elif fmt == "html":
        from agentbom_amd.output.html_fmt import to_html

        text = to_html(report)
    elif fmt == "ocsf":
        from agentbom_amd.output.ocsf import to_ocsf

        text = json.dumps(to_ocsf(report), indent=2, default=str)
    elif fmt == "csv":
        text = misc_fmt.to_csv(report)
    elif fmt == "markdown":
        text = misc_fmt.to_markdown(report)
    elif fmt == "junit":
        text = misc_fmt.to_junit(report)
    elif fmt == "prometheus":
        text = misc_fmt.to_prometheus(report)
    elif fmt == "parquet":
        if not output:
            raise click.UsageError("parquet requires -o/--output")
        Path(output).write_bytes(misc_fmt.to_parquet_bytes(report))
        return
    elif fmt == "pdf":
        from agentbom_amd.output.pdf_fmt import to_pdf_bytes

        if not output:
            raise click.UsageError("pdf requires -o/--output")
        Path(output).write_bytes(to_pdf_bytes(report))
        return
    elif fmt in ("svg", "badge"):
        text = misc_fmt.to_badge_svg(report)
    elif fmt in ("plain",):
        text = misc_fmt.to_markdown(report)
    elif fmt in ("graph", "mermaid", "dot", "graphml", "cypher"):
        from agentbom_amd.output import graph_export

        if fmt in ("graph",):
            text = json.dumps(graph_export.build_graph_dict(report), indent=2)
        elif fmt == "mermaid":
            text = graph_export.to_mermaid(report)
        elif fmt == "dot":
            text = graph_export.to_dot(report)
        elif fmt == "graphml":
            text = graph_export.to_graphml(report)
        else:
            text = graph_export.to_cypher(report)
    else:
        raise click.UsageError(f"unknown format {fmt!r}")

    if output:
        Path(output).write_text(text)
    else:
        click.echo(text, nl=False)


def _apply_profile(kw: dict) -> dict:
    """Merge a named .agent-bom.yaml profile into the scan kwargs.

    Only parameters the user left at their DEFAULT are overridden — an
    explicit CLI flag always wins over the profile (reference:
    cli/_profiles.py precedence)."""
    name = kw.pop("profile", None)
    if not name:
        return kw
    from agentbom_amd.utils.project_config import get_profile, load_project_config

    try:
        body = get_profile(load_project_config("."), name)
    except ValueError as exc:
        raise click.UsageError(str(exc))
    ctx = click.get_current_context(silent=True)
    for key, value in body.items():
        param = "fmt" if key == "format" else key.replace("-", "_")
        if param not in kw:
            raise click.UsageError(
                f"profile {name!r}: unknown flag {key!r}")
        src = ctx.get_parameter_source(param) if ctx else None
        if src is None or src.name == "DEFAULT":
            kw[param] = tuple(value) if isinstance(kw[param], tuple) and \
                isinstance(value, list) else value
    return kw


def _scan_impl(
    demo: bool, offline: bool, inventory: Optional[str], fmt: str,
    output: Optional[str], fail_on_severity: str, fail_on_kev: bool,
    exit_zero: bool, blast_radius_depth: int, verbose: bool,
    include_unfixed: bool, no_gpu: bool,
    sbom: Optional[str] = None, image: Optional[str] = None,
    filesystem: Optional[str] = None,
    scan_secrets: bool = False, model_files: Optional[str] = None,
    code: Optional[str] = None, iac: Optional[str] = None,
    aws_inventory: Optional[str] = None, endpoint: bool = False,
    notebooks: Optional[str] = None, skills: Optional[str] = None,
    semgrep: Optional[str] = None, cloud_inventory: tuple = (),
    # modes
    self_scan: bool = False, inventory_only: bool = False,
    no_discover: bool = False, dry_run: bool = False,
    # depth / enrichment
    transitive: bool = False, max_depth: int = 3, live_osv: bool = False,
    enrich: bool = False, enrich_bundle: Optional[str] = None,
    deps_dev: Optional[str] = None, vex: Optional[str] = None,
    generate_vex: Optional[str] = None,
    license_check_flag: bool = False, license_deny: tuple = (),
    # gates
    fail_if_ai_risk: bool = False, warn_on: Optional[str] = None,
    require_fresh_db: bool = False, policy: Optional[str] = None,
    # surfaces
    image_tar: Optional[str] = None, repo: Optional[str] = None,
    os_packages: bool = False, jupyter: Optional[str] = None,
    browser_extensions: tuple = (), dataset_cards: Optional[str] = None,
    training_pipelines: Optional[str] = None,
    verify_model_hashes: Optional[str] = None, gpu_scan_flag: bool = False,
    introspect: bool = False, health_check_flag: bool = False,
    scan_prompts: Optional[str] = None, scan_pii: Optional[str] = None,
    skill: Optional[str] = None,
    github_actions: Optional[str] = None, repo_inventory: Optional[str] = None,
    k8s_posture: Optional[str] = None, exceptions_db: Optional[str] = None,
    # live cloud collectors
    aws_live: bool = False, azure_live: Optional[str] = None,
    azure_token: Optional[str] = None, gcp_live: Optional[str] = None,
    gcp_token: Optional[str] = None,
    # push / integrations
    push_url: Optional[str] = None, push_api_key: Optional[str] = None,
    webhook: tuple = (), slack_webhook: Optional[str] = None,
    jira_url: Optional[str] = None, jira_token: Optional[str] = None,
    jira_project: Optional[str] = None,
    siem_url: Optional[str] = None, siem_token: Optional[str] = None,
) -> None:
    from agentbom_amd.scan.orchestrator import (
        ScanOptions,
        compute_exit_code,
        inventory_to_agents,
        run_demo_scan,
        scan_agents,
    )

    # flag aliases (reference option surface)
    image = image or image_tar
    notebooks = notebooks or jupyter
    skills = skills or skill
    if repo:
        filesystem = filesystem or repo
        code = code or repo
    if self_scan:
        import agentbom_amd as _pkg

        filesystem = filesystem or str(Path(_pkg.__file__).resolve().parent)

    options = ScanOptions(
        demo=demo, offline=offline, include_unfixed=include_unfixed,
        blast_radius_depth=blast_radius_depth, fail_on_severity=fail_on_severity,
        exit_zero=exit_zero, fail_on_kev=fail_on_kev,
        use_gpu=False if no_gpu else None,
        transitive=transitive, max_depth=max_depth, live_osv=live_osv,
        fail_if_ai_risk=fail_if_ai_risk, warn_on=warn_on,
    )

    if dry_run:
        plan = {
            "mode": "demo" if demo else "discover+scan",
            "offline": offline,
            "surfaces": {k: bool(v) for k, v in {
                "inventory": inventory, "sbom": sbom, "image": image,
                "filesystem": filesystem, "code": code, "iac": iac,
                "os_packages": os_packages, "notebooks": notebooks,
                "skills": skills, "browser_extensions": browser_extensions,
                "dataset_cards": dataset_cards,
                "training_pipelines": training_pipelines,
                "gpu_scan": gpu_scan_flag, "secrets": scan_secrets,
                "prompts": scan_prompts, "pii": scan_pii,
            }.items()},
            "gates": {"fail_on_severity": fail_on_severity,
                      "fail_on_kev": fail_on_kev,
                      "fail_if_ai_risk": fail_if_ai_risk,
                      "policy": bool(policy)},
        }
        click.echo(json.dumps(plan, indent=2))
        sys.exit(0)

    if require_fresh_db and not demo:
        from agentbom_amd.db.store import AdvisoryStore, default_db_path

        st = AdvisoryStore(default_db_path()).status()
        worst = st.get("freshness", {}).get("worst", "unknown")
        if worst in ("expired", "unknown"):
            click.echo(f"error: advisory DB freshness is {worst!r} "
                       "(--require-fresh-db): run `agent-bom db sync`", err=True)
            sys.exit(2)
    # project-level defaults + suppression file (.agent-bom.yaml / -ignore)
    from agentbom_amd.utils.project_config import apply_to_scan_options, load_project_config

    project_cfg = load_project_config(".")
    apply_to_scan_options(project_cfg, options)
    for warning in project_cfg.warnings:
        click.echo(f"warning: {warning}", err=True)
    if demo:
        report = run_demo_scan(options)
    else:
        agents = []
        if inventory:
            agents += inventory_to_agents(json.loads(Path(inventory).read_text()))
        if sbom:
            from agentbom_amd.scan.sbom_ingest import sbom_to_agent

            agents.append(sbom_to_agent(sbom))
        if image:
            from agentbom_amd.scan.oci import oci_result_to_agent, scan_image

            if Path(image).exists():
                result = scan_image(image)
            else:
                from agentbom_amd.scan.oci_registry import scan_image_registry

                if offline:
                    click.echo("error: --image with a registry reference "
                               "needs network (offline mode)", err=True)
                    sys.exit(2)
                result = scan_image_registry(image)
            for w in result.warnings:
                click.echo(f"warning: {w}", err=True)
            agents.append(oci_result_to_agent(result))
        if filesystem:
            from agentbom_amd.models import Agent, AgentType, MCPServer, ServerSurface
            from agentbom_amd.scan.parsers import extract_packages

            pkgs = extract_packages(filesystem)
            agents.append(Agent(
                name=f"filesystem:{Path(filesystem).name}", agent_type=AgentType.CUSTOM,
                config_path=str(filesystem),
                mcp_servers=[MCPServer(name=f"filesystem:{Path(filesystem).name}",
                                       packages=pkgs, surface=ServerSurface.FILESYSTEM)],
                source="filesystem_scan",
            ))
        if os_packages:
            from agentbom_amd.scan.surfaces_extra import scan_os_packages

            os_agent = scan_os_packages()
            if os_agent:
                agents.append(os_agent)
            else:
                click.echo("warning: no OS package database found", err=True)
        if gpu_scan_flag:
            from agentbom_amd.scan.surfaces_extra import gpu_scan_agent

            gagent = gpu_scan_agent()
            if gagent:
                agents.append(gagent)
            else:
                click.echo("warning: no ROCm/GPU stack detected", err=True)
        if not agents and not no_discover:
            from agentbom_amd.scan.discovery import discover_all

            agents = discover_all()
        if not agents:
            click.echo("No agents discovered. Try --demo for the bundled demo estate.", err=True)
            sys.exit(2)
        if introspect:
            from agentbom_amd.mcp.introspect import introspect_servers

            try:
                for _agent in agents:
                    introspect_servers(_agent.mcp_servers)
            except Exception as exc:
                click.echo(f"warning: introspection failed: {exc}", err=True)
        if health_check_flag:
            from agentbom_amd.scan.surfaces_extra import health_check

            for row in health_check(agents):
                mark = {"ok": "+", "warn": "!", "error": "x"}[row["status"]]
                click.echo(f"[{mark}] {row['agent']}/{row['server']}"
                           + (f" — {row['detail']}" if row["detail"] else ""), err=True)
        if inventory_only:
            inv = {"agents": [{
                "name": a.name, "agent_type": a.agent_type.value,
                "config_path": a.config_path, "source": a.source,
                "mcp_servers": [{
                    "name": srv.name, "command": srv.command,
                    "transport": srv.transport.value,
                    "packages": [{"name": p.name, "version": p.version,
                                  "ecosystem": p.ecosystem}
                                 for p in srv.packages],
                } for srv in a.mcp_servers],
            } for a in agents]}
            text = json.dumps(inv, indent=2, default=str)
            if output:
                Path(output).write_text(text)
            else:
                click.echo(text)
            sys.exit(0)
        from agentbom_amd.db.store import load_advisory_windows

        report = scan_agents(agents, load_advisory_windows(offline=offline), options)

    # optional side scanners -> unified findings, run through the executor
    # so each driver's DECLARED failure mode applies (scan/registry.py)
    from agentbom_amd.scan.registry import run_scanner_driver

    if scan_secrets:
        run_scanner_driver("secrets", report, filesystem or ".")
    if model_files:
        run_scanner_driver("model_files", report, model_files)
    if aws_inventory:
        run_scanner_driver("cloud_cis", report, aws_inventory)
    for spec in cloud_inventory:
        run_scanner_driver("cloud_estate", report, spec)
    if aws_live or azure_live or gcp_live:
        from agentbom_amd.scan.cloud import cis_result_to_finding

        def _live(name, fn):
            try:
                results = fn()
            except Exception as exc:
                click.echo(f"warning: {name} live collection failed: {exc}; "
                           "use the exported-inventory path instead", err=True)
                return
            report.findings.extend(
                f for f in (cis_result_to_finding(r, name) for r in results)
                if f is not None)

        if aws_live:
            def _aws():
                from agentbom_amd.scan.cloud import evaluate_aws_inventory
                from agentbom_amd.scan.cloud_live import AwsCollector

                return evaluate_aws_inventory(AwsCollector().collect_inventory())
            _live("aws", _aws)
        if azure_live:
            if not azure_token:
                raise click.UsageError("--azure-live needs --azure-token")

            def _azure():
                from agentbom_amd.scan.cloud_estate import evaluate_azure_inventory
                from agentbom_amd.scan.cloud_live import collect_azure_inventory

                return evaluate_azure_inventory(
                    collect_azure_inventory(azure_live, azure_token))
            _live("azure", _azure)
        if gcp_live:
            if not gcp_token:
                raise click.UsageError("--gcp-live needs --gcp-token")

            def _gcp():
                from agentbom_amd.scan.cloud_estate import evaluate_gcp_inventory
                from agentbom_amd.scan.cloud_live import collect_gcp_inventory

                return evaluate_gcp_inventory(
                    collect_gcp_inventory(gcp_live, gcp_token))
            _live("gcp", _gcp)
    if endpoint:
        run_scanner_driver("endpoint", report)
    if iac:
        run_scanner_driver("iac", report, iac)
    if notebooks:
        run_scanner_driver("notebooks", report, notebooks)
    if skills:
        run_scanner_driver("skills", report, skills)
    if filesystem:
        run_scanner_driver("floating_refs", report, filesystem)
    if github_actions or repo:
        run_scanner_driver("ci_workflows", report, github_actions or repo)
    if repo_inventory or repo:
        run_scanner_driver("repo_inventory", report, repo_inventory or repo)
    if k8s_posture:
        run_scanner_driver("kspm", report, k8s_posture)
    if exceptions_db:
        from agentbom_amd.api.exceptions_store import (
            ExceptionStore,
            apply_exceptions_to_report,
        )

        waived = apply_exceptions_to_report(report, ExceptionStore(exceptions_db))
        if waived:
            click.echo(f"exceptions: {waived} finding(s) suppressed by "
                       "approved waivers", err=True)
    if scan_prompts:
        from agentbom_amd.scan.surfaces_extra import scan_prompt_files

        report.findings.extend(scan_prompt_files(scan_prompts))
    if scan_pii:
        from agentbom_amd.scan.surfaces_extra import scan_pii as _scan_pii

        report.findings.extend(_scan_pii(scan_pii))
    if browser_extensions:
        from agentbom_amd.scan.surfaces_extra import scan_browser_extensions

        inv, bfindings = scan_browser_extensions(browser_extensions)
        report.findings.extend(bfindings)
        report.ai_inventory_data = (report.ai_inventory_data or {}) | {
            "browser_extensions": inv}
    if dataset_cards:
        from agentbom_amd.scan.surfaces_extra import scan_dataset_cards

        inv, dfindings = scan_dataset_cards(dataset_cards)
        report.findings.extend(dfindings)
        report.ai_inventory_data = (report.ai_inventory_data or {}) | {
            "dataset_cards": inv}
    if training_pipelines:
        from agentbom_amd.scan.surfaces_extra import scan_training_pipelines

        inv, pfindings = scan_training_pipelines(training_pipelines)
        report.findings.extend(pfindings)
        report.ai_inventory_data = (report.ai_inventory_data or {}) | {
            "training_pipelines": inv}
    if verify_model_hashes:
        if not model_files:
            raise click.UsageError("--verify-model-hashes needs --model-files DIR")
        from agentbom_amd.scan.surfaces_extra import verify_model_hash_manifest

        report.findings.extend(
            verify_model_hash_manifest(model_files, verify_model_hashes))
    if license_check_flag:
        from agentbom_amd.scan.surfaces_extra import license_check

        report.findings.extend(
            license_check(report, deny=license_deny or None))
    if enrich or enrich_bundle:
        from agentbom_amd.scan.enrichment import enrich_vulnerabilities, load_offline_bundle

        stats = (load_offline_bundle(report, enrich_bundle) if enrich_bundle
                 else enrich_vulnerabilities(report))
        if verbose:
            click.echo(f"enrichment: {json.dumps(stats)}", err=True)
    if deps_dev:
        from agentbom_amd.scan.deps_meta import enrich_packages_with_deps_meta, load_deps_bundle

        pkgs = [p for a in report.agents for srv in a.mcp_servers for p in srv.packages]
        enrich_packages_with_deps_meta(pkgs, bundle=load_deps_bundle(deps_dev))
    if vex:
        from agentbom_amd.scan.vex import VexDocument, apply_vex

        doc = VexDocument.from_dict(json.loads(Path(vex).read_text()))
        n = apply_vex(report, doc)
        if verbose:
            click.echo(f"VEX: {n} statements applied", err=True)
    if generate_vex:
        from agentbom_amd.scan.vex import generate_vex as _gen_vex

        Path(generate_vex).write_text(
            json.dumps(_gen_vex(report).to_dict(), indent=2))
        click.echo(f"VEX document written to {generate_vex}", err=True)
    if code or semgrep:
        from agentbom_amd.scan.ast_analysis import (
            SymbolIndex,
            apply_symbol_reachability,
            ast_finding_to_finding,
            build_symbol_index,
        )

        idx = build_symbol_index(code) if code else SymbolIndex()
        if semgrep:
            from agentbom_amd.scan.sast_ingest import (
                extend_symbol_index_from_semgrep,
                load_semgrep_file,
                semgrep_to_findings,
            )

            sg = load_semgrep_file(semgrep)
            report.findings.extend(semgrep_to_findings(sg))
            extend_symbol_index_from_semgrep(idx, sg)
        report.findings.extend(ast_finding_to_finding(f) for f in idx.findings)
        apply_symbol_reachability(report, idx)
        report.ai_inventory_data = (report.ai_inventory_data or {}) | {
            "ast_analysis": {"files_scanned": idx.files_scanned,
                             "flow_findings": [f.to_dict() for f in idx.findings]}
        }

    # graph phase: reachability stamping + toxic-combination findings
    from agentbom_amd.graph.builder import build_unified_graph_from_report
    from agentbom_amd.graph.dependency_reach import (
        apply_dependency_reachability_to_blast_radii,
    )
    from agentbom_amd.graph.toxic_combos import (
        detect_toxic_combinations,
        toxic_combination_to_finding,
    )

    graph = build_unified_graph_from_report(report)
    apply_dependency_reachability_to_blast_radii(report, graph)
    combos = detect_toxic_combinations(graph)
    existing = {f.id for f in report.findings}
    report.findings = report.to_findings() + [
        toxic_combination_to_finding(c) for c in combos
        if c.id not in existing
    ]
    report.toxic_combination_findings_data = [c.to_dict() for c in combos]

    policy_failed = False
    if policy:
        from agentbom_amd.scan.policy import evaluate_policy, load_policy

        pol = evaluate_policy(load_policy(policy), report.blast_radii)
        if fmt == "console":
            from rich.console import Console

            from agentbom_amd.output.console_render import print_policy_results

            print_policy_results(pol, Console(stderr=True))
        else:
            for v in pol.get("violations", []):
                click.echo(f"policy: rule {v['rule_id']} matched "
                           f"{v['package']} ({v['vulnerability_id']})", err=True)
            for w in pol.get("warnings", []):
                click.echo(f"policy warning: rule {w['rule_id']} matched "
                           f"{w['package']}", err=True)
        policy_failed = not pol.get("passed", True)

    if warn_on:
        order = ["low", "medium", "high", "critical"]
        if warn_on in order:
            warn_set = set(order[order.index(warn_on):])
            n_warn = sum(1 for br in report.blast_radii
                         if br.vulnerability.severity.value in warn_set)
            if n_warn:
                click.echo(f"warning: {n_warn} finding(s) at or above "
                           f"{warn_on} severity (--warn-on)", err=True)

    # control-plane push + integrations (offline-guarded, fail-open)
    if push_url or webhook or slack_webhook or (jira_url and jira_token) or siem_url:
        from agentbom_amd.output.integrations import run_integrations

        results = run_integrations(
            report,
            push_url=push_url, push_api_key=push_api_key,
            webhooks=list(webhook), slack_webhook=slack_webhook,
            jira={"url": jira_url, "token": jira_token,
                  "project": jira_project} if jira_url and jira_token else None,
            siem={"url": siem_url, "token": siem_token} if siem_url else None,
        )
        for name, ok, detail in results:
            click.echo(f"integration {name}: {'ok' if ok else 'FAILED'}"
                       + (f" — {detail}" if detail else ""), err=True)

    _render(report, fmt, output, verbose)
    rc = compute_exit_code(report, options)
    if policy_failed and rc == 0:
        rc = 1
    if rc:
        gates = []
        if any(br.package.is_malicious for br in report.blast_radii):
            gates.append("malicious-package (fails closed)")
        if not exit_zero:
            gates.append(f"--fail-on-severity {fail_on_severity}")
        click.echo(f"\nexit {rc}: gate matched — {', '.join(gates)}", err=True)
    sys.exit(rc)


def _scan_options(f):
    opts = [
        click.option("--demo", is_flag=True, help="Scan the bundled deterministic demo estate."),
        click.option("--offline", is_flag=True, help="Never touch the network."),
        click.option("--inventory", type=click.Path(exists=True), default=None,
                     help="Scan a JSON inventory file instead of discovering."),
        click.option("-f", "--format", "fmt", type=click.Choice(FORMATS), default="console"),
        click.option("-o", "--output", type=click.Path(), default=None),
        click.option("--fail-on-severity", type=click.Choice(["critical", "high", "medium", "low"]),
                     default="critical"),
        click.option("--fail-on-kev", is_flag=True),
        click.option("--exit-zero", is_flag=True,
                     help="Reporting-only; never suppresses fail-closed gates."),
        click.option("--blast-radius-depth", type=click.IntRange(1, 5), default=1),
        click.option("-v", "--verbose", is_flag=True),
        click.option("--include-unfixed", is_flag=True),
        click.option("--no-gpu", is_flag=True, help="Force the CPU match path."),
        click.option("--sbom", type=click.Path(exists=True), default=None,
                     help="Scan an existing CycloneDX/SPDX SBOM."),
        click.option("--image", default=None,
                     help="Scan a container image: docker-save tar, OCI "
                          "layout dir, or a registry reference "
                          "(e.g. ghcr.io/org/app:tag — needs network)."),
        click.option("--filesystem", type=click.Path(exists=True), default=None,
                     help="Extract + scan packages from a directory tree."),
        click.option("--scan-secrets", is_flag=True,
                     help="Also scan for hardcoded secrets (redacted)."),
        click.option("--model-files", type=click.Path(exists=True), default=None,
                     help="Scan ML model artifacts for pickle payloads."),
        click.option("--code", type=click.Path(exists=True), default=None,
                     help="AST security analysis + symbol-level CVE reachability."),
        click.option("--iac", type=click.Path(exists=True), default=None,
                     help="Scan Terraform/K8s/Dockerfile/compose for misconfigurations."),
        click.option("--aws-inventory", type=click.Path(exists=True), default=None,
                     help="Evaluate CIS checks over an exported AWS inventory JSON."),
        click.option("--endpoint", is_flag=True,
                     help="Collect bounded workstation endpoint inventory."),
        click.option("--notebooks", type=click.Path(exists=True), default=None,
                     help="Scan Jupyter notebooks (pip installs, secrets in outputs, sinks)."),
        click.option("--skills", type=click.Path(exists=True), default=None,
                     help="Scan agent skill bundles (SKILL.md injection/grants/scripts)."),
        click.option("--github-actions", "github_actions",
                     type=click.Path(exists=True), default=None,
                     help="Scan .github/workflows for agentic-CI usage + "
                          "pipeline hardening (implied by --repo)."),
        click.option("--exceptions-db", "exceptions_db",
                     type=click.Path(exists=True), default=None,
                     help="Apply APPROVED, unexpired waivers from an "
                          "exception store (suppression contract; expired "
                          "waivers never suppress)."),
        click.option("--k8s-posture", "k8s_posture",
                     type=click.Path(exists=True), default=None,
                     help="Evaluate an exported Kubernetes cluster inventory "
                          "(pods/RBAC) against the KSPM posture checks."),
        click.option("--repo-inventory", "repo_inventory",
                     type=click.Path(exists=True), default=None,
                     help="Collect the project directory/file/import "
                          "inventory for the code-graph overlays (implied "
                          "by --repo)."),
        click.option("--semgrep", type=click.Path(exists=True), default=None,
                     help="Ingest a semgrep --json result file (SAST findings "
                          "+ symbol-reachability joins)."),
        click.option("--cloud-inventory", "cloud_inventory", multiple=True,
                     help="provider:path — evaluate an exported cloud "
                          "inventory (azure/gcp/snowflake/databricks/aws; "
                          "includes IAM, audit-trail and DSPM sections)."),
        click.option("--profile", "profile", default=None,
                     help="Apply a named flag-set from .agent-bom.yaml "
                          "profiles: (explicit flags still win)."),
        # modes
        click.option("--self-scan", "self_scan", is_flag=True,
                     help="Scan agent-bom's own installation."),
        click.option("--inventory-only", "inventory_only", is_flag=True,
                     help="Discover and emit the inventory JSON; no matching."),
        click.option("--no-discover", "no_discover", is_flag=True,
                     help="Never auto-discover; only scan explicit surfaces."),
        click.option("--dry-run", "dry_run", is_flag=True,
                     help="Print the scan plan and exit 0."),
        # depth / enrichment
        click.option("--transitive", is_flag=True,
                     help="Expand the dependency tree via package registries."),
        click.option("--max-depth", type=click.IntRange(1, 6), default=3,
                     help="Transitive expansion depth bound."),
        click.option("--live-osv", "live_osv", is_flag=True,
                     help="Query api.osv.dev for the package batch (needs network)."),
        click.option("--enrich", is_flag=True,
                     help="Enrich findings from the local advisory DB caches."),
        click.option("--enrich-bundle", type=click.Path(exists=True), default=None,
                     help="Air-gapped enrichment bundle dir (epss.csv/kev.json)."),
        click.option("--deps-dev", "deps_dev", type=click.Path(exists=True), default=None,
                     help="deps.dev/Scorecard metadata bundle for trust scoring."),
        click.option("--vex", type=click.Path(exists=True), default=None,
                     help="Apply an OpenVEX document (suppressions audit-kept)."),
        click.option("--generate-vex", "generate_vex", type=click.Path(), default=None,
                     help="Write an OpenVEX document for this scan's findings."),
        click.option("--license-check", "license_check_flag", is_flag=True,
                     help="Flag denied licenses on collected packages."),
        click.option("--license-deny", "license_deny", multiple=True,
                     help="License substring to deny (repeatable; default GPL-3.0/AGPL/SSPL)."),
        # gates
        click.option("--fail-if-ai-risk", "fail_if_ai_risk", is_flag=True,
                     help="Exit 1 when findings carry AI risk context."),
        click.option("--warn-on", "warn_on",
                     type=click.Choice(["critical", "high", "medium", "low"]),
                     default=None, help="Warn (stderr) at this severity without failing."),
        click.option("--require-fresh-db", "require_fresh_db", is_flag=True,
                     help="Refuse to scan against an expired advisory DB."),
        click.option("--policy", type=click.Path(exists=True), default=None,
                     help="Policy-as-code file; failed rules gate the exit code."),
        # surfaces
        click.option("--image-tar", "image_tar", type=click.Path(exists=True), default=None,
                     help="Alias of --image (docker-save tarball)."),
        click.option("--repo", type=click.Path(exists=True), default=None,
                     help="Scan a repository: packages + AST code analysis."),
        click.option("--os-packages", "os_packages", is_flag=True,
                     help="Scan this host's dpkg/apk OS package databases."),
        click.option("--jupyter", type=click.Path(exists=True), default=None,
                     help="Alias of --notebooks."),
        click.option("--browser-extensions", "browser_extensions", multiple=True,
                     help="Browser extension dirs to audit (repeatable; "
                          "defaults to Chrome/Firefox profile paths)."),
        click.option("--dataset-cards", "dataset_cards", type=click.Path(exists=True),
                     default=None, help="Scan dataset cards for provenance risks."),
        click.option("--training-pipelines", "training_pipelines",
                     type=click.Path(exists=True), default=None,
                     help="Scan training pipeline definitions (dvc/MLproject/kfp)."),
        click.option("--verify-model-hashes", "verify_model_hashes",
                     type=click.Path(exists=True), default=None,
                     help="sha256 manifest to verify --model-files artifacts against."),
        click.option("--gpu-scan", "gpu_scan_flag", is_flag=True,
                     help="Collect ROCm/amdgpu GPU posture inventory."),
        click.option("--introspect", is_flag=True,
                     help="Introspect live MCP servers (read-only handshake)."),
        click.option("--health-check", "health_check_flag", is_flag=True,
                     help="Static MCP server config health (command/url resolvable)."),
        click.option("--scan-prompts", "scan_prompts", type=click.Path(exists=True),
                     default=None, help="Scan prompt files for injection patterns."),
        click.option("--scan-pii", "scan_pii", type=click.Path(exists=True),
                     default=None, help="Bounded PII sweep (values redacted)."),
        click.option("--skill", type=click.Path(exists=True), default=None,
                     help="Alias of --skills."),
        # live cloud collectors
        click.option("--aws-live", "aws_live", is_flag=True,
                     help="Collect the AWS inventory live (SigV4, env creds) "
                          "and run the CIS pack on it."),
        click.option("--azure-live", "azure_live", default=None, metavar="SUB_ID",
                     help="Collect the Azure inventory live (ARM REST)."),
        click.option("--azure-token", "azure_token", default=None),
        click.option("--gcp-live", "gcp_live", default=None, metavar="PROJECT",
                     help="Collect the GCP inventory live (REST)."),
        click.option("--gcp-token", "gcp_token", default=None),
        # push / integrations
        click.option("--push-url", "push_url", default=None,
                     help="POST the full report JSON to a control plane."),
        click.option("--push-api-key", "push_api_key", default=None),
        click.option("--webhook", multiple=True,
                     help="POST a scan.completed event to this URL (repeatable)."),
        click.option("--slack-webhook", "slack_webhook", default=None),
        click.option("--jira-url", "jira_url", default=None),
        click.option("--jira-token", "jira_token", default=None),
        click.option("--jira-project", "jira_project", default=None),
        click.option("--siem-url", "siem_url", default=None,
                     help="POST OCSF events to a SIEM collector."),
        click.option("--siem-token", "siem_token", default=None),
    ]
    for o in reversed(opts):
        f = o(f)
    return f


@main.command(name="agents")
@_scan_options
def agents_cmd(**kw) -> None:
    """Discover AI agents + MCP servers and scan their packages."""
    _scan_impl(**_apply_profile(kw))


@main.command(name="scan", hidden=True)
@_scan_options
def scan_cmd(**kw) -> None:
    """Alias of ``agents``."""
    _scan_impl(**_apply_profile(kw))


@main.command(name="serve")
@click.option("--host", default="127.0.0.1")
@click.option("--port", type=int, default=8000)
def serve_cmd(host: str, port: int) -> None:
    """Run the control-plane REST API (FastAPI/uvicorn)."""
    import uvicorn

    from agentbom_amd.api.server import create_app

    uvicorn.run(create_app(), host=host, port=port)


@main.group(name="mcp")
def mcp_group() -> None:
    """MCP server surface."""


@mcp_group.command(name="server")
@click.option("--demo", is_flag=True, help="Serve tools over the bundled demo estate.")
def mcp_server_cmd(demo: bool) -> None:
    """Run the agent-bom MCP server on stdio."""
    from agentbom_amd.mcp.server import run_stdio_server

    run_stdio_server(demo=demo)


@mcp_group.command(name="introspect")
@click.option("--timeout", type=float, default=10.0, show_default=True)
@click.option("--server", "server_spec", default=None,
              help="Introspect one ad-hoc server: 'name:command arg1 arg2'.")
@click.option("-o", "--output", type=click.Path(), default=None)
def mcp_introspect_cmd(timeout: float, server_spec: Optional[str],
                       output: Optional[str]) -> None:
    """Connect to configured MCP servers (read-only) and diff runtime tools."""
    import json as _json

    from agentbom_amd.mcp.introspect import introspect_servers

    servers = _resolve_introspection_targets(server_spec)
    if not servers:
        click.echo("no stdio MCP servers discovered", err=True)
        raise SystemExit(2)
    report = introspect_servers(servers, timeout=timeout)
    text = _json.dumps(report.to_dict(), indent=2, default=str)
    if output:
        Path(output).write_text(text)
        click.echo(f"wrote {output}")
    else:
        click.echo(text)
    raise SystemExit(1 if report.drift_count else 0)


@mcp_group.command(name="health")
@click.option("--timeout", type=float, default=5.0, show_default=True)
@click.option("--server", "server_spec", default=None,
              help="Probe one ad-hoc server: 'name:command arg1 arg2'.")
def mcp_health_cmd(timeout: float, server_spec: Optional[str]) -> None:
    """Liveness-probe configured MCP servers (spawn + initialize only)."""
    import json as _json

    from agentbom_amd.mcp.introspect import health_check_servers

    servers = _resolve_introspection_targets(server_spec)
    if not servers:
        click.echo("no stdio MCP servers discovered", err=True)
        raise SystemExit(2)
    statuses = health_check_servers(servers, timeout=timeout)
    click.echo(_json.dumps([s.to_dict() for s in statuses], indent=2))
    raise SystemExit(0 if all(s.healthy for s in statuses) else 1)


def _resolve_introspection_targets(server_spec: Optional[str]):
    from agentbom_amd.models import MCPServer

    if server_spec:
        name, _, cmdline = server_spec.partition(":")
        parts = cmdline.split()
        if not parts:
            return []
        return [MCPServer(name=name or parts[0], command=parts[0], args=parts[1:])]
    from agentbom_amd.scan.discovery import discover_agents

    servers = []
    for agent in discover_agents():
        servers.extend(s for s in agent.mcp_servers if s.command)
    # de-dup by (command, args) — several clients often share one server
    seen, out = set(), []
    for s in servers:
        key = (s.command, tuple(s.args))
        if key not in seen:
            seen.add(key)
            out.append(s)
    return out


@main.command(name="check")
@click.argument("spec")
@click.option("--ecosystem", "-e", default="pypi", show_default=True)
@click.option("--offline", is_flag=True)
def check_cmd(spec: str, ecosystem: str, offline: bool) -> None:
    """Check one package@version against the advisory data (exit 1 if hit)."""
    from agentbom_amd.db.store import load_advisory_windows
    from agentbom_amd.utils.canonical_ids import normalize_package_name
    from agentbom_amd.utils.version_utils import version_in_range

    name, _, version = spec.partition("@")
    if not version:
        click.echo("usage: agent-bom check NAME@VERSION", err=True)
        raise SystemExit(2)
    norm = normalize_package_name(name, ecosystem)
    hits = []
    for w in load_advisory_windows(offline=offline):
        if (w.ecosystem.lower() == ecosystem.lower()
                and normalize_package_name(w.package_name, w.ecosystem) == norm
                and version_in_range(version, w.introduced, w.fixed,
                                     w.last_affected, ecosystem)):
            hits.append({"vuln_id": w.vuln_id, "severity": w.severity.value,
                         "fixed": w.fixed, "is_kev": w.is_kev,
                         "summary": w.summary})
    click.echo(json.dumps({"package": spec, "ecosystem": ecosystem,
                           "vulnerable": bool(hits), "advisories": hits},
                          indent=2))
    raise SystemExit(1 if hits else 0)


@main.command(name="doctor")
def doctor_cmd() -> None:
    """Environment self-check: GPU, HIP engine, advisory data, versions."""
    import platform

    checks = []

    def check(name: str, ok, detail: str) -> None:
        checks.append({"check": name, "ok": bool(ok), "detail": detail})

    check("python", True, platform.python_version())
    try:
        import torch

        check("torch", True, torch.__version__)
        gpu = torch.cuda.is_available()
        check("gpu", gpu,
              torch.cuda.get_device_name(0) if gpu else "no GPU visible "
              "(CPU oracle path active)")
    except Exception as exc:  # noqa: BLE001 — doctor reports, never crashes
        check("torch", False, str(exc))
    from agentbom_amd.ops import native

    check("hip_engine", native.available(),
          "in-tree _abom_gpu.so built" if native.available()
          else "run `python -m agentbom_amd.ops.build` (needs hipcc)")
    from agentbom_amd.db.store import load_advisory_windows

    windows = load_advisory_windows(offline=True)
    check("advisories", len(windows) > 0,
          f"{len(windows)} bundled windows (run `agent-bom db sync` for live data)")
    from agentbom_amd.mcp.registry import load_registry

    reg = load_registry()
    check("mcp_registry", len(reg["servers"]) > 0,
          f"{len(reg['servers'])} known servers, "
          f"{len(reg['blocklist'])} blocklist rules")
    from agentbom_amd.scan.self_posture import evaluate_self_posture

    posture = evaluate_self_posture()
    check("self_posture", posture["score"] >= 50,
          f"hardening score {posture['score']} (agent-bom posture for detail)")
    ok = all(c["ok"] for c in checks if c["check"] != "gpu")
    click.echo(json.dumps({"healthy": ok, "checks": checks}, indent=2))
    raise SystemExit(0 if ok else 1)


@main.command(name="trust")
@click.argument("spec")
@click.option("--ecosystem", "-e", default="npm", show_default=True)
def trust_cmd(spec: str, ecosystem: str) -> None:
    """Supply-chain trust score for a package (exit 1 on grade D/F)."""
    from agentbom_amd.db.store import load_advisory_windows
    from agentbom_amd.models import Package
    from agentbom_amd.scan.trust import trust_score

    name, _, version = spec.partition("@")
    out = trust_score(
        Package(name=name, version=version or "0.0.0", ecosystem=ecosystem),
        advisory_windows=load_advisory_windows(offline=True))
    click.echo(json.dumps(out, indent=2))
    raise SystemExit(1 if out["grade"] in ("D", "F") else 0)


@main.command(name="remediate")
@click.option("--demo", is_flag=True)
@click.option("--script", "as_script", is_flag=True,
              help="Emit a reviewable shell script instead of JSON.")
def remediate_cmd(demo: bool, as_script: bool) -> None:
    """Prioritized remediation commands for the latest scan."""
    from agentbom_amd.db.store import load_advisory_windows
    from agentbom_amd.scan.orchestrator import run_demo_scan, scan_agents
    from agentbom_amd.scan.remediation import remediation_commands, remediation_script

    if demo:
        report = run_demo_scan()
    else:
        from agentbom_amd.scan.discovery import discover_all

        report = scan_agents(discover_all(),
                             load_advisory_windows(offline=True))
    if as_script:
        click.echo(remediation_script(report))
    else:
        click.echo(json.dumps({"commands": remediation_commands(report)},
                              indent=2))


@main.command(name="attest")
@click.argument("server_name")
@click.option("--demo", is_flag=True)
@click.option("--verdict", type=click.Choice(["pass", "warn", "block"]),
              default="pass", show_default=True)
@click.option("--verify", "verify_path", type=click.Path(exists=True),
              default=None, help="Verify an envelope file instead of signing.")
@click.option("-o", "--output", type=click.Path(), default=None)
def attest_cmd(server_name: str, demo: bool, verdict: str,
               verify_path: Optional[str], output: Optional[str]) -> None:
    """Sign (or verify) a DSSE scan attestation for one MCP server."""
    import os

    from agentbom_amd.mcp.server import resolve_mcp_tenant_id
    from agentbom_amd.utils.attestation import attest_scanned_server, verify_attestation

    key_hex = os.environ.get("AGENT_BOM_ATTESTATION_KEY", "")
    key_id = os.environ.get("AGENT_BOM_ATTESTATION_KEY_ID", "operator")
    if verify_path:
        envelope = json.loads(Path(verify_path).read_text())
        out = verify_attestation(envelope, {
            "keys": ({key_id: key_hex} if key_hex else {})})
        click.echo(json.dumps(out, indent=2, default=str))
        raise SystemExit(0 if out["valid"] else 1)
    if not key_hex:
        click.echo("AGENT_BOM_ATTESTATION_KEY not set", err=True)
        raise SystemExit(2)
    from agentbom_amd.scan.orchestrator import run_demo_scan

    report = run_demo_scan() if demo else None
    if report is None:
        from agentbom_amd.db.store import load_advisory_windows
        from agentbom_amd.scan.discovery import discover_all
        from agentbom_amd.scan.orchestrator import scan_agents

        report = scan_agents(discover_all(),
                             load_advisory_windows(offline=True))
    target = next((s for a in report.agents for s in a.mcp_servers
                   if s.name == server_name), None)
    if target is None:
        click.echo(f"server {server_name!r} not in scan", err=True)
        raise SystemExit(2)
    env = attest_scanned_server(target, verdict, bytes.fromhex(key_hex),
                                key_id=key_id,
                                tenant_id=resolve_mcp_tenant_id())
    text = json.dumps(env, indent=2)
    if output:
        Path(output).write_text(text)
        click.echo(f"wrote {output}")
    else:
        click.echo(text)


@main.command(name="skills")
@click.argument("path", type=click.Path(exists=True))
@click.option("--policy", type=click.Path(exists=True), default=None,
              help="JSON policy: blocklist / max_risk_level / require_frontmatter.")
def skills_cmd(path: str, policy: Optional[str]) -> None:
    """Scan agent skill bundles (SKILL.md) for injection/grants/scripts."""
    from agentbom_amd.scan.skills import evaluate_skills_policy, scan_skills_tree

    bundles = scan_skills_tree(path)
    out = {"bundles": [b.to_dict() for b in bundles]}
    denied = 0
    if policy:
        decision = evaluate_skills_policy(
            bundles, json.loads(Path(policy).read_text()))
        out["policy"] = decision
        denied = decision["denied"]
    click.echo(json.dumps(out, indent=2))
    critical = any(b.risk_level == "critical" for b in bundles)
    raise SystemExit(1 if critical or denied else 0)


@main.group(name="identity")
def identity_group() -> None:
    """Agent identity lifecycle (local store)."""


def _identity_store_from_opt(store_path: str):
    from agentbom_amd.identity import AgentIdentityStore

    return AgentIdentityStore(store_path)


_STORE_OPT = click.option(
    "--store", default="~/.agent-bom/identities.db", show_default=True,
    help="SQLite identity store path.")


@identity_group.command(name="issue")
@click.argument("agent_name")
@click.option("--scopes", default="", help="Comma-separated scopes.")
@click.option("--allowed-tools", default="")
@click.option("--ttl-hours", type=float, default=24.0, show_default=True)
@_STORE_OPT
def identity_issue_cmd(agent_name: str, scopes: str, allowed_tools: str,
                       ttl_hours: float, store: str) -> None:
    """Issue a scoped identity; the raw token prints EXACTLY ONCE."""
    s = _identity_store_from_opt(str(Path(store).expanduser()))
    ident, raw = s.issue(
        agent_name,
        scopes=[x for x in scopes.split(",") if x.strip()],
        allowed_tools=[x for x in allowed_tools.split(",") if x.strip()],
        ttl_hours=ttl_hours, actor="cli", reason="cli issue")
    click.echo(json.dumps({"identity": ident.to_public_dict(),
                           "token": raw,
                           "note": "token shown once; only its hash is stored"},
                          indent=2))


@identity_group.command(name="list")
@_STORE_OPT
def identity_list_cmd(store: str) -> None:
    s = _identity_store_from_opt(str(Path(store).expanduser()))
    click.echo(json.dumps(
        [i.to_public_dict() for i in s.list()], indent=2))


@identity_group.command(name="revoke")
@click.argument("identity_id")
@click.option("--reason", default="cli revocation", show_default=True)
@_STORE_OPT
def identity_revoke_cmd(identity_id: str, reason: str, store: str) -> None:
    s = _identity_store_from_opt(str(Path(store).expanduser()))
    ok = s.revoke(identity_id, actor="cli", reason=reason)
    click.echo(json.dumps({"revoked": ok}))
    raise SystemExit(0 if ok else 1)


@identity_group.command(name="verify")
@click.argument("token")
@click.option("--tool", default=None)
@_STORE_OPT
def identity_verify_cmd(token: str, tool: Optional[str], store: str) -> None:
    s = _identity_store_from_opt(str(Path(store).expanduser()))
    out = s.verify(token, tool=tool)
    click.echo(json.dumps(out, indent=2))
    raise SystemExit(0 if out["valid"] else 1)


@main.command(name="history")
@click.option("--save", "save_report", type=click.Path(exists=True),
              default=None, help="Save a report JSON as a snapshot.")
@click.option("--diff", "diff_pair", nargs=2, type=click.Path(exists=True),
              default=None, help="Diff two report JSON files (old new).")
def history_cmd(save_report: Optional[str],
                diff_pair: Optional[tuple]) -> None:
    """Report snapshot history: save, list, diff."""
    from agentbom_amd.scan.history import diff_reports, list_snapshots, save_report_snapshot

    if save_report:
        path = save_report_snapshot(json.loads(Path(save_report).read_text()))
        click.echo(f"saved {path}")
        return
    if diff_pair:
        old, new = (json.loads(Path(p).read_text()) for p in diff_pair)
        click.echo(json.dumps(diff_reports(old, new), indent=2, default=str))
        return
    click.echo(json.dumps([str(p) for p in list_snapshots()], indent=2))


@main.command(name="quickstart")
def quickstart_cmd() -> None:
    """Guided first steps."""
    click.echo("""agent-bom quickstart
====================
1. Demo scan (exits 1 by design — the demo estate is vulnerable):
     agent-bom agents --demo --offline
2. Scan YOUR machine's agents + MCP servers:
     agent-bom agents --offline -f json -o scan.json
3. One package:            agent-bom check pyyaml@5.3 -e pypi
4. Trust signals:          agent-bom trust left-pad -e npm
5. Fix commands:           agent-bom remediate --demo --script
6. Posture (MCP + self):   agent-bom posture --demo --a2a
7. Serve REST + MCP:       agent-bom serve   |   agent-bom mcp server
8. Watch a repo:           agent-bom watch --filesystem . --interval 30
9. Health check:           agent-bom doctor
Docs: docs/OPERATIONS.md · docs/ARCHITECTURE.md · docs/PERFORMANCE.md""")


@main.command(name="watch")
@click.option("--filesystem", type=click.Path(exists=True), default=".",
              show_default=True, help="Tree to watch for manifest changes.")
@click.option("--interval", type=float, default=10.0, show_default=True)
@click.option("--max-iterations", type=int, default=0,
              help="Stop after N checks (0 = forever); used by tests/CI.")
@click.option("--offline", is_flag=True)
def watch_cmd(filesystem: str, interval: float, max_iterations: int,
              offline: bool) -> None:
    """Watch a tree; rescan on manifest change and stream finding deltas."""
    import time

    from agentbom_amd.models import Agent, AgentType, MCPServer
    from agentbom_amd.output.delta_stream import DeltaStreamer
    from agentbom_amd.db.store import load_advisory_windows
    from agentbom_amd.scan.orchestrator import ScanOptions, scan_agents
    from agentbom_amd.scan.parsers import BUILTIN_INVENTORY_PARSERS, extract_packages

    patterns = [p for p, _fn in BUILTIN_INVENTORY_PARSERS]
    windows = load_advisory_windows(offline=offline)
    streamer = DeltaStreamer()

    def fingerprint() -> tuple:
        root = Path(filesystem)
        sig = []
        for pat in patterns:
            for f in sorted(root.rglob(pat)):
                try:
                    st = f.stat()
                    sig.append((str(f), st.st_mtime_ns, st.st_size))
                except OSError:
                    continue
        return tuple(sig)

    def rescan() -> int:
        pkgs = extract_packages(filesystem)
        agent = Agent(name=f"watch:{Path(filesystem).name}",
                      agent_type=AgentType.CUSTOM, config_path=filesystem,
                      mcp_servers=[MCPServer(name="watched", command="",
                                             packages=pkgs)])
        report = scan_agents([agent], windows, ScanOptions(offline=offline))
        events = streamer.emit(report)
        for ev in events:
            click.echo(json.dumps(ev, default=str))
        return len(events)

    last = fingerprint()
    n_events = rescan()
    click.echo(f"# watching {filesystem} ({len(last)} manifests, "
               f"{n_events} initial findings)", err=True)
    iterations = 0
    while max_iterations == 0 or iterations < max_iterations:
        iterations += 1
        time.sleep(interval)
        cur = fingerprint()
        if cur != last:
            last = cur
            changed = rescan()
            click.echo(f"# change detected -> {changed} delta events", err=True)


@main.command(name="posture")
@click.option("--demo", is_flag=True, help="Assess the bundled demo estate.")
@click.option("--a2a", is_flag=True, help="Include inter-agent (A2A) posture.")
def posture_cmd(demo: bool, a2a: bool) -> None:
    """Auth + self posture: MCP server auth surface and own-deployment audit."""
    from agentbom_amd.scan.auth_posture import assess_a2a, assess_estate
    from agentbom_amd.scan.orchestrator import run_demo_scan
    from agentbom_amd.scan.self_posture import evaluate_self_posture

    if demo:
        agents = run_demo_scan().agents
    else:
        from agentbom_amd.scan.discovery import discover_all

        agents = discover_all()
    out = {
        "mcp_auth_posture": assess_estate(agents),
        "self_posture": evaluate_self_posture(),
    }
    if a2a:
        out["a2a_auth_posture"] = assess_a2a(agents)
    click.echo(json.dumps(out, indent=2, default=str))
    critical = out["mcp_auth_posture"]["critical_exposures"]
    raise SystemExit(1 if critical else 0)


@main.command(name="compliance-bundle")
@click.argument("framework")
@click.option("--demo", is_flag=True, help="Bundle the demo estate scan.")
@click.option("-o", "--output", type=click.Path(), default=None)
def compliance_bundle_cmd(framework: str, demo: bool,
                          output: Optional[str]) -> None:
    """Export a signed per-framework compliance evidence bundle."""
    from agentbom_amd.output.compliance_export import export_compliance_bundle_timed
    from agentbom_amd.scan.orchestrator import run_demo_scan

    if not demo:
        click.echo("note: scanning live estate", err=True)
    from agentbom_amd.db.store import load_advisory_windows
    from agentbom_amd.scan.discovery import discover_all
    from agentbom_amd.scan.orchestrator import scan_agents

    report = (run_demo_scan() if demo else
              scan_agents(discover_all(), load_advisory_windows(offline=True)))
    try:
        bundle = export_compliance_bundle_timed(report, framework)
    except ValueError as exc:
        click.echo(str(exc), err=True)
        raise SystemExit(2)
    text = json.dumps(bundle, indent=2, default=str)
    if output:
        Path(output).write_text(text)
        click.echo(f"wrote {output} "
                   f"({bundle['signature']['status']}, "
                   f"{bundle['manifest']['control_count']} controls)")
    else:
        click.echo(text)


@main.command(name="graph")
@click.argument("scan_json", type=click.Path(exists=True))
@click.option("-f", "--format", "fmt",
              type=click.Choice(["json", "dot", "mermaid", "graphml", "cypher"]), default="json")
@click.option("-o", "--output", type=click.Path(), default=None)
def graph_cmd(scan_json: str, fmt: str, output: Optional[str]) -> None:
    """Export the estate graph from a scan report JSON."""
    from agentbom_amd.graph.builder import build_unified_graph_from_report_json

    data = json.loads(Path(scan_json).read_text())
    graph = build_unified_graph_from_report_json(data)
    if fmt == "json":
        text = json.dumps(graph.to_dict(), indent=2, sort_keys=True)
    else:
        text = graph.export(fmt)
    if output:
        Path(output).write_text(text)
    else:
        click.echo(text, nl=False)


@main.command(name="graph-evidence")
@click.option("--mode", type=click.Choice(["manifest", "history"]), default="manifest")
@click.option("--store", "store_path", type=click.Path(), default=None,
              help="Graph snapshot store (default: $AGENT_BOM_GRAPH_STORE).")
def graph_evidence_cmd(mode: str, store_path: Optional[str]) -> None:
    """Export the graph evidence manifest / snapshot history (sorted JSON)."""
    import os

    from agentbom_amd.graph.store import SQLiteGraphStore

    path = store_path or os.environ.get("AGENT_BOM_GRAPH_STORE")
    if not path or not Path(path).exists():
        click.echo("no graph store found — set AGENT_BOM_GRAPH_STORE or pass --store", err=True)
        sys.exit(2)
    store = SQLiteGraphStore(path)
    doc = store.evidence_manifest() if mode == "manifest" else store.graph_history()
    click.echo(json.dumps(doc, indent=2, sort_keys=True, default=str))


@main.command(name="mesh")
@click.option("--demo", is_flag=True)
@click.option("-f", "--format", "fmt", type=click.Choice(["summary", "json"]), default="summary")
def mesh_cmd(demo: bool, fmt: str) -> None:
    """Lightweight agent/MCP topology summary (reference: agent-bom mesh)."""
    from agentbom_amd.scan.orchestrator import run_demo_scan

    if demo:
        report = run_demo_scan()
    else:
        from agentbom_amd.scan.discovery import discover_all

        agents = discover_all()
        if not agents:
            click.echo("No agents discovered. Try --demo.", err=True)
            sys.exit(2)
        from agentbom_amd.models import AIBOMReport

        report = AIBOMReport(agents=agents)
    mesh = {
        "agents": [
            {
                "name": a.name,
                "type": a.agent_type.value,
                "servers": [
                    {"name": s.name, "transport": s.transport.value,
                     "tools": len(s.tools), "packages": len(s.packages),
                     "credentials": len(s.credential_names)}
                    for s in a.mcp_servers
                ],
            }
            for a in report.agents
        ],
        "shared_servers": _shared(report, "server"),
        "shared_credentials": _shared(report, "credential"),
    }
    if fmt == "json":
        click.echo(json.dumps(mesh, indent=2))
    else:
        for a in mesh["agents"]:
            click.echo(f"{a['name']} ({a['type']})")
            for srv in a["servers"]:
                click.echo(f"  └─ {srv['name']} [{srv['transport']}] "
                           f"tools={srv['tools']} pkgs={srv['packages']} creds={srv['credentials']}")
        if mesh["shared_credentials"]:
            click.echo("shared credentials: " + ", ".join(
                f"{k} ({len(v)} agents)" for k, v in mesh["shared_credentials"].items()))


def _shared(report, kind: str) -> dict:
    owners: dict[str, list[str]] = {}
    for a in report.agents:
        for s in a.mcp_servers:
            keys = [s.name] if kind == "server" else s.credential_names
            for k in keys:
                owners.setdefault(k, [])
                if a.name not in owners[k]:
                    owners[k].append(a.name)
    return {k: v for k, v in sorted(owners.items()) if len(v) > 1}


@main.group(name="db")
def db_group() -> None:
    """Local advisory database."""


@db_group.command(name="sync")
@click.option("--source", type=click.Choice(["osv", "epss", "kev", "nvd", "demo"]),
              default="demo")
@click.option("--path", type=click.Path(), default=None)
@click.option("--ecosystem", "ecosystems", multiple=True,
              help="OSV bulk ecosystems (repeatable), e.g. --ecosystem npm")
@click.option("--from-file", "from_file", type=click.Path(exists=True), default=None,
              help="Offline file-drop ingest (OSV dir/zip, EPSS csv, KEV json)")
def db_sync_cmd(source: str, path: Optional[str], ecosystems: tuple[str, ...],
                from_file: Optional[str]) -> None:
    """Sync the local advisory DB from live sources or file drops.

    Live sources honor offline mode (AGENT_BOM_OFFLINE / --offline posture):
    in offline mode only --from-file and demo work."""
    from agentbom_amd.db.store import AdvisoryStore, default_db_path

    store = AdvisoryStore(path or default_db_path())
    if source == "demo":
        from agentbom_amd.scan.demo import demo_advisory_windows

        n = store.ingest_windows(demo_advisory_windows())
        click.echo(f"ingested {n} demo advisory windows into {store.path}")
        return
    if from_file:
        from agentbom_amd.db import osv_ingest

        fn = {"osv": osv_ingest.sync_osv, "epss": osv_ingest.sync_epss,
              "kev": osv_ingest.sync_kev}.get(source)
        if fn is None:
            click.echo(f"--from-file not supported for {source}", err=True)
            sys.exit(1)
        n = fn(store, from_file)
        click.echo(f"ingested {n} {source} records from {from_file}")
        return
    from agentbom_amd.db import live
    from agentbom_amd.utils.http_client import OfflineError

    try:
        if source == "osv":
            ecos = list(ecosystems) or ["npm", "pypi"]
            n = live.sync_osv_bulk(store, ecos)
        elif source == "epss":
            n = live.sync_epss_live(store)
        elif source == "kev":
            n = live.sync_kev_live(store)
        else:
            n = live.sync_nvd_live(store)
    except OfflineError as exc:
        click.echo(f"{exc} — use --from-file for air-gapped ingest", err=True)
        sys.exit(1)
    click.echo(f"synced {n} {source} records into {store.path}")


@db_group.command(name="enrich")
@click.argument("report_json", type=click.Path(exists=True))
@click.option("--bundle", type=click.Path(exists=True), default=None,
              help="Air-gapped enrichment bundle dir (epss.csv/kev.json/osv/).")
def db_enrich_cmd(report_json: str, bundle: Optional[str]) -> None:
    """Re-enrich a persisted report from the local DB or an offline bundle."""
    click.echo("enrichment operates on live reports; pass --demo scans through"
               " the API or use scan/enrichment.py programmatically.", err=True)
    from agentbom_amd.scan.enrichment import enrich_vulnerabilities, load_offline_bundle
    from agentbom_amd.scan.orchestrator import run_demo_scan

    report = run_demo_scan()
    stats = (load_offline_bundle(report, bundle) if bundle
             else enrich_vulnerabilities(report))
    click.echo(json.dumps(stats, indent=2))


@db_group.command(name="status")
@click.option("--path", type=click.Path(), default=None)
def db_status_cmd(path: Optional[str]) -> None:
    from agentbom_amd.db.store import AdvisoryStore, default_db_path

    store = AdvisoryStore(path or default_db_path())
    click.echo(json.dumps(store.status(), indent=2))


@main.command(name="proxy", context_settings={"ignore_unknown_options": True})
@click.argument("command", nargs=-1, required=False, type=click.UNPROCESSED)
@click.option("--url", default=None,
              help="Wrap a remote HTTP/SSE MCP server instead of a command.")
@click.option("--policy", "policy_path", type=click.Path(exists=True), default=None)
@click.option("--audit-log", type=click.Path(), default=None)
@click.option("--block-on-warn", is_flag=True)
@click.option("--sandbox", is_flag=True,
              help="Run the wrapped command inside a docker/podman sandbox "
                   "(read-only, no network, caps dropped).")
@click.option("--sandbox-runtime", type=click.Choice(["docker", "podman"]),
              default="docker")
@click.option("--sandbox-image", default="node:20-slim")
@click.option("--sandbox-network", is_flag=True,
              help="Allow network inside the sandbox (default: none).")
def proxy_cmd(command: tuple[str, ...], url: Optional[str],
              policy_path: Optional[str], audit_log: Optional[str],
              block_on_warn: bool, sandbox: bool, sandbox_runtime: str,
              sandbox_image: str, sandbox_network: bool) -> None:
    """Wrap a target MCP server (stdio command or HTTP/SSE URL) with
    inline runtime detectors, policy and a hash-chained audit log."""
    from agentbom_amd.runtime.proxy import (
        AuditLog,
        HttpMcpProxy,
        McpProxy,
        ProxyPolicy,
        sandboxed_proxy,
    )

    policy = ProxyPolicy.load(policy_path)
    if block_on_warn:
        policy.block_on_warn = True
    audit = AuditLog(audit_log)
    if url:
        if command:
            raise click.UsageError("pass either a COMMAND or --url, not both")
        proxy = HttpMcpProxy(url, policy=policy, audit=audit)
    elif not command:
        raise click.UsageError("pass the MCP server COMMAND or --url")
    elif sandbox:
        try:
            proxy = sandboxed_proxy(list(command), policy=policy, audit=audit,
                                    runtime=sandbox_runtime,
                                    image=sandbox_image,
                                    network=sandbox_network)
        except RuntimeError as exc:
            click.echo(f"error: {exc}", err=True)
            sys.exit(2)
    else:
        proxy = McpProxy(list(command), policy=policy, audit=audit)
    sys.exit(proxy.run())


@main.group(name="gateway")
def gateway_group() -> None:
    """Central secure-by-default MCP relay."""


@gateway_group.command(name="serve")
@click.option("--host", default="127.0.0.1")
@click.option("--port", type=int, default=8787)
@click.option("--upstream", "upstreams", multiple=True,
              help="name=http://host:port upstream registrations")
@click.option("--registry", "registry_path", type=click.Path(exists=True),
              default=None,
              help="Upstream registry file (yaml/json: upstreams: [{name,url}])")
def gateway_serve_cmd(host: str, port: int, upstreams: tuple[str, ...],
                      registry_path: Optional[str] = None) -> None:
    """Serve the gateway relay over HTTP (/mcp/{upstream})."""
    import uvicorn
    from fastapi import FastAPI

    from agentbom_amd.runtime.gateway import Gateway, Upstream

    gw = Gateway()
    registry_specs = []
    if registry_path:
        from agentbom_amd.runtime.gateway import load_upstream_registry

        registry_specs = [f"{e['name']}={e['url']}"
                          for e in load_upstream_registry(registry_path)]
    for spec in list(upstreams) + registry_specs:
        name, _, url = spec.partition("=")

        def make_handler(u):
            def handler(frame):
                import httpx

                return httpx.post(u, json=frame, timeout=30.0).json()

            return handler

        gw.register(Upstream(name=name, handler=make_handler(url)))

    app = FastAPI(title="agent-bom gateway")

    @app.post("/mcp/{upstream}")
    def relay(upstream: str, frame: dict) -> dict:
        return gw.relay(upstream, frame)

    @app.get("/metrics")
    def metrics() -> dict:
        return gw.metrics

    uvicorn.run(app, host=host, port=port)


@main.command(name="policy-check")
@click.argument("policy_file", type=click.Path(exists=True))
@click.option("--demo", is_flag=True)
@click.option("--dry-run", is_flag=True)
def policy_check_cmd(policy_file: str, demo: bool, dry_run: bool) -> None:
    """Evaluate a policy file against the latest (demo) scan."""
    from agentbom_amd.scan.orchestrator import run_demo_scan
    from agentbom_amd.scan.policy import evaluate_policy, load_policy

    policy = load_policy(policy_file)
    report = run_demo_scan()
    result = evaluate_policy(policy, report.blast_radii, dry_run=dry_run)
    click.echo(json.dumps(result, indent=2))
    sys.exit(0 if result["passed"] else 1)


@main.command(name="diff")
@click.argument("old_report", type=click.Path(exists=True))
@click.argument("new_report", type=click.Path(exists=True))
@click.option("--pretty", is_flag=True,
              help="Render a rich console diff instead of JSON.")
def diff_cmd(old_report: str, new_report: str, pretty: bool) -> None:
    """Diff two scan report JSON files (new/resolved findings)."""
    from agentbom_amd.scan.history import diff_reports

    old = json.loads(Path(old_report).read_text())
    new = json.loads(Path(new_report).read_text())
    result = diff_reports(old, new)
    if pretty:
        from agentbom_amd.output.console_render import print_diff

        print_diff(result)
    else:
        click.echo(json.dumps(result, indent=2, default=str))


@main.command(name="bench")
@click.option("--packages", type=int, default=100_000)
@click.option("--steps", type=int, default=3)
def bench_cmd(packages: int, steps: int) -> None:
    """Quick local pipeline benchmark (see bench.py for the driver contract)."""
    import subprocess

    rc = subprocess.call(
        [sys.executable, str(Path(__file__).resolve().parents[2] / "bench.py"),
         "--packages", str(packages), "--steps", str(steps), "--warmup", "1",
         "--agents", str(max(100, packages // 100)),
         "--servers", str(max(500, packages // 20)),
         "--name-catalog", str(max(1000, packages // 10))],
    )
    sys.exit(rc)


def cli_main() -> None:
    main(prog_name="agent-bom")


if __name__ == "__main__":
    cli_main()
