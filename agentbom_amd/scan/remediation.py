"""Executable remediation commands per ecosystem.

Reference parity: src/agent_bom/remediation_commands.py +
remediation_apply.py — the remediation plan says WHAT to fix; this module
renders the exact upgrade command per ecosystem (and an optional shell
script) so an operator or CI job can apply it.  Commands are generated,
never executed, by this module.
"""

from __future__ import annotations

import shlex
from typing import Any, Optional

from agentbom_amd.models import AIBOMReport

_COMMANDS = {
    "pypi": "pip install --upgrade {name}=={fix}",
    "npm": "npm install {name}@{fix}",
    "cargo": "cargo update -p {name} --precise {fix}",
    "go": "go get {name}@v{fix}",
    "maven": ("mvn versions:use-dep-version -Dincludes={name} "
              "-DdepVersion={fix}"),
    "rubygems": "bundle update {name} --conservative",
    "composer": "composer require {name}:{fix}",
    "nuget": "dotnet add package {name} --version {fix}",
    "hex": "mix deps.update {name}",
    "pub": "dart pub upgrade {name}",
    "deb": "apt-get install --only-upgrade {name}",
    "apk": "apk upgrade {name}",
    "conda": "conda install {name}={fix}",
}

_REMOVAL = {
    "pypi": "pip uninstall -y {name}",
    "npm": "npm uninstall {name}",
    "cargo": "cargo remove {name}",
    "rubygems": "gem uninstall {name}",
    "composer": "composer remove {name}",
}


def command_for(name: str, ecosystem: str, fix: Optional[str],
                malicious: bool = False) -> Optional[str]:
    eco = ecosystem.lower()
    name_q = shlex.quote(name)
    if malicious:
        tpl = _REMOVAL.get(eco)
        return tpl.format(name=name_q) if tpl else f"# remove {name_q} ({eco}) manually"
    if not fix:
        return None
    tpl = _COMMANDS.get(eco)
    if tpl is None:
        return None
    return tpl.format(name=name_q, fix=shlex.quote(str(fix)))


def remediation_commands(report: AIBOMReport) -> list[dict[str, Any]]:
    """One command per remediation item, risk-ranked."""
    from agentbom_amd.output.json_fmt import _build_remediation_json

    out = []
    for item in _build_remediation_json(report):
        name = item["package"].split("@")[0]
        malicious = "malicious" in item["action"]
        cmd = command_for(name, item["ecosystem"], item.get("fix_version"),
                          malicious=malicious)
        out.append({
            "package": item["package"],
            "ecosystem": item["ecosystem"],
            "fix_version": item.get("fix_version"),
            "max_risk_score": item["max_risk_score"],
            "vulns": item["vulns"],
            "command": cmd,
            "manual_review": cmd is None,
        })
    return out


def remediation_script(report: AIBOMReport) -> str:
    """A reviewable shell script applying every automatable remediation."""
    lines = [
        "#!/bin/sh",
        "# agent-bom remediation script — review before running.",
        "# Generated from the latest scan; one command per remediation item.",
        "set -e",
        "",
    ]
    for item in remediation_commands(report):
        vulns = ", ".join(item["vulns"][:5])
        lines.append(f"# {item['package']} (risk "
                     f"{item['max_risk_score']:.1f}): {vulns}")
        if item["command"]:
            lines.append(item["command"])
        else:
            lines.append(f"# MANUAL: review advisories for {item['package']}")
        lines.append("")
    return "\n".join(lines)
