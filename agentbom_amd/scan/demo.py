"""Bundled deterministic demo estate for ``agent-bom agents --demo``.

Same curated content as the reference demo (reference: src/agent_bom/demo.py,
demo_advisories.py): five agents, ten MCP servers, real published CVE ids
(PyYAML RCE, LangChain RCE, KEV Pillow), credential-bearing servers, and the
``reqeusts`` typosquat — a closed evidence set that reproduces identical
findings on every machine with no network and no local vuln DB.

Stored compactly: each server row is (name, command, transport, packages,
env-credential names, tool names); packages are "eco:name@version".
"""

from __future__ import annotations

from typing import Any

from agentbom_amd.db.arena import AdvisoryWindow
from agentbom_amd.models.core import Severity

_S = "stdio"


def _srv(name: str, command: str, transport: str, pkgs: list[str],
         env: list[str], tools: list[str]) -> dict[str, Any]:
    packages = []
    for spec in pkgs:
        eco, rest = spec.split(":", 1)
        pname, ver = rest.rsplit("@", 1)
        packages.append({"name": pname, "version": ver, "ecosystem": eco})
    return {
        "name": name,
        "command": command.split()[0],
        "args": command.split()[1:],
        "transport": transport,
        "packages": packages,
        "env": {k: "***" for k in env},
        "tools": [{"name": t} for t in tools],
    }


DEMO_INVENTORY: dict[str, Any] = {
    "agents": [
        {
            "name": "cursor",
            "agent_type": "cursor",
            "source": "agent-bom --demo",
            "mcp_servers": [
                _srv("filesystem-server", "npx @modelcontextprotocol/server-filesystem /", _S,
                     ["npm:express@4.17.1", "npm:node-fetch@2.6.1", "npm:ws@8.5.0"],
                     [], ["read_file", "write_file", "list_directory"]),
                # Hero chain: AWS creds + run_shell tool + PyYAML CRITICAL RCE.
                _srv("shell-runner-server", "python -m mcp_shell_runner", _S,
                     ["pypi:pyyaml@5.3", "pypi:requests@2.28.0"],
                     ["AWS_ACCESS_KEY_ID", "AWS_SECRET_ACCESS_KEY"],
                     ["run_shell", "exec_command", "read_file"]),
            ],
        },
        {
            "name": "langchain-service",
            "agent_type": "custom",
            "source": "agent-bom --demo",
            "mcp_servers": [
                _srv("llm-orchestrator-server", "python -m mcp_orchestrator", "streamable-http",
                     ["pypi:langchain@0.0.150", "pypi:jinja2@3.0.0"],
                     ["OPENAI_API_KEY", "ANTHROPIC_API_KEY"],
                     ["run_chain", "eval_expression", "http_get"]),
                _srv("vector-db-server", "python -m mcp_vectors", _S,
                     ["pypi:cryptography@39.0.0", "pypi:requests@2.28.0"],
                     ["PINECONE_API_KEY", "DATABASE_URL"],
                     ["query_vectors", "upsert_vectors"]),
            ],
        },
        {
            "name": "support-copilot",
            "agent_type": "custom",
            "source": "agent-bom --demo",
            "mcp_servers": [
                _srv("helpdesk-server", "python -m mcp_helpdesk", "sse",
                     ["npm:axios@1.4.0", "npm:jsonwebtoken@8.5.1"],
                     ["HELPDESK_API_TOKEN", "JWT_SECRET"],
                     ["create_ticket", "search_tickets", "send_reply"]),
                _srv("email-server", "python -m mcp_email", _S,
                     ["npm:node-fetch@2.6.1", "pypi:certifi@2022.12.7"],
                     ["SMTP_PASSWORD"], ["send_email", "list_inbox"]),
            ],
        },
        {
            "name": "data-pipeline",
            "agent_type": "custom",
            "source": "agent-bom --demo",
            "mcp_servers": [
                _srv("warehouse-server", "python -m mcp_warehouse", _S,
                     ["pypi:pyyaml@5.3", "pypi:cryptography@39.0.0"],
                     ["SNOWFLAKE_PASSWORD", "DATABASE_URL"],
                     ["run_query", "execute_sql", "export_csv"]),
                # KEV Pillow + the "reqeusts" typosquat differentiator.
                _srv("etl-server", "python -m mcp_etl", _S,
                     ["pypi:pillow@9.0.0", "pypi:reqeusts@2.99.0"],
                     ["GCS_SERVICE_ACCOUNT_KEY"], ["transform_image", "load_data"]),
            ],
        },
        {
            "name": "claude-desktop",
            "agent_type": "claude-desktop",
            "source": "agent-bom --demo",
            "mcp_servers": [
                _srv("github-server", "npx @modelcontextprotocol/server-github", _S,
                     ["npm:axios@1.4.0", "npm:lodash@4.17.20", "npm:semver@7.5.2"],
                     ["GITHUB_TOKEN"], ["create_issue", "search_repos", "push_files"]),
                _srv("team-chat-server", "python -m slack_mcp", _S,
                     ["pypi:flask@2.2.0", "pypi:werkzeug@2.2.2", "pypi:jinja2@3.0.0"],
                     ["SLACK_BOT_TOKEN", "SLACK_SIGNING_SECRET"],
                     ["send_message", "list_channels"]),
            ],
        },
    ],
}


# (eco, package, introduced, fixed, vuln_id, severity, cvss, cwe, kev, summary)
_ADVISORY_ROWS: tuple[tuple, ...] = (
    ("npm", "express", "0", "4.19.2", "CVE-2024-29041", "medium", 6.1, "CWE-601", False,
     "Express open redirect via malformed URLs passed to res.location/redirect"),
    ("npm", "jsonwebtoken", "0", "9.0.0", "CVE-2022-23529", "high", 7.6, "CWE-347", False,
     "jsonwebtoken insecure key handling allows signature verification bypass"),
    ("npm", "node-fetch", "0", "3.1.1", "CVE-2022-0235", "high", 6.1, "CWE-200", False,
     "node-fetch leaks Cookie/Authorization headers on cross-origin redirect"),
    ("npm", "axios", "0", "1.6.0", "CVE-2023-45857", "high", 6.5, "CWE-918", False,
     "axios SSRF and credential leak via follow-redirects proxy handling"),
    ("npm", "ws", "0", "8.17.1", "CVE-2024-37890", "high", 7.5, "CWE-400", False,
     "ws denial of service when handling a request with many HTTP headers"),
    ("npm", "lodash", "0", "4.17.21", "CVE-2021-23337", "high", 7.2, "CWE-77", False,
     "lodash command injection via template() with tainted options"),
    ("pypi", "pyyaml", "0", "5.4", "CVE-2020-14343", "critical", 9.8, "CWE-20", False,
     "PyYAML arbitrary code execution via yaml.full_load on untrusted input"),
    ("pypi", "langchain", "0", "0.0.247", "CVE-2023-36258", "critical", 9.8, "CWE-94", False,
     "LangChain arbitrary code execution via PALChain prompt-to-Python evaluation"),
    ("pypi", "pillow", "0", "10.0.1", "CVE-2023-4863", "high", 8.8, "CWE-787", True,
     "Pillow bundled libwebp heap buffer overflow — exploited in the wild (CISA KEV)"),
    ("pypi", "requests", "0", "2.31.0", "CVE-2023-32681", "medium", 6.1, "CWE-200", False,
     "Requests leaks Proxy-Authorization header to destination on redirect"),
    ("pypi", "cryptography", "0", "42.0.0", "CVE-2023-50782", "high", 7.5, "CWE-208", False,
     "pyca/cryptography Bleichenbacher timing oracle in RSA PKCS#1 v1.5 decryption"),
    ("pypi", "flask", "0", "2.3.2", "CVE-2023-30861", "high", 7.5, "CWE-539", False,
     "Flask session cookie disclosed to other clients via a caching proxy"),
    ("pypi", "werkzeug", "0", "2.2.3", "CVE-2023-25577", "high", 7.5, "CWE-400", False,
     "Werkzeug multipart form-data parsing denial of service"),
    ("pypi", "jinja2", "0", "3.1.3", "CVE-2024-22195", "medium", 5.4, "CWE-79", False,
     "Jinja2 cross-site scripting via the xmlattr filter with attacker-controlled keys"),
    ("pypi", "certifi", "0", "2023.7.22", "CVE-2023-37920", "high", 7.5, "CWE-345", False,
     "certifi trusted a compromised e-Tugra root certificate authority"),
    # Coverage sentinel: semver@7.5.2 stays clean but the package is covered.
    ("npm", "semver", "0", "7.5.0", "DEMO-CLEAN-semver", "unknown", 0.0, "", False,
     "Non-matching coverage sentinel for clean demo package semver@7.5.2"),
)


def demo_advisory_windows() -> list[AdvisoryWindow]:
    out = []
    for eco, pkg, intro, fixed, vid, sev, cvss, cwe, kev, summary in _ADVISORY_ROWS:
        out.append(
            AdvisoryWindow(
                ecosystem=eco,
                package_name=pkg,
                vuln_id=vid,
                introduced=intro,
                fixed=fixed,
                severity=Severity(sev),
                cvss_score=cvss or None,
                is_kev=kev,
                summary=summary,
                cwe_ids=(cwe,) if cwe else (),
                fixed_version=fixed,
            )
        )
    return out
