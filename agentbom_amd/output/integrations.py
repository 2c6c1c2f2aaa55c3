"""Outbound integrations: control-plane push, webhooks, Slack, Jira, SIEM.

Reference surface: the scan command's --push-*/--siem-*/--jira-*/--slack-*
flags (src/agent_bom/cli/agents/scan_cmd.py:3182 run_integrations).  All
senders route through the offline-guarded retry client and FAIL OPEN: an
unreachable sink becomes a reported failure, never a crashed scan.
"""

from __future__ import annotations

import json
from typing import Any, Optional

from agentbom_amd.output.json_fmt import to_json
from agentbom_amd.utils.http_client import (
    OfflineError,
    create_client,
    request_with_retry,
    sanitize_url,
)


def _post(client, url: str, payload: dict, headers: Optional[dict] = None):
    resp = request_with_retry(client, "POST", url, json=payload,
                              headers=headers or {})
    ok = resp is not None and 200 <= resp.status_code < 300
    return ok, ("" if ok else f"status {resp.status_code if resp else 'unreachable'}")


def run_integrations(report, push_url=None, push_api_key=None, webhooks=(),
                     slack_webhook=None, jira=None, siem=None,
                     client=None) -> list[tuple[str, bool, str]]:
    """Send the report to every configured sink; returns (name, ok, detail)."""
    results: list[tuple[str, bool, str]] = []
    try:
        client = client or create_client()
    except Exception as exc:  # pragma: no cover - client construction
        return [("client", False, str(exc))]
    doc = to_json(report)
    summary = doc.get("summary", {})

    def guard(name: str, fn) -> None:
        try:
            ok, detail = fn()
        except OfflineError as exc:
            ok, detail = False, str(exc)
        except Exception as exc:  # fail open: report, never crash the scan
            ok, detail = False, str(exc)
        results.append((name, ok, detail))

    if push_url:
        headers = {"X-API-Key": push_api_key} if push_api_key else {}
        guard(f"push:{sanitize_url(push_url)}",
              lambda: _post(client, push_url, doc, headers))
    for hook in webhooks or ():
        guard(f"webhook:{sanitize_url(hook)}",
              lambda h=hook: _post(client, h, {
                  "event": "scan.completed", "summary": summary}))
    if slack_webhook:
        crit = summary.get("critical", 0)
        high = summary.get("high", 0)
        text = (f"agent-bom scan: {summary.get('total_vulnerabilities', 0)} "
                f"findings ({crit} critical / {high} high) across "
                f"{summary.get('total_agents', 0)} agents")
        guard("slack", lambda: _post(client, slack_webhook, {"text": text}))
    if jira:
        issue = {
            "fields": {
                "project": {"key": jira.get("project") or "SEC"},
                "issuetype": {"name": "Bug"},
                "summary": f"agent-bom: {summary.get('critical', 0)} critical "
                           f"finding(s)",
                "description": json.dumps(summary, indent=2),
            }
        }
        guard("jira", lambda: _post(
            client, jira["url"].rstrip("/") + "/rest/api/2/issue", issue,
            {"Authorization": f"Bearer {jira['token']}"}))
    if siem:
        from agentbom_amd.output.ocsf import to_ocsf

        events = to_ocsf(report)
        headers = ({"Authorization": f"Bearer {siem['token']}"}
                   if siem.get("token") else {})
        guard("siem", lambda: _post(client, siem["url"],
                                    {"events": events}, headers))
    return results
