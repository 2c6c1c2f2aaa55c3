"""Known-MCP-server security-metadata registry + blocklist.

Reference surface: src/agent_bom/mcp_official_registry.py +
mcp_server_catalog.py + mcp_blocklist.py over data/mcp-registry.yaml
(1,099 servers).  This build ships its own curated registry
(``agentbom_amd/data/mcp_registry.json``, same schema v2.0) and can load
an operator-supplied registry via ``AGENT_BOM_MCP_REGISTRY``.

The registry powers:
- **discovery stamping** — a discovered server matching a registry entry
  gets ``registry_verified``/``registry_id`` + inferred tool/credential
  surfaces BEFORE the server is ever spawned (blast-radius pre-compute);
- **pre-install trust checks** (``marketplace_check``);
- **the blocklist** — known-malicious package patterns fail closed.
"""

from __future__ import annotations

import json
import os
from functools import lru_cache
from pathlib import Path
from typing import Any, Optional

_BUNDLED = Path(__file__).resolve().parents[1] / "data" / "mcp_registry.json"


@lru_cache(maxsize=4)
def _load(path_str: str) -> dict[str, Any]:
    try:
        data = json.loads(Path(path_str).read_text())
    except (OSError, json.JSONDecodeError):
        return {"servers": {}, "blocklist": []}
    if not isinstance(data, dict):
        return {"servers": {}, "blocklist": []}
    data.setdefault("servers", {})
    data.setdefault("blocklist", [])
    return data


def load_registry(path: Optional[str] = None) -> dict[str, Any]:
    return _load(path or os.environ.get("AGENT_BOM_MCP_REGISTRY") or str(_BUNDLED))


def lookup_package(package: str, registry: Optional[dict] = None) -> Optional[dict]:
    """Exact package-name lookup (the pre-install trust check key)."""
    reg = registry or load_registry()
    entry = reg["servers"].get(package)
    if entry:
        return dict(entry)
    # npm scoped names are often configured unscoped
    for key, e in reg["servers"].items():
        if key.split("/")[-1] == package:
            return dict(e)
    return None


def match_command(command: str, args: list[str],
                  registry: Optional[dict] = None) -> Optional[dict]:
    """Match a configured server command line against command_patterns."""
    reg = registry or load_registry()
    haystack = " ".join([command, *args]).lower()
    best = None
    best_len = 0
    for entry in reg["servers"].values():
        for pattern in entry.get("command_patterns", []):
            p = str(pattern).lower()
            if p and p in haystack and len(p) > best_len:
                best, best_len = entry, len(p)
    return dict(best) if best else None


def check_blocklist(name_or_package: str,
                    registry: Optional[dict] = None) -> Optional[dict]:
    """Fail-closed known-malicious check; substring patterns."""
    reg = registry or load_registry()
    lowered = name_or_package.lower()
    for rule in reg["blocklist"]:
        if str(rule.get("pattern", "")).lower() in lowered:
            return dict(rule)
    return None


def stamp_server_from_registry(server, registry: Optional[dict] = None) -> bool:
    """Enrich a discovered MCPServer from the registry (pre-spawn metadata).

    Returns True when a registry entry matched.  Declared tools are only
    filled in when discovery found none (runtime introspection and actual
    config always win over registry inference).
    """
    from agentbom_amd.models import MCPTool

    entry = match_command(server.command, server.args, registry)
    if entry is None:
        return False
    server.registry_verified = bool(entry.get("verified", False))
    server.registry_id = entry.get("package")
    if not server.tools and entry.get("tools"):
        server.tools = [
            MCPTool(name=t, description="", discovery_source="registry",
                    discovery_confidence="inferred")
            for t in entry["tools"]
        ]
        server.stamp_child_identities()
    for var in entry.get("credential_env_vars", []):
        # record the NAME as an expected credential surface (values are
        # never retained anywhere in discovery — "***" is the convention)
        server.env.setdefault(var, "***")
    block = check_blocklist(server.registry_id or server.name, registry)
    if block:
        server.security_blocked = True
        server.security_warnings.append(
            f"registry blocklist: {block['reason']}")
    elif entry.get("risk_level") in ("high", "critical"):
        server.security_warnings.append(
            f"registry risk {entry['risk_level']}: "
            f"{entry.get('risk_justification', entry.get('description', ''))}")
    return True
