"""Agent / MCP-server discovery from 30 client config surfaces.

Reference: src/agent_bom/discovery/__init__.py:65-290 (AGENT_CONFIGS map +
AGENT_BINARIES), :422 discover_global_configs, :530 discover_project_configs,
:759 discover_running_processes, :1228 discover_all.

The config paths below are the published locations of each client's MCP
configuration.  Parsing accepts the common ``mcpServers`` JSON shape plus
the TOML (Codex CLI) and YAML (Goose) variants.
"""

from __future__ import annotations

import glob as globmod
import json
import os
import platform
import shutil
from pathlib import Path
from typing import Any, Optional

from agentbom_amd.models import (
    Agent,
    AgentStatus,
    AgentType,
    MCPServer,
    Package,
    TransportType,
)

# ~/.x paths identical across OSes use the "all" key; VS Code-extension-style
# paths expand per-OS via _VSX.
_VSX = {
    "Darwin": "~/Library/Application Support/{app}/User/{tail}",
    "Linux": "~/.config/{app}/User/{tail}",
    "Windows": "~/AppData/Roaming/{app}/User/{tail}",
}
_APPDIR = {
    "Darwin": "~/Library/Application Support/{app}/{tail}",
    "Linux": "~/.config/{app}/{tail}",
    "Windows": "~/AppData/Roaming/{app}/{tail}",
}


def _vsx(app: str, tail: str) -> dict[str, list[str]]:
    return {os_: [tmpl.format(app=app, tail=tail)] for os_, tmpl in _VSX.items()}


def _appdir(app: str, tail: str) -> dict[str, list[str]]:
    return {os_: [tmpl.format(app=app, tail=tail)] for os_, tmpl in _APPDIR.items()}


def _same(*paths: str) -> dict[str, list[str]]:
    return {"Darwin": list(paths), "Linux": list(paths), "Windows": list(paths)}


def _merge(*maps: dict[str, list[str]]) -> dict[str, list[str]]:
    out: dict[str, list[str]] = {"Darwin": [], "Linux": [], "Windows": []}
    for m in maps:
        for k, v in m.items():
            out[k].extend(v)
    return out


AGENT_CONFIGS: dict[AgentType, dict[str, list[str]]] = {
    AgentType.CLAUDE_DESKTOP: _appdir("Claude", "claude_desktop_config.json"),
    AgentType.CLAUDE_CODE: _same("~/.claude/settings.json", "~/.claude.json"),
    AgentType.CURSOR: _merge(
        _vsx("Cursor", "globalStorage/cursor.mcp/mcp.json"), _same("~/.cursor/mcp.json")
    ),
    AgentType.WINDSURF: _merge(
        _same("~/.windsurf/mcp.json"),
        {"Darwin": ["~/Library/Application Support/Windsurf/User/globalStorage/windsurf.mcp/mcp.json"],
         "Linux": [], "Windows": []},
    ),
    AgentType.CLINE: _vsx("Code", "globalStorage/saoudrizwan.claude-dev/settings/cline_mcp_settings.json"),
    AgentType.VSCODE_COPILOT: _vsx("Code", "mcp.json"),
    AgentType.CORTEX_CODE: _same(
        "~/.snowflake/cortex/mcp.json", "~/.snowflake/cortex/settings.json",
        "~/.snowflake/cortex/permissions.json", "~/.snowflake/cortex/hooks.json",
    ),
    AgentType.CODEX_CLI: _same("~/.codex/config.toml"),
    AgentType.GEMINI_CLI: _same("~/.gemini/settings.json"),
    AgentType.GOOSE: {
        "Darwin": ["~/.config/goose/config.yaml"],
        "Linux": ["~/.config/goose/config.yaml"],
        "Windows": ["~/AppData/Roaming/Block/goose/config/config.yaml"],
    },
    AgentType.SNOWFLAKE_CLI: _same("~/.snowflake/connections.toml", "~/.snowflake/config.toml"),
    AgentType.CONTINUE: _merge(
        _same("~/.continue/config.json"),
        _vsx("Code", "globalStorage/continue.continue/config.json"),
    ),
    AgentType.ZED: {
        "Darwin": ["~/.config/zed/settings.json"],
        "Linux": ["~/.config/zed/settings.json"],
        "Windows": ["~/AppData/Roaming/Zed/settings.json"],
    },
    AgentType.OPENCLAW: _same("~/.openclaw/openclaw.json"),
    AgentType.ROO_CODE: _vsx("Code", "globalStorage/rooveterinaryinc.roo-cline/settings/cline_mcp_settings.json"),
    AgentType.AMAZON_Q: _vsx("Code", "globalStorage/amazonwebservices.amazon-q-vscode/mcp.json"),
    AgentType.DOCKER_MCP: _same(),
    AgentType.JETBRAINS_AI: _merge(
        {"Darwin": ["~/Library/Application Support/JetBrains/*/mcp.json"],
         "Linux": ["~/.config/JetBrains/*/mcp.json"],
         "Windows": ["~/AppData/Roaming/JetBrains/*/mcp.json"]},
        _same("~/.config/github-copilot/intellij/mcp.json"),
    ),
    AgentType.JUNIE: _same("~/.junie/mcp/mcp.json"),
    AgentType.COPILOT_CLI: _same("~/.copilot/mcp-config.json"),
    AgentType.TABNINE: _same("~/.tabnine/mcp_servers.json"),
    AgentType.SOURCEGRAPH_CODY: _merge(
        _vsx("Code", "globalStorage/sourcegraph.cody-ai/mcp.json"), _same("~/.cody/mcp.json")
    ),
    AgentType.AIDER: _same("~/.aider/mcp.json", "~/.aider.conf.yml"),
    AgentType.REPLIT_AGENT: _same(),
    AgentType.VOID_EDITOR: _merge(
        _same("~/.void/mcp.json"), _vsx("Void", "globalStorage/void.mcp/mcp.json")
    ),
    AgentType.AIDE: _merge(_same("~/.aide/mcp.json"), _vsx("Aide", "mcp.json")),
    AgentType.TRAE: _merge(_same("~/.trae/mcp.json"), _vsx("Trae", "mcp.json")),
    AgentType.PIECES: {
        "Darwin": ["~/Library/Application Support/com.pieces.os/mcp.json", "~/.pieces/mcp.json"],
        "Linux": ["~/.pieces/mcp.json"],
        "Windows": ["~/AppData/Roaming/Pieces/mcp.json", "~/.pieces/mcp.json"],
    },
    AgentType.MCP_CLI: _same("~/.mcp/config.json"),
}

AGENT_BINARIES: dict[AgentType, str] = {
    AgentType.CLAUDE_CODE: "claude",
    AgentType.OPENCLAW: "openclaw",
    AgentType.ZED: "zed",
    AgentType.CURSOR: "cursor",
    AgentType.WINDSURF: "windsurf",
    AgentType.CORTEX_CODE: "cortex",
    AgentType.CODEX_CLI: "codex",
    AgentType.GEMINI_CLI: "gemini",
    AgentType.GOOSE: "goose",
    AgentType.SNOWFLAKE_CLI: "snow",
    AgentType.JUNIE: "junie",
    AgentType.AIDER: "aider",
    AgentType.VOID_EDITOR: "void",
    AgentType.AIDE: "aide",
    AgentType.TRAE: "trae",
    AgentType.MCP_CLI: "mcp",
}


def _parse_servers_dict(servers: dict[str, Any], config_path: str) -> list[MCPServer]:
    out = []
    for name, spec in servers.items():
        if not isinstance(spec, dict):
            continue
        transport = TransportType.STDIO
        url = spec.get("url") or spec.get("serverUrl")
        if url:
            transport = (
                TransportType.SSE if str(spec.get("type", "")).lower() == "sse"
                else TransportType.STREAMABLE_HTTP
            )
        out.append(
            MCPServer(
                name=name,
                command=spec.get("command", ""),
                args=[str(a) for a in spec.get("args", [])],
                env={k: "***" for k in (spec.get("env") or {})},  # values never retained
                transport=transport,
                url=url,
                config_path=config_path,
                discovery_sources=["global_config"],
            )
        )
    # pre-spawn enrichment from the known-server registry (verification,
    # inferred tool/credential surface, blocklist fail-closed)
    from agentbom_amd.mcp.registry import stamp_server_from_registry

    for server in out:
        if server.command:
            stamp_server_from_registry(server)
    return out


def parse_mcp_config(path: Path) -> list[MCPServer]:
    """Extract MCP servers from a client config (JSON / TOML / YAML)."""
    try:
        text = path.read_text()
    except OSError:
        return []
    suffix = path.suffix.lower()
    config_path = str(path)
    try:
        if suffix == ".toml":
            from agentbom_amd.utils.compat import tomllib

            data = tomllib.loads(text)
            servers = data.get("mcp_servers") or data.get("mcpServers") or {}
            return _parse_servers_dict(servers, config_path)
        if suffix in (".yaml", ".yml"):
            import yaml

            data = yaml.safe_load(text) or {}
            servers = data.get("extensions") or data.get("mcpServers") or {}
            if isinstance(servers, dict):
                norm = {}
                for name, spec in servers.items():
                    if isinstance(spec, dict):
                        norm[name] = {
                            "command": spec.get("cmd") or spec.get("command", ""),
                            "args": spec.get("args", []),
                            "env": spec.get("envs") or spec.get("env", {}),
                        }
                return _parse_servers_dict(norm, config_path)
            return []
        data = json.loads(text)
    except Exception:
        return []
    if not isinstance(data, dict):
        return []
    servers = data.get("mcpServers") or data.get("mcp_servers") or {}
    if not servers and "context_servers" in data:  # Zed
        servers = data["context_servers"]
    if not servers and "experimental" in data and isinstance(data["experimental"], dict):
        servers = data["experimental"].get("modelContextProtocolServers", {})
    if isinstance(servers, list):  # some clients use a list form
        servers = {s.get("name", f"server-{i}"): s for i, s in enumerate(servers)}
    return _parse_servers_dict(servers if isinstance(servers, dict) else {}, config_path)


def discover_global_configs(
    agent_types: Optional[list[AgentType]] = None, home: Optional[str] = None
) -> list[Agent]:
    """Walk every client's published config locations on this machine."""
    system = platform.system()
    agents: list[Agent] = []
    for agent_type, os_paths in AGENT_CONFIGS.items():
        if agent_types and agent_type not in agent_types:
            continue
        for raw in os_paths.get(system, []):
            pattern = os.path.expanduser(raw if home is None else raw.replace("~", home))
            paths = globmod.glob(pattern) if "*" in pattern else (
                [pattern] if os.path.exists(pattern) else []
            )
            for p in paths:
                servers = parse_mcp_config(Path(p))
                if servers:
                    agents.append(
                        Agent(
                            name=agent_type.value,
                            agent_type=agent_type,
                            config_path=p,
                            mcp_servers=servers,
                            source="local",
                        )
                    )
    return agents


def discover_installed_not_configured(agent_types: Optional[list[AgentType]] = None) -> list[Agent]:
    """Clients whose binary is on PATH but with no MCP config found."""
    out = []
    for agent_type, binary in AGENT_BINARIES.items():
        if agent_types and agent_type not in agent_types:
            continue
        if shutil.which(binary):
            out.append(
                Agent(
                    name=agent_type.value,
                    agent_type=agent_type,
                    config_path="",
                    status=AgentStatus.INSTALLED_NOT_CONFIGURED,
                    source="binary_on_path",
                )
            )
    return out


def discover_project_configs(root: Optional[str] = None) -> list[Agent]:
    """Project-scoped configs: .cursor/mcp.json, .vscode/mcp.json, .mcp.json."""
    base = Path(root or os.getcwd())
    candidates = [
        (base / ".cursor" / "mcp.json", AgentType.CURSOR),
        (base / ".vscode" / "mcp.json", AgentType.VSCODE_COPILOT),
        (base / ".mcp.json", AgentType.CLAUDE_CODE),
        (base / ".windsurf" / "mcp.json", AgentType.WINDSURF),
    ]
    out = []
    for path, agent_type in candidates:
        if path.exists():
            servers = parse_mcp_config(path)
            if servers:
                out.append(
                    Agent(
                        name=f"{agent_type.value} (project)",
                        agent_type=agent_type,
                        config_path=str(path),
                        mcp_servers=servers,
                        source="project_config",
                    )
                )
    return out


def discover_all(
    agent_types: Optional[list[AgentType]] = None,
    project_root: Optional[str] = None,
    include_installed: bool = True,
    include_runtime: bool = True,
) -> list[Agent]:
    """Union of global + project config surfaces, deduped by canonical id.

    ``include_runtime`` adds the live surfaces: Docker MCP Toolkit,
    compose services, running processes, Kubernetes (each best-effort —
    absent substrate yields nothing, never an error)."""
    agents = discover_global_configs(agent_types)
    agents += discover_project_configs(project_root)
    if include_installed:
        configured = {a.agent_type for a in agents}
        agents += [
            a for a in discover_installed_not_configured(agent_types)
            if a.agent_type not in configured
        ]
    if include_runtime:
        from agentbom_amd.scan.discovery_runtime import (
            discover_compose_mcp_servers,
            discover_docker_mcp,
            discover_k8s_mcp_servers,
            discover_running_processes,
        )

        for collect in (discover_docker_mcp,
                        lambda: discover_compose_mcp_servers(project_root),
                        discover_running_processes,
                        discover_k8s_mcp_servers):
            try:
                found = collect()
            except Exception:
                found = None  # runtime surfaces are strictly best-effort
            if found is not None:
                agents.append(found)
    seen: set[str] = set()
    unique = []
    for a in agents:
        if a.stable_id not in seen:
            seen.add(a.stable_id)
            unique.append(a)
    return unique
