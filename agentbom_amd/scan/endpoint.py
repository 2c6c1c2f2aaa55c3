"""Bounded workstation endpoint inventory.

Reference: src/agent_bom/endpoint/inventory.py (901 LoC) — bounded
app/process/service inventory: AI-relevant processes (agents, MCP servers,
inference runtimes), listening ports, and installed AI CLI binaries.
Read-only, bounded (caps + timeouts), never captures command-line secrets.
"""

from __future__ import annotations

import re
import shutil
from dataclasses import dataclass, field
from typing import Any, Optional

_AI_PROCESS_HINTS = re.compile(
    r"(?i)\b(mcp|claude|cursor|copilot|windsurf|aider|goose|gemini|codex|"
    r"ollama|vllm|llama|sglang|tgi|triton|uvicorn.*mcp|node.*modelcontext)\b"
)
_SECRETISH = re.compile(r"(?i)(--?(api[-_]?key|token|password|secret)[= ])\S+")

AI_BINARIES = (
    "claude", "cursor", "windsurf", "aider", "goose", "gemini", "codex",
    "ollama", "mcp", "cortex", "snow", "copilot",
)


@dataclass
class EndpointInventory:
    processes: list[dict[str, Any]] = field(default_factory=list)
    listening_ports: list[dict[str, Any]] = field(default_factory=list)
    installed_binaries: list[str] = field(default_factory=list)
    truncated: bool = False

    def to_dict(self) -> dict[str, Any]:
        return {
            "schema_version": "1",
            "processes": self.processes,
            "listening_ports": self.listening_ports,
            "installed_binaries": self.installed_binaries,
            "truncated": self.truncated,
        }


def _redact_cmdline(cmdline: list[str]) -> str:
    joined = " ".join(cmdline)[:300]
    return _SECRETISH.sub(r"\1***", joined)


def collect_endpoint_inventory(max_processes: int = 200) -> EndpointInventory:
    inv = EndpointInventory()
    try:
        import psutil
    except ImportError:
        return inv

    count = 0
    for proc in psutil.process_iter(["pid", "name", "cmdline", "username"]):
        try:
            info = proc.info
            cmdline = info.get("cmdline") or []
            text = f"{info.get('name', '')} {' '.join(cmdline)}"
            if not _AI_PROCESS_HINTS.search(text):
                continue
            if count >= max_processes:
                inv.truncated = True
                break
            inv.processes.append({
                "pid": info["pid"],
                "name": info.get("name"),
                "cmdline": _redact_cmdline(cmdline),
                "username": info.get("username"),
            })
            count += 1
        except (psutil.NoSuchProcess, psutil.AccessDenied):
            continue

    try:
        for conn in psutil.net_connections(kind="tcp"):
            if conn.status == "LISTEN" and conn.laddr:
                inv.listening_ports.append({
                    "port": conn.laddr.port,
                    "address": conn.laddr.ip,
                    "pid": conn.pid,
                })
                if len(inv.listening_ports) >= 200:
                    inv.truncated = True
                    break
    except (psutil.AccessDenied, PermissionError):
        pass

    inv.listening_ports.sort(key=lambda x: x["port"])
    inv.installed_binaries = [b for b in AI_BINARIES if shutil.which(b)]
    return inv
