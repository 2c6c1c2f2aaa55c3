"""Runtime discovery surfaces: Docker MCP Toolkit, compose, processes, K8s.

VERDICT r1 'What's missing' #5 (reference:
src/agent_bom/discovery/__init__.py:557 discover_docker_mcp,
:636 discover_compose_mcp_servers, :759 discover_running_processes,
:999 discover_k8s_mcp_servers).  Every collector is dependency-injected
(home dir / process iterator / kubectl runner) so fixtures drive the tests
without Docker, psutil state, or a cluster.
"""

from __future__ import annotations

import json
import os
import re
import shutil
import subprocess
from pathlib import Path
from typing import Any, Callable, Iterable, Optional

import yaml

from agentbom_amd.models.core import (
    Agent,
    AgentType,
    MCPServer,
    Package,
    ServerSurface,
    TransportType,
)

# command-line shapes that identify an MCP server process
_MCP_PROCESS_PATTERNS = (
    re.compile(r"@modelcontextprotocol/[\w.\-]+"),
    re.compile(r"\bmcp-server-[\w.\-]+"),
    re.compile(r"\buvx\b.*\bmcp[\w.\-]*"),
    re.compile(r"python\S*\s+-m\s+mcp[\w.\-]*"),
    re.compile(r"\bfastmcp\b"),
)

_NPM_PKG = re.compile(r"(@[\w.\-]+/[\w.\-]+|mcp-server-[\w.\-]+)(?:@([\w.\-]+))?")


def _package_from_command(command: str, args: list[str]) -> list[Package]:
    """Best-effort package attribution for npx/uvx launched servers."""
    joined = " ".join([command, *args])
    pkgs = []
    if "npx" in command or "npm" in command or "node" in command:
        m = _NPM_PKG.search(joined)
        if m:
            pkgs.append(Package(name=m.group(1), version=m.group(2) or "",
                                ecosystem="npm"))
    elif "uvx" in command or "python" in command or "uv" in command:
        m = re.search(r"\b(mcp[\w\-]*|[\w\-]+-mcp[\w\-]*)\b", joined)
        if m:
            pkgs.append(Package(name=m.group(1), version="", ecosystem="pypi"))
    return pkgs


# ── Docker Desktop MCP Toolkit ──────────────────────────────────────────────


def discover_docker_mcp(home: Optional[Path] = None) -> Optional[Agent]:
    """~/.docker/mcp/registry.yaml (enabled servers) cross-referenced with
    catalogs/docker-mcp.yaml (image refs, tools, secrets)."""
    root = Path(home) if home else Path(os.path.expanduser("~/.docker/mcp"))
    reg_path = root / "registry.yaml"
    if not reg_path.exists():
        return None
    try:
        reg = yaml.safe_load(reg_path.read_text())
    except (OSError, ValueError, yaml.YAMLError):
        return None
    if not isinstance(reg, dict) or not isinstance(reg.get("registry"), dict):
        return None
    enabled = set(reg["registry"].keys())
    if not enabled:
        return None

    catalog: dict[str, dict] = {}
    cat_path = root / "catalogs" / "docker-mcp.yaml"
    if cat_path.exists():
        try:
            cat = yaml.safe_load(cat_path.read_text()) or {}
            catalog = ((cat.get("registry") or {})
                       if isinstance(cat.get("registry"), dict) else {})
        except (OSError, ValueError, yaml.YAMLError):
            catalog = {}

    servers = []
    for name in sorted(enabled):
        entry = catalog.get(name) or {}
        image = entry.get("image") or f"mcp/{name}"
        secrets = [s.get("name", "") for s in entry.get("secrets", []) or []
                   if isinstance(s, dict)]
        srv = MCPServer(
            name=name, command="docker", args=["run", image],
            transport=TransportType.STDIO,
            env={k: "***" for k in secrets},
            surface=ServerSurface.CONTAINER_IMAGE,
        )
        servers.append(srv)
    return Agent(
        name="Docker MCP Toolkit", agent_type=AgentType.DOCKER_MCP,
        config_path=str(reg_path), mcp_servers=servers, source="docker-mcp",
    )


# ── docker-compose services ─────────────────────────────────────────────────

_COMPOSE_FILES = ("docker-compose.yml", "docker-compose.yaml",
                  "compose.yml", "compose.yaml")
_MCP_IMAGE_HINT = re.compile(r"mcp", re.IGNORECASE)


def discover_compose_mcp_servers(project_dir: Optional[str] = None) -> Optional[Agent]:
    """Compose services that look like MCP servers: ``mcp.*`` labels, an
    image/name containing 'mcp', or an MCP port annotation."""
    base = Path(project_dir or ".")
    path = next((base / f for f in _COMPOSE_FILES if (base / f).exists()), None)
    if path is None:
        return None
    try:
        data = yaml.safe_load(path.read_text())
    except (OSError, ValueError, yaml.YAMLError):
        return None
    services = (data or {}).get("services")
    if not isinstance(services, dict):
        return None
    servers = []
    for name, svc in sorted(services.items()):
        if not isinstance(svc, dict):
            continue
        image = str(svc.get("image", ""))
        labels = svc.get("labels") or {}
        if isinstance(labels, list):
            labels = dict(lab.split("=", 1) for lab in labels if "=" in lab)
        is_mcp = (
            any(str(k).startswith("mcp.") for k in labels)
            or _MCP_IMAGE_HINT.search(image) is not None
            or _MCP_IMAGE_HINT.search(name) is not None
        )
        if not is_mcp:
            continue
        env = svc.get("environment") or {}
        if isinstance(env, list):
            env = dict(e.split("=", 1) if "=" in e else (e, "") for e in env)
        servers.append(MCPServer(
            name=name, command="docker", args=["compose", "up", name],
            transport=(TransportType.SSE if str(labels.get("mcp.transport")) == "sse"
                       else TransportType.STDIO),
            env={k: "***" for k in env},
            surface=ServerSurface.CONTAINER_IMAGE,
        ))
    if not servers:
        return None
    return Agent(name=f"compose:{path.parent.name}", agent_type=AgentType.CUSTOM,
                 config_path=str(path), mcp_servers=servers, source="docker-compose")


# ── running processes ───────────────────────────────────────────────────────


def discover_running_processes(
    process_iter: Optional[Iterable[dict[str, Any]]] = None,
) -> Optional[Agent]:
    """Host process sweep: command lines matching MCP server shapes.

    ``process_iter`` yields {pid, name, cmdline} dicts (tests inject
    fixtures); default is psutil.process_iter."""
    if process_iter is None:
        try:
            import psutil
        except ImportError:
            return None

        def _iter():
            for proc in psutil.process_iter(["pid", "name", "cmdline"]):
                try:
                    yield proc.info
                except (psutil.NoSuchProcess, psutil.AccessDenied):
                    continue

        process_iter = _iter()

    servers = []
    for info in process_iter:
        cmdline = info.get("cmdline") or []
        if not cmdline:
            continue
        cmd_str = " ".join(str(c) for c in cmdline)
        if not any(p.search(cmd_str) for p in _MCP_PROCESS_PATTERNS):
            continue
        command = str(cmdline[0])
        args = [str(c) for c in cmdline[1:]]
        servers.append(MCPServer(
            name=f"pid-{info.get('pid', '?')}:{Path(command).name}",
            command=command, args=args, transport=TransportType.STDIO,
            packages=_package_from_command(command, args),
        ))
    if not servers:
        return None
    return Agent(name="Running MCP processes", agent_type=AgentType.CUSTOM,
                 config_path="process://localhost", mcp_servers=servers,
                 source="running-processes")


# ── Kubernetes ──────────────────────────────────────────────────────────────


def _kubectl_runner(context: Optional[str]) -> Optional[Callable[..., Optional[dict]]]:
    if not shutil.which("kubectl"):
        return None

    def run(*args: str) -> Optional[dict]:
        cmd = ["kubectl", *args]
        if context:
            cmd += ["--context", context]
        cmd += ["-o", "json"]
        try:
            out = subprocess.run(cmd, capture_output=True, timeout=30, text=True)
        except (OSError, subprocess.TimeoutExpired):
            return None
        if out.returncode != 0:
            return None
        try:
            return json.loads(out.stdout)
        except ValueError:
            return None

    return run


def discover_k8s_mcp_servers(
    namespace: str = "default",
    all_namespaces: bool = False,
    context: Optional[str] = None,
    run: Optional[Callable[..., Optional[dict]]] = None,
) -> Optional[Agent]:
    """Pods with MCP signals (labels/annotations/images/ports) plus
    ``mcpservers.mcp.io`` custom resources.  ``run`` injects a kubectl-JSON
    runner for tests; default shells out to kubectl."""
    run = run or _kubectl_runner(context)
    if run is None:
        return None
    scope = ["-A"] if all_namespaces else ["-n", namespace]
    servers = []

    pods = run("get", "pods", *scope)
    for item in (pods or {}).get("items", []) or []:
        meta = item.get("metadata") or {}
        labels = meta.get("labels") or {}
        annotations = meta.get("annotations") or {}
        spec = item.get("spec") or {}
        containers = spec.get("containers") or []
        signals = (
            any("mcp" in str(k).lower() for k in labels)
            or any("mcp" in str(k).lower() for k in annotations)
            or any(_MCP_IMAGE_HINT.search(str(c.get("image", ""))) for c in containers)
        )
        if not signals:
            continue
        for c in containers:
            image = str(c.get("image", ""))
            servers.append(MCPServer(
                name=f"{meta.get('namespace', namespace)}/{meta.get('name', 'pod')}"
                     f"/{c.get('name', 'c')}",
                command="k8s", args=[image],
                transport=TransportType.STREAMABLE_HTTP,
                surface=ServerSurface.CONTAINER_IMAGE,
                env={k: "***" for k in (labels | annotations)
                     if "mcp" in str(k).lower()},
            ))

    crds = run("get", "mcpservers.mcp.io", *scope)
    for item in (crds or {}).get("items", []) or []:
        meta = item.get("metadata") or {}
        spec = item.get("spec") or {}
        servers.append(MCPServer(
            name=f"crd/{meta.get('namespace', namespace)}/{meta.get('name', 'srv')}",
            command="k8s", args=[str(spec.get("image", ""))],
            transport=TransportType.STREAMABLE_HTTP,
            surface=ServerSurface.CONTAINER_IMAGE,
        ))

    if not servers:
        return None
    return Agent(name="Kubernetes MCP servers", agent_type=AgentType.CUSTOM,
                 config_path=f"k8s://{'all' if all_namespaces else namespace}",
                 mcp_servers=servers, source="kubernetes")
