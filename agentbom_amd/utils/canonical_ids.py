"""Deterministic UUIDv5 identities — the join key across scans, graph, stores.

Byte-compatible with the reference (src/agent_bom/canonical_ids.py,
src/agent_bom/package_utils.py): identical namespace UUID and fingerprint
normalization so canonical ids produced here equal the reference's for the
same entities.
"""

from __future__ import annotations

import json
import os
import re
import uuid
from pathlib import Path
from typing import Any, Mapping, Optional, Sequence
from urllib.parse import urlparse, urlunparse

AGENT_BOM_ID_NAMESPACE = uuid.UUID("7f3e4b2a-9c1d-5f8e-a0b4-12c3d4e5f6a7")
CANONICAL_ID_SCHEMA_VERSION = "2"

_PY_NAME_RE = re.compile(r"[-_.]+")
_ECO_ALIASES = {"golang": "go"}


def normalize_package_name(name: str, ecosystem: str = "") -> str:
    """Lowercase; PyPI additionally folds ``[-_.]+`` runs to ``-`` (PEP 503)."""
    if not name:
        return name
    if ecosystem.lower() == "pypi":
        return _PY_NAME_RE.sub("-", name).lower()
    return name.lower()


def normalize_package_ecosystem(ecosystem: str) -> str:
    eco = (ecosystem or "").strip().lower()
    return _ECO_ALIASES.get(eco, eco)


def _part_to_text(value: Any) -> str:
    if value is None:
        return ""
    if isinstance(value, Mapping):
        return json.dumps(value, sort_keys=True, separators=(",", ":"), default=str)
    if isinstance(value, Sequence) and not isinstance(value, (str, bytes, bytearray)):
        return json.dumps(list(value), sort_keys=True, separators=(",", ":"), default=str)
    return str(value)


def canonical_fingerprint(*parts: Any) -> str:
    return ":".join(t.lower().strip() for t in (_part_to_text(p) for p in parts) if t)


def canonical_id(*parts: Any) -> str:
    return str(uuid.uuid5(AGENT_BOM_ID_NAMESPACE, canonical_fingerprint(*parts)))


def source_ids(**values: Any) -> dict[str, str]:
    out: dict[str, str] = {}
    for key, value in values.items():
        if value in (None, "", [], {}):
            continue
        out[str(key)] = _part_to_text(value)
    return out


def canonical_package_key(name: str, version: str, ecosystem: str, purl: Optional[str] = None) -> str:
    eco = normalize_package_ecosystem(ecosystem)
    pname = normalize_package_name((name or "").strip(), eco)
    ver = (version or "").strip()
    return f"{eco}:{pname}@{ver}" if ver else f"{eco}:{pname}"


def canonical_package_id(name: str, version: str, ecosystem: str, purl: Optional[str] = None) -> str:
    return canonical_id("package", canonical_package_key(name, version, ecosystem, purl))


def canonical_agent_id(
    agent_type: str,
    name: str,
    *,
    source_id: str = "",
    device_fingerprint: str = "",
    config_path: str = "",
) -> str:
    """Agent identity: device fingerprint > source id > config path > name."""
    fp = (device_fingerprint or "").strip()
    if fp:
        return canonical_id("agent", agent_type, f"device:{fp}")
    src = (source_id or "").strip()
    if src:
        return canonical_id("agent", agent_type, f"source:{src}", f"name:{name}")
    loc = (config_path or "").strip()
    if loc:
        return canonical_id("agent", agent_type, f"config:{loc}", f"name:{name}")
    return canonical_id("agent", agent_type, name)


def normalize_command_arg(arg: str) -> str:
    text = str(arg).strip()
    if not text:
        return ""
    if text.startswith(("/", "~", ".")):
        try:
            return os.path.normpath(os.path.expanduser(text)).lower()
        except (OSError, ValueError):
            return text.lower()
    return text.lower()


def mcp_server_identity_discriminator(
    name: str,
    command: str = "",
    *,
    url: Optional[str] = None,
    args: Optional[Sequence[str]] = None,
) -> str:
    """url > command+args > name — matches discovery dedup granularity."""
    if url and url.strip():
        p = urlparse(url.strip())
        return f"url:{urlunparse((p.scheme.lower(), p.netloc.lower(), p.path.rstrip('/'), '', '', ''))}"
    cmd = Path(command or "").name.lower().strip()
    norm_args = tuple(a for a in (normalize_command_arg(x) for x in (args or [])) if a)
    if cmd or norm_args:
        arg_str = " ".join(norm_args)
        if not norm_args:
            return f"cmd:{cmd}:{arg_str}:{name.strip().lower()}"
        return f"cmd:{cmd}:{arg_str}"
    return f"name:{name.strip().lower()}"


def canonical_mcp_server_id(
    name: str,
    command: str = "",
    *,
    registry_id: Optional[str] = None,
    url: Optional[str] = None,
    args: Optional[Sequence[str]] = None,
) -> str:
    ident = registry_id if registry_id else mcp_server_identity_discriminator(name, command, url=url, args=args)
    return canonical_id("mcp_server", ident)


def canonical_mcp_tool_id(name: str, input_schema: Optional[Mapping[str, Any]] = None, *, server_id: Optional[str] = None) -> str:
    schema = json.dumps(input_schema or {}, sort_keys=True, separators=(",", ":"))
    return canonical_id("mcp_tool", server_id or "", name, schema)


def canonical_mcp_resource_id(uri: str, mime_type: Optional[str] = None, *, server_id: Optional[str] = None) -> str:
    return canonical_id("mcp_resource", server_id or "", uri, mime_type or "")


def canonical_mcp_prompt_id(
    name: str,
    arguments: Optional[Sequence[Mapping[str, Any]]] = None,
    *,
    server_id: Optional[str] = None,
) -> str:
    args = json.dumps(list(arguments or []), sort_keys=True, separators=(",", ":"))
    return canonical_id("mcp_prompt", server_id or "", name, args)


def canonical_finding_id(asset_canonical_id: str, finding_key: str, *qualifiers: Any) -> str:
    return canonical_id(asset_canonical_id, finding_key, *qualifiers)


def canonical_graph_node_id(entity_type: str, graph_id: str) -> str:
    return canonical_id("graph_node", entity_type, graph_id)


def canonical_graph_edge_id(source: str, target: str, relationship: str) -> str:
    return canonical_id("graph_edge", relationship, source, target)


def vulnerability_occurrence_key(vulnerability_id: str, package_ref: str, ecosystem: str = "") -> tuple[str, str, str]:
    """Stable finding identity across package-version remediation (version-free)."""
    raw = (package_ref or "").strip()
    pkg_name = raw
    if raw.startswith("@"):
        sep = raw.rfind("@")
        if sep > 0:
            pkg_name = raw[:sep]
    elif "@" in raw:
        pkg_name = raw.rsplit("@", 1)[0]
    eco = normalize_package_ecosystem(ecosystem)
    return ((vulnerability_id or "").strip().upper(), eco, normalize_package_name(pkg_name, eco))
