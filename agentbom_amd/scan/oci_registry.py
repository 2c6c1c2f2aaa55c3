"""Container registry pulls: scan an image straight from a registry.

Closes the §2.2 containers/images gap (reference: src/agent_bom/image.py
registry pulls) using the Docker Registry HTTP API v2 through the
offline-guarded retry client:

  1. anonymous/token auth (WWW-Authenticate: Bearer realm=... flow);
  2. GET /v2/{name}/manifests/{ref} (manifest list -> linux/amd64 pick);
  3. GET /v2/{name}/blobs/{digest} per layer (size-capped);
  4. the downloaded layers feed the SAME layer walker scan/oci.py uses
     for docker-save tars (whiteouts, newest-first dedup, provenance).

Transport is injected for tests (httpx.MockTransport serves a synthetic
registry); offline mode refuses at the first request.
"""

from __future__ import annotations

import gzip
import io
import re
import tarfile
from typing import Any, Optional

from agentbom_amd.scan.oci import OciScanResult, _scan_layers
from agentbom_amd.utils.http_client import check_offline, create_client, request_with_retry

_MANIFEST_TYPES = ", ".join([
    "application/vnd.docker.distribution.manifest.v2+json",
    "application/vnd.oci.image.manifest.v1+json",
    "application/vnd.docker.distribution.manifest.list.v2+json",
    "application/vnd.oci.image.index.v1+json",
])

_MAX_LAYER_BYTES = 512 * 1024 * 1024  # refuse silly layers; surfaced as warning


def parse_image_ref(ref: str) -> tuple[str, str, str]:
    """'nginx:1.25' -> (registry-2 host, repository, tag/digest)."""
    tag = "latest"
    name = ref
    if "@" in ref:
        name, _, tag = ref.partition("@")
    elif ":" in ref.rsplit("/", 1)[-1]:
        name, _, tag = ref.rpartition(":")
    host = "registry-1.docker.io"
    if "/" in name and ("." in name.split("/")[0] or ":" in name.split("/")[0]
                        or name.split("/")[0] == "localhost"):
        host, _, name = name.partition("/")
    elif "/" not in name:
        name = f"library/{name}"
    return host, name, tag


class RegistryClient:
    """Minimal registry v2 client with bearer-token auth."""

    def __init__(self, host: str, client=None, token: Optional[str] = None):
        self.base = f"https://{host}"
        check_offline(self.base)
        self.client = client or create_client(timeout=120.0)
        self.token = token

    def _headers(self, accept: str) -> dict[str, str]:
        h = {"Accept": accept}
        if self.token:
            h["Authorization"] = f"Bearer {self.token}"
        return h

    def _get(self, path: str, accept: str):
        url = f"{self.base}{path}"
        resp = request_with_retry(self.client, "GET", url,
                                  headers=self._headers(accept))
        if resp is not None and resp.status_code == 401 and self.token is None:
            challenge = resp.headers.get("WWW-Authenticate", "")
            token = self._fetch_token(challenge)
            if token:
                self.token = token
                resp = request_with_retry(self.client, "GET", url,
                                          headers=self._headers(accept))
        return resp

    def _fetch_token(self, challenge: str) -> Optional[str]:
        """Bearer realm="https://auth...",service="...",scope="..." flow."""
        fields = dict(re.findall(r'(\w+)="([^"]*)"', challenge))
        realm = fields.get("realm")
        if not realm:
            return None
        params = {k: v for k, v in fields.items() if k in ("service", "scope")}
        resp = request_with_retry(self.client, "GET", realm, params=params)
        if resp is None or resp.status_code != 200:
            return None
        return resp.json().get("token") or resp.json().get("access_token")

    def manifest(self, name: str, ref: str) -> Optional[dict]:
        resp = self._get(f"/v2/{name}/manifests/{ref}", _MANIFEST_TYPES)
        if resp is None or resp.status_code != 200:
            return None
        return resp.json()

    def blob(self, name: str, digest: str) -> Optional[bytes]:
        resp = self._get(f"/v2/{name}/blobs/{digest}", "application/octet-stream")
        if resp is None or resp.status_code != 200:
            return None
        return resp.content


def _pick_platform_manifest(client: RegistryClient, name: str,
                            doc: dict) -> Optional[dict]:
    if "layers" in doc:
        return doc
    for entry in doc.get("manifests", []) or []:
        plat = entry.get("platform") or {}
        if plat.get("os") == "linux" and plat.get("architecture") == "amd64":
            return client.manifest(name, entry["digest"])
    # fall back to the first referenced manifest
    for entry in doc.get("manifests", []) or []:
        return client.manifest(name, entry["digest"])
    return None


def scan_image_registry(ref: str, client=None,
                        token: Optional[str] = None) -> OciScanResult:
    """Pull + scan an image from its registry (network required)."""
    host, name, tag = parse_image_ref(ref)
    reg = RegistryClient(host, client=client, token=token)
    result = OciScanResult(image_ref=ref)

    doc = reg.manifest(name, tag)
    if doc is None:
        result.warnings.append(f"manifest fetch failed for {ref} "
                               "(auth, name, or registry unreachable)")
        return result
    manifest = _pick_platform_manifest(reg, name, doc)
    if manifest is None or not manifest.get("layers"):
        result.warnings.append(f"no usable linux/amd64 manifest for {ref}")
        return result

    layer_blobs: dict[str, bytes] = {}
    layer_ids: list[str] = []
    for layer in manifest["layers"]:
        digest = layer.get("digest", "")
        size = int(layer.get("size", 0))
        layer_ids.append(digest)
        if size > _MAX_LAYER_BYTES:
            result.warnings.append(
                f"layer {digest[:19]} skipped: {size} bytes exceeds the "
                f"{_MAX_LAYER_BYTES}-byte pull cap")
            continue
        blob = reg.blob(name, digest)
        if blob is None:
            result.warnings.append(f"layer {digest[:19]} blob fetch failed")
            continue
        layer_blobs[digest] = blob

    def open_layer(layer_id: str) -> tarfile.TarFile:
        raw = layer_blobs.get(layer_id)
        if raw is None:
            raise OSError("blob not fetched")
        if raw[:2] == b"\x1f\x8b":
            raw = gzip.decompress(raw)
        return tarfile.open(fileobj=io.BytesIO(raw), mode="r")

    scanned = _scan_layers(open_layer, layer_ids, {}, ref)
    scanned.warnings = result.warnings + scanned.warnings
    return scanned
