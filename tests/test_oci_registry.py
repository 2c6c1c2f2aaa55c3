"""Registry pulls against a synthetic Docker Registry v2 (MockTransport)."""

from __future__ import annotations

import gzip
import io
import json
import tarfile

import httpx
import pytest

from agentbom_amd.scan.oci_registry import (
    parse_image_ref,
    scan_image_registry,
)
from agentbom_amd.utils.http_client import OfflineError, set_offline


@pytest.fixture(autouse=True)
def _online():
    set_offline(False)
    yield
    set_offline(False)


class TestRefParsing:
    @pytest.mark.parametrize("ref,expect", [
        ("nginx", ("registry-1.docker.io", "library/nginx", "latest")),
        ("nginx:1.25", ("registry-1.docker.io", "library/nginx", "1.25")),
        ("org/app:2", ("registry-1.docker.io", "org/app", "2")),
        ("ghcr.io/org/app:v3", ("ghcr.io", "org/app", "v3")),
        ("localhost:5000/x:1", ("localhost:5000", "x", "1")),
        ("repo/app@sha256:abc", ("registry-1.docker.io", "repo/app",
                                 "sha256:abc")),
    ])
    def test_parse(self, ref, expect):
        assert parse_image_ref(ref) == expect


def _layer_tar(files: dict[str, str]) -> bytes:
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w") as tf:
        for name, content in files.items():
            data = content.encode()
            ti = tarfile.TarInfo(name)
            ti.size = len(data)
            tf.addfile(ti, io.BytesIO(data))
    return gzip.compress(buf.getvalue())


@pytest.fixture
def registry():
    """Synthetic registry: token auth + manifest list + two layers."""
    layer1 = _layer_tar({"app/package.json": json.dumps({
        "name": "webapp", "version": "1.0.0",
        "dependencies": {"express": "4.17.1"}})})
    layer2 = _layer_tar({"app/requirements.txt": "pyyaml==5.3\n"})
    d1, d2 = "sha256:layer1", "sha256:layer2"
    manifest = {"schemaVersion": 2,
                "layers": [{"digest": d1, "size": len(layer1)},
                           {"digest": d2, "size": len(layer2)}]}
    index = {"manifests": [
        {"digest": "sha256:amd64", "platform": {"os": "linux",
                                                "architecture": "amd64"}},
        {"digest": "sha256:arm", "platform": {"os": "linux",
                                              "architecture": "arm64"}}]}
    state = {"token_fetches": 0, "authed": []}

    def handler(request: httpx.Request) -> httpx.Response:
        path = request.url.path
        if request.url.host == "auth.example":
            state["token_fetches"] += 1
            assert request.url.params["service"] == "registry.example"
            return httpx.Response(200, json={"token": "tok-123"})
        auth = request.headers.get("Authorization", "")
        if not auth:
            return httpx.Response(401, headers={
                "WWW-Authenticate": ('Bearer realm="https://auth.example/token",'
                                     'service="registry.example",'
                                     'scope="repository:org/app:pull"')})
        state["authed"].append(path)
        if path == "/v2/org/app/manifests/1.0":
            return httpx.Response(200, json=index)
        if path == "/v2/org/app/manifests/sha256:amd64":
            return httpx.Response(200, json=manifest)
        if path == f"/v2/org/app/blobs/{d1}":
            return httpx.Response(200, content=layer1)
        if path == f"/v2/org/app/blobs/{d2}":
            return httpx.Response(200, content=layer2)
        return httpx.Response(404)

    return handler, state


def test_registry_scan_end_to_end(registry):
    handler, state = registry
    client = httpx.Client(transport=httpx.MockTransport(handler))
    result = scan_image_registry("registry.example/org/app:1.0", client=client)
    names = {(p.ecosystem, p.name, p.version) for p in result.packages}
    assert ("npm", "express", "4.17.1") in names
    assert ("pypi", "pyyaml", "5.3") in names
    assert state["token_fetches"] == 1  # token reused across requests
    assert not result.warnings


def test_manifest_failure_warns_not_raises():
    def handler(request):
        return httpx.Response(500)

    client = httpx.Client(transport=httpx.MockTransport(handler))
    result = scan_image_registry("registry.example/org/app:1.0", client=client)
    assert result.packages == []
    assert any("manifest fetch failed" in w for w in result.warnings)


def test_oversize_layer_skipped(registry, monkeypatch):
    handler, _ = registry
    monkeypatch.setattr("agentbom_amd.scan.oci_registry._MAX_LAYER_BYTES", 10)
    client = httpx.Client(transport=httpx.MockTransport(handler))
    result = scan_image_registry("registry.example/org/app:1.0", client=client)
    assert any("exceeds" in w for w in result.warnings)
    assert result.packages == []


def test_offline_refused():
    set_offline(True)
    with pytest.raises(OfflineError):
        scan_image_registry("nginx:1.25")
