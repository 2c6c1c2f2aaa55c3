"""OCI / docker-save image scanning tests (synthetic images, no daemon)."""

import io
import json
import tarfile
from pathlib import Path

import pytest

from agentbom_amd.scan.oci import (
    oci_result_to_agent,
    scan_docker_save,
    scan_image,
    scan_oci_layout,
)


def _tar_bytes(files: dict[str, bytes]) -> bytes:
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w") as tf:
        for name, data in files.items():
            info = tarfile.TarInfo(name)
            info.size = len(data)
            tf.addfile(info, io.BytesIO(data))
    return buf.getvalue()


LAYER1 = _tar_bytes({
    "app/requirements.txt": b"pyyaml==5.3\nrequests==2.25.0\n",
    "var/lib/dpkg/status": (
        b"Package: openssl\nVersion: 1.1.1k-1\nStatus: install ok installed\n\n"
        b"Package: doomed\nVersion: 1.0\nStatus: install ok installed\n\n"),
})
# layer 2: removes app/requirements.txt via whiteout, adds package.json
LAYER2 = _tar_bytes({
    "app/.wh.requirements.txt": b"",
    "srv/package.json": json.dumps(
        {"dependencies": {"left-pad": "1.3.0"}}).encode(),
})


def _docker_save(tmp_path: Path, layers: dict[str, bytes],
                 history: list[str]) -> Path:
    config = {"history": [{"created_by": h} for h in history]}
    manifest = [{"Config": "config.json",
                 "RepoTags": ["myapp:1.0"],
                 "Layers": list(layers)}]
    files = {
        "manifest.json": json.dumps(manifest).encode(),
        "config.json": json.dumps(config).encode(),
        **layers,
    }
    path = tmp_path / "image.tar"
    path.write_bytes(_tar_bytes(files))
    return path


class TestDockerSave:
    @pytest.fixture()
    def image(self, tmp_path):
        return _docker_save(
            tmp_path,
            {"l1/layer.tar": LAYER1, "l2/layer.tar": LAYER2},
            ["/bin/sh -c pip install -r requirements.txt",
             "/bin/sh -c #(nop) COPY srv /srv"])

    def test_packages_and_attribution(self, image):
        result = scan_docker_save(image)
        names = {p.name for p in result.packages}
        # dpkg packages survive; left-pad from upper layer
        assert {"openssl", "doomed", "left-pad"} <= names
        lp = next(p for p in result.packages if p.name == "left-pad")
        assert lp.occurrences[0].layer_id == "l2/layer.tar"
        ossl = next(p for p in result.packages if p.name == "openssl")
        assert "pip install" in ossl.occurrences[0].created_by

    def test_whiteout_deletes_lower_layer_file(self, image):
        result = scan_docker_save(image)
        # requirements.txt was whiteouted by layer 2 -> its pkgs must not appear
        assert not any(p.name in ("pyyaml", "requests")
                       for p in result.packages)

    def test_layer_metadata(self, image):
        result = scan_docker_save(image)
        assert [l.digest for l in result.layers] == ["l1/layer.tar",
                                                     "l2/layer.tar"]
        assert result.layers[0].created_by.startswith("RUN pip install")

    def test_agent_wrapper_scans_end_to_end(self, image):
        from agentbom_amd.db.store import load_advisory_windows
        from agentbom_amd.scan.orchestrator import scan_agents

        agent = oci_result_to_agent(scan_docker_save(image))
        report = scan_agents([agent], load_advisory_windows(offline=True))
        assert report.total_packages > 0

    def test_not_an_image(self, tmp_path):
        bogus = tmp_path / "x.tar"
        bogus.write_bytes(_tar_bytes({"foo": b"bar"}))
        with pytest.raises(ValueError):
            scan_docker_save(bogus)

    def test_traversal_members_skipped(self, tmp_path):
        evil = _tar_bytes({"../../etc/requirements.txt": b"evil==1.0\n"})
        image = _docker_save(tmp_path, {"l/layer.tar": evil}, ["RUN x"])
        result = scan_docker_save(image)
        assert not any(p.name == "evil" for p in result.packages)


class TestOciLayout:
    def test_layout_scan(self, tmp_path):
        blobs = tmp_path / "layout" / "blobs" / "sha256"
        blobs.mkdir(parents=True)
        (blobs / "aaa").write_bytes(LAYER1)
        config = {"history": [{"created_by": "RUN apt-get install"}]}
        (blobs / "cfg").write_text(json.dumps(config))
        manifest = {"config": {"digest": "sha256:cfg"},
                    "layers": [{"digest": "sha256:aaa"}]}
        (blobs / "mmm").write_text(json.dumps(manifest))
        (tmp_path / "layout" / "index.json").write_text(json.dumps(
            {"manifests": [{"digest": "sha256:mmm"}],
             "annotations": {"org.opencontainers.image.ref.name": "base:12"}}))
        result = scan_oci_layout(tmp_path / "layout")
        assert result.image_ref == "base:12"
        assert any(p.name == "openssl" for p in result.packages)
        # dispatcher picks layout for dirs
        assert scan_image(tmp_path / "layout").image_ref == "base:12"


class TestCliImage:
    def test_scan_image_flag(self, tmp_path):
        from click.testing import CliRunner

        from agentbom_amd.cli import main

        image = _docker_save(tmp_path, {"l/layer.tar": LAYER1}, ["RUN x"])
        runner = CliRunner()
        out = runner.invoke(main, ["scan", "--image", str(image), "--offline",
                                   "-f", "json", "-o",
                                   str(tmp_path / "r.json"), "--exit-zero"])
        assert out.exit_code in (0, 1), out.output
        doc = json.loads((tmp_path / "r.json").read_text())
        assert any(p["name"] == "openssl" for p in doc["packages"])
