"""IaC scanning + Python SDK tests."""

import sys
import textwrap
from pathlib import Path

import pytest
from fastapi.testclient import TestClient

sys.path.insert(0, str(Path(__file__).resolve().parents[1] / "sdks" / "python"))

from agentbom_amd.scan.iac import iac_finding_to_finding, scan_iac_text, scan_iac_tree


class TestIac:
    def test_terraform_rules(self):
        tf = textwrap.dedent('''
            resource "aws_security_group" "open" {
              ingress { cidr_blocks = ["0.0.0.0/0"] }
            }
            resource "aws_db_instance" "db" {
              publicly_accessible = true
              password = "supersecret99"
            }
        ''')
        hits = scan_iac_text(tf, "main.tf")
        rules = {h.rule_id for h in hits}
        assert {"TF001", "TF003", "TF004"} <= rules
        assert any(h.severity == "critical" for h in hits)

    def test_k8s_rules(self):
        yaml_text = textwrap.dedent("""
            apiVersion: v1
            kind: Pod
            spec:
              hostNetwork: true
              containers:
              - name: x
                securityContext:
                  privileged: true
                  runAsUser: 0
        """)
        hits = scan_iac_text(yaml_text, "pod.yaml")
        assert {"K8S001", "K8S002", "K8S004"} <= {h.rule_id for h in hits}

    def test_dockerfile_rules(self):
        df = "FROM ubuntu\nUSER root\nRUN curl https://x.sh | bash\nENV API_KEY=abc123456\n"
        hits = scan_iac_text(df, "Dockerfile")
        assert {"DKR001", "DKR002", "DKR003", "DKR004"} <= {h.rule_id for h in hits}

    def test_compose_rules(self):
        text = "services:\n  app:\n    privileged: true\n    volumes:\n      - /var/run/docker.sock:/var/run/docker.sock\n"
        hits = scan_iac_text(text, "docker-compose.yml")
        assert {"CMP001", "CMP002"} <= {h.rule_id for h in hits}

    def test_tree_walk_and_finding(self, tmp_path):
        (tmp_path / "main.tf").write_text('acl = "public-read"\n')
        (tmp_path / "node_modules").mkdir()
        (tmp_path / "node_modules" / "x.tf").write_text('acl = "public-read"\n')
        hits = scan_iac_tree(tmp_path)
        assert len(hits) == 1
        f = iac_finding_to_finding(hits[0])
        assert f.finding_type.value == "CIS_FAIL"
        assert f.attack_tags == ["T1530"]


class TestSdk:
    @pytest.fixture(scope="class")
    def client(self):
        import httpx
        from agentbom_client import AgentBomClient

        from agentbom_amd.api.server import create_app

        tc = TestClient(create_app())

        class _Bridge(httpx.BaseTransport):
            """Route the SDK's sync httpx calls through the in-process app."""

            def handle_request(self, request: httpx.Request) -> httpx.Response:
                resp = tc.request(
                    request.method,
                    str(request.url.copy_with(scheme="http", host="testserver")),
                    content=request.read(),
                    headers=dict(request.headers),
                )
                return httpx.Response(resp.status_code, headers=resp.headers,
                                      content=resp.content)

        return AgentBomClient(base_url="http://testserver", transport=_Bridge())

    def test_full_flow(self, client):
        assert client.health()["status"] == "ok"
        job = client.scan(demo=True)
        assert job["status"] == "done"
        report = client.scan_report(job["job_id"])
        assert report["schema_version"] == "1.0"
        assert client.findings(severity="critical")["findings"]
        assert client.graph_search("pyyaml")["total"] == 1
        assert client.should_i_deploy()["verdict"] == "block"
        assert client.rollup()["containers"]
        assert client.compliance_report("soc2")["tagged_findings"] > 0
        hb = client.heartbeat("sdk-host", agents=1)
        assert hb["status"] == "healthy"

    def test_error_mapping(self, client):
        from agentbom_client import AgentBomError

        with pytest.raises(AgentBomError) as exc:
            client.scan_job("missing")
        assert exc.value.status_code == 404


class TestGoSdk:
    """The Go SDK is source-shipped (no Go toolchain in this image) — these
    tests pin its route surface against the live app so path drift is caught
    the same way the python/ts clients are."""

    GO_SRC = (Path(__file__).resolve().parents[1] / "sdks" / "go" / "client.go").read_text()

    def test_paths_exist_on_app(self):
        import re

        from agentbom_amd.api.server import create_app

        app = create_app()
        known = set()
        for r in app.routes:
            p = getattr(r, "path", "")
            known.add(re.sub(r"\{[^}]+\}", "{}", p))
        for quoted in re.findall(r'"(/(?:healthz|v1)[^"]*)"', self.GO_SRC):
            path = quoted.split("?")[0]
            norm = re.sub(r"\{[^}]+\}", "{}", path)
            # Go builds parameterized paths by concatenation — normalize the
            # literal prefix and check SOME app route starts with it
            assert any(k == norm or k.startswith(norm) for k in known), path

    def test_mirrors_python_client_methods(self):
        """Every /v1 path the python SDK uses appears in the Go source."""
        import re

        py_src = (Path(__file__).resolve().parents[1] / "sdks" / "python"
                  / "agentbom_client.py").read_text()
        py_paths = set(re.findall(r'"(/v1/[a-z\-]+(?:/[a-z\-]+)*)"', py_src))
        for p in py_paths:
            assert p in self.GO_SRC, f"Go SDK missing {p}"

    def test_go_mod_present(self):
        mod = (Path(__file__).resolve().parents[1] / "sdks" / "go" / "go.mod").read_text()
        assert "module " in mod and "go 1." in mod

    def test_balanced_braces_and_no_todo(self):
        assert self.GO_SRC.count("{") == self.GO_SRC.count("}")
        assert "TODO" not in self.GO_SRC
        assert self.GO_SRC.startswith("// Package agentbom")


class TestExtendedIacRules:
    """Round-2 rule-pack depth: sampled checks per family."""

    def _scan(self, text, path):
        from agentbom_amd.scan.iac import scan_iac_text

        return {f.rule_id for f in scan_iac_text(text, path)}

    def test_terraform_new_rules(self):
        tf = '''
resource "aws_s3_bucket_public_access_block" "b" {
  block_public_acls = false
}
resource "aws_db_instance" "d" {
  skip_final_snapshot = true
}
resource "aws_instance" "i" {
  associate_public_ip_address = true
}
resource "aws_security_group" "sg" {
  ingress {
    from_port = 22
    to_port = 22
    cidr_blocks = ["0.0.0.0/0"]
  }
}
'''
        ids = self._scan(tf, "main.tf")
        assert {"TF008", "TF009", "TF012", "TF015"} <= ids

    def test_k8s_new_rules(self):
        manifest = '''
apiVersion: v1
kind: Pod
spec:
  hostPID: true
  automountServiceAccountToken: true
  containers:
    - name: c
      securityContext:
        capabilities:
          add: ["SYS_ADMIN"]
---
apiVersion: v1
kind: Service
spec:
  type: NodePort
'''
        ids = self._scan(manifest, "pod.yaml")
        assert {"K8S009", "K8S010", "K8S011", "K8S014"} <= ids

    def test_dockerfile_new_rules(self):
        df = '''
FROM ubuntu:22.04
ADD https://example.com/tool.tar.gz /opt/
RUN apt-get update && apt-get install -y curl
RUN chmod -R 777 /app
EXPOSE 22
'''
        ids = self._scan(df, "Dockerfile")
        assert {"DKR006", "DKR007", "DKR008", "DKR010"} <= ids

    def test_compose_new_rules(self):
        compose = '''
services:
  app:
    image: corp/app:1
    pid: host
    cap_add:
      - NET_ADMIN
    environment:
      - DB_PASSWORD=hunter2
'''
        ids = self._scan(compose, "docker-compose.yml")
        assert {"CMP005", "CMP006", "CMP007"} <= ids
