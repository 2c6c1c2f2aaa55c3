"""Lockfile/manifest parser tests across ecosystems."""

import json
import textwrap

import pytest

from agentbom_amd.scan import parsers


def _by_name(pkgs):
    return {p.name: p for p in pkgs}


class TestNpm:
    def test_package_lock_v3(self):
        data = {
            "lockfileVersion": 3,
            "packages": {
                "": {"dependencies": {"express": "^4.17.1"}},
                "node_modules/express": {"version": "4.17.1"},
                "node_modules/lodash": {"version": "4.17.20", "dev": True},
                "node_modules/@scope/pkg": {"version": "1.0.0"},
            },
        }
        pkgs = _by_name(parsers.parse_package_lock(json.dumps(data), "package-lock.json"))
        assert pkgs["express"].version == "4.17.1"
        assert pkgs["express"].is_direct
        assert pkgs["lodash"].dependency_scope == "dev"
        assert not pkgs["lodash"].is_direct
        assert "@scope/pkg" in pkgs

    def test_package_lock_v1(self):
        data = {"dependencies": {"ws": {"version": "8.5.0"}}}
        pkgs = parsers.parse_package_lock(json.dumps(data), "x")
        assert pkgs[0].name == "ws" and pkgs[0].version == "8.5.0"

    def test_yarn_lock(self):
        text = textwrap.dedent('''
            # yarn lockfile v1

            express@^4.17.1:
              version "4.17.1"

            "@scope/pkg@^1.0.0":
              version "1.0.0"
        ''')
        pkgs = _by_name(parsers.parse_yarn_lock(text, "yarn.lock"))
        assert pkgs["express"].version == "4.17.1"
        assert pkgs["@scope/pkg"].version == "1.0.0"

    def test_pnpm_lock(self):
        text = "packages:\n  /express@4.17.1:\n    resolution: {}\n  /@scope/pkg@1.0.0(peer@2):\n    resolution: {}\n"
        pkgs = _by_name(parsers.parse_pnpm_lock(text, "pnpm-lock.yaml"))
        assert pkgs["express"].version == "4.17.1"
        assert pkgs["@scope/pkg"].version == "1.0.0"

    def test_package_json_declaration_only(self):
        data = {"dependencies": {"axios": "^1.4.0"}, "devDependencies": {"jest": "~29.0.0"}}
        pkgs = _by_name(parsers.parse_package_json(json.dumps(data), "package.json"))
        assert pkgs["axios"].reachability_evidence == "declaration_only"
        assert pkgs["jest"].dependency_scope == "dev"


class TestPython:
    def test_requirements(self):
        text = "requests==2.31.0\n# comment\npyyaml[full]==5.4.1\nflask>=2.0\n-r other.txt\n"
        pkgs = _by_name(parsers.parse_requirements_txt(text, "requirements.txt"))
        assert pkgs["requests"].version == "2.31.0"
        assert pkgs["pyyaml"].version == "5.4.1"
        assert pkgs["flask"].reachability_evidence == "declaration_only"

    def test_poetry_lock(self):
        text = '[[package]]\nname = "requests"\nversion = "2.31.0"\ncategory = "main"\n'
        pkgs = parsers.parse_poetry_lock(text, "poetry.lock")
        assert pkgs[0].name == "requests"

    def test_uv_lock(self):
        text = '[[package]]\nname = "httpx"\nversion = "0.27.0"\n'
        assert parsers.parse_uv_lock(text, "uv.lock")[0].version == "0.27.0"

    def test_pipfile_lock(self):
        data = {"default": {"requests": {"version": "==2.31.0"}},
                "develop": {"pytest": {"version": "==8.0.0"}}}
        pkgs = _by_name(parsers.parse_pipfile_lock(json.dumps(data), "Pipfile.lock"))
        assert pkgs["requests"].version == "2.31.0"
        assert pkgs["pytest"].dependency_scope == "dev"

    def test_conda_env(self):
        text = "dependencies:\n  - numpy=1.26.0=py310\n  - pip:\n    - requests==2.31.0\n"
        pkgs = _by_name(parsers.parse_conda_env(text, "environment.yml"))
        assert pkgs["numpy"].ecosystem == "conda"
        assert pkgs["requests"].ecosystem == "pypi"


class TestGoRust:
    def test_go_mod(self):
        text = textwrap.dedent("""
            module example.com/app
            require (
                github.com/gin-gonic/gin v1.9.1
                golang.org/x/text v0.3.7 // indirect
            )
            require github.com/stretchr/testify v1.8.0
        """)
        pkgs = _by_name(parsers.parse_go_mod(text, "go.mod"))
        assert pkgs["github.com/gin-gonic/gin"].version == "v1.9.1"
        assert not pkgs["golang.org/x/text"].is_direct
        assert pkgs["github.com/stretchr/testify"].version == "v1.8.0"

    def test_go_sum(self):
        text = ("github.com/gin-gonic/gin v1.9.1 h1:abc=\n"
                "github.com/gin-gonic/gin v1.9.1/go.mod h1:def=\n")
        pkgs = parsers.parse_go_sum(text, "go.sum")
        assert len(pkgs) == 1 and pkgs[0].version == "v1.9.1"

    def test_cargo_lock(self):
        text = '[[package]]\nname = "serde"\nversion = "1.0.190"\n'
        assert parsers.parse_cargo_lock(text, "Cargo.lock")[0].name == "serde"


class TestJvmDotnet:
    def test_pom_xml(self):
        text = textwrap.dedent("""<?xml version="1.0"?>
            <project xmlns="http://maven.apache.org/POM/4.0.0">
              <dependencies>
                <dependency>
                  <groupId>org.springframework</groupId>
                  <artifactId>spring-core</artifactId>
                  <version>5.3.20</version>
                </dependency>
              </dependencies>
            </project>""")
        pkgs = parsers.parse_pom_xml(text, "pom.xml")
        assert pkgs[0].name == "org.springframework:spring-core"
        assert pkgs[0].version == "5.3.20"

    def test_gradle_lockfile(self):
        text = "# comment\norg.slf4j:slf4j-api:1.7.36=runtimeClasspath\n"
        pkgs = parsers.parse_gradle_lockfile(text, "gradle.lockfile")
        assert pkgs[0].name == "org.slf4j:slf4j-api"

    def test_packages_lock_json(self):
        data = {"dependencies": {"net6.0": {
            "Newtonsoft.Json": {"type": "Direct", "resolved": "13.0.1"}}}}
        pkgs = parsers.parse_packages_lock_json(json.dumps(data), "packages.lock.json")
        assert pkgs[0].version == "13.0.1" and pkgs[0].is_direct

    def test_csproj(self):
        text = '<ItemGroup><PackageReference Include="Serilog" Version="3.0.1" /></ItemGroup>'
        assert parsers.parse_csproj(text, "x.csproj")[0].name == "Serilog"


class TestOthers:
    def test_gemfile_lock(self):
        text = "GEM\n  remote: https://rubygems.org/\n  specs:\n    rails (7.0.4)\n    rake (13.0.6)\n\nPLATFORMS\n  ruby\n"
        pkgs = _by_name(parsers.parse_gemfile_lock(text, "Gemfile.lock"))
        assert pkgs["rails"].version == "7.0.4"

    def test_composer_lock(self):
        data = {"packages": [{"name": "monolog/monolog", "version": "v2.8.0"}],
                "packages-dev": [{"name": "phpunit/phpunit", "version": "9.5.0"}]}
        pkgs = _by_name(parsers.parse_composer_lock(json.dumps(data), "composer.lock"))
        assert pkgs["monolog/monolog"].version == "2.8.0"  # v stripped
        assert pkgs["phpunit/phpunit"].dependency_scope == "dev"

    def test_package_resolved(self):
        data = {"pins": [{"identity": "swift-nio", "state": {"version": "2.40.0"}}]}
        assert parsers.parse_package_resolved(json.dumps(data), "Package.resolved")[0].name == "swift-nio"

    def test_mix_lock(self):
        text = '%{\n  "phoenix": {:hex, :phoenix, "1.7.2", "hash", [:mix]},\n}\n'
        assert parsers.parse_mix_lock(text, "mix.lock")[0].version == "1.7.2"

    def test_pubspec_lock(self):
        text = "packages:\n  http:\n    dependency: \"direct main\"\n    version: \"0.13.5\"\n"
        pkgs = parsers.parse_pubspec_lock(text, "pubspec.lock")
        assert pkgs[0].version == "0.13.5" and pkgs[0].is_direct


class TestOsDatabases:
    def test_dpkg_status(self):
        text = textwrap.dedent("""\
            Package: openssl
            Status: install ok installed
            Version: 3.0.2-0ubuntu1.10
            Source: openssl-src

            Package: removed-pkg
            Status: deinstall ok config-files
            Version: 1.0
        """)
        pkgs = parsers.parse_dpkg_status(text, "status")
        assert len(pkgs) == 1
        assert pkgs[0].name == "openssl"
        assert pkgs[0].source_package == "openssl-src"
        assert pkgs[0].ecosystem == "deb"

    def test_apk_installed(self):
        text = "P:musl\nV:1.2.4-r2\nA:x86_64\n\nP:busybox\nV:1.36.1-r0\n\n"
        pkgs = _by_name(parsers.parse_apk_installed(text, "installed"))
        assert pkgs["musl"].version == "1.2.4-r2"


class TestExtractPackages:
    def test_walk_and_dedup(self, tmp_path):
        (tmp_path / "package-lock.json").write_text(json.dumps({
            "lockfileVersion": 3,
            "packages": {"node_modules/express": {"version": "4.17.1"}},
        }))
        (tmp_path / "package.json").write_text(json.dumps({
            "dependencies": {"express": "^4.17.1"}}))
        (tmp_path / "requirements.txt").write_text("requests==2.31.0\n")
        sub = tmp_path / "svc"
        sub.mkdir()
        (sub / "go.mod").write_text("module m\nrequire github.com/x/y v1.0.0\n")
        (tmp_path / "node_modules").mkdir()
        (tmp_path / "node_modules" / "requirements.txt").write_text("evil==1.0\n")

        pkgs = parsers.extract_packages(tmp_path)
        by_key = {f"{p.ecosystem}:{p.name}": p for p in pkgs}
        assert by_key["npm:express"].reachability_evidence == "lockfile"  # lock wins
        assert "pypi:requests" in by_key
        assert "go:github.com/x/y" in by_key
        assert "pypi:evil" not in by_key  # node_modules skipped
