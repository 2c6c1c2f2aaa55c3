"""Package extraction: lockfile/manifest parsers across 15 ecosystems.

Reference: src/agent_bom/parsers/ — 22 registered inventory parsers
(_BUILTIN_INVENTORY_PARSERS, parsers/__init__.py:74-97) over npm/yarn/pnpm/
bun, pip/poetry/uv/pipenv/conda, go, cargo, maven/gradle, nuget, rubygems,
composer, swift, hex, pub and the dpkg/rpm/apk OS databases.

Each parser returns Package objects; lockfiles yield resolved versions
(reachability_evidence="lockfile"), manifests yield declared ranges
(declaration_only).  ``extract_packages`` walks a directory and runs every
matching parser.
"""

from __future__ import annotations

import json
import re
from pathlib import Path
from typing import Callable, Optional

from agentbom_amd.models import Package

ParserFn = Callable[[str, str], list[Package]]  # (text, path) -> packages


def _pkg(name: str, version: str, eco: str, direct: bool = True,
         scope: str = "runtime", evidence: str = "lockfile") -> Package:
    return Package(name=name, version=version, ecosystem=eco, is_direct=direct,
                   dependency_scope=scope, reachability_evidence=evidence)


# ── npm family ─────────────────────────────────────────────────────────────


def parse_package_lock(text: str, path: str) -> list[Package]:
    """npm package-lock.json v1 (dependencies) and v2/v3 (packages)."""
    data = json.loads(text)
    out: dict[tuple, Package] = {}
    packages = data.get("packages")
    if isinstance(packages, dict):  # lockfile v2/v3
        direct = set((data.get("packages", {}).get("", {}) or {}).get("dependencies", {}) or {})
        for key, meta in packages.items():
            if not key or not isinstance(meta, dict):
                continue
            name = meta.get("name") or key.rsplit("node_modules/", 1)[-1]
            version = meta.get("version", "")
            if not version:
                continue
            scope = "dev" if meta.get("dev") else ("optional" if meta.get("optional") else "runtime")
            out[(name, version)] = _pkg(name, version, "npm", direct=name in direct, scope=scope)
    for name, meta in (data.get("dependencies") or {}).items():  # v1
        if isinstance(meta, dict) and meta.get("version"):
            out.setdefault((name, meta["version"]),
                           _pkg(name, meta["version"], "npm", scope="dev" if meta.get("dev") else "runtime"))
    return list(out.values())


_YARN_ENTRY = re.compile(r'^"?([^@"\s]+(?:@[^@"\s/]+/[^@"\s]+)?)@', re.MULTILINE)


def parse_yarn_lock(text: str, path: str) -> list[Package]:
    out: dict[tuple, Package] = {}
    current: Optional[str] = None
    for line in text.splitlines():
        if line and not line.startswith(("#", " ", "\t")) and line.rstrip().endswith(":"):
            spec = line.rstrip().rstrip(":").strip('"').split(",")[0].strip().strip('"')
            # name@range (scoped names keep their first @)
            if spec.startswith("@"):
                current = "@" + spec[1:].split("@", 1)[0]
            else:
                current = spec.split("@", 1)[0]
        elif current and line.strip().startswith(("version ", 'version: ', '"version"')):
            version = line.split()[-1].strip('"').rstrip(":")
            if version:
                out[(current, version)] = _pkg(current, version, "npm")
            current = None
    return list(out.values())


def parse_pnpm_lock(text: str, path: str) -> list[Package]:
    import yaml

    try:
        data = yaml.safe_load(text) or {}
    except yaml.YAMLError:
        return []
    if not isinstance(data, dict):
        return []
    out: dict[tuple, Package] = {}
    for key in (data.get("packages") or {}):
        # "/name@version(peer)" or "/@scope/name@version"
        k = key.lstrip("/")
        k = k.split("(", 1)[0]
        if "@" not in k:
            continue
        if k.startswith("@"):
            name, _, version = k.rpartition("@")
        else:
            name, _, version = k.rpartition("@")
        if name and version:
            out[(name, version)] = _pkg(name, version, "npm")
    return list(out.values())


def parse_package_json(text: str, path: str) -> list[Package]:
    data = json.loads(text)
    out = []
    for field, scope in (("dependencies", "runtime"), ("devDependencies", "dev"),
                         ("optionalDependencies", "optional"), ("peerDependencies", "peer")):
        for name, rng in (data.get(field) or {}).items():
            out.append(_pkg(name, str(rng).lstrip("^~=<> "), "npm", scope=scope,
                            evidence="declaration_only"))
    return out


# ── python family ──────────────────────────────────────────────────────────

_REQ_LINE = re.compile(r"^([A-Za-z0-9._-]+)(?:\[[^\]]*\])?\s*==\s*([^\s;#]+)")


def parse_requirements_txt(text: str, path: str) -> list[Package]:
    out = []
    for line in text.splitlines():
        line = line.strip()
        if not line or line.startswith(("#", "-")):
            continue
        m = _REQ_LINE.match(line)
        if m:
            out.append(_pkg(m.group(1), m.group(2), "pypi"))
        else:
            m2 = re.match(r"^([A-Za-z0-9._-]+)(?:\[[^\]]*\])?\s*([<>~!=].*)?$", line.split(";")[0].strip())
            if m2 and m2.group(1):
                out.append(_pkg(m2.group(1), (m2.group(2) or "").strip(), "pypi",
                                evidence="declaration_only"))
    return out


def parse_poetry_lock(text: str, path: str) -> list[Package]:
    from agentbom_amd.utils.compat import tomllib

    data = tomllib.loads(text)
    out = []
    for entry in data.get("package", []):
        name, version = entry.get("name"), entry.get("version")
        if name and version:
            scope = "dev" if entry.get("category") == "dev" else "runtime"
            out.append(_pkg(name, version, "pypi", scope=scope))
    return out


def parse_uv_lock(text: str, path: str) -> list[Package]:
    from agentbom_amd.utils.compat import tomllib

    data = tomllib.loads(text)
    return [
        _pkg(e["name"], e.get("version", ""), "pypi")
        for e in data.get("package", []) if e.get("name") and e.get("version")
    ]


def parse_pipfile_lock(text: str, path: str) -> list[Package]:
    data = json.loads(text)
    out = []
    for section, scope in (("default", "runtime"), ("develop", "dev")):
        for name, meta in (data.get(section) or {}).items():
            version = str(meta.get("version", "")).lstrip("=") if isinstance(meta, dict) else ""
            if version:
                out.append(_pkg(name, version, "pypi", scope=scope))
    return out


def parse_pyproject_toml(text: str, path: str) -> list[Package]:
    from agentbom_amd.utils.compat import tomllib

    data = tomllib.loads(text)
    out = []
    for dep in (data.get("project", {}).get("dependencies") or []):
        m = re.match(r"^([A-Za-z0-9._-]+)", dep)
        if m:
            out.append(_pkg(m.group(1), "", "pypi", evidence="declaration_only"))
    poetry = data.get("tool", {}).get("poetry", {})
    for name, spec in (poetry.get("dependencies") or {}).items():
        if name.lower() == "python":
            continue
        version = spec if isinstance(spec, str) else (spec.get("version", "") if isinstance(spec, dict) else "")
        out.append(_pkg(name, str(version).lstrip("^~"), "pypi", evidence="declaration_only"))
    return out


def parse_conda_env(text: str, path: str) -> list[Package]:
    import yaml

    try:
        data = yaml.safe_load(text) or {}
    except yaml.YAMLError:
        return []
    if not isinstance(data, dict):
        return []
    out = []
    deps = data.get("dependencies")
    for dep in (deps if isinstance(deps, list) else []):
        if isinstance(dep, str):
            parts = dep.split("=")
            if len(parts) >= 2:
                out.append(_pkg(parts[0], parts[1], "conda"))
        elif isinstance(dep, dict) and "pip" in dep:
            for pip_dep in (dep["pip"] if isinstance(dep["pip"], list) else []):
                m = _REQ_LINE.match(pip_dep)
                if m:
                    out.append(_pkg(m.group(1), m.group(2), "pypi"))
    return out


# ── go ─────────────────────────────────────────────────────────────────────


def parse_go_mod(text: str, path: str) -> list[Package]:
    out = []
    in_require = False
    for line in text.splitlines():
        line = line.strip()
        if line.startswith("require ("):
            in_require = True
            continue
        if in_require and line == ")":
            in_require = False
            continue
        m = None
        if in_require:
            m = re.match(r"^([^\s]+)\s+(v[^\s]+)", line)
        elif line.startswith("require "):
            m = re.match(r"^require\s+([^\s]+)\s+(v[^\s]+)", line)
        if m:
            indirect = "// indirect" in line
            out.append(_pkg(m.group(1), m.group(2), "go", direct=not indirect,
                            evidence="declaration_only"))
    return out


def parse_go_sum(text: str, path: str) -> list[Package]:
    out: dict[tuple, Package] = {}
    for line in text.splitlines():
        parts = line.split()
        if len(parts) >= 2 and parts[1].startswith("v") and not parts[1].endswith("/go.mod"):
            out[(parts[0], parts[1])] = _pkg(parts[0], parts[1], "go", direct=False)
    return list(out.values())


# ── rust ───────────────────────────────────────────────────────────────────


def parse_cargo_lock(text: str, path: str) -> list[Package]:
    from agentbom_amd.utils.compat import tomllib

    data = tomllib.loads(text)
    return [
        _pkg(e["name"], e.get("version", ""), "cargo")
        for e in data.get("package", []) if e.get("name") and e.get("version")
    ]


# ── jvm ────────────────────────────────────────────────────────────────────


def parse_pom_xml(text: str, path: str) -> list[Package]:
    import xml.etree.ElementTree as ET

    try:
        root = ET.fromstring(text)
    except ET.ParseError:
        return []
    ns = {"m": root.tag.split("}")[0].strip("{")} if root.tag.startswith("{") else {}
    pfx = "m:" if ns else ""
    out = []
    for dep in root.iter(f"{{{ns['m']}}}dependency" if ns else "dependency"):
        gid = dep.findtext(f"{pfx}groupId", "", ns)
        aid = dep.findtext(f"{pfx}artifactId", "", ns)
        ver = dep.findtext(f"{pfx}version", "", ns)
        scope = dep.findtext(f"{pfx}scope", "runtime", ns) or "runtime"
        if gid and aid and ver and not ver.startswith("${"):
            out.append(_pkg(f"{gid}:{aid}", ver, "maven", scope=scope,
                            evidence="declaration_only"))
    return out


def parse_gradle_lockfile(text: str, path: str) -> list[Package]:
    out = []
    for line in text.splitlines():
        line = line.strip()
        if line.startswith("#") or "=" not in line:
            continue
        coord = line.split("=", 1)[0]
        parts = coord.split(":")
        if len(parts) == 3:
            out.append(_pkg(f"{parts[0]}:{parts[1]}", parts[2], "maven"))
    return out


# ── dotnet ─────────────────────────────────────────────────────────────────


def parse_packages_lock_json(text: str, path: str) -> list[Package]:
    data = json.loads(text)
    out: dict[tuple, Package] = {}
    for _fw, deps in (data.get("dependencies") or {}).items():
        for name, meta in deps.items():
            version = meta.get("resolved", "") if isinstance(meta, dict) else ""
            if version:
                direct = (meta.get("type") == "Direct")
                out[(name, version)] = _pkg(name, version, "nuget", direct=direct)
    return list(out.values())


def parse_csproj(text: str, path: str) -> list[Package]:
    out = []
    for m in re.finditer(r'<PackageReference\s+Include="([^"]+)"\s+Version="([^"]+)"', text):
        out.append(_pkg(m.group(1), m.group(2), "nuget", evidence="declaration_only"))
    return out


# ── ruby / php / swift / hex / pub ────────────────────────────────────────


def parse_gemfile_lock(text: str, path: str) -> list[Package]:
    out = []
    in_specs = False
    for line in text.splitlines():
        if line.strip() == "specs:":
            in_specs = True
            continue
        if in_specs:
            if line and not line.startswith(" "):
                in_specs = False
                continue
            m = re.match(r"^    ([A-Za-z0-9._-]+) \(([^)]+)\)$", line)
            if m:
                out.append(_pkg(m.group(1), m.group(2), "rubygems"))
    return out


def parse_composer_lock(text: str, path: str) -> list[Package]:
    data = json.loads(text)
    out = []
    for section, scope in (("packages", "runtime"), ("packages-dev", "dev")):
        for e in data.get(section) or []:
            if e.get("name") and e.get("version"):
                out.append(_pkg(e["name"], e["version"].lstrip("v"), "composer", scope=scope))
    return out


def parse_package_resolved(text: str, path: str) -> list[Package]:
    data = json.loads(text)
    pins = data.get("pins") or data.get("object", {}).get("pins") or []
    out = []
    for pin in pins:
        name = pin.get("identity") or pin.get("package", "")
        version = (pin.get("state") or {}).get("version", "")
        if name and version:
            out.append(_pkg(name, version, "swifturl"))
    return out


def parse_mix_lock(text: str, path: str) -> list[Package]:
    out = []
    for m in re.finditer(r'"([a-z0-9_]+)":\s*\{:hex,\s*:[a-z0-9_]+,\s*"([^"]+)"', text):
        out.append(_pkg(m.group(1), m.group(2), "hex"))
    return out


def parse_pubspec_lock(text: str, path: str) -> list[Package]:
    import yaml

    try:
        data = yaml.safe_load(text) or {}
    except yaml.YAMLError:
        return []
    if not isinstance(data, dict):
        return []
    out = []
    for name, meta in (data.get("packages") or {}).items():
        version = meta.get("version", "") if isinstance(meta, dict) else ""
        if version:
            direct = (meta.get("dependency", "") or "").startswith("direct")
            out.append(_pkg(name, version, "pub", direct=direct))
    return out


# ── OS package databases ───────────────────────────────────────────────────


def parse_dpkg_status(text: str, path: str) -> list[Package]:
    out = []
    name = version = source = None
    installed = False
    for line in text.splitlines() + [""]:
        if line.startswith("Package:"):
            name = line.split(":", 1)[1].strip()
        elif line.startswith("Status:"):
            installed = "installed" in line
        elif line.startswith("Version:"):
            version = line.split(":", 1)[1].strip()
        elif line.startswith("Source:"):
            source = line.split(":", 1)[1].strip().split()[0]
        elif not line:
            if name and version and installed:
                p = _pkg(name, version, "deb")
                p.source_package = source
                p.distro_name = "debian"
                out.append(p)
            name = version = source = None
            installed = False
    return out


def parse_apk_installed(text: str, path: str) -> list[Package]:
    out = []
    name = version = None
    for line in text.splitlines() + [""]:
        if line.startswith("P:"):
            name = line[2:].strip()
        elif line.startswith("V:"):
            version = line[2:].strip()
        elif not line:
            if name and version:
                p = _pkg(name, version, "apk")
                p.distro_name = "alpine"
                out.append(p)
            name = version = None
    return out


# ── registry ───────────────────────────────────────────────────────────────

# (glob pattern, parser) — order matters: lockfiles before manifests.
BUILTIN_INVENTORY_PARSERS: list[tuple[str, ParserFn]] = [
    ("package-lock.json", parse_package_lock),
    ("npm-shrinkwrap.json", parse_package_lock),
    ("yarn.lock", parse_yarn_lock),
    ("pnpm-lock.yaml", parse_pnpm_lock),
    ("package.json", parse_package_json),
    ("requirements*.txt", parse_requirements_txt),
    ("poetry.lock", parse_poetry_lock),
    ("uv.lock", parse_uv_lock),
    ("Pipfile.lock", parse_pipfile_lock),
    ("pyproject.toml", parse_pyproject_toml),
    ("environment.yml", parse_conda_env),
    ("environment.yaml", parse_conda_env),
    ("go.mod", parse_go_mod),
    ("go.sum", parse_go_sum),
    ("Cargo.lock", parse_cargo_lock),
    ("pom.xml", parse_pom_xml),
    ("gradle.lockfile", parse_gradle_lockfile),
    ("packages.lock.json", parse_packages_lock_json),
    ("*.csproj", parse_csproj),
    ("Gemfile.lock", parse_gemfile_lock),
    ("composer.lock", parse_composer_lock),
    ("Package.resolved", parse_package_resolved),
    ("mix.lock", parse_mix_lock),
    ("pubspec.lock", parse_pubspec_lock),
    ("status", parse_dpkg_status),  # /var/lib/dpkg/status
    ("installed", parse_apk_installed),  # /lib/apk/db/installed
]

_SKIP_DIRS = {"node_modules", ".git", ".venv", "venv", "__pycache__", "dist", "build",
              "target", ".tox", "vendor"}


def extract_packages(root: str | Path, max_depth: int = 6) -> list[Package]:
    """Walk a tree and run every matching parser; dedup by identity.

    Lockfile-resolved packages win over manifest declarations of the same
    name (lockfiles are ordered first in the registry)."""
    root = Path(root)
    seen: dict[str, Package] = {}
    if root.is_file():
        candidates = [root]
    else:
        candidates = [
            p for p in root.rglob("*")
            if p.is_file()
            and len(p.relative_to(root).parts) <= max_depth
            and not (_SKIP_DIRS & set(p.relative_to(root).parts[:-1]))
        ]
    for pattern, parser in BUILTIN_INVENTORY_PARSERS:
        for path in candidates:
            if not path.match(pattern):
                continue
            # the bare names "status"/"installed" only apply at their db paths
            if pattern in ("status", "installed") and "dpkg" not in str(path) and "apk" not in str(path):
                continue
            try:
                for pkg in parser(path.read_text(errors="replace"), str(path)):
                    key = f"{pkg.ecosystem}:{pkg.name}@{pkg.version}"
                    if key not in seen:
                        seen[key] = pkg
            except Exception:  # noqa: BLE001 — per-file parse boundary
                continue
    return list(seen.values())
