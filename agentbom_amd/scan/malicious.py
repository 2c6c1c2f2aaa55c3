"""Typosquat / dependency-confusion / known-malicious package detection.

Reference: src/agent_bom/malicious.py (curated typosquat lists + heuristics).
Fails closed: a malicious hit always forces a non-zero scan exit regardless
of --exit-zero (site-docs/reference/exit-codes.md, "fails closed" gates).
"""

from __future__ import annotations

from typing import Optional

# Curated known typosquats -> the legitimate package they imitate.
KNOWN_TYPOSQUATS: dict[tuple[str, str], str] = {
    ("pypi", "reqeusts"): "requests",
    ("pypi", "requets"): "requests",
    ("pypi", "request"): "requests",
    ("pypi", "urlib3"): "urllib3",
    ("pypi", "urllib"): "urllib3",
    ("pypi", "python-sqlite"): "pysqlite3",
    ("pypi", "beautifulsoup"): "beautifulsoup4",
    ("pypi", "pytorch"): "torch",
    ("pypi", "tensorflowjs"): "tensorflow",
    ("pypi", "colourama"): "colorama",
    ("pypi", "djanga"): "django",
    ("pypi", "crypt"): "cryptography",
    ("npm", "crossenv"): "cross-env",
    ("npm", "lodahs"): "lodash",
    ("npm", "loadsh"): "lodash",
    ("npm", "expres"): "express",
    ("npm", "mangoose"): "mongoose",
    ("npm", "babelcli"): "babel-cli",
    ("npm", "momnet"): "moment",
    ("npm", "jquery.js"): "jquery",
}

# High-download packages used for edit-distance-1 screening.
POPULAR_PACKAGES: dict[str, frozenset[str]] = {
    "pypi": frozenset({
        "requests", "urllib3", "numpy", "pandas", "django", "flask", "boto3",
        "cryptography", "pyyaml", "pillow", "setuptools", "pip", "torch",
        "scipy", "matplotlib", "sqlalchemy", "celery", "pydantic", "httpx",
    }),
    "npm": frozenset({
        "react", "lodash", "express", "axios", "moment", "chalk", "webpack",
        "typescript", "jquery", "vue", "next", "eslint", "jest", "mongoose",
    }),
}

# Internal-looking scope/prefix patterns for dependency-confusion screening.
_INTERNAL_HINTS = ("internal", "corp", "private", "intranet")


def _edit_distance_1(a: str, b: str) -> bool:
    """True when a and b differ by one substitution/insertion/deletion/swap."""
    if a == b:
        return False
    la, lb = len(a), len(b)
    if abs(la - lb) > 1:
        return False
    if la == lb:
        diffs = [i for i in range(la) if a[i] != b[i]]
        if len(diffs) == 1:
            return True
        if len(diffs) == 2 and diffs[1] == diffs[0] + 1:
            i, j = diffs
            return a[i] == b[j] and a[j] == b[i]  # adjacent transposition
        return False
    if la > lb:
        a, b = b, a
        la, lb = lb, la
    # a shorter: one deletion from b
    i = 0
    while i < la and a[i] == b[i]:
        i += 1
    return a[i:] == b[i + 1:]


def check_typosquat(name: str, ecosystem: str) -> Optional[str]:
    """Return the imitated package name when ``name`` looks like a typosquat."""
    eco = (ecosystem or "").lower()
    lname = (name or "").lower()
    hit = KNOWN_TYPOSQUATS.get((eco, lname))
    if hit:
        return hit
    popular = POPULAR_PACKAGES.get(eco, frozenset())
    if lname in popular:
        return None
    for target in popular:
        if _edit_distance_1(lname, target):
            return target
    return None


def check_dependency_confusion(name: str, ecosystem: str) -> Optional[str]:
    """Flag unscoped names that look like internal packages (npm heuristic)."""
    lname = (name or "").lower()
    if any(h in lname for h in _INTERNAL_HINTS) and not lname.startswith("@"):
        return f"unscoped internal-looking name {name!r} is a dependency-confusion risk"
    return None


def flag_malicious_packages(packages) -> int:
    """Stamp is_malicious/malicious_reason in place; returns hit count."""
    hits = 0
    for pkg in packages:
        target = check_typosquat(pkg.name, pkg.ecosystem)
        if target:
            pkg.is_malicious = True
            pkg.malicious_reason = (
                f"typosquat of {target!r} ({pkg.ecosystem}) — likely malicious package"
            )
            hits += 1
            continue
        confusion = check_dependency_confusion(pkg.name, pkg.ecosystem)
        if confusion:
            pkg.is_malicious = True
            pkg.malicious_reason = confusion
            hits += 1
    return hits
