"""Per-ecosystem version ordering + fail-closed advisory range resolution.

Semantics-parity rebuild of the reference's comparator stack
(reference: src/agent_bom/version_utils.py:26-1315).  Implements the
published ordering algorithms directly:

- Debian ``dpkg --compare-versions`` (epoch / ``~`` / alpha-vs-digit runs)
- RPM ``rpmvercmp`` incl. ``~`` pre-release and ``^`` post-release markers
- Alpine apk (_alpha < _beta < _pre < _rc < release < _cvs.._p, -rN revision)
- Go modules (pseudo-version timestamps; pseudo < its base tag)
- Maven ComparableVersion (qualifier ranks, nested lists, null trimming)
- PHP ``version_compare`` for Packagist/Composer
- NuGet (4-part numeric, case-insensitive labels)
- RubyGems ``Gem::Version`` canonical segments
- strict SemVer for npm-family, PEP 440 (``packaging``) for PyPI and fallback

Matching invariants preserved: ``introduced: "0"``/missing is
unbounded; commit-SHA bounds are undecidable → window dropped with a warning;
unparseable bounds fail CLOSED; npm prerelease sorts strictly below its
release; version-vs-bound comparisons that cannot be performed never match.

The GPU bulk matcher (ops/csrc/match.hip) consumes order-preserving u128 keys
produced by :mod:`agentbom_amd.utils.version_keys`; any version this module
can compare but the encoder cannot encode falls back to this CPU comparator —
bit-identical decisions either way (tested in tests/test_version_keys.py).
"""

from __future__ import annotations

import logging
import re
from functools import lru_cache
from typing import Optional

_log = logging.getLogger(__name__)

# ── Validation patterns ─────────────────────────────────────────────────────

_SEMVER_RE = re.compile(
    r"^v?(0|[1-9]\d*)\.(0|[1-9]\d*)\.(0|[1-9]\d*)"
    r"(?:-([0-9A-Za-z\-.]+))?(?:\+([0-9A-Za-z\-.]+))?$"
)
_PEP440_RE = re.compile(
    r"^v?(\d+!)?(\d+)(?:\.(\d+))?(?:\.(\d+))?"
    r"(?:(a|alpha|b|beta|c|rc|pre|preview)\d*)?"
    r"(?:\.?(post|rev|r)\d*)?(?:\.?(dev)\d*)?$",
    re.IGNORECASE,
)
_GO_VERSION_RE = re.compile(r"^v?\d+\.\d+\.\d+(?:-[0-9A-Za-z\-.]+)?(?:\+[0-9A-Za-z\-.]+)?$")
_MAVEN_RE = re.compile(r"^\d+(?:\.\d+){0,3}(?:[.-][A-Za-z0-9\-.]+)?$")
# All three Go pseudo-version forms (go.dev/ref/mod#pseudo-versions).
_GO_PSEUDO_RE = re.compile(r"^v?\d+\.\d+\.\d+-(?:[0-9A-Za-z.\-]+\.)?(\d{14})-[0-9a-f]{12}$")
_HEXISH_RE = re.compile(r"^[0-9a-f]{7,40}$")


def validate_version(version: str, ecosystem: str) -> bool:
    if not version or version in ("latest", "unknown"):
        return False
    if ecosystem in ("npm", "cargo", "nuget"):
        return _SEMVER_RE.match(version) is not None
    if ecosystem == "pypi":
        return _PEP440_RE.match(version) is not None
    if ecosystem == "go":
        return _GO_VERSION_RE.match(version) is not None
    if ecosystem == "maven":
        return _MAVEN_RE.match(version) is not None
    if ecosystem in ("deb", "apk", "rpm"):
        return bool(version.strip())
    return True


def normalize_version(version: str, ecosystem: str) -> str:
    """Trim, strip leading 'v' (non-Go), normalize PyPI pre/post/dev tags."""
    version = version.strip()
    if not version or version in ("latest", "unknown"):
        return version
    if ecosystem != "go" and version.startswith("v"):
        version = version[1:]
    if ecosystem == "pypi":
        version = re.sub(r"\.?(alpha|a)(\d+)?", r"a\2", version, flags=re.IGNORECASE)
        version = re.sub(r"\.?(beta|b)(\d+)?", r"b\2", version, flags=re.IGNORECASE)
        version = re.sub(r"\.?(preview|rc)(\d+)?", r"rc\2", version, flags=re.IGNORECASE)
        version = re.sub(r"(?<![a-z])c(\d+)", r"rc\1", version, flags=re.IGNORECASE)
        version = re.sub(r"\.?(post|rev)(\d+)?", r".post\2", version, flags=re.IGNORECASE)
        version = re.sub(r"\.?r(\d+)", r".post\1", version, flags=re.IGNORECASE)
        version = re.sub(r"\.?(dev)(\d+)?", r".dev\2", version, flags=re.IGNORECASE)
    return version


def strip_pip_extras(name: str) -> tuple[str, str]:
    """``requests[security]==2.31.0`` → (``requests``, ``2.31.0``)."""
    name = re.sub(r"\[.*?\]", "", name)
    m = re.match(r"^([a-zA-Z0-9._-]+)\s*(?:[>=<~!]+\s*)?(.*)$", name)
    if m:
        return m.group(1).strip(), m.group(2).strip()
    return name.strip(), ""


def _looks_like_commit_sha(version: str) -> bool:
    s = version.strip().lower().lstrip("v")
    if not _HEXISH_RE.fullmatch(s):
        return False
    # All-digit tokens (date-stamped OS packages: 20230311) are versions, not
    # abbreviated SHAs — failing them closed would hide real CVEs.
    if s.isdigit() and len(s) != 40:
        return False
    return True


# ── Go ──────────────────────────────────────────────────────────────────────


def _go_pseudo_timestamp(version: str) -> Optional[str]:
    m = _GO_PSEUDO_RE.match(version)
    return m.group(1) if m else None


def _go_base(version: str) -> str:
    if _go_pseudo_timestamp(version):
        version = version.split("-", 1)[0]
    return version[1:] if version.startswith("v") else version


def _cmp(a, b) -> int:
    return (a > b) - (a < b)


def _compare_go(left: str, right: str) -> Optional[int]:
    lts, rts = _go_pseudo_timestamp(left), _go_pseudo_timestamp(right)
    if lts and rts:
        return _cmp(lts, rts)
    try:
        from packaging.version import Version

        base = _cmp(Version(_go_base(left)), Version(_go_base(right)))
        if base:
            return base
        # pseudo-version is a pre-release of its base: strictly below the tag
        if bool(lts) != bool(rts):
            return -1 if lts else 1
        return 0
    except Exception:
        return None


# ── Debian (dpkg) ───────────────────────────────────────────────────────────


def _deb_char_order(ch: Optional[str]) -> int:
    if ch is None:
        return 0
    if ch == "~":
        return -1
    if ch.isalpha():
        return ord(ch)
    return ord(ch) + 256


def _deb_cmp_part(left: str, right: str) -> int:
    i = j = 0
    while i < len(left) or j < len(right):
        # non-digit run
        while (i < len(left) and not left[i].isdigit()) or (j < len(right) and not right[j].isdigit()):
            lc = left[i] if i < len(left) and not left[i].isdigit() else None
            rc = right[j] if j < len(right) and not right[j].isdigit() else None
            if lc == rc:
                if lc is not None:
                    i += 1
                if rc is not None:
                    j += 1
                continue
            d = _cmp(_deb_char_order(lc), _deb_char_order(rc))
            if d:
                return d
            if lc is not None:
                i += 1
            if rc is not None:
                j += 1
        # digit run
        ld = ""
        while i < len(left) and left[i].isdigit():
            ld += left[i]
            i += 1
        rd = ""
        while j < len(right) and right[j].isdigit():
            rd += right[j]
            j += 1
        ld = ld.lstrip("0") or "0"
        rd = rd.lstrip("0") or "0"
        if len(ld) != len(rd):
            return _cmp(len(ld), len(rd))
        if ld != rd:
            return _cmp(ld, rd)
    return 0


def _deb_split(version: str) -> tuple[int, str, str]:
    epoch_str, _, rest = version.partition(":")
    if rest:
        try:
            epoch = int(epoch_str)
        except ValueError:
            epoch = 0
    else:
        epoch, rest = 0, version
    if "-" in rest:
        upstream, revision = rest.rsplit("-", 1)
    else:
        upstream, revision = rest, "0"
    return epoch, upstream, revision


def _compare_deb(left: str, right: str) -> int:
    le, lu, lr = _deb_split(left)
    re_, ru, rr = _deb_split(right)
    if le != re_:
        return _cmp(le, re_)
    d = _deb_cmp_part(lu, ru)
    if d:
        return d
    return _deb_cmp_part(lr, rr)


# ── RPM (rpmvercmp) ─────────────────────────────────────────────────────────


def _rpm_segment(value: str, start: int) -> tuple[str, int]:
    end = start
    kind = value[start].isdigit()
    while end < len(value) and value[end].isdigit() == kind and value[end].isalnum():
        end += 1
    return value[start:end], end


def _rpmvercmp(left: str, right: str) -> int:
    i = j = 0
    while True:
        while i < len(left) and not left[i].isalnum() and left[i] not in "~^":
            i += 1
        while j < len(right) and not right[j].isalnum() and right[j] not in "~^":
            j += 1

        l_tilde = i < len(left) and left[i] == "~"
        r_tilde = j < len(right) and right[j] == "~"
        if l_tilde or r_tilde:
            if not l_tilde:
                return 1
            if not r_tilde:
                return -1
            i += 1
            j += 1
            continue

        l_caret = i < len(left) and left[i] == "^"
        r_caret = j < len(right) and right[j] == "^"
        if l_caret or r_caret:
            if i >= len(left):
                return -1
            if j >= len(right):
                return 1
            if not l_caret:
                return 1
            if not r_caret:
                return -1
            i += 1
            j += 1
            continue

        if i >= len(left) or j >= len(right):
            break

        lseg, i = _rpm_segment(left, i)
        rseg, j = _rpm_segment(right, j)
        l_num = lseg[0].isdigit()
        r_num = rseg[0].isdigit()
        if l_num != r_num:
            return 1 if l_num else -1
        if l_num:
            ln = lseg.lstrip("0") or "0"
            rn = rseg.lstrip("0") or "0"
            if len(ln) != len(rn):
                return _cmp(len(ln), len(rn))
            if ln != rn:
                return _cmp(ln, rn)
        else:
            if lseg != rseg:
                return _cmp(lseg, rseg)

    if i >= len(left) and j >= len(right):
        return 0
    return -1 if i >= len(left) else 1


def _split_epoch(version: str) -> tuple[int, str]:
    epoch_str, sep, rest = version.partition(":")
    if not sep:
        return 0, version
    try:
        return int(epoch_str), rest
    except ValueError:
        return 0, version


def _compare_rpm(left: str, right: str) -> int:
    le, lrest = _split_epoch(left)
    re_, rrest = _split_epoch(right)
    if le != re_:
        return _cmp(le, re_)
    return _rpmvercmp(lrest, rrest)


# ── Alpine apk ──────────────────────────────────────────────────────────────

_APK_PRE = ("alpha", "beta", "pre", "rc")
_APK_POST = ("cvs", "svn", "git", "hg", "p")
_APK_SUFFIX_RE = re.compile(r"_([a-z]+)(\d*)")


def _apk_suffix_rank(name: str) -> int:
    if name in _APK_PRE:
        return _APK_PRE.index(name) - len(_APK_PRE)
    if name in _APK_POST:
        return _APK_POST.index(name) + 1
    return 0


def _apk_suffix_key(suffix: str) -> list[tuple[int, int]]:
    return [(_apk_suffix_rank(n), int(num) if num else 0) for n, num in _APK_SUFFIX_RE.findall(suffix)]


def _apk_cmp_suffix_keys(left: list[tuple[int, int]], right: list[tuple[int, int]]) -> int:
    # An exhausted side IS the release: outranks pre-, outranked by post-.
    for i in range(max(len(left), len(right))):
        if i >= len(left):
            rank = right[i][0]
            return 1 if rank < 0 else -1 if rank > 0 else 0
        if i >= len(right):
            rank = left[i][0]
            return -1 if rank < 0 else 1 if rank > 0 else 0
        if left[i] != right[i]:
            return _cmp(left[i], right[i])
    return 0


def _compare_apk(left: str, right: str) -> int:
    def split_rev(v: str) -> tuple[str, int]:
        if "-r" in v:
            base, rev = v.rsplit("-r", 1)
            try:
                return base, int(rev)
            except ValueError:
                return base, 0
        return v, 0

    lb, lrev = split_rev(left)
    rb, rrev = split_rev(right)
    lcore, _, lhash = lb.partition("~")
    rcore, _, rhash = rb.partition("~")
    li = lcore.find("_")
    ri = rcore.find("_")
    lmain, lsuf = (lcore, "") if li == -1 else (lcore[:li], lcore[li:])
    rmain, rsuf = (rcore, "") if ri == -1 else (rcore[:ri], rcore[ri:])
    d = _rpmvercmp(lmain, rmain)
    if d:
        return d
    d = _apk_cmp_suffix_keys(_apk_suffix_key(lsuf), _apk_suffix_key(rsuf))
    if d:
        return d
    if lhash != rhash:
        return _cmp(lhash, rhash)
    return _cmp(lrev, rrev)


# ── Maven ComparableVersion ─────────────────────────────────────────────────

_MVN_QUALIFIERS = ("alpha", "beta", "milestone", "rc", "snapshot", "", "sp")
_MVN_RELEASE_KEY = str(_MVN_QUALIFIERS.index(""))
_MVN_ALIASES = {"ga": "", "final": "", "release": "", "cr": "rc"}
_MVN_SHORT = {"a": "alpha", "b": "beta", "m": "milestone"}


def _mvn_qkey(q: str) -> str:
    try:
        return str(_MVN_QUALIFIERS.index(q))
    except ValueError:
        return f"{len(_MVN_QUALIFIERS)}-{q}"


class _MvnInt:
    __slots__ = ("value",)

    def __init__(self, raw: str) -> None:
        self.value = int(raw or "0")

    def is_null(self) -> bool:
        return self.value == 0


class _MvnStr:
    __slots__ = ("value",)

    def __init__(self, raw: str, followed_by_digit: bool) -> None:
        v = raw
        if followed_by_digit and len(v) == 1:
            v = _MVN_SHORT.get(v, v)
        self.value = _MVN_ALIASES.get(v, v)

    def is_null(self) -> bool:
        return _mvn_qkey(self.value) == _MVN_RELEASE_KEY


class _MvnList(list):
    def is_null(self) -> bool:
        return len(self) == 0

    def trim_nulls(self) -> None:
        for i in range(len(self) - 1, -1, -1):
            item = self[i]
            if item.is_null():
                del self[i]
            elif not isinstance(item, _MvnList):
                break


def _mvn_rank(item) -> int:
    # cross-type order: qualifier < list < numeric
    if isinstance(item, _MvnStr):
        return 0
    if isinstance(item, _MvnList):
        return 1
    return 2


def _mvn_cmp(left, right) -> int:
    if left is None and right is None:
        return 0
    if left is None:
        return -_mvn_cmp(right, None)
    if right is None:
        if isinstance(left, _MvnInt):
            return 0 if left.value == 0 else 1
        if isinstance(left, _MvnStr):
            k = _mvn_qkey(left.value)
            return _cmp(k, _MVN_RELEASE_KEY)
        if isinstance(left, _MvnList):
            return 0 if not left else _mvn_cmp(left[0], None)
        return 0
    lr, rr = _mvn_rank(left), _mvn_rank(right)
    if lr != rr:
        return _cmp(lr, rr)
    if isinstance(left, _MvnInt):
        return _cmp(left.value, right.value)
    if isinstance(left, _MvnStr):
        return _cmp(_mvn_qkey(left.value), _mvn_qkey(right.value))
    for i in range(max(len(left), len(right))):
        li = left[i] if i < len(left) else None
        ri = right[i] if i < len(right) else None
        d = _mvn_cmp(li, ri)
        if d:
            return d
    return 0


def _mvn_parse(version: str) -> _MvnList:
    version = version.strip().lower()
    root = _MvnList()
    current = root
    stack = [root]
    is_digit = False
    start = 0

    def make(digit: bool, buf: str):
        return _MvnInt(buf.lstrip("0") or "0") if digit else _MvnStr(buf, False)

    def descend() -> None:
        nonlocal current
        nested = _MvnList()
        current.append(nested)
        current = nested
        stack.append(nested)

    for idx, ch in enumerate(version):
        if ch == ".":
            current.append(_MvnInt("0") if idx == start else make(is_digit, version[start:idx]))
            start = idx + 1
        elif ch in "-_":
            current.append(_MvnInt("0") if idx == start else make(is_digit, version[start:idx]))
            start = idx + 1
            descend()
        elif ch.isdigit():
            if not is_digit and idx > start:
                # char→digit transition: qualifier token, short-form aware
                if current:
                    descend()
                current.append(_MvnStr(version[start:idx], True))
                start = idx
                descend()
            is_digit = True
        else:
            if is_digit and idx > start:
                # digit→char transition ≡ hyphen
                current.append(make(True, version[start:idx]))
                start = idx
                descend()
            is_digit = False

    if len(version) > start:
        # Trailing qualifier nests one level (``2.0.a`` is ``2, [a]``).
        if not is_digit and current:
            descend()
        current.append(make(is_digit, version[start:]))

    while stack:
        stack.pop().trim_nulls()
    return root


def _compare_maven(left: str, right: str) -> int:
    return _mvn_cmp(_mvn_parse(left), _mvn_parse(right))


# ── PHP version_compare (Packagist/Composer) ────────────────────────────────

_PHP_ORDER = {"dev": 0, "alpha": 1, "a": 1, "beta": 2, "b": 2, "RC": 3, "rc": 3, "#": 4, "pl": 5, "p": 5}
_PHP_UNLISTED = -1


def _php_canon(version: str) -> str:
    if version[:1] in ("v", "V"):
        version = version[1:]
    version = re.sub(r"[-_+.]+", ".", version)
    version = re.sub(r"([^\d.])(\d)", r"\1.\2", version)
    return re.sub(r"(\d)([^\d.])", r"\1.\2", version)


def _php_part_cmp(left: str, right: str) -> int:
    return _cmp(_PHP_ORDER.get(left, _PHP_UNLISTED), _PHP_ORDER.get(right, _PHP_UNLISTED))


def _php_slices_cmp(left: list[str], right: list[str]) -> int:
    for lp, rp in zip(left, right):
        ln, rn = lp.isdecimal(), rp.isdecimal()
        if ln and rn:
            d = _cmp(int(lp), int(rp))
        elif not ln and not rn:
            d = _php_part_cmp(lp, rp)
        elif ln:
            d = _php_part_cmp("#", rp)
        else:
            d = _php_part_cmp(lp, "#")
        if d:
            return d
    if len(left) > len(right):
        tail = left[len(right):]
        return 1 if tail[0].isdecimal() else _php_slices_cmp(tail, ["#"])
    if len(left) < len(right):
        tail = right[len(left):]
        return -1 if tail[0].isdecimal() else _php_slices_cmp(["#"], tail)
    return 0


def _compare_php(left: str, right: str) -> int:
    return _php_slices_cmp(_php_canon(left).split("."), _php_canon(right).split("."))


# ── NuGet ───────────────────────────────────────────────────────────────────

_NUGET_RE = re.compile(r"^(\d+(?:\.\d+){0,3})(?:-([0-9A-Za-z.\-]+))?(?:\+[0-9A-Za-z.\-]+)?$")


def _nuget_parse(version: str):
    v = version.strip()
    if v[:1] in ("v", "V"):
        v = v[1:]
    m = _NUGET_RE.match(v)
    if m is None:
        return None
    nums = [int(p) for p in m.group(1).split(".")]
    nums.extend([0] * (4 - len(nums)))
    pre = m.group(2)
    labels = tuple(lbl.lower() for lbl in pre.split(".")) if pre else ()
    return tuple(nums), labels


def _nuget_label_cmp(left: str, right: str) -> int:
    ln, rn = left.isdigit(), right.isdigit()
    if ln and rn:
        return _cmp(int(left), int(right))
    if ln != rn:
        return -1 if ln else 1
    return _cmp(left, right)


def _compare_nuget(left: str, right: str) -> Optional[int]:
    lp, rp = _nuget_parse(left), _nuget_parse(right)
    if lp is None or rp is None:
        return None
    if lp[0] != rp[0]:
        return _cmp(lp[0], rp[0])
    ll, rl = lp[1], rp[1]
    if not ll or not rl:
        if ll == rl:
            return 0
        return 1 if not ll else -1
    for a, b in zip(ll, rl):
        d = _nuget_label_cmp(a, b)
        if d:
            return d
    return _cmp(len(ll), len(rl))


# ── RubyGems ────────────────────────────────────────────────────────────────

_GEM_RE = re.compile(r"^\s*(?:[0-9]+(?:\.[0-9a-zA-Z]+)*(?:-[0-9A-Za-z-]+(?:\.[0-9A-Za-z-]+)*)?)?\s*$")
_GEM_SEG_RE = re.compile(r"[0-9]+|[a-z]+", re.IGNORECASE)


def _gem_canonical(version: str):
    if _GEM_RE.match(version) is None:
        return None
    expanded = version.strip().replace("-", ".pre.")
    segs: list = [int(s) if s.isdigit() else s for s in _GEM_SEG_RE.findall(expanded)]
    split_at = next((i for i, s in enumerate(segs) if isinstance(s, str)), len(segs))
    canonical: list = []
    for group in (segs[:split_at], segs[split_at:]):
        end = len(group)
        while end and group[end - 1] == 0:
            end -= 1
        canonical.extend(group[:end])
    return canonical


def _compare_gem(left: str, right: str) -> Optional[int]:
    ls, rs = _gem_canonical(left), _gem_canonical(right)
    if ls is None or rs is None:
        return None
    if ls == rs:
        return 0
    for i in range(max(len(ls), len(rs))):
        a = ls[i] if i < len(ls) else 0
        b = rs[i] if i < len(rs) else 0
        if a == b:
            continue
        if isinstance(a, str) and isinstance(b, int):
            return -1
        if isinstance(a, int) and isinstance(b, str):
            return 1
        return 1 if a > b else -1
    return 0


# ── Strict SemVer (npm family) ──────────────────────────────────────────────

_STRICT_SEMVER = re.compile(
    r"^v?(\d+)\.(\d+)\.(\d+)(?:-([0-9A-Za-z-]+(?:\.[0-9A-Za-z-]+)*))?"
    r"(?:\+[0-9A-Za-z-]+(?:\.[0-9A-Za-z-]+)*)?$"
)


def _compare_strict_semver(left: str, right: str) -> Optional[int]:
    lm = _STRICT_SEMVER.fullmatch(left.strip())
    rm = _STRICT_SEMVER.fullmatch(right.strip())
    if lm is None or rm is None:
        return None
    lb = tuple(int(lm.group(i)) for i in (1, 2, 3))
    rb = tuple(int(rm.group(i)) for i in (1, 2, 3))
    if lb != rb:
        return _cmp(lb, rb)
    lpre, rpre = lm.group(4), rm.group(4)
    if lpre is None or rpre is None:
        if lpre == rpre:
            return 0
        return -1 if lpre is not None else 1  # prerelease < release
    lparts, rparts = lpre.split("."), rpre.split(".")
    for a, b in zip(lparts, rparts):
        if a == b:
            continue
        an, bn = a.isdigit(), b.isdigit()
        if an and bn:
            return _cmp(int(a), int(b))
        if an != bn:
            return -1 if an else 1
        return _cmp(a, b)
    return _cmp(len(lparts), len(rparts))


_SEMVER_PRE_TAGS = frozenset(
    {"canary", "beta", "alpha", "rc", "pre", "dev", "nightly", "next", "snapshot", "m", "preview"}
)


def _strip_semver_pre_tag(version: str) -> str:
    base, sep, suffix = version.partition("-")
    if not sep:
        return version
    tag = suffix.split(".", 1)[0].lower()
    return base if tag in _SEMVER_PRE_TAGS else version


def _split_local_suffix(version: str, ecosystem: str):
    """Split a local-version-style suffix (``2.6.0-cu124``/``-NA``) off."""
    from packaging.version import Version

    try:
        Version(normalize_version(version, ecosystem))
        return version, False
    except Exception:
        pass
    for sep in ("+", "-"):
        base, s, _ = version.partition(sep)
        if not s or not base:
            continue
        try:
            Version(normalize_version(base, ecosystem))
        except Exception:
            continue
        return base, True
    return None


def _compare_local_suffix_strip(left: str, right: str, eco: str) -> Optional[int]:
    ls = _split_local_suffix(left, eco)
    rs = _split_local_suffix(right, eco)
    if ls is None or rs is None:
        return None
    lbase, lsuf = ls
    rbase, rsuf = rs
    if not (lsuf or rsuf):
        return None
    from packaging.version import Version

    lv = Version(normalize_version(lbase, eco))
    rv = Version(normalize_version(rbase, eco))
    if lv != rv:
        return 1 if lv > rv else -1
    # Equal base: the suffixed side orders ABOVE the bare release (PEP 440
    # local-version ordering) — a bare version never reads past a suffixed fix.
    if lsuf and not rsuf:
        return 1
    if rsuf and not lsuf:
        return -1
    return 0


# ── Dispatch ────────────────────────────────────────────────────────────────

_PACKAGIST = frozenset({"packagist", "composer", "php"})
_RUBY = frozenset({"rubygems", "gem", "gems"})
_NPMISH = frozenset({"npm", "npmjs", "yarn", "pnpm", "node", "javascript", "js"})

ECO_ALIASES = {"debian": "deb", "alpine": "apk", "linux": "rpm", "golang": "go"}


@lru_cache(maxsize=65536)
def compare_version_order(left: str, right: str, ecosystem: str) -> Optional[int]:
    """-1 / 0 / 1, or ``None`` when the pair cannot be ordered (fail-closed)."""
    eco = ECO_ALIASES.get((ecosystem or "").lower(), (ecosystem or "").lower())
    left = (left or "").strip()
    right = (right or "").strip()
    if not left or not right:
        return None
    if _looks_like_commit_sha(left) or _looks_like_commit_sha(right):
        return None

    if eco == "deb":
        return _compare_deb(left, right)
    if eco == "rpm":
        return _compare_rpm(left, right)
    if eco == "apk":
        return _compare_apk(left, right)
    if eco == "go":
        return _compare_go(left, right)
    if eco == "maven":
        return _compare_maven(left, right)
    if eco in _PACKAGIST:
        return _compare_php(left, right)
    if eco == "nuget":
        return _compare_nuget(left, right)
    if eco in _RUBY:
        return _compare_gem(left, right)

    if eco in _NPMISH:
        sv = _compare_strict_semver(left, right)
        if sv is not None:
            return sv

    # PEP 440 primary path; SemVer-prerelease and local-suffix fallbacks keep
    # npm-style tags (13.4.20-canary.13) and wheel-local bounds comparable.
    try:
        from packaging.version import Version

        ln = normalize_version(left, eco)
        rn = normalize_version(right, eco)
        return _cmp(Version(ln), Version(rn))
    except Exception:
        try:
            from packaging.version import Version

            lstrip = _strip_semver_pre_tag(left)
            rstrip = _strip_semver_pre_tag(right)
            lpre, rpre = lstrip != left, rstrip != right
            if not lpre and not rpre:
                return _compare_local_suffix_strip(left, right, eco)
            base = _cmp(
                Version(normalize_version(lstrip, eco)), Version(normalize_version(rstrip, eco))
            )
            if base != 0:
                return base
            # Prerelease sorts strictly below its release.
            if lpre and not rpre:
                return -1
            if rpre and not lpre:
                return 1
            return 0
        except Exception:
            return _compare_local_suffix_strip(left, right, eco)


def compare_versions(current: str, fixed: str, ecosystem: str) -> bool:
    """True when ``fixed`` is newer than ``current`` (upgrade needed)."""
    order = compare_version_order(current, fixed, ecosystem)
    if order is not None:
        return order < 0

    def vt(v: str) -> tuple[tuple[int, ...], bool]:
        is_pre = bool(re.search(r"(alpha|beta|rc|dev|pre|preview|[ab]\d)", v, re.IGNORECASE))
        parts = re.findall(
            r"\d+", re.split(r"[-]|(?:alpha|beta|rc|dev|pre|preview)", v, flags=re.IGNORECASE)[0]
        )
        return (tuple(int(p) for p in parts) if parts else (0,)), is_pre

    try:
        cn, cp = vt(current)
        fn, fp = vt(fixed)
        if fn != cn:
            return fn > cn
        if cp and not fp:
            return True
        return False
    except (ValueError, TypeError):
        return False


def is_prerelease_version(version: str, ecosystem: str) -> bool:
    if not version:
        return False
    eco = ecosystem.lower()
    candidate = version if eco == "go" else version.lstrip("v")
    try:
        from packaging.version import Version

        return Version(candidate).is_prerelease
    except Exception:
        pass
    normalized = normalize_version(version, ecosystem)
    candidate = normalized if eco == "go" else normalized.lstrip("v")
    try:
        from packaging.version import Version

        return Version(candidate).is_prerelease
    except Exception:
        pass
    if eco == "maven":
        return bool(re.search(r"-(snapshot|rc\d*|m\d+|alpha|beta|pre|preview|canary)", candidate, re.IGNORECASE))
    return bool(re.search(r"(?:-|\.)(alpha|beta|rc|pre|preview|canary|dev)\d*(?:$|\+)", candidate, re.IGNORECASE))


# ── Range resolution (fail-closed) ──────────────────────────────────────────

# Scan-boundary warning sink: the scan orchestrator registers a callback so
# dropped bounds surface in the report, not only the log.
_scan_warning_sink = None


def set_scan_warning_sink(fn) -> None:
    global _scan_warning_sink
    _scan_warning_sink = fn


def _dropped_bound_message(bound: str, ecosystem: str) -> str:
    return (
        f"advisory version bound {bound!r} ({ecosystem}) could not be compared; "
        "the affected range was dropped, so results may under-report"
    )


@lru_cache(maxsize=4096)
def _log_unparseable_bound(bound: str, ecosystem: str) -> None:
    _log.warning(
        "Advisory version bound %r (%s) could not be compared; failing closed — the bound "
        "cannot establish a match, so affected-range accuracy may be reduced",
        bound,
        ecosystem,
    )


def _warn_unparseable_bound(bound: str, ecosystem: str) -> None:
    _log_unparseable_bound(bound, ecosystem)
    if _scan_warning_sink is not None:
        _scan_warning_sink(_dropped_bound_message(bound, ecosystem))


def normalize_introduced(introduced: Optional[str]) -> Optional[str]:
    """OSV ``introduced: "0"`` is a sentinel (sorts before everything), not a
    version — treat as unbounded so Go pseudo-versions below 0.0.0 match."""
    if introduced is None:
        return None
    bound = introduced.strip()
    if not bound or bound == "0":
        return None
    return bound


def version_in_range(
    version: str,
    introduced: Optional[str],
    fixed: Optional[str],
    last_affected: Optional[str],
    ecosystem: str,
) -> bool:
    """Whether ``version`` falls in the advisory window.  Fail-closed: a bound
    that cannot be compared never establishes a match; dropped bounds are
    reported every scan (warning sink) but decisions are memoised."""
    affected, dropped = _resolve_version_range(version, introduced, fixed, last_affected, ecosystem)
    for bound in dropped:
        _warn_unparseable_bound(bound, ecosystem)
    return affected


@lru_cache(maxsize=65536)
def _resolve_version_range(
    version: str,
    introduced: Optional[str],
    fixed: Optional[str],
    last_affected: Optional[str],
    ecosystem: str,
) -> tuple[bool, tuple[str, ...]]:
    dropped: list[str] = []
    intro = normalize_introduced(introduced)
    fix = fixed or None
    last = last_affected or None

    # Git-commit bounds cannot establish range membership — drop the window.
    if any(b and _looks_like_commit_sha(b) for b in (intro, fix, last)):
        return False, ()

    if ecosystem.lower() == "go":
        ver_ts = _go_pseudo_timestamp(version)
        if ver_ts:
            for boundary, is_lower in ((intro, True), (fix, False), (last, False)):
                if not boundary:
                    continue
                bts = _go_pseudo_timestamp(boundary)
                cmp_: Optional[int]
                if bts:
                    cmp_ = _cmp(ver_ts, bts)
                else:
                    cmp_ = compare_version_order(version, boundary, ecosystem)
                if cmp_ is None:
                    continue
                if is_lower and cmp_ < 0:
                    return False, ()
                if not is_lower and boundary == fix and cmp_ >= 0:
                    return False, ()
                if not is_lower and boundary == last and cmp_ > 0:
                    return False, ()

    intro_unperformed = False
    if intro:
        c = compare_version_order(version, intro, ecosystem)
        if c is not None and c < 0:
            return False, tuple(dropped)
        if c is None:
            dropped.append(intro)
            intro_unperformed = True
    if fix:
        c = compare_version_order(version, fix, ecosystem)
        if c is None:
            dropped.append(fix)
            return False, tuple(dropped)
        if c >= 0:
            return False, tuple(dropped)
    if last:
        c = compare_version_order(version, last, ecosystem)
        if c is None:
            dropped.append(last)
            return False, tuple(dropped)
        if c > 0:
            return False, tuple(dropped)
    if intro_unperformed and not fix and not last:
        return False, tuple(dropped)
    return True, tuple(dropped)
