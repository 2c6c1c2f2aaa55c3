"""GPU-resident columnar advisory arena.

The local advisory store (db/store.py, SQLite — schema parity with the
reference's db/schema.py) exports its ``affected`` windows into sorted
columnar arrays that live in HBM for the bulk matcher:

- one row per (advisory, ecosystem, package, introduced/fixed/last_affected)
  window — multi-branch advisories are never collapsed (reference fixed this
  as schema v5; src/agent_bom/db/schema.py:73-142);
- rows grouped by ``group_key = hash64(ecosystem, normalized_name)`` and
  sorted, with a CSR-style ``group_off`` index;
- per-window u128 version-key bounds (utils/version_keys) + flags; windows
  whose bounds the encoder cannot represent carry WF_CPU_FALLBACK and are
  resolved host-side with the exact comparator (fail-closed either way);
- per-window metadata (severity code, cvss, epss, kev, vuln row index) for
  device-side scoring joins.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional, Sequence

import numpy as np

from agentbom_amd.models.core import SEVERITY_CODE, Severity
from agentbom_amd.models.cwe_impact import IMPACT_CODE, classify_cwe_impact
from agentbom_amd.ops.cpu_ref import (
    WF_CPU_FALLBACK,
    WF_HAS_FIXED,
    WF_HAS_INTRO,
    WF_HAS_LAST,
    WF_UNFIXED_SUPPRESSED,
)
from agentbom_amd.utils.canonical_ids import normalize_package_ecosystem, normalize_package_name
from agentbom_amd.utils.version_keys import encode_version
from agentbom_amd.utils.version_utils import _looks_like_commit_sha, normalize_introduced

_FNV_OFFSET = np.uint64(0xCBF29CE484222325)
_FNV_PRIME = np.uint64(0x100000001B3)

# Unfixed-advisory suppression applies ONLY to OS distro ecosystems: app
# ecosystems (npm/PyPI/...) keep unfixed windows matched by default (the
# reference gates on package_scan._OS_DISTRO_ECOSYSTEMS, package_scan.py:203,244).
OS_DISTRO_ECOSYSTEMS = frozenset({"deb", "apk", "rpm"})


def _suppress_unfixed(ecosystem: str) -> bool:
    return normalize_package_ecosystem(ecosystem) in OS_DISTRO_ECOSYSTEMS


def hash_name(ecosystem: str, name: str) -> int:
    """Stable FNV-1a 64 over ``eco:name`` (normalized) — the group key."""
    eco = normalize_package_ecosystem(ecosystem)
    data = f"{eco}:{normalize_package_name(name, eco)}".encode()
    h = int(_FNV_OFFSET)
    for b in data:
        h = ((h ^ b) * int(_FNV_PRIME)) & 0xFFFFFFFFFFFFFFFF
    return h


@dataclass
class AdvisoryWindow:
    """One affected window of one advisory for one package."""

    ecosystem: str
    package_name: str
    vuln_id: str
    introduced: Optional[str] = None
    fixed: Optional[str] = None
    last_affected: Optional[str] = None
    severity: Severity = Severity.UNKNOWN
    cvss_score: Optional[float] = None
    epss_score: Optional[float] = None
    is_kev: bool = False
    unfixed: bool = False  # distro advisory with no fix (suppressed by default)
    summary: str = ""
    cwe_ids: tuple[str, ...] = ()
    aliases: tuple[str, ...] = ()
    fixed_version: Optional[str] = None


@dataclass
class AdvisoryArena:
    """Columnar arena (numpy host-side; .to_torch moves it to HBM)."""

    group_keys: np.ndarray  # u64 [G] sorted
    group_off: np.ndarray  # u32 [G+1]
    intro_hi: np.ndarray  # u64 [W] (stored as int64 bit pattern for torch)
    intro_lo: np.ndarray
    fixed_hi: np.ndarray
    fixed_lo: np.ndarray
    last_hi: np.ndarray
    last_lo: np.ndarray
    flags: np.ndarray  # u8 [W]
    vuln_idx: np.ndarray  # u32 [W] row into vulns list
    severity: np.ndarray  # u8 [W] SEVERITY_CODE
    cvss: np.ndarray  # f32 [W]
    epss: np.ndarray  # f32 [W] (-1 = none)
    kev: np.ndarray  # u8 [W]
    impact: np.ndarray  # u8 [W] cwe_impact.IMPACT_CODE (8 = unknown)
    windows: list[AdvisoryWindow] = field(default_factory=list)
    # windows needing exact CPU comparison, as indices into ``windows``
    cpu_fallback_idx: np.ndarray = field(default_factory=lambda: np.array([], dtype=np.int64))

    @property
    def num_windows(self) -> int:
        return len(self.flags)

    @property
    def num_groups(self) -> int:
        return len(self.group_keys)

    def to_torch(self, device) -> dict:
        """Device tensors for ops.native.match (u64 views as int64)."""
        import torch

        def t(a, dtype):
            return torch.from_numpy(np.ascontiguousarray(a)).to(device=device, dtype=dtype)

        # AoS mirror: one 64-byte line per window [ihi ilo fhi flo lhi llo
        # flags pad] — the match walk reads ONE cache line per window
        # instead of touching 7 column streams (tail groups are short runs,
        # so SoA pays ~7 lines for 56 bytes of data)
        W = self.num_windows
        packed = np.zeros((W, 8), dtype=np.uint64)
        packed[:, 0] = self.intro_hi
        packed[:, 1] = self.intro_lo
        packed[:, 2] = self.fixed_hi
        packed[:, 3] = self.fixed_lo
        packed[:, 4] = self.last_hi
        packed[:, 5] = self.last_lo
        packed[:, 6] = self.flags.astype(np.uint64)
        return {
            "group_keys": t(self.group_keys.view(np.int64), torch.int64),
            "group_off": t(self.group_off.view(np.int32), torch.int32),
            "windows": {
                "intro_hi": t(self.intro_hi.view(np.int64), torch.int64),
                "intro_lo": t(self.intro_lo.view(np.int64), torch.int64),
                "fixed_hi": t(self.fixed_hi.view(np.int64), torch.int64),
                "fixed_lo": t(self.fixed_lo.view(np.int64), torch.int64),
                "last_hi": t(self.last_hi.view(np.int64), torch.int64),
                "last_lo": t(self.last_lo.view(np.int64), torch.int64),
                "flags": t(self.flags, torch.uint8),
                "packed": t(packed.view(np.int64), torch.int64),
            },
            "severity": t(self.severity, torch.uint8),
            "cvss": t(self.cvss, torch.float32),
            "epss": t(self.epss, torch.float32),
            "kev": t(self.kev, torch.uint8),
            "impact": t(self.impact, torch.uint8),
            "vuln_idx": t(self.vuln_idx.view(np.int32), torch.int32),
        }


def build_arena(windows: Sequence[AdvisoryWindow], include_unfixed: bool = False) -> AdvisoryArena:
    """Build the sorted columnar arena from advisory windows."""
    W = len(windows)
    gkeys = np.empty(W, dtype=np.uint64)
    ihi = np.zeros(W, dtype=np.uint64)
    ilo = np.zeros(W, dtype=np.uint64)
    fhi = np.zeros(W, dtype=np.uint64)
    flo = np.zeros(W, dtype=np.uint64)
    lhi = np.zeros(W, dtype=np.uint64)
    llo = np.zeros(W, dtype=np.uint64)
    flags = np.zeros(W, dtype=np.uint8)
    sev = np.zeros(W, dtype=np.uint8)
    cvss = np.zeros(W, dtype=np.float32)
    epss = np.full(W, -1.0, dtype=np.float32)
    kev = np.zeros(W, dtype=np.uint8)
    impact = np.full(W, 8, dtype=np.uint8)  # IMPACT_CODE["unknown"]

    for i, w in enumerate(windows):
        gkeys[i] = np.uint64(hash_name(w.ecosystem, w.package_name))
        f = 0
        # Commit-SHA bounds: the whole window is undecidable -> dropped
        # (fail-closed, reference package_scan.py:666-691).  Represent as a
        # CPU-fallback window the host also drops, so it never matches.
        sha_bound = any(
            b and _looks_like_commit_sha(b) for b in (w.introduced, w.fixed, w.last_affected)
        )
        intro = normalize_introduced(w.introduced)
        encodable = True
        if intro:
            hi, lo, ok = encode_version(intro, w.ecosystem)
            ihi[i], ilo[i] = hi, lo
            f |= WF_HAS_INTRO
            encodable &= ok
        if w.fixed:
            hi, lo, ok = encode_version(w.fixed, w.ecosystem)
            fhi[i], flo[i] = hi, lo
            f |= WF_HAS_FIXED
            encodable &= ok
        if w.last_affected:
            hi, lo, ok = encode_version(w.last_affected, w.ecosystem)
            lhi[i], llo[i] = hi, lo
            f |= WF_HAS_LAST
            encodable &= ok
        if sha_bound:
            f |= WF_CPU_FALLBACK  # host drops these windows entirely
        elif not encodable:
            f |= WF_CPU_FALLBACK  # host resolves with the exact comparator
        if w.unfixed and not include_unfixed and _suppress_unfixed(w.ecosystem):
            f |= WF_UNFIXED_SUPPRESSED
        flags[i] = f
        sev[i] = SEVERITY_CODE.get(w.severity, 1)
        cvss[i] = w.cvss_score if w.cvss_score is not None else 0.0
        epss[i] = w.epss_score if w.epss_score is not None else -1.0
        kev[i] = 1 if w.is_kev else 0
        impact[i] = IMPACT_CODE[classify_cwe_impact(list(w.cwe_ids))]

    order = np.argsort(gkeys, kind="stable")
    gkeys = gkeys[order]
    windows_sorted = [windows[j] for j in order]
    uniq, starts = np.unique(gkeys, return_index=True)
    group_off = np.append(starts, W).astype(np.uint32)

    cpu_fb = np.nonzero(
        (flags[order] & WF_CPU_FALLBACK) != 0
    )[0].astype(np.int64)

    return AdvisoryArena(
        group_keys=uniq,
        group_off=group_off,
        intro_hi=ihi[order], intro_lo=ilo[order],
        fixed_hi=fhi[order], fixed_lo=flo[order],
        last_hi=lhi[order], last_lo=llo[order],
        flags=flags[order],
        # sorted window row -> original `windows` argument index
        vuln_idx=np.asarray(order, dtype=np.uint32),
        severity=sev[order],
        cvss=cvss[order],
        epss=epss[order],
        kev=kev[order],
        impact=impact[order],
        windows=windows_sorted,
        cpu_fallback_idx=cpu_fb,
    )


def build_arena_from_columns(
    group_keys_unsorted: np.ndarray,
    intro_hi, intro_lo, fixed_hi, fixed_lo, last_hi, last_lo, flags,
    severity, cvss, epss, kev, impact=None,
) -> AdvisoryArena:
    """Fast vectorized arena build for pre-encoded columns (synthetic path)."""
    order = np.argsort(group_keys_unsorted, kind="stable")
    gkeys = group_keys_unsorted[order]
    W = len(gkeys)
    uniq, starts = np.unique(gkeys, return_index=True)
    group_off = np.append(starts, W).astype(np.uint32)
    if impact is None:
        impact = np.full(W, 8, dtype=np.uint8)
    return AdvisoryArena(
        group_keys=uniq,
        group_off=group_off,
        intro_hi=intro_hi[order], intro_lo=intro_lo[order],
        fixed_hi=fixed_hi[order], fixed_lo=fixed_lo[order],
        last_hi=last_hi[order], last_lo=last_lo[order],
        flags=flags[order],
        vuln_idx=np.asarray(order, dtype=np.uint32),
        severity=severity[order],
        cvss=cvss[order],
        epss=epss[order],
        kev=kev[order],
        impact=impact[order],
        windows=[],
        cpu_fallback_idx=np.nonzero((flags[order] & WF_CPU_FALLBACK) != 0)[0].astype(np.int64),
    )


def _group_range(arena: AdvisoryArena, gkey: int) -> range:
    g = np.searchsorted(arena.group_keys, np.uint64(gkey))
    if g >= len(arena.group_keys) or arena.group_keys[g] != np.uint64(gkey):
        return range(0)
    return range(int(arena.group_off[g]), int(arena.group_off[g + 1]))


def match_cpu_fallback(
    arena: AdvisoryArena,
    all_packages: Sequence,
    unencodable_idx: set[int],
    include_unfixed: bool = False,
) -> list[tuple[int, int]]:
    """Exact-comparator resolution for what the GPU kernel skipped:

    1. windows flagged WF_CPU_FALLBACK (unencodable bounds) x every package
       in their group (commit-SHA windows stay dropped — fail-closed);
    2. packages with unencodable versions x every window of their group.

    ``all_packages`` is a sequence of (idx, ecosystem, name, version);
    ``unencodable_idx`` selects case-2 packages.  Returns (pkg_idx,
    window_idx-in-sorted-arena) pairs, deduplicated.
    """
    from agentbom_amd.utils.version_utils import version_in_range

    if not arena.windows:
        return []
    fb_windows = set(int(x) for x in arena.cpu_fallback_idx)
    out: set[tuple[int, int]] = set()
    for idx, eco, name, version in all_packages:
        unenc = idx in unencodable_idx
        grange = _group_range(arena, hash_name(eco, name))
        for w_sorted in grange:
            is_fb = w_sorted in fb_windows
            if not (unenc or is_fb):
                continue  # GPU already decided this pair
            w = arena.windows[w_sorted]
            sha_bound = any(
                b and _looks_like_commit_sha(b) for b in (w.introduced, w.fixed, w.last_affected)
            )
            if sha_bound:
                continue  # dropped window, never matches
            if w.unfixed and not include_unfixed and _suppress_unfixed(w.ecosystem):
                continue
            if version_in_range(version, w.introduced, w.fixed, w.last_affected, eco):
                out.add((idx, w_sorted))
    return sorted(out)
