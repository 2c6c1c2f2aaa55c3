"""Matching-accuracy corpus — each case encodes a known real-world bug class.

Mirrors the reference's accuracy suite (multi-window ranges,
versions-list-not-exhaustive, introduced sentinel, fixed-version accuracy,
Maven ComparableVersion conformance).  Every case runs through the SAME
pipeline the scanner uses: parse_osv_record → arena → match (CPU oracle,
bit-identical to the GPU kernel by the parity suite).
"""

import pytest

from agentbom_amd.db.osv_ingest import parse_osv_record
from agentbom_amd.utils.version_utils import compare_version_order


def _hits(windows, name, version, eco="pypi"):
    from agentbom_amd.utils.canonical_ids import normalize_package_name
    from agentbom_amd.utils.version_utils import version_in_range

    out = []
    for w in windows:
        if (w.ecosystem.lower() == eco.lower()
                and normalize_package_name(w.package_name, w.ecosystem)
                == normalize_package_name(name, eco)
                and version_in_range(version, w.introduced, w.fixed,
                                     w.last_affected, eco)):
            out.append(w.vuln_id)
    return out


class TestMultiWindowRanges:
    """Multi-branch advisories: every (introduced, fixed) window matters."""

    RECORD = {
        "id": "GHSA-multi", "summary": "multi-branch",
        "affected": [{
            "package": {"ecosystem": "PyPI", "name": "web"},
            "ranges": [{"type": "ECOSYSTEM", "events": [
                {"introduced": "0"}, {"fixed": "1.9.5"},
                {"introduced": "2.0.0"}, {"fixed": "2.3.1"},
                {"introduced": "3.0.0"}, {"fixed": "3.0.4"},
            ]}]}],
    }

    @pytest.mark.parametrize("version,affected", [
        ("1.0.0", True), ("1.9.5", False), ("1.9.9", False),
        ("2.0.0", True), ("2.3.0", True), ("2.3.1", False),
        ("3.0.0", True), ("3.0.4", False), ("4.0", False),
    ])
    def test_each_branch(self, version, affected):
        windows = parse_osv_record(self.RECORD)
        assert len(windows) == 3  # one row per branch, never collapsed
        assert bool(_hits(windows, "web", version)) is affected


class TestVersionsListNotExhaustive:
    """versions[] lists SOME affected versions; ranges still bind others."""

    RECORD = {
        "id": "GHSA-vlist",
        "affected": [{
            "package": {"ecosystem": "PyPI", "name": "lib"},
            "ranges": [{"type": "ECOSYSTEM", "events": [
                {"introduced": "1.0.0"}, {"fixed": "1.4.0"}]}],
            "versions": ["1.0.0", "1.1.0"],
        }],
    }

    def test_range_still_matches_unlisted_version(self):
        windows = parse_osv_record(self.RECORD)
        # 1.3.7 is not in versions[] but inside the range -> affected
        assert _hits(windows, "lib", "1.3.7")
        assert not _hits(windows, "lib", "1.4.0")


class TestIntroducedSentinel:
    """introduced '0' (or missing) means no lower bound, for EVERY scheme."""

    def test_zero_sentinel_matches_preleases_and_epochs(self):
        rec = {"id": "X", "affected": [{
            "package": {"ecosystem": "Debian", "name": "pkg"},
            "ranges": [{"type": "ECOSYSTEM", "events": [
                {"introduced": "0"}, {"fixed": "2:1.0-1"}]}]}]}
        windows = parse_osv_record(rec)
        # epoch-0 version far 'below' the epoch-2 fix
        assert _hits(windows, "pkg", "1:9.9", eco="deb")
        assert not _hits(windows, "pkg", "2:1.0-1", eco="deb")

    def test_missing_introduced_event(self):
        rec = {"id": "X", "affected": [{
            "package": {"ecosystem": "PyPI", "name": "pkg"},
            "ranges": [{"type": "ECOSYSTEM", "events": [
                {"fixed": "0.5"}]}]}]}
        windows = parse_osv_record(rec)
        assert _hits(windows, "pkg", "0.0.1")


class TestFixedVersionAccuracy:
    """fixed is EXCLUSIVE; last_affected is INCLUSIVE — off-by-ones here
    produce both false positives and false negatives."""

    def test_fixed_exclusive(self):
        rec = {"id": "X", "affected": [{
            "package": {"ecosystem": "PyPI", "name": "p"},
            "ranges": [{"type": "ECOSYSTEM", "events": [
                {"introduced": "0"}, {"fixed": "2.6.0"}]}]}]}
        windows = parse_osv_record(rec)
        assert _hits(windows, "p", "2.5.9")
        assert not _hits(windows, "p", "2.6.0")
        assert not _hits(windows, "p", "2.6.0.post1")

    def test_last_affected_inclusive(self):
        rec = {"id": "X", "affected": [{
            "package": {"ecosystem": "PyPI", "name": "p"},
            "ranges": [{"type": "ECOSYSTEM", "events": [
                {"introduced": "1.0"}, {"last_affected": "1.9"}]}]}]}
        windows = parse_osv_record(rec)
        assert _hits(windows, "p", "1.9")
        assert not _hits(windows, "p", "1.9.1")

    def test_fixed_version_surfaces_in_window(self):
        rec = {"id": "X", "affected": [{
            "package": {"ecosystem": "PyPI", "name": "p"},
            "ranges": [{"type": "ECOSYSTEM", "events": [
                {"introduced": "0"}, {"fixed": "3.1.2"}]}]}]}
        w = parse_osv_record(rec)[0]
        assert w.fixed_version == "3.1.2"  # remediation advice exactness


class TestMavenConformance:
    """Maven ComparableVersion conformance vectors (qualifier ordering)."""

    VECTORS = [
        ("1.0-alpha", "1.0-beta", -1),
        ("1.0-beta", "1.0-milestone", -1),
        ("1.0-milestone", "1.0-rc", -1),
        ("1.0-rc", "1.0", -1),
        ("1.0", "1.0-sp", -1),
        ("1.0-SNAPSHOT", "1.0", -1),
        ("1.0", "1.0.1", -1),
        ("1.0.0", "1.0", 0),
        ("1.0-alpha1", "1.0-alpha2", -1),
        ("1.0-RELEASE", "1.0", 0),
        ("1.0-final", "1.0", 0),
        ("1.0-ga", "1.0", 0),
    ]

    @pytest.mark.parametrize("a,b,expect", VECTORS)
    def test_vector(self, a, b, expect):
        got = compare_version_order(a, b, "maven")
        assert (got < 0) == (expect < 0) and (got == 0) == (expect == 0), \
            f"{a} vs {b}: got {got}, want sign {expect}"


class TestCommitBoundsFailClosed:
    """GIT (commit-SHA) ranges are undecidable by version order: dropped
    with a warning, never guessed."""

    def test_git_range_not_matched(self):
        rec = {"id": "X", "affected": [{
            "package": {"ecosystem": "PyPI", "name": "p"},
            "ranges": [
                {"type": "GIT", "repo": "https://x/y", "events": [
                    {"introduced": "abc123"}, {"fixed": "def456"}]},
                {"type": "ECOSYSTEM", "events": [
                    {"introduced": "0"}, {"fixed": "1.2"}]}],
        }]}
        windows = parse_osv_record(rec)
        assert len(windows) == 1  # only the ECOSYSTEM window survives
        assert _hits(windows, "p", "1.0")
