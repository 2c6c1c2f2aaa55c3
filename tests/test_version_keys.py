"""Order-preserving key encoding vs the CPU comparator oracle.

For every pair of encodable versions of one ecosystem, u128 key order must
equal compare_version_order.  Unencodable forms must be flagged (never a
wrong key).  This is the contract the GPU bulk matcher relies on.
"""

import itertools

import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from agentbom_amd.utils.version_keys import compare_keys, encode_version
from agentbom_amd.utils.version_utils import compare_version_order

CORPUS = {
    "npm": [
        "0.0.1", "1.0.0", "1.0.1", "1.2.3", "1.10.0", "2.0.0",
        "1.0.0-alpha", "1.0.0-alpha.1", "1.0.0-beta", "1.0.0-beta.2",
        "1.0.0-beta.11", "1.0.0-rc.1", "13.4.20-canary.13", "16.2.4",
        "1.0.0-pre.7", "5.0.0-next.3",
    ],
    "pypi": [
        "0.1", "1.0", "1.0.0", "1.0.1", "2.31.0", "1.0a1", "1.0a2", "1.0b1",
        "1.0rc1", "1.0rc2", "1.0.post1", "1.0.post2", "1.0.dev1", "1.0.dev2",
        "2!1.0", "1!2.0", "5.3.1", "5.4", "6.0.1",
    ],
    "maven": [
        "1", "1.0", "1.0.1", "2.5.6", "1.0-alpha1", "1.0-alpha2", "1.0-beta1",
        "1.0-rc1", "1.0-SNAPSHOT", "1.0-sp1", "5.0.6", "1.0-ga", "1.0-m2",
    ],
    "composer": ["1.0.0", "1.0.0-beta1", "1.0.0-rc1", "2.4.5", "2.4.5-p1", "2.4.5-p2", "1.0.0-dev1"],
    "nuget": ["1.0.0", "1.0.0.1", "1.0.0.2", "1.0.0-alpha", "1.0.0-alpha.2", "2.1.0", "1.0"],
    "rubygems": ["1.0.0", "1.0.1", "5.0.0", "1.0.0-pre.1", "2.3.4"],
    "apk": ["1.2.3", "1.2.3_alpha", "1.2.3_alpha1", "1.2.3_beta", "1.2.3_rc1",
            "1.2.3_p1", "1.2.3_p2", "1.2.3-r1", "1.2.3-r10", "1.2.4"],
    "deb": ["1.2.3", "1.2.3-1", "1.2.3-2", "1.2.4", "1:1.0", "2:0.5", "0.9"],
    "rpm": ["1.0", "1.0.1", "2.3.4", "1:1.0"],
    "go": ["v1.0.0", "v1.2.3", "v1.10.0", "v2.0.0", "v1.2.3-pre.1"],
    "cargo": ["0.8.5", "1.0.0", "1.0.0-alpha.6"],
}


@pytest.mark.parametrize("eco", sorted(CORPUS))
def test_key_order_matches_comparator(eco):
    versions = CORPUS[eco]
    encoded = {v: encode_version(v, eco) for v in versions}
    for left, right in itertools.combinations(versions, 2):
        lhi, llo, lok = encoded[left]
        rhi, rlo, rok = encoded[right]
        if not (lok and rok):
            continue
        oracle = compare_version_order(left, right, eco)
        if oracle is None:
            continue
        got = compare_keys((lhi, llo), (rhi, rlo))
        assert got == oracle, f"{eco}: {left} vs {right}: key {got} != oracle {oracle}"


def test_unencodable_forms_flagged():
    unencodable = [
        ("deadbeefdeadbeefdeadbeefdeadbeefdeadbeef", "npm"),
        ("v0.0.0-20200622213623-75b288015ac9", "go"),  # pseudo-version
        ("2.6.0+cu124", "pypi"),  # local version
        ("1.2~rc1-1", "deb"),  # tilde
        ("2.5.6.SEC03", "maven"),  # unknown qualifier
        ("1.0.0-build.1.2.3", "npm"),  # multi-identifier prerelease
        ("1.2.3.4.5", "npm"),  # too many components
        ("300000000.0.0", "npm"),  # component overflow (>2^28)
        ("1.0a1.dev1", "pypi"),  # stacked suffix classes
    ]
    for v, eco in unencodable:
        _, _, ok = encode_version(v, eco)
        assert not ok, f"{v} ({eco}) should be unencodable"


def test_sentinel_and_release_ordering():
    # introduced "0" handling happens at window build; key for "0" is just 0.0.0
    hi0, lo0, ok0 = encode_version("0", "npm")
    hi1, lo1, ok1 = encode_version("0.0.1-alpha", "npm")
    assert ok0 and ok1
    # release 0.0.0 > prerelease of 0.0.1? No: 0.0.0 < 0.0.1-alpha (base wins)
    assert compare_keys((hi0, lo0), (hi1, lo1)) == -1


def test_separator_sensitive_forms():
    """Separator variants that the exact comparator distinguishes must either
    encode order-preservingly or be unencodable ('1.0.0-rc.2' and
    '1.0.0-rc2' previously shared one key while the oracle says rc.2 < rc2)."""
    # strict-SemVer family: tagN / tag-N / tag_N are lexical -> unencodable
    for v in ("1.0.0-rc2", "1.0.0-rc-2", "1.0.0-rc_2", "1.0.0-RC.2",
              "1.0.0.rc.2", "1.0.0-rc."):
        for eco in ("npm", "nuget"):
            assert not encode_version(v, eco)[2], (v, eco)
    assert encode_version("1.0.0-rc.2", "npm")[2]
    assert encode_version("1.0.0-rc", "npm")[2]
    # maven: dot-joined number orders differently -> unencodable; tagN/tag-N fine
    assert not encode_version("1.0-alpha.1", "maven")[2]
    assert encode_version("1.0-alpha1", "maven")[2]
    assert encode_version("1.0-alpha-1", "maven")[2]
    # gem: hyphen-prefixed tag canonicalizes via an extra "pre" segment
    assert not encode_version("1.0.0-rc2", "rubygems")[2]
    assert encode_version("1.0.0.rc2", "rubygems")[2]
    # cross-check every encodable separator variant against the oracle
    variants = {
        "npm": ["1.0.0-rc.2", "1.0.0-rc.10", "1.0.0-rc", "1.0.0-alpha.1", "1.0.0"],
        "maven": ["1.0-alpha1", "1.0-alpha-2", "1.0-rc1", "1.0-rc10", "1.0"],
        "rubygems": ["1.0.0.rc2", "1.0.0.rc10", "1.0.0.beta1", "1.0.0"],
        "cargo": ["1.0.0-rc.2", "1.0.0-rc2", "1.0.0-rc10", "1.0.0"],
        "composer": ["1.0.0-rc.2", "1.0.0-rc2", "1.0.0-rc10", "1.0.0"],
    }
    for eco, versions in variants.items():
        for left, right in itertools.combinations(versions, 2):
            lhi, llo, lok = encode_version(left, eco)
            rhi, rlo, rok = encode_version(right, eco)
            if not (lok and rok):
                continue
            oracle = compare_version_order(left, right, eco)
            if oracle is None:
                continue
            got = compare_keys((lhi, llo), (rhi, rlo))
            assert got == oracle, f"{eco}: {left} vs {right}: key {got} != oracle {oracle}"


_num = st.integers(min_value=0, max_value=400)
_tag = st.sampled_from(["alpha", "beta", "rc", "pre", ""])


@settings(max_examples=300, deadline=None)
@given(
    a=st.tuples(_num, _num, _num, _tag, st.integers(0, 40)),
    b=st.tuples(_num, _num, _num, _tag, st.integers(0, 40)),
)
def test_hypothesis_semver_pairs(a, b):
    def mk(t):
        n1, n2, n3, tag, tn = t
        v = f"{n1}.{n2}.{n3}"
        if tag:
            v += f"-{tag}.{tn}"
        return v

    left, right = mk(a), mk(b)
    lhi, llo, lok = encode_version(left, "npm")
    rhi, rlo, rok = encode_version(right, "npm")
    assert lok and rok
    oracle = compare_version_order(left, right, "npm")
    assert compare_keys((lhi, llo), (rhi, rlo)) == oracle


@settings(max_examples=300, deadline=None)
@given(
    a=st.tuples(_num, _num, _num, st.sampled_from(["a", "b", "rc", "post", "dev", ""]), st.integers(0, 40)),
    b=st.tuples(_num, _num, _num, st.sampled_from(["a", "b", "rc", "post", "dev", ""]), st.integers(0, 40)),
)
def test_hypothesis_pypi_pairs(a, b):
    def mk(t):
        n1, n2, n3, tag, tn = t
        v = f"{n1}.{n2}.{n3}"
        if tag in ("a", "b", "rc"):
            v += f"{tag}{tn}"
        elif tag in ("post", "dev"):
            v += f".{tag}{tn}"
        return v

    left, right = mk(a), mk(b)
    lhi, llo, lok = encode_version(left, "pypi")
    rhi, rlo, rok = encode_version(right, "pypi")
    assert lok and rok
    oracle = compare_version_order(left, right, "pypi")
    assert compare_keys((lhi, llo), (rhi, rlo)) == oracle
