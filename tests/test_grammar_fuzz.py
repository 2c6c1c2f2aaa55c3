"""Hypothesis fuzz over the round-2 grammars: npm ranges, GHSA ranges,
SSE streams, pooled version determinism — parsers must never crash and
resolutions must respect their bounds."""

from __future__ import annotations

import json

from hypothesis import given, settings
from hypothesis import strategies as st

from agentbom_amd.db.live import _parse_ghsa_range
from agentbom_amd.runtime.proxy import parse_sse_stream
from agentbom_amd.scan.transitive import resolve_npm_range
from agentbom_amd.utils.version_utils import compare_version_order

_ver = st.builds(lambda a, b, c: f"{a}.{b}.{c}",
                 st.integers(0, 30), st.integers(0, 30), st.integers(0, 30))
_spec_text = st.text(
    alphabet="0123456789.^~<>=| x*-vrcalphabet", min_size=0, max_size=24)


@settings(max_examples=300, deadline=None)
@given(spec=_spec_text, versions=st.lists(_ver, min_size=1, max_size=12,
                                          unique=True))
def test_npm_range_never_crashes_and_respects_membership(spec, versions):
    pick = resolve_npm_range(spec, versions)
    if pick is not None and spec not in ("", "*", "latest", "x"):
        assert pick in versions


@settings(max_examples=200, deadline=None)
@given(lo=_ver, hi=_ver, versions=st.lists(_ver, min_size=1, max_size=16,
                                           unique=True))
def test_npm_compound_range_bounds(lo, hi, versions):
    pick = resolve_npm_range(f">={lo} <{hi}", versions)
    if pick is not None:
        assert (compare_version_order(pick, lo, "npm") or 0) >= 0
        assert (compare_version_order(pick, hi, "npm") or 0) < 0


@settings(max_examples=300, deadline=None)
@given(text=st.text(alphabet="<>=!,. 0123456789abcdefv", max_size=40))
def test_ghsa_range_never_crashes(text):
    intro, fixed, last = _parse_ghsa_range(text)
    assert intro is not None  # defaults to "0"


@settings(max_examples=200, deadline=None)
@given(chunks=st.lists(st.one_of(
    st.just(": comment"),
    st.just("event: message"),
    st.builds(lambda i: f'data: {{"jsonrpc":"2.0","id":{i}}}', st.integers(0, 99)),
    st.text(alphabet="datev:{} \n", max_size=20),
), max_size=12))
def test_sse_parser_never_crashes(chunks):
    body = "\n".join(chunks) + "\n\n"
    frames = parse_sse_stream(body)
    for f in frames:
        assert isinstance(f, dict)


@settings(max_examples=50, deadline=None)
@given(seed=st.integers(0, 2**31 - 1))
def test_pooled_versions_deterministic(seed):
    from agentbom_amd.scan.synth import generate_estate

    a = generate_estate(n_agents=5, n_servers=20, n_packages=300,
                        name_catalog=50, seed=seed, arena_windows=100)
    b = generate_estate(n_agents=5, n_servers=20, n_packages=300,
                        name_catalog=50, seed=seed, arena_windows=100)
    import numpy as np

    assert np.array_equal(a.pkg_key_hi, b.pkg_key_hi)
    assert np.array_equal(a.arena.group_keys, b.arena.group_keys)
    assert np.array_equal(a.arena.flags, b.arena.flags)
