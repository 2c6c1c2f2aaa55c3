"""Property-based tests (hypothesis) for the version total orders.

The match kernel's correctness rests on two contracts:

1. ``compare_version_order`` is a TOTAL ORDER on comparable pairs per
   ecosystem (antisymmetry, reflexivity, transitivity) — fail-closed
   ``None`` is allowed, but an inconsistent ordering is never;
2. ``encode_version`` keys are ORDER-PRESERVING: whenever two versions
   are both encodable, key order must equal comparator order — a key
   inversion would make the GPU match disagree with the CPU oracle.

Generators produce structured versions (numeric triples, pre-release
tags, epochs, build metadata) rather than raw text so the interesting
comparator branches are actually exercised.
"""

from __future__ import annotations

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from agentbom_amd.utils.version_keys import encode_version
from agentbom_amd.utils.version_utils import compare_version_order

_SETTINGS = dict(max_examples=200, deadline=None,
                 suppress_health_check=[HealthCheck.too_slow])

_num = st.integers(min_value=0, max_value=400)
_tag = st.sampled_from(["alpha", "beta", "rc", "dev", "pre"])


@st.composite
def semver(draw):
    base = f"{draw(_num)}.{draw(_num)}.{draw(_num)}"
    if draw(st.booleans()):
        base += f"-{draw(_tag)}.{draw(st.integers(0, 20))}"
    return base


@st.composite
def pep440(draw):
    base = ".".join(str(draw(_num)) for _ in range(draw(st.integers(1, 4))))
    kind = draw(st.sampled_from(["", "a", "b", "rc", ".post", ".dev"]))
    if kind:
        base += f"{kind}{draw(st.integers(0, 20))}"
    return base


@st.composite
def debver(draw):
    v = f"{draw(_num)}.{draw(_num)}.{draw(_num)}"
    if draw(st.booleans()):
        v = f"{draw(st.integers(0, 3))}:{v}"
    if draw(st.booleans()):
        v += f"-{draw(st.integers(0, 30))}"
    if draw(st.booleans()):
        v += f"~{draw(_tag)}"
    return v


_CASES = [("npm", semver()), ("pypi", pep440()), ("deb", debver()),
          ("rpm", debver()), ("cargo", semver())]


@pytest.mark.parametrize("eco,strategy", _CASES,
                         ids=[c[0] for c in _CASES])
class TestTotalOrder:
    @settings(**_SETTINGS)
    @given(data=st.data())
    def test_antisymmetry(self, eco, strategy, data):
        a, b = data.draw(strategy), data.draw(strategy)
        ab = compare_version_order(a, b, eco)
        ba = compare_version_order(b, a, eco)
        if ab is None or ba is None:
            assert ab is None and ba is None, (a, b)
        else:
            assert ab == -ba, (a, b)

    @settings(**_SETTINGS)
    @given(data=st.data())
    def test_reflexive_equal(self, eco, strategy, data):
        a = data.draw(strategy)
        r = compare_version_order(a, a, eco)
        assert r in (0, None), a

    @settings(**_SETTINGS)
    @given(data=st.data())
    def test_transitivity(self, eco, strategy, data):
        a, b, c = (data.draw(strategy) for _ in range(3))
        ab = compare_version_order(a, b, eco)
        bc = compare_version_order(b, c, eco)
        ac = compare_version_order(a, c, eco)
        if ab == bc == -1 and None not in (ab, bc, ac):
            assert ac == -1, (a, b, c)
        if ab == bc == 1 and None not in (ab, bc, ac):
            assert ac == 1, (a, b, c)


@pytest.mark.parametrize("eco,strategy", _CASES,
                         ids=[c[0] for c in _CASES])
class TestKeyOrderPreservation:
    @settings(**_SETTINGS)
    @given(data=st.data())
    def test_key_agrees_with_comparator(self, eco, strategy, data):
        a, b = data.draw(strategy), data.draw(strategy)
        (ah, al, a_ok) = encode_version(a, eco)
        (bh, bl, b_ok) = encode_version(b, eco)
        cmp = compare_version_order(a, b, eco)
        if not (a_ok and b_ok) or cmp is None:
            return  # fail-closed rows are matched host-side
        key_cmp = -1 if (ah, al) < (bh, bl) else (1 if (ah, al) > (bh, bl) else 0)
        # keys may COLLIDE only for comparator-equal versions; an
        # INVERSION is the bug class this guards against
        if cmp == 0:
            assert key_cmp == 0, (a, b)
        else:
            assert key_cmp in (0, cmp), (a, b)
            if key_cmp == 0:
                # equal keys for unequal versions would hide a boundary:
                # the encoder must not be coarser than the comparator on
                # the grammar it claims to encode
                pytest.fail(f"key collision for unequal versions {a!r} vs {b!r}")
