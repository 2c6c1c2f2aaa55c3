"""deps.dev / Scorecard metadata enrichment."""

import json

from agentbom_amd.models.core import Package
from agentbom_amd.scan.deps_meta import (
    enrich_packages_with_deps_meta,
    load_deps_bundle,
    supply_chain_risk_notes,
)


def _pkg(name="requests", eco="PyPI"):
    return Package(name=name, version="1.0.0", ecosystem=eco)


BUNDLE = {
    "pypi:requests": {
        "scorecard_score": 9.1,
        "scorecard_checks": {"Maintained": 10, "Signed-Releases": 1},
        "scorecard_repo": "github.com/psf/requests",
        "maintainer_count": 5, "license": "Apache-2.0",
    },
    "npm:leftpad": {"scorecard_score": 2.5, "maintainer_count": 1,
                    "deprecated": True},
}


class TestEnrichment:
    def test_bundle_hit_stamps_fields(self):
        p = _pkg()
        n = enrich_packages_with_deps_meta([p], bundle=BUNDLE)
        assert n == 1
        assert p.scorecard_score == 9.1
        assert p.scorecard_checks["Maintained"] == 10
        assert p.deps_dev_resolved
        assert p.scorecard_lookup_state == "found"
        assert p.license == "Apache-2.0"

    def test_miss_records_state_never_fakes_score(self):
        p = _pkg("unknown-thing")
        enrich_packages_with_deps_meta([p], bundle=BUNDLE)
        assert p.scorecard_score is None
        assert p.scorecard_lookup_state == "not_in_bundle"

    def test_deprecated_escalates_auto_risk(self):
        p = _pkg("leftpad", "npm")
        enrich_packages_with_deps_meta([p], bundle=BUNDLE)
        assert p.auto_risk_level == "elevated"
        assert "deprecated" in p.auto_risk_justification

    def test_injected_fetcher_and_failure(self):
        calls = []

        def fetcher(eco, name):
            calls.append(name)
            if name == "boom":
                raise RuntimeError("api down")
            return {"scorecard_score": 5.0}

        good, bad = _pkg("fresh"), _pkg("boom")
        enrich_packages_with_deps_meta([good, bad], bundle={}, fetcher=fetcher)
        assert good.scorecard_score == 5.0
        assert bad.scorecard_lookup_state == "error"
        assert calls == ["fresh", "boom"]

    def test_existing_license_not_clobbered(self):
        p = _pkg()
        p.license = "MIT"
        enrich_packages_with_deps_meta([p], bundle=BUNDLE)
        assert p.license == "MIT"

    def test_env_bundle_loading(self, tmp_path, monkeypatch):
        f = tmp_path / "bundle.json"
        f.write_text(json.dumps({"packages": BUNDLE}))
        monkeypatch.setenv("AGENT_BOM_DEPS_BUNDLE", str(f))
        assert "pypi:requests" in load_deps_bundle()
        monkeypatch.setenv("AGENT_BOM_DEPS_BUNDLE", str(tmp_path / "nope"))
        assert load_deps_bundle() == {}

    def test_trust_score_consumes_scorecard(self):
        from agentbom_amd.scan.trust import trust_score

        p = _pkg()
        enrich_packages_with_deps_meta([p], bundle=BUNDLE)
        high = trust_score(p)
        q = _pkg("leftpad", "npm")
        enrich_packages_with_deps_meta([q], bundle=BUNDLE)
        low = trust_score(q)
        assert high["score"] > low["score"]


class TestRiskNotes:
    def test_notes(self):
        p = _pkg("leftpad", "npm")
        enrich_packages_with_deps_meta([p], bundle=BUNDLE)
        notes = supply_chain_risk_notes(p)
        assert any("scorecard" in n for n in notes)
        assert any("single-maintainer" in n for n in notes)
        assert any("deprecated" in n for n in notes)

    def test_unsigned_releases_note(self):
        p = _pkg()
        enrich_packages_with_deps_meta([p], bundle=BUNDLE)
        assert any("unsigned" in n for n in supply_chain_risk_notes(p))

    def test_healthy_package_no_notes(self):
        p = _pkg()
        p.scorecard_score = 9.0
        p.maintainer_count = 10
        assert supply_chain_risk_notes(p) == []
