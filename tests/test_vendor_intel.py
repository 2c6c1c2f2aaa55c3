"""Vendor-bulletin supplemental matching + local threat-intel lookup.

Reference parity: scanners/{amd,nvidia,ghsa}_advisory.py and
intel_lookup.py — bulletins that never reach OSV and local IOC feeds.
"""

import json

import pytest

from agentbom_amd.scan.intel import IntelIndicator, IntelStore, enrich_report_with_intel
from agentbom_amd.scan.orchestrator import ScanOptions, run_demo_scan
from agentbom_amd.scan.vendor_advisories import (
    VendorAdvisory,
    VendorAffected,
    check_vendor_advisories,
    load_vendor_feed,
    parse_csaf_document,
    vendor_for_package,
)


class TestVendorPrefixes:
    def test_rocm_stack_is_amd(self):
        for name in ("rocblas", "hipblaslt", "miopen", "rccl", "rocm-smi",
                     "composable-kernel", "amdsmi"):
            assert vendor_for_package(name) == "amd", name

    def test_cuda_stack_is_nvidia(self):
        for name in ("cuda-toolkit", "nvidia-cudnn-cu12", "tensorrt", "nccl"):
            assert vendor_for_package(name) == "nvidia", name

    def test_non_vendor_packages(self):
        for name in ("requests", "left-pad", "flask", ""):
            assert vendor_for_package(name) is None


class TestFeedLoading:
    def test_bundled_feed_examples_gated_off_by_default(self, monkeypatch):
        monkeypatch.delenv("AGENT_BOM_VENDOR_EXAMPLES", raising=False)
        monkeypatch.delenv("AGENT_BOM_VENDOR_FEED", raising=False)
        assert load_vendor_feed() == []

    def test_bundled_feed_examples_opt_in(self):
        feed = load_vendor_feed(include_examples=True)
        assert len(feed) == 3
        assert {a.vendor for a in feed} == {"amd", "nvidia", "intel"}

    def test_operator_feed_file(self, tmp_path):
        f = tmp_path / "feed.json"
        f.write_text(json.dumps({"advisories": [
            {"vendor": "amd", "advisory_id": "AMD-SB-1000", "severity": "high",
             "affected": [{"ecosystem": "PyPI", "name_prefix": "rocm",
                           "fixed": "6.0.0"}]},
            {"bogus": True},
            "not-a-dict",
        ]}))
        feed = load_vendor_feed(str(f))
        assert len(feed) == 1 and feed[0].advisory_id == "AMD-SB-1000"

    def test_unreadable_feed_fail_soft(self, tmp_path):
        f = tmp_path / "bad.json"
        f.write_text("{not json")
        assert load_vendor_feed(str(f)) == []

    def test_csaf_directory(self, tmp_path):
        doc = {
            "document": {
                "title": "ROCm runtime out-of-bounds write",
                "publisher": {"name": "AMD"},
                "tracking": {"id": "AMD-SB-7001"},
                "aggregate_severity": {"text": "High"},
            },
            "product_tree": {"full_product_names": [
                {"product_id": "P1", "name": "rocblas 4.x"}]},
            "vulnerabilities": [{"cve": "CVE-2025-12345"}],
        }
        (tmp_path / "a.json").write_text(json.dumps(doc))
        (tmp_path / "junk.json").write_text("[1,2,3]")
        feed = load_vendor_feed(str(tmp_path))
        assert len(feed) == 1
        adv = feed[0]
        assert adv.advisory_id == "AMD-SB-7001"
        assert adv.vendor == "amd"
        assert adv.severity == "high"
        assert adv.cve_ids == ["CVE-2025-12345"]
        assert adv.affected and adv.affected[0].name_prefix == "rocblas"

    def test_csaf_malformed_shapes_yield_empty(self):
        for doc in (None, [], {}, {"document": []}, {"document": {"tracking": {}}}):
            assert parse_csaf_document(doc) == []


class TestVendorMatching:
    FEED = [VendorAdvisory(
        vendor="amd", advisory_id="AMD-SB-2000", title="t", severity="high",
        affected=[VendorAffected(ecosystem="PyPI", name_prefix="rocm",
                                 introduced="5.0.0", fixed="6.1.0")],
    )]

    def test_bounded_match_tier(self):
        hits = check_vendor_advisories([("PyPI", "rocm-smi", "5.7.0")], self.FEED)
        (vulns,) = hits.values()
        assert vulns[0].match_confidence_tier == "vendor_range"
        assert vulns[0].advisory_sources == ["vendor:amd"]
        assert vulns[0].fixed_version == "6.1.0"

    def test_version_outside_window_no_match(self):
        assert check_vendor_advisories([("PyPI", "rocm-smi", "6.1.0")], self.FEED) == {}
        assert check_vendor_advisories([("PyPI", "rocm-smi", "4.9")], self.FEED) == {}

    def test_wrong_ecosystem_no_match(self):
        assert check_vendor_advisories([("npm", "rocm-smi", "5.7.0")], self.FEED) == {}

    def test_unbounded_prefix_tier_lower_confidence(self):
        feed = [VendorAdvisory(
            vendor="intel", advisory_id="INTEL-SA-1", severity="low",
            affected=[VendorAffected(ecosystem="*", name_prefix="openvino")])]
        hits = check_vendor_advisories([("PyPI", "openvino", "2024.1")], feed)
        (vulns,) = hits.values()
        assert vulns[0].match_confidence_tier == "vendor_prefix"
        bounded = check_vendor_advisories(
            [("PyPI", "rocm-smi", "5.7.0")], self.FEED)
        (bv,) = bounded.values()
        assert vulns[0].confidence < bv[0].confidence

    def test_unparseable_bound_fails_closed(self):
        feed = [VendorAdvisory(
            vendor="amd", advisory_id="AMD-SB-3000",
            affected=[VendorAffected(ecosystem="PyPI", name_prefix="rocm",
                                     introduced="deadbeefcafe")])]
        assert check_vendor_advisories([("PyPI", "rocm-smi", "5.7.0")], feed) == {}

    def test_orchestrator_wiring(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_VENDOR_EXAMPLES", "1")
        report = run_demo_scan(ScanOptions(demo=True, vendor_advisories=True))
        stages = report.scan_performance_data["stages"]
        assert "vendor_advisories_ms" in stages

    def test_orchestrator_opt_out(self):
        report = run_demo_scan(ScanOptions(demo=True, vendor_advisories=False))
        for br in report.blast_radii:
            assert not any(s.startswith("vendor:")
                           for s in br.vulnerability.advisory_sources or [])


class TestIntelStore:
    def test_bundled_store_loads(self):
        store = IntelStore.load()
        assert len(store) >= 4
        hit = store.lookup_cve("CVE-2021-44228")
        assert hit is not None and hit.exploited

    def test_env_overlay(self, tmp_path, monkeypatch):
        f = tmp_path / "intel.json"
        f.write_text(json.dumps({"indicators": [
            {"type": "cve", "key": "CVE-2099-1", "exploited": True,
             "note": "zero day"},
            {"type": "bogus", "key": "x"},
            {"type": "package"},
        ]}))
        monkeypatch.setenv("AGENT_BOM_INTEL_DB", str(f))
        store = IntelStore.load()
        assert store.lookup_cve("cve-2099-1") is not None
        assert store.lookup("bogus", "x") is None

    def test_package_lookup_eco_qualified_and_bare(self):
        store = IntelStore(
            [IntelIndicator(type="package", key="pypi:evil"),
             IntelIndicator(type="package", key="anywhere")])
        assert store.lookup_package("PyPI", "evil") is not None
        assert store.lookup_package("npm", "evil") is None
        assert store.lookup_package("npm", "anywhere") is not None

    def test_search(self):
        store = IntelStore.load()
        assert any("log4shell" in i.note for i in store.search("44228"))
        assert store.search("zzz-no-such") == []

    def test_unreadable_db_fail_soft(self, tmp_path, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_INTEL_DB", str(tmp_path / "missing.json"))
        IntelStore.load()  # must not raise


class TestIntelEnrichment:
    def _store_for(self, report):
        vid = report.blast_radii[0].vulnerability.id
        pkg = report.blast_radii[0].package
        return vid, pkg, IntelStore([
            IntelIndicator(type="cve", key=vid.lower(), exploited=True,
                           note="exploited in the wild", confidence=1.0),
            IntelIndicator(type="package",
                           key=f"{pkg.ecosystem.lower()}:{pkg.name.lower()}",
                           note="observed in campaign"),
        ])

    def test_cve_hit_sets_active_exploitation(self):
        report = run_demo_scan()
        vid, _, store = self._store_for(report)
        matches = enrich_report_with_intel(report, store)
        vuln = report.blast_radii[0].vulnerability
        assert vuln.exploitability == "active_exploitation"
        assert "intel" in vuln.advisory_sources
        assert any(m["entity"] == "vulnerability" and m["id"] == vid
                   for m in matches)
        assert report.intel_matches == matches

    def test_malicious_package_indicator_fails_closed(self):
        report = run_demo_scan()
        pkg = report.blast_radii[0].package
        store = IntelStore([IntelIndicator(
            type="package", key=f"{pkg.ecosystem.lower()}:{pkg.name.lower()}",
            malicious=True, note="backdoored release")])
        enrich_report_with_intel(report, store)
        assert pkg.is_malicious
        assert "backdoored" in pkg.malicious_reason

    def test_live_search_injectable_and_fail_soft(self):
        report = run_demo_scan()
        calls = []

        def live(q):
            calls.append(q)
            if len(calls) == 1:
                raise RuntimeError("egress down")
            return [{"title": "writeup", "url": "https://x", "snippet": "…"}]

        matches = enrich_report_with_intel(report, IntelStore(), live_search=live)
        assert calls  # invoked only because explicitly provided
        assert any(m["entity"] == "live" for m in matches) or len(calls) == 1

    def test_no_live_search_by_default(self):
        report = run_demo_scan()
        matches = enrich_report_with_intel(report, IntelStore())
        assert all(m["entity"] != "live" for m in matches)


class TestOrchestratorIntelWiring:
    def test_scan_runs_intel_stage_and_report_carries_matches(self):
        report = run_demo_scan()
        assert "threat_intel_ms" in report.scan_performance_data["stages"]
        assert report.intel_matches is not None  # list, possibly empty

    def test_json_contract_has_intel_matches(self):
        from agentbom_amd.output.json_fmt import to_json

        doc = to_json(run_demo_scan())
        assert "intel_matches" in doc

    def test_opt_out(self):
        report = run_demo_scan(ScanOptions(demo=True, threat_intel=False))
        assert "threat_intel_ms" not in report.scan_performance_data["stages"]
