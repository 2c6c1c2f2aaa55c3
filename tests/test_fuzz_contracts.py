"""Property/fuzz tests for the fail-soft contracts.

Three boundaries must never crash on hostile input (SURVEY.md §4 test
strategy — the reference fuzzes the same seams):
- lockfile/manifest parsing (`extract_packages` catches per-file),
- OSV record ingestion (`parse_osv_record` tolerates arbitrary shapes),
- version comparators (`version_in_range` fails closed, never raises),
- the policy expression language (bool or ValueError, nothing else).
"""

import json
import string

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from agentbom_amd.db.osv_ingest import parse_osv_record
from agentbom_amd.scan.parsers import BUILTIN_INVENTORY_PARSERS, extract_packages
from agentbom_amd.scan.policy import evaluate_expression
from agentbom_amd.utils.version_utils import version_in_range

FAST = settings(max_examples=60, deadline=None,
                suppress_health_check=[HealthCheck.function_scoped_fixture])

text_junk = st.text(
    alphabet=string.printable + "é中﻿", max_size=400)

# JSON-ish structures for OSV fuzzing
json_scalars = st.one_of(st.none(), st.booleans(), st.integers(),
                         st.floats(allow_nan=False), st.text(max_size=30))
json_values = st.recursive(
    json_scalars,
    lambda inner: st.one_of(st.lists(inner, max_size=4),
                            st.dictionaries(st.text(max_size=12), inner,
                                            max_size=4)),
    max_leaves=12)


class TestParserFuzz:
    @FAST
    @given(junk=text_junk)
    def test_extract_packages_never_raises_on_junk_trees(self, junk, tmp_path):
        for fname in ("package-lock.json", "requirements.txt", "go.mod",
                      "Gemfile.lock", "pyproject.toml", "pom.xml",
                      "pubspec.lock", "Cargo.lock"):
            (tmp_path / fname).write_text(junk)
        pkgs = extract_packages(tmp_path)
        for p in pkgs:
            assert p.name and isinstance(p.name, str)
            assert isinstance(p.version, str)

    @FAST
    @given(junk=text_junk)
    def test_line_parsers_accept_any_text(self, junk):
        """Text-format parsers (no strict syntax) must not crash on any text."""
        from agentbom_amd.scan import parsers as P

        for fn in (P.parse_requirements_txt, P.parse_yarn_lock, P.parse_go_mod,
                   P.parse_go_sum, P.parse_gemfile_lock, P.parse_dpkg_status,
                   P.parse_apk_installed, P.parse_conda_env,
                   P.parse_gradle_lockfile):
            out = fn(junk, "<fuzz>")
            assert isinstance(out, list)
            for p in out:
                assert p.name

    @FAST
    @given(payload=json_values)
    def test_json_parsers_tolerate_arbitrary_json(self, payload):
        """Structured parsers: any *valid JSON* parses to a list or raises
        nothing worse than a controlled skip inside extract_packages."""
        from agentbom_amd.scan import parsers as P

        text = json.dumps(payload)
        for fn in (P.parse_package_lock, P.parse_package_json,
                   P.parse_pipfile_lock, P.parse_packages_lock_json,
                   P.parse_composer_lock, P.parse_package_resolved):
            try:
                out = fn(text, "<fuzz>")
            except (AttributeError, TypeError, KeyError, ValueError):
                # malformed-but-valid JSON may be rejected; the tree walker
                # catches these — what matters is no hang / no corruption
                continue
            assert isinstance(out, list)

    def test_truncated_real_lockfile_fail_soft(self, tmp_path):
        doc = {"lockfileVersion": 3, "packages": {
            "node_modules/left-pad": {"version": "1.3.0"}}}
        text = json.dumps(doc)
        for cut in range(0, len(text), 7):
            (tmp_path / "package-lock.json").write_text(text[:cut])
            extract_packages(tmp_path)  # must never raise


class TestOsvFuzz:
    @FAST
    @given(record=st.dictionaries(
        st.sampled_from(["id", "summary", "details", "aliases", "affected",
                         "severity", "database_specific", "references"]),
        json_values, max_size=6))
    def test_parse_osv_record_never_raises(self, record):
        try:
            windows = parse_osv_record(record)
        except (AttributeError, TypeError):
            pytest.fail(f"parse_osv_record crashed on {record!r}")
        for w in windows:
            assert w.vuln_id
            assert w.package_name

    @FAST
    @given(events=st.lists(st.dictionaries(
        st.sampled_from(["introduced", "fixed", "last_affected", "limit"]),
        st.text(max_size=12), max_size=2), max_size=5))
    def test_range_events_any_order(self, events):
        record = {"id": "FUZZ-1", "affected": [{
            "package": {"ecosystem": "PyPI", "name": "x"},
            "ranges": [{"type": "ECOSYSTEM", "events": events}]}]}
        windows = parse_osv_record(record)
        for w in windows:
            assert w.vuln_id == "FUZZ-1"


class TestVersionFuzz:
    @FAST
    @given(version=text_junk, lo=text_junk, hi=text_junk,
           eco=st.sampled_from(["PyPI", "npm", "Go", "Maven", "RubyGems",
                                "Debian", "Alpine", "NuGet", "Packagist",
                                "crates.io", "???"]))
    def test_version_in_range_never_raises(self, version, lo, hi, eco):
        out = version_in_range(version, lo or None, hi or None, None, eco)
        assert isinstance(out, bool)

    @FAST
    @given(version=st.from_regex(r"[0-9]{1,3}(\.[0-9]{1,3}){0,3}",
                                 fullmatch=True))
    def test_self_inclusion_invariant(self, version):
        """v is always in [v, None] and never in [None, fixed=v)."""
        assert version_in_range(version, version, None, None, "PyPI")
        assert not version_in_range(version, None, version, None, "PyPI")


class TestPolicyFuzz:
    class _Br:
        class vulnerability:
            severity = type("S", (), {"value": "high"})()
            is_kev = True
            cvss_score = 8.1
            epss_score = 0.5
            id = "CVE-X"

        class package:
            is_malicious = False
            name = "p"
            ecosystem = "PyPI"

        risk_score = 7.5
        reachability = "reachable"
        impact_category = "code-execution"

    @FAST
    @given(expr=st.text(alphabet=string.ascii_letters + string.digits +
                        " .<>=!()'\"&|_", max_size=60))
    def test_expression_bool_or_valueerror(self, expr):
        try:
            out = evaluate_expression(expr, self._Br())
        except ValueError:
            return
        assert isinstance(out, bool)

    def test_known_expressions_still_work(self):
        br = self._Br()
        assert evaluate_expression("severity >= high and is_kev", br)
        assert not evaluate_expression("risk_score > 9", br)


class TestGraphBuilderFuzz:
    """The graph builder consumes persisted report JSON — a seam where
    hand-edited or cross-version documents arrive."""

    @FAST
    @given(mutation=st.dictionaries(
        st.sampled_from(["agents", "blast_radius", "packages", "summary",
                         "findings", "exposure_paths"]),
        json_values, max_size=4))
    def test_builder_from_mutated_doc(self, mutation):
        from agentbom_amd.graph.builder import build_unified_graph_from_report_json
        from agentbom_amd.output.json_fmt import to_json
        from agentbom_amd.scan.orchestrator import run_demo_scan

        doc = to_json(run_demo_scan())
        doc.update(mutation)
        try:
            graph = build_unified_graph_from_report_json(doc)
        except (KeyError, TypeError, AttributeError, ValueError) as exc:
            pytest.fail(f"builder crashed on mutated doc keys "
                        f"{sorted(mutation)}: {exc!r}")
        assert graph.nodes is not None


class TestNewSeamFuzz:
    """Round-1 late additions parse operator/attacker-supplied files too:
    cloud inventories, OTLP traces, vendor feeds, deps bundles, semgrep."""

    @FAST
    @given(inv=json_values)
    def test_cloud_estate_evaluators_never_raise(self, inv):
        from agentbom_amd.scan import cloud_estate as ce

        doc = inv if isinstance(inv, dict) else {"buckets": inv}
        for fn in (ce.evaluate_azure_inventory, ce.evaluate_gcp_inventory,
                   ce.evaluate_snowflake_inventory,
                   ce.evaluate_databricks_inventory):
            try:
                out = fn(doc)
            except (AttributeError, TypeError, KeyError) as exc:
                pytest.fail(f"{fn.__name__} crashed on {doc!r}: {exc!r}")
            assert isinstance(out, list)

    @FAST
    @given(rows=st.lists(json_values, max_size=6))
    def test_iam_and_audit_never_raise(self, rows):
        from agentbom_amd.scan.cloud_estate import (
            evaluate_iam_policies,
            ingest_audit_trail,
        )

        assert isinstance(evaluate_iam_policies(rows), list)
        assert isinstance(ingest_audit_trail(rows), list)

    @FAST
    @given(doc=json_values)
    def test_otlp_and_langfuse_never_raise(self, doc):
        from agentbom_amd.utils.otel_ingest import (
            parse_langfuse_export,
            parse_otlp_json,
            summarize_agent_activity,
        )

        spans = parse_otlp_json(doc)
        spans += parse_langfuse_export(doc)
        assert isinstance(summarize_agent_activity(spans), dict)

    @FAST
    @given(doc=json_values)
    def test_vendor_feed_and_csaf_never_raise(self, doc, tmp_path):
        from agentbom_amd.scan.vendor_advisories import (
            load_vendor_feed,
            parse_csaf_document,
        )

        assert isinstance(parse_csaf_document(doc), list)
        f = tmp_path / "feed.json"
        f.write_text(json.dumps(doc))
        assert isinstance(load_vendor_feed(str(f)), list)

    @FAST
    @given(doc=json_values)
    def test_semgrep_parse_never_raises(self, doc):
        from agentbom_amd.scan.sast_ingest import parse_semgrep_json

        rows = parse_semgrep_json(doc)
        assert isinstance(rows, list)

    @FAST
    @given(doc=json_values)
    def test_deps_bundle_never_raises(self, doc, tmp_path):
        from agentbom_amd.models.core import Package
        from agentbom_amd.scan.deps_meta import (
            enrich_packages_with_deps_meta,
            load_deps_bundle,
        )

        f = tmp_path / "b.json"
        f.write_text(json.dumps(doc))
        bundle = load_deps_bundle(str(f))
        p = Package(name="x", version="1", ecosystem="PyPI")
        enrich_packages_with_deps_meta([p], bundle=bundle)
