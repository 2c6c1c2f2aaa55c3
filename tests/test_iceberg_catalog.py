"""Iceberg lake registration over an injected catalog protocol."""

from __future__ import annotations

import pytest

from agentbom_amd.output.iceberg_catalog import (
    IcebergCatalogConfig,
    display_catalog_url,
    register_findings,
    to_arrow_table,
)


@pytest.fixture(scope="module")
def report():
    from agentbom_amd.scan.orchestrator import run_demo_scan

    return run_demo_scan()


class _FakeUpdate:
    def __init__(self, table):
        self.table = table

    def union_by_name(self, schema):
        self.schema = schema
        return self

    def commit(self):
        self.table._schema = self.schema
        self.table.evolutions += 1


class _FakeTable:
    def __init__(self, schema):
        self._schema = schema
        self.appended = []
        self.evolutions = 0
        self._snap = 0

    def schema(self):
        return self._schema

    def update_schema(self):
        return _FakeUpdate(self)

    def append(self, arrow):
        self.appended.append(arrow)
        self._snap += 1

    def current_snapshot(self):
        class S:  # noqa: N801 — tiny stub
            snapshot_id = self._snap
        return S()


class _FakeCatalog:
    def __init__(self):
        self.namespaces = []
        self.tables = {}

    def create_namespace_if_not_exists(self, ns):
        self.namespaces.append(ns)

    def create_table_if_not_exists(self, identifier, schema):
        return self.tables.setdefault(identifier, _FakeTable(schema))


class TestConfig:
    def test_from_env_and_properties(self):
        cfg = IcebergCatalogConfig.from_env({
            "AGENT_BOM_ICEBERG_CATALOG_URL": "https://cat.example/api",
            "AGENT_BOM_ICEBERG_TOKEN": "tok",
            "AGENT_BOM_ICEBERG_WAREHOUSE": "wh"})
        assert cfg.enabled and cfg.identifier == ("agent_bom", "findings")
        props = cfg.catalog_properties()
        assert props["uri"].startswith("https://") and props["token"] == "tok"
        assert not IcebergCatalogConfig.from_env({}).enabled

    def test_display_url_strips_secrets(self):
        assert display_catalog_url(
            "https://user:pass@cat.example:8181/v1?token=x") == \
            "https://cat.example:8181"
        assert display_catalog_url("nonsense") == "<invalid url>"


class TestRegistration:
    def test_append_creates_and_snapshots(self, report):
        cfg = IcebergCatalogConfig(catalog_url="https://cat.example")
        cat = _FakeCatalog()
        out = register_findings(report, cfg, catalog=cat)
        assert out["rows"] > 0 and out["snapshot_id"] == 1
        assert out["schema_evolved"] is False  # fresh table, same schema
        assert cat.namespaces == [("agent_bom",)]
        # second append on the same table -> snapshot advances
        out2 = register_findings(report, cfg, catalog=cat)
        assert out2["snapshot_id"] == 2

    def test_additive_evolution(self, report):
        import pyarrow as pa

        cfg = IcebergCatalogConfig(catalog_url="https://cat.example")
        cat = _FakeCatalog()
        # pre-create the table with a NARROWER v1 schema
        old = pa.schema([("finding_id", pa.string())])
        cat.tables[cfg.identifier] = _FakeTable(old)
        out = register_findings(report, cfg, catalog=cat)
        assert out["schema_evolved"] is True
        assert cat.tables[cfg.identifier].evolutions == 1

    def test_unconfigured_and_missing_dependency(self, report):
        with pytest.raises(RuntimeError, match="not configured"):
            register_findings(report, IcebergCatalogConfig())
        with pytest.raises(RuntimeError, match="pyiceberg"):
            register_findings(report,
                              IcebergCatalogConfig(catalog_url="https://x"))

    def test_arrow_table_matches_parquet_rows(self, report):
        t = to_arrow_table(report)
        assert t.num_rows == len(list(report.to_findings()))
        assert "finding_id" in t.schema.names
