"""Iceberg REST-catalog registration for the findings lake table.

Reference parity: src/agent_bom/output/iceberg_catalog.py — beyond flat
``.parquet`` files, the same findings rows land as an Apache Iceberg
table snapshot so lake consumers query ONE versioned table.

The catalog client is a PROTOCOL here (``create_namespace_if_not_exists``,
``create_table_if_not_exists``, table ``append``/``schema``/
``update_schema().union_by_name().commit()``) — production deployments
hand in ``pyiceberg``'s RestCatalog (not installed in this offline
build, imported lazily and gated with a clear error); tests inject a
fake.  Schema evolution is ADDITIVE ONLY: ``union_by_name`` preserves
field ids and refuses type rewrites, so a v(n+1) batch never silently
rewrites consumers' columns.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Any, Optional
from urllib.parse import urlsplit

DEFAULT_NAMESPACE = "agent_bom"
DEFAULT_TABLE = "findings"


@dataclass
class IcebergCatalogConfig:
    catalog_url: str = ""
    namespace: str = DEFAULT_NAMESPACE
    table: str = DEFAULT_TABLE
    token: str = ""
    credential: str = ""     # OAuth2 client_id:client_secret
    warehouse: str = ""

    @property
    def enabled(self) -> bool:
        return bool(self.catalog_url)

    @property
    def identifier(self) -> tuple[str, str]:
        return (self.namespace, self.table)

    @classmethod
    def from_env(cls, env: Optional[dict] = None) -> "IcebergCatalogConfig":
        e = env if env is not None else os.environ
        return cls(
            catalog_url=e.get("AGENT_BOM_ICEBERG_CATALOG_URL", ""),
            namespace=e.get("AGENT_BOM_ICEBERG_NAMESPACE", DEFAULT_NAMESPACE),
            table=e.get("AGENT_BOM_ICEBERG_TABLE", DEFAULT_TABLE),
            token=e.get("AGENT_BOM_ICEBERG_TOKEN", ""),
            credential=e.get("AGENT_BOM_ICEBERG_CREDENTIAL", ""),
            warehouse=e.get("AGENT_BOM_ICEBERG_WAREHOUSE", ""))

    def catalog_properties(self) -> dict[str, str]:
        """RestCatalog property map (credentials stay out of the URL)."""
        props = {"uri": self.catalog_url}
        if self.token:
            props["token"] = self.token
        if self.credential:
            props["credential"] = self.credential
        if self.warehouse:
            props["warehouse"] = self.warehouse
        return props


def display_catalog_url(value: str) -> str:
    """Credential-, path- and query-free endpoint for user-facing output."""
    try:
        parts = urlsplit(value)
    except ValueError:
        return "<invalid url>"
    host = parts.hostname or ""
    port = f":{parts.port}" if parts.port else ""
    return f"{parts.scheme}://{host}{port}" if parts.scheme and host \
        else "<invalid url>"


def to_arrow_table(report):
    """The unified findings rows as one Arrow table (shared lake schema —
    identical rows to output/misc_fmt.to_parquet_bytes)."""
    import io

    import pyarrow.parquet as pq

    from agentbom_amd.output.misc_fmt import to_parquet_bytes

    return pq.read_table(io.BytesIO(to_parquet_bytes(report)))


def _schema_names(schema: Any) -> set[str]:
    names = getattr(schema, "names", None)
    if names is not None:
        return set(names)
    return {f.name for f in schema}


def _evolve_additive(table: Any, arrow_schema: Any) -> bool:
    """Commit an additive schema union when the batch adds columns."""
    current = table.schema() if callable(getattr(table, "schema", None)) \
        else table.schema
    if current is None:
        raise RuntimeError("Iceberg table did not expose its schema")
    if _schema_names(arrow_schema) <= _schema_names(current):
        return False
    update = table.update_schema()
    update.union_by_name(arrow_schema)
    update.commit()
    return True


def _build_catalog(config: IcebergCatalogConfig) -> Any:
    try:
        from pyiceberg.catalog.rest import RestCatalog  # type: ignore
    except ImportError as exc:
        raise RuntimeError(
            "Iceberg export needs the optional pyiceberg dependency "
            "(pip install pyiceberg) — it speaks the REST catalog protocol "
            "and writes the Avro manifests a snapshot commit requires"
        ) from exc
    return RestCatalog("agent_bom", **config.catalog_properties())


def register_findings(report, config: IcebergCatalogConfig,
                      catalog: Any = None) -> dict[str, Any]:
    """Append the findings as a new table snapshot.

    Creates namespace + table when absent, evolves the schema additively,
    appends, and reports the resulting snapshot id.  ``catalog`` is
    injectable for tests; production builds a RestCatalog from config.
    """
    if not config.enabled:
        raise RuntimeError(
            "Iceberg catalog export is not configured — set "
            "AGENT_BOM_ICEBERG_CATALOG_URL (or --iceberg-catalog-url)")
    arrow = to_arrow_table(report)
    catalog = catalog if catalog is not None else _build_catalog(config)
    catalog.create_namespace_if_not_exists((config.namespace,))
    table = catalog.create_table_if_not_exists(config.identifier,
                                               schema=arrow.schema)
    evolved = _evolve_additive(table, arrow.schema)
    table.append(arrow)
    snapshot = None
    current = getattr(table, "current_snapshot", None)
    if callable(current):
        snap = current()
        snapshot = getattr(snap, "snapshot_id", None) if snap else None
    return {"identifier": config.identifier, "rows": arrow.num_rows,
            "schema_evolved": evolved, "snapshot_id": snapshot,
            "catalog_url": display_catalog_url(config.catalog_url)}
