"""OSV / GHSA / EPSS / KEV record ingestion into the advisory store.

Reference: src/agent_bom/db/sync.py (sync_osv :602, sync_ghsa :1341,
sync_epss :1076, sync_kev :1147) — this build ingests the same record
formats from local files/offline bundles (the container has no egress;
the network fetch layer plugs in where ``iter_records`` is called).

OSV semantics preserved:
- ranges[].events alternate introduced/fixed/last_affected -> one window
  per (introduced, terminator) pair, several windows per range;
- type GIT ranges are skipped (commit bounds are undecidable — the
  matcher fails closed on SHA bounds anyway);
- versions[] lists emit degenerate [v, v] windows (explicit hit =>
  affected; a miss still falls through to the SEMVER/ECOSYSTEM windows).
"""

from __future__ import annotations

import csv
import io
import json
from pathlib import Path
from typing import Any, Iterable, Optional

from agentbom_amd.db.arena import AdvisoryWindow
from agentbom_amd.db.store import AdvisoryStore
from agentbom_amd.models.core import Severity

_ECO_MAP = {
    "pypi": "pypi", "npm": "npm", "go": "go", "crates.io": "cargo",
    "maven": "maven", "nuget": "nuget", "rubygems": "rubygems",
    "packagist": "composer", "hex": "hex", "pub": "pub",
    "debian": "deb", "alpine": "apk", "swifturl": "swifturl",
}


def _dict(value: Any) -> dict:
    """Coerce an untrusted field to a dict (fail-soft ingestion contract)."""
    return value if isinstance(value, dict) else {}


def _list(value: Any) -> list:
    return value if isinstance(value, list) else []


def _severity_from_osv(record: dict[str, Any]) -> tuple[Severity, Optional[float]]:
    cvss = None
    for sev in _list(record.get("severity")):
        if not isinstance(sev, dict):
            continue
        if str(sev.get("type", "")).startswith("CVSS"):
            score = sev.get("score", "")
            # score may be a vector string or a number
            try:
                cvss = float(score)
            except (TypeError, ValueError):
                cvss = _cvss_base_from_vector(str(score))
    db_sev = _dict(record.get("database_specific")).get("severity", "")
    label = str(db_sev).lower()
    if label in ("critical", "high", "medium", "low"):
        return Severity(label), cvss
    if cvss is not None:
        if cvss >= 9.0:
            return Severity.CRITICAL, cvss
        if cvss >= 7.0:
            return Severity.HIGH, cvss
        if cvss >= 4.0:
            return Severity.MEDIUM, cvss
        return Severity.LOW, cvss
    return Severity.UNKNOWN, cvss


def _cvss_base_from_vector(vector: str) -> Optional[float]:
    """CVSS 3.x base-score computation from a vector string (first 8 metrics)."""
    if not vector.startswith("CVSS:3"):
        return None
    try:
        m = dict(p.split(":", 1) for p in vector.split("/")[1:])
        av = {"N": 0.85, "A": 0.62, "L": 0.55, "P": 0.2}[m["AV"]]
        ac = {"L": 0.77, "H": 0.44}[m["AC"]]
        ui = {"N": 0.85, "R": 0.62}[m["UI"]]
        scope_changed = m["S"] == "C"
        pr_map = ({"N": 0.85, "L": 0.68, "H": 0.5} if scope_changed
                  else {"N": 0.85, "L": 0.62, "H": 0.27})
        pr = pr_map[m["PR"]]
        cia = {"H": 0.56, "L": 0.22, "N": 0.0}
        c, i, a = cia[m["C"]], cia[m["I"]], cia[m["A"]]
        iss = 1 - (1 - c) * (1 - i) * (1 - a)
        impact = (7.52 * (iss - 0.029) - 3.25 * (iss - 0.02) ** 15
                  if scope_changed else 6.42 * iss)
        exploitability = 8.22 * av * ac * pr * ui
        if impact <= 0:
            return 0.0
        import math

        raw = min(1.08 * (impact + exploitability) if scope_changed
                  else impact + exploitability, 10.0)
        return math.ceil(raw * 10) / 10
    except (KeyError, ValueError):
        return None


def parse_osv_record(record: dict[str, Any]) -> list[AdvisoryWindow]:
    """One OSV advisory -> windows (one per affected range segment)."""
    vuln_id = str(record.get("id") or "")
    if not vuln_id:
        return []
    details = record.get("details")
    summary = str(record.get("summary")
                  or (details if isinstance(details, str) else ""))[:200]
    severity, cvss = _severity_from_osv(record)
    aliases = tuple(str(a) for a in _list(record.get("aliases")))
    cwes = tuple(str(c) for c in
                 _list(_dict(record.get("database_specific")).get("cwe_ids")))
    out: list[AdvisoryWindow] = []

    for affected in _list(record.get("affected")):
        if not isinstance(affected, dict):
            continue
        pkg = _dict(affected.get("package"))
        eco_raw = str(pkg.get("ecosystem", "")).split(":")[0].lower()
        eco = _ECO_MAP.get(eco_raw, eco_raw)
        name = str(pkg.get("name") or "")
        if not name:
            continue
        aff_cwes = cwes or tuple(str(c) for c in
                                 _list(_dict(affected.get("database_specific"))
                                       .get("cwes")))

        def mk(intro, fixed, last, unfixed=False):
            out.append(AdvisoryWindow(
                ecosystem=eco, package_name=name, vuln_id=vuln_id,
                introduced=intro, fixed=fixed, last_affected=last,
                severity=severity, cvss_score=cvss, summary=summary,
                cwe_ids=aff_cwes, aliases=aliases,
                fixed_version=fixed, unfixed=unfixed,
            ))

        emitted = False
        for rng in _list(affected.get("ranges")):
            if not isinstance(rng, dict):
                continue
            if rng.get("type") == "GIT":
                continue  # commit bounds: undecidable, fail closed
            intro: Optional[str] = None
            has_terminator = False
            for event in _list(rng.get("events")):
                if not isinstance(event, dict):
                    continue
                if "introduced" in event:
                    # a new introduced before a terminator closes the prior
                    # window as unfixed
                    if intro is not None and not has_terminator:
                        mk(intro, None, None, unfixed=True)
                    intro = str(event["introduced"])
                    has_terminator = False
                elif "fixed" in event:
                    mk(intro if intro is not None else "0", str(event["fixed"]), None)
                    has_terminator = True
                    emitted = True
                elif "last_affected" in event:
                    mk(intro if intro is not None else "0", None,
                       str(event["last_affected"]))
                    has_terminator = True
                    emitted = True
            if intro is not None and not has_terminator:
                mk(intro, None, None, unfixed=True)
                emitted = True

        # explicit versions[] list: degenerate [v, v] windows
        for v in _list(affected.get("versions")):
            mk(str(v), None, str(v))
            emitted = True

        if not emitted:
            # affected entry with no ranges at all: unfixed advisory
            mk("0", None, None, unfixed=True)
    return out


def iter_osv_files(path: str | Path) -> Iterable[dict[str, Any]]:
    """Yield OSV records from a file / directory of .json files / .ndjson."""
    p = Path(path)
    files = [p] if p.is_file() else sorted(p.rglob("*.json"))
    for f in files:
        text = f.read_text()
        if f.suffix == ".ndjson" or "\n{" in text[:10000] and not text.lstrip().startswith("["):
            for line in text.splitlines():
                line = line.strip()
                if line.startswith("{"):
                    yield json.loads(line)
            continue
        data = json.loads(text)
        if isinstance(data, list):
            yield from data
        else:
            yield data


def sync_osv(store: AdvisoryStore, source: str | Path) -> int:
    windows: list[AdvisoryWindow] = []
    for record in iter_osv_files(source):
        windows.extend(parse_osv_record(record))
    return store.ingest_windows(windows, source="osv")


def sync_epss(store: AdvisoryStore, csv_path: str | Path) -> int:
    """EPSS bulk CSV: cve,epss,percentile (comment lines skipped)."""
    text = Path(csv_path).read_text()
    n = 0
    reader = csv.reader(io.StringIO(text))
    from agentbom_amd.db.store import _now

    for row in reader:
        if not row or row[0].startswith("#") or row[0] == "cve":
            continue
        try:
            cve, epss, pct = row[0], float(row[1]), float(row[2])
        except (IndexError, ValueError):
            continue
        store.conn.execute(
            "INSERT OR REPLACE INTO epss_scores(cve_id, probability, percentile,"
            " updated_at) VALUES (?,?,?,?)", (cve, epss, pct * 100.0, _now()),
        )
        n += 1
    store.conn.execute(
        "INSERT OR REPLACE INTO sync_meta(source, last_synced, record_count)"
        " VALUES ('epss', ?, ?)", (_now(), n),
    )
    store.conn.commit()
    return n


def sync_kev(store: AdvisoryStore, json_path: str | Path) -> int:
    """CISA KEV catalog JSON (vulnerabilities[] with cveID/dateAdded/dueDate)."""
    data = json.loads(Path(json_path).read_text())
    n = 0
    from agentbom_amd.db.store import _now

    for entry in data.get("vulnerabilities", []) or []:
        cve = entry.get("cveID")
        if not cve:
            continue
        store.conn.execute(
            "INSERT OR REPLACE INTO kev_entries(cve_id, date_added, due_date, product,"
            " vendor_project) VALUES (?,?,?,?,?)",
            (cve, entry.get("dateAdded"), entry.get("dueDate"),
             entry.get("product"), entry.get("vendorProject")),
        )
        n += 1
    store.conn.execute(
        "INSERT OR REPLACE INTO sync_meta(source, last_synced, record_count)"
        " VALUES ('kev', ?, ?)", (_now(), n),
    )
    store.conn.commit()
    return n
