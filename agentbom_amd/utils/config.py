"""Environment-driven configuration knobs.

Every tunable is an ``AGENT_BOM_*`` environment variable parsed once at import
time, with defaults identical to the reference implementation so that risk
scores and thresholds are numerically reproducible
(reference: src/agent_bom/config.py:22-260).
"""

from __future__ import annotations

import os


def _float(name: str, default: float) -> float:
    raw = os.environ.get(name)
    if raw is None or raw == "":
        return default
    try:
        return float(raw)
    except ValueError:
        return default


def _int(name: str, default: int) -> int:
    raw = os.environ.get(name)
    if raw is None or raw == "":
        return default
    try:
        return int(raw)
    except ValueError:
        return default


def _bool(name: str, default: bool) -> bool:
    raw = os.environ.get(name)
    if raw is None or raw == "":
        return default
    return raw.strip().lower() in ("1", "true", "yes", "on")


def _str(name: str, default: str) -> str:
    raw = os.environ.get(name)
    return default if raw is None else raw


# ── EPSS thresholds ─────────────────────────────────────────────────────────
EPSS_CRITICAL_THRESHOLD = _float("AGENT_BOM_EPSS_CRITICAL_THRESHOLD", 0.7)
EPSS_HIGH_LIKELY_THRESHOLD = _float("AGENT_BOM_EPSS_HIGH_THRESHOLD", 0.3)

# ── Blast-radius risk scoring (BlastRadius.calculate_risk_score) ────────────
# Base severity starts at 80% of max (CRITICAL = 8.0/10) leaving headroom for
# reach amplifiers; each severity step drops 2 points.
RISK_BASE_CRITICAL = _float("AGENT_BOM_RISK_BASE_CRITICAL", 8.0)
RISK_BASE_HIGH = _float("AGENT_BOM_RISK_BASE_HIGH", 6.0)
RISK_BASE_MEDIUM = _float("AGENT_BOM_RISK_BASE_MEDIUM", 4.0)
RISK_BASE_LOW = _float("AGENT_BOM_RISK_BASE_LOW", 2.0)

# Reach amplifiers: weight x count, capped.
RISK_AGENT_WEIGHT = _float("AGENT_BOM_RISK_AGENT_WEIGHT", 0.5)
RISK_AGENT_CAP = _float("AGENT_BOM_RISK_AGENT_CAP", 2.0)
RISK_CRED_WEIGHT = _float("AGENT_BOM_RISK_CRED_WEIGHT", 0.3)
RISK_CRED_CAP = _float("AGENT_BOM_RISK_CRED_CAP", 1.5)
RISK_TOOL_WEIGHT = _float("AGENT_BOM_RISK_TOOL_WEIGHT", 0.1)
RISK_TOOL_CAP = _float("AGENT_BOM_RISK_TOOL_CAP", 1.0)

# Conditional boosts.
RISK_AI_BOOST = _float("AGENT_BOM_RISK_AI_BOOST", 0.5)
RISK_KEV_BOOST = _float("AGENT_BOM_RISK_KEV_BOOST", 1.0)
RISK_EPSS_BOOST = _float("AGENT_BOM_RISK_EPSS_BOOST", 0.5)

# OpenSSF Scorecard tiers: poorly-maintained packages amplify risk.
RISK_SCORECARD_TIER1_THRESHOLD = _float("AGENT_BOM_RISK_SCORECARD_T1", 3.0)
RISK_SCORECARD_TIER1_BOOST = _float("AGENT_BOM_RISK_SCORECARD_B1", 0.75)
RISK_SCORECARD_TIER2_THRESHOLD = _float("AGENT_BOM_RISK_SCORECARD_T2", 5.0)
RISK_SCORECARD_TIER2_BOOST = _float("AGENT_BOM_RISK_SCORECARD_B2", 0.5)
RISK_SCORECARD_TIER3_THRESHOLD = _float("AGENT_BOM_RISK_SCORECARD_T3", 7.0)
RISK_SCORECARD_TIER3_BOOST = _float("AGENT_BOM_RISK_SCORECARD_B3", 0.25)

# Graph-walk reachability nudge (only when the graph engine produced a
# definitive answer; None leaves scoring unchanged).
RISK_REACHABLE_BOOST = _float("AGENT_BOM_RISK_REACHABLE_BOOST", 0.5)
RISK_UNREACHABLE_PENALTY = _float("AGENT_BOM_RISK_UNREACHABLE_PENALTY", 0.5)

# ── MCP server risk scoring (risk_analyzer) ────────────────────────────────
SERVER_RISK_BASE_CEILING = _float("AGENT_BOM_SERVER_RISK_CEILING", 7.0)
SERVER_RISK_TOOL_WEIGHT = _float("AGENT_BOM_SERVER_TOOL_WEIGHT", 0.15)
SERVER_RISK_TOOL_CAP = _float("AGENT_BOM_SERVER_TOOL_CAP", 1.5)
SERVER_RISK_CRED_WEIGHT = _float("AGENT_BOM_SERVER_CRED_WEIGHT", 0.5)
SERVER_RISK_CRED_CAP = _float("AGENT_BOM_SERVER_CRED_CAP", 2.0)
SERVER_RISK_COMBO_WEIGHT = _float("AGENT_BOM_SERVER_COMBO_WEIGHT", 0.3)
SERVER_RISK_COMBO_CAP = _float("AGENT_BOM_SERVER_COMBO_CAP", 1.5)
SERVER_RISK_REGISTRY_HIGH_FLOOR = _float("AGENT_BOM_SERVER_REG_HIGH", 6.0)
SERVER_RISK_REGISTRY_MEDIUM_FLOOR = _float("AGENT_BOM_SERVER_REG_MEDIUM", 3.0)
SERVER_RISK_CRITICAL_THRESHOLD = _float("AGENT_BOM_SERVER_CRITICAL", 9.0)
SERVER_RISK_HIGH_THRESHOLD = _float("AGENT_BOM_SERVER_HIGH", 7.0)
SERVER_RISK_MEDIUM_THRESHOLD = _float("AGENT_BOM_SERVER_MEDIUM", 4.0)

# ── HTTP client ─────────────────────────────────────────────────────────────
HTTP_MAX_RETRIES = _int("AGENT_BOM_HTTP_MAX_RETRIES", 3)
HTTP_INITIAL_BACKOFF = _float("AGENT_BOM_HTTP_INITIAL_BACKOFF", 1.0)
HTTP_MAX_BACKOFF = _float("AGENT_BOM_HTTP_MAX_BACKOFF", 30.0)
HTTP_DEFAULT_TIMEOUT = _float("AGENT_BOM_HTTP_DEFAULT_TIMEOUT", 30.0)
HTTP_RATE_LIMIT_BREAKER_THRESHOLD = _int("AGENT_BOM_HTTP_RATE_LIMIT_BREAKER_THRESHOLD", 3)
CLOUD_DISCOVERY_TIMEOUT = _float("AGENT_BOM_CLOUD_DISCOVERY_TIMEOUT", 45.0)

# ── Scanner batching ────────────────────────────────────────────────────────
SCANNER_MAX_CONCURRENT = _int("AGENT_BOM_SCANNER_MAX_CONCURRENT", 10)
SCANNER_OSV_BATCH_CONCURRENCY = _int("AGENT_BOM_SCANNER_OSV_BATCH_CONCURRENCY", 3)
SCANNER_BATCH_DELAY = _float("AGENT_BOM_SCANNER_BATCH_DELAY", 0.5)
SCANNER_BATCH_SIZE = _int("AGENT_BOM_SCANNER_BATCH_SIZE", 1000)
GHSA_UNAUTH_PACKAGE_BUDGET = _int("AGENT_BOM_GHSA_UNAUTH_PACKAGE_BUDGET", 25)
ENABLE_CPE_MATCH = _bool("AGENT_BOM_ENABLE_CPE_MATCH", False)

# ── Matching policy ─────────────────────────────────────────────────────────
INCLUDE_UNFIXED = _bool("AGENT_BOM_INCLUDE_UNFIXED", False)

# ── Deploy-decision thresholds (MCP should_i_deploy) ────────────────────────
DEPLOY_WARN_RISK = _float("AGENT_BOM_DEPLOY_WARN_RISK", 40.0)
DEPLOY_BLOCK_RISK = _float("AGENT_BOM_DEPLOY_BLOCK_RISK", 80.0)

# ── GPU engine ──────────────────────────────────────────────────────────────
# Whether the HIP engine is required.  On a machine with a visible GPU the
# native path is mandatory and a missing extension raises; set to "0" only
# for CPU-only development boxes (the CPU reference engine is then used).
GPU_REQUIRED = _bool("AGENT_BOM_GPU_REQUIRED", True)
# Entity-count threshold above which the estate graph is built GPU-resident.
GPU_GRAPH_THRESHOLD = _int("AGENT_BOM_GPU_GRAPH_THRESHOLD", 5000)

# ── Offline mode ────────────────────────────────────────────────────────────
OFFLINE = _bool("AGENT_BOM_OFFLINE", False)

# ── Gateway / proxy runtime knobs ───────────────────────────────────────────
GATEWAY_BREAKER_THRESHOLD = _int("AGENT_BOM_GATEWAY_BREAKER_THRESHOLD", 5)
GATEWAY_BREAKER_COOLDOWN_S = _float("AGENT_BOM_GATEWAY_BREAKER_COOLDOWN_S", 30.0)
GATEWAY_COST_BUDGET = _float("AGENT_BOM_GATEWAY_COST_BUDGET", 1000.0)
GATEWAY_COST_WINDOW_S = _float("AGENT_BOM_GATEWAY_COST_WINDOW_S", 3600.0)
PROXY_BLOCK_ON_WARN = _bool("AGENT_BOM_PROXY_BLOCK_ON_WARN", False)
PROXY_AUDIT_PATH = _str("AGENT_BOM_PROXY_AUDIT_PATH", "")
SANDBOX_RUNTIME = _str("AGENT_BOM_SANDBOX_RUNTIME", "docker")
SANDBOX_IMAGE = _str("AGENT_BOM_SANDBOX_IMAGE", "node:20-slim")
SANDBOX_MEMORY = _str("AGENT_BOM_SANDBOX_MEMORY", "512m")
SANDBOX_PIDS_LIMIT = _int("AGENT_BOM_SANDBOX_PIDS_LIMIT", 256)

# ── Runtime detector thresholds ─────────────────────────────────────────────
DETECTOR_RATE_LIMIT_WINDOW_S = _float("AGENT_BOM_DETECTOR_RATE_WINDOW_S", 60.0)
DETECTOR_RATE_LIMIT_MAX_CALLS = _int("AGENT_BOM_DETECTOR_RATE_MAX_CALLS", 120)
DETECTOR_ARG_MAX_BYTES = _int("AGENT_BOM_DETECTOR_ARG_MAX_BYTES", 65536)
DETECTOR_ENTROPY_THRESHOLD = _float("AGENT_BOM_DETECTOR_ENTROPY_THRESHOLD", 4.5)
DETECTOR_REPLAY_WINDOW_S = _float("AGENT_BOM_DETECTOR_REPLAY_WINDOW_S", 300.0)

# ── API / control plane ─────────────────────────────────────────────────────
API_MAX_BODY_BYTES = _int("AGENT_BOM_API_MAX_BODY_BYTES", 10 * 1024 * 1024)
API_SCAN_QUOTA_PER_HOUR = _int("AGENT_BOM_API_SCAN_QUOTA_PER_HOUR", 60)
API_SCAN_CONCURRENCY = _int("AGENT_BOM_API_SCAN_CONCURRENCY", 4)
API_STUCK_JOB_AGE_S = _float("AGENT_BOM_API_STUCK_JOB_AGE_S", 600.0)
GRAPH_SNAPSHOT_RETENTION = _int("AGENT_BOM_GRAPH_SNAPSHOT_RETENTION", 10)
GRAPH_NODE_BUDGET = _int("AGENT_BOM_GRAPH_NODE_BUDGET", 250_000)
OIDC_CLOCK_LEEWAY_S = _float("AGENT_BOM_OIDC_CLOCK_LEEWAY_S", 60.0)
DELEGATION_TOKEN_TTL_S = _float("AGENT_BOM_DELEGATION_TOKEN_TTL_S", 3600.0)
WEBHOOK_TIMEOUT_S = _float("AGENT_BOM_WEBHOOK_TIMEOUT_S", 10.0)

# ── Graph / path engine ─────────────────────────────────────────────────────
PATH_DP_THRESHOLD = _int("AGENT_BOM_PATH_DP_THRESHOLD", 20_000)
PATH_MAX_DEPTH = _int("AGENT_BOM_PATH_MAX_DEPTH", 6)
PATH_MAX_PATHS = _int("AGENT_BOM_PATH_MAX_PATHS", 100)
BLAST_RADIUS_MAX_DEPTH = _int("AGENT_BOM_BLAST_RADIUS_MAX_DEPTH", 5)
IMPACT_QUERY_MAX_NODES = _int("AGENT_BOM_IMPACT_QUERY_MAX_NODES", 4096)
ROLLUP_MAX_CONTAINERS = _int("AGENT_BOM_ROLLUP_MAX_CONTAINERS", 200)

# ── Transitive expansion / registries ───────────────────────────────────────
TRANSITIVE_MAX_DEPTH = _int("AGENT_BOM_TRANSITIVE_MAX_DEPTH", 3)
TRANSITIVE_MAX_NODES = _int("AGENT_BOM_TRANSITIVE_MAX_NODES", 2000)
TRANSITIVE_CACHE_SIZE = _int("AGENT_BOM_TRANSITIVE_CACHE_SIZE", 5000)
NPM_REGISTRY_URL = _str("AGENT_BOM_NPM_REGISTRY_URL", "https://registry.npmjs.org")
PYPI_API_URL = _str("AGENT_BOM_PYPI_API_URL", "https://pypi.org/pypi")
GO_PROXY_URL = _str("AGENT_BOM_GO_PROXY_URL", "https://proxy.golang.org")

# ── Live advisory acquisition ───────────────────────────────────────────────
OSV_CACHE_TTL_S = _float("AGENT_BOM_OSV_CACHE_TTL_S", 6 * 3600.0)
EPSS_PAGE_SIZE = _int("AGENT_BOM_EPSS_PAGE_SIZE", 10_000)
NVD_PAGE_SIZE = _int("AGENT_BOM_NVD_PAGE_SIZE", 2000)
NVD_API_KEY = _str("AGENT_BOM_NVD_API_KEY", "")
GITHUB_TOKEN = _str("AGENT_BOM_GITHUB_TOKEN", "")
DB_FRESH_DAYS = _float("AGENT_BOM_DB_FRESH_DAYS", 3.0)
DB_STALE_DAYS = _float("AGENT_BOM_DB_STALE_DAYS", 30.0)

# ── Scan surface bounds ─────────────────────────────────────────────────────
PROMPT_SCAN_MAX_FILES = _int("AGENT_BOM_PROMPT_SCAN_MAX_FILES", 500)
PII_SCAN_MAX_FILES = _int("AGENT_BOM_PII_SCAN_MAX_FILES", 300)
BROWSER_EXT_MAX_MANIFESTS = _int("AGENT_BOM_BROWSER_EXT_MAX_MANIFESTS", 500)
SECRETS_MAX_FILE_BYTES = _int("AGENT_BOM_SECRETS_MAX_FILE_BYTES", 2_000_000)
UNTRUSTED_TEXT_MAX_LEN = _int("AGENT_BOM_UNTRUSTED_TEXT_MAX_LEN", 500)
LICENSE_DENYLIST = _str("AGENT_BOM_LICENSE_DENYLIST", "GPL-3.0,AGPL-3.0,SSPL-1.0")

# ── Multi-GPU / distributed engine ──────────────────────────────────────────
DIST_EXCHANGE_CAP = _int("AGENT_BOM_DIST_EXCHANGE_CAP", 0)  # 0 = auto
DIST_MAX_BFS_LEVELS = _int("AGENT_BOM_DIST_MAX_BFS_LEVELS", 64)

# ── Live feed endpoints (override for mirrors / air-gapped relays) ─────────
OSV_API_URL = _str("AGENT_BOM_OSV_API_URL", "https://api.osv.dev/v1")
OSV_BULK_URL = _str("AGENT_BOM_OSV_BULK_URL",
                    "https://osv-vulnerabilities.storage.googleapis.com")
GHSA_API_URL = _str("AGENT_BOM_GHSA_API_URL", "https://api.github.com/advisories")
EPSS_API_URL = _str("AGENT_BOM_EPSS_API_URL", "https://api.first.org/data/v1/epss")
KEV_URL = _str("AGENT_BOM_KEV_URL",
               "https://www.cisa.gov/sites/default/files/feeds/"
               "known_exploited_vulnerabilities.json")
NVD_API_URL = _str("AGENT_BOM_NVD_API_URL",
                   "https://services.nvd.nist.gov/rest/json/cves/2.0")
OSV_BATCH_CHUNK = _int("AGENT_BOM_OSV_BATCH_CHUNK", 1000)

# ── Identity lifecycle ──────────────────────────────────────────────────────
IDENTITY_TOKEN_TTL_HOURS = _float("AGENT_BOM_IDENTITY_TOKEN_TTL_HOURS", 24.0)
IDENTITY_ROTATE_OVERLAP_MIN = _float("AGENT_BOM_IDENTITY_ROTATE_OVERLAP_MIN", 15.0)
JIT_GRANT_TTL_MINUTES = _float("AGENT_BOM_JIT_GRANT_TTL_MINUTES", 60.0)

# ── Repo inventory / CI scanning bounds ────────────────────────────────────
REPO_INVENTORY_MAX_FILES = _int("AGENT_BOM_REPO_INVENTORY_MAX_FILES", 4000)
REPO_INVENTORY_MAX_DEPTH = _int("AGENT_BOM_REPO_INVENTORY_MAX_DEPTH", 6)
REPO_INVENTORY_MAX_DIR_ENTRIES = _int(
    "AGENT_BOM_REPO_INVENTORY_MAX_DIR_ENTRIES", 400)
TAINT_PROPAGATION_ROUNDS = _int("AGENT_BOM_TAINT_PROPAGATION_ROUNDS", 2)

# ── Optional analytics stores ───────────────────────────────────────────────
CLICKHOUSE_TIMEOUT_S = _float("AGENT_BOM_CLICKHOUSE_TIMEOUT_S", 30.0)
SNOWFLAKE_TIMEOUT_S = _float("AGENT_BOM_SNOWFLAKE_TIMEOUT_S", 60.0)
SNOWFLAKE_JWT_LIFETIME_S = _int("AGENT_BOM_SNOWFLAKE_JWT_LIFETIME_S", 3600)
