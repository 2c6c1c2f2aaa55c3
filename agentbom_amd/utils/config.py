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
