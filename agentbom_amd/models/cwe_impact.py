"""CWE-aware impact classification for blast-radius accuracy.

Maps CWE weakness ids to impact categories that bound which credentials and
tools a vulnerability can realistically reach (reference:
src/agent_bom/cwe_impact.py).  Unknown CWE data never infers RCE.

The numeric IMPACT_CODE table at the bottom is the device-side encoding used
by the GPU blast-radius kernels (ops/csrc/blast.hip): impact category travels
as a u8 and the credential/tool filter becomes a bitmask test.
"""

from __future__ import annotations

from typing import Optional

IMPACT_CODE_EXECUTION = "code-execution"
IMPACT_CREDENTIAL_ACCESS = "credential-access"
IMPACT_FILE_ACCESS = "file-access"
IMPACT_INJECTION = "injection"
IMPACT_SSRF = "ssrf"
IMPACT_DATA_LEAK = "data-leak"
IMPACT_AVAILABILITY = "availability"
IMPACT_CLIENT_SIDE = "client-side"
IMPACT_UNKNOWN = "unknown"

# Most → least severe; worst-case selection over a finding's CWE list.
IMPACT_SEVERITY_ORDER = [
    IMPACT_CODE_EXECUTION,
    IMPACT_CREDENTIAL_ACCESS,
    IMPACT_FILE_ACCESS,
    IMPACT_SSRF,
    IMPACT_INJECTION,
    IMPACT_DATA_LEAK,
    IMPACT_AVAILABILITY,
    IMPACT_CLIENT_SIDE,
]

_CE = IMPACT_CODE_EXECUTION
_CA = IMPACT_CREDENTIAL_ACCESS
_FA = IMPACT_FILE_ACCESS
_IN = IMPACT_INJECTION
_SS = IMPACT_SSRF
_DL = IMPACT_DATA_LEAK
_AV = IMPACT_AVAILABILITY
_CS = IMPACT_CLIENT_SIDE

CWE_IMPACT_CATEGORIES: dict[str, str] = {
    # code execution
    "CWE-77": _CE, "CWE-78": _CE, "CWE-94": _CE, "CWE-95": _CE, "CWE-96": _CE,
    "CWE-98": _CE, "CWE-502": _CE, "CWE-787": _CE, "CWE-788": _CE, "CWE-416": _CE,
    "CWE-119": _CE, "CWE-120": _CE, "CWE-122": _CE, "CWE-125": _CE, "CWE-190": _CE,
    "CWE-434": _CE, "CWE-917": _CE, "CWE-1321": _CE, "CWE-913": _CE,
    # credential access
    "CWE-287": _CA, "CWE-306": _CA, "CWE-307": _CA, "CWE-347": _CA, "CWE-384": _CA,
    "CWE-522": _CA, "CWE-798": _CA, "CWE-862": _CA, "CWE-863": _CA, "CWE-1259": _CA,
    # file access
    "CWE-22": _FA, "CWE-23": _FA, "CWE-36": _FA, "CWE-59": _FA, "CWE-73": _FA, "CWE-67": _FA,
    # injection
    "CWE-89": _IN, "CWE-90": _IN, "CWE-91": _IN, "CWE-943": _IN, "CWE-1236": _IN,
    # ssrf
    "CWE-918": _SS,
    # data leak
    "CWE-200": _DL, "CWE-209": _DL, "CWE-215": _DL, "CWE-532": _DL, "CWE-538": _DL,
    "CWE-312": _DL, "CWE-319": _DL, "CWE-327": _DL, "CWE-326": _DL, "CWE-330": _DL,
    "CWE-331": _DL, "CWE-338": _DL,
    # availability
    "CWE-400": _AV, "CWE-770": _AV, "CWE-674": _AV, "CWE-834": _AV, "CWE-835": _AV,
    "CWE-1333": _AV, "CWE-410": _AV, "CWE-404": _AV, "CWE-407": _AV, "CWE-409": _AV,
    # client side
    "CWE-79": _CS, "CWE-80": _CS, "CWE-352": _CS, "CWE-601": _CS, "CWE-1021": _CS,
    "CWE-524": _CS, "CWE-539": _CS, "CWE-614": _CS, "CWE-1004": _CS, "CWE-1275": _CS,
    # ambiguous input-validation class → conservative data-leak
    "CWE-20": _DL, "CWE-116": _DL, "CWE-173": _DL, "CWE-670": _DL, "CWE-754": _DL,
    "CWE-1286": _DL,
}


def classify_cwe_impact(cwe_ids: list[str]) -> str:
    """Worst-case impact category over CWE ids; ``unknown`` when unmapped."""
    if not cwe_ids:
        return IMPACT_UNKNOWN
    best = len(IMPACT_SEVERITY_ORDER)
    for cwe in cwe_ids:
        key = cwe if cwe.startswith("CWE-") else cwe.upper()
        cat = CWE_IMPACT_CATEGORIES.get(key)
        if cat is not None:
            best = min(best, IMPACT_SEVERITY_ORDER.index(cat))
    return IMPACT_SEVERITY_ORDER[best] if best < len(IMPACT_SEVERITY_ORDER) else IMPACT_UNKNOWN


_DB_CREDENTIAL_PATTERNS = frozenset(
    {"database", "db_", "mysql", "postgres", "mongo", "redis", "dsn", "sql",
     "clickhouse", "snowflake", "supabase"}
)


def _is_db_credential(name: str) -> bool:
    lower = name.lower()
    return any(p in lower for p in _DB_CREDENTIAL_PATTERNS)


_FULL_CRED_REACH = {IMPACT_CODE_EXECUTION, IMPACT_CREDENTIAL_ACCESS, IMPACT_DATA_LEAK,
                    IMPACT_FILE_ACCESS, IMPACT_SSRF}
_NO_REACH = {IMPACT_AVAILABILITY, IMPACT_CLIENT_SIDE}


def filter_credentials_by_impact(category: str, all_credentials: list[str]) -> list[str]:
    """Only credentials the vulnerability class can realistically reach."""
    if not all_credentials:
        return []
    if category in _FULL_CRED_REACH:
        return list(all_credentials)
    if category == IMPACT_INJECTION:
        return [c for c in all_credentials if _is_db_credential(c)]
    return []  # availability / client-side / unknown: no asserted reach


_DB_TOOL_KEYWORDS = {"query", "sql", "execute", "database", "db", "select", "insert"}


def filter_tools_by_impact(category: str, all_tools: list) -> list:
    """Only tools the vulnerability class can realistically invoke."""
    if not all_tools:
        return []
    if category in (IMPACT_CODE_EXECUTION, IMPACT_CREDENTIAL_ACCESS,
                    IMPACT_FILE_ACCESS, IMPACT_SSRF, IMPACT_DATA_LEAK):
        return list(all_tools)
    if category == IMPACT_INJECTION:
        return [t for t in all_tools if any(kw in t.name.lower() for kw in _DB_TOOL_KEYWORDS)]
    return []


def build_attack_vector_summary(
    cwe_ids: list[str],
    category: str,
    filtered_creds: list[str],
    filtered_tools: list,
    severity: Optional[str] = None,
    is_kev: bool = False,
) -> str:
    """One-sentence description of what this vulnerability enables in context."""
    cwe_str = cwe_ids[0] if cwe_ids else "Unknown CWE"
    n_creds = len(filtered_creds)
    n_tools = len(filtered_tools)
    kev = "Actively exploited. " if is_kev else ""
    if category == IMPACT_CODE_EXECUTION:
        tail = f": {n_creds} credential(s) and {n_tools} tool(s) reachable." if n_creds or n_tools else "."
        return f"{kev}Code execution ({cwe_str}) grants full server access{tail}"
    if category == IMPACT_CREDENTIAL_ACCESS:
        tail = f": {n_creds} credential(s) at risk." if n_creds else "."
        return f"{kev}Authentication bypass ({cwe_str}) enables direct credential compromise{tail}"
    if category == IMPACT_FILE_ACCESS:
        tail = f": {n_creds} credential(s) potentially readable." if n_creds else "."
        return f"{kev}File access ({cwe_str}) may expose configuration and credentials{tail}"
    if category == IMPACT_INJECTION:
        tail = f": {n_creds} database credential(s) in scope." if n_creds else "."
        return f"{kev}Injection ({cwe_str}) targets data stores{tail}"
    if category == IMPACT_SSRF:
        tail = f": {n_creds} credential(s) reachable." if n_creds else "."
        return f"{kev}SSRF ({cwe_str}) enables internal service access{tail}"
    if category == IMPACT_DATA_LEAK:
        tail = f": {n_creds} credential(s) potentially visible." if n_creds else "."
        return f"{kev}Information disclosure ({cwe_str}) may expose sensitive data{tail}"
    if category == IMPACT_AVAILABILITY:
        return f"{kev}Denial of service ({cwe_str}) may disrupt service availability. Does not expose credentials or tools."
    if category == IMPACT_CLIENT_SIDE:
        return (f"{kev}Client-side vulnerability ({cwe_str}) affects end-user browsers. "
                "Does not expose server-side credentials or tools.")
    if category == IMPACT_UNKNOWN:
        return (f"{kev}Vulnerability ({cwe_str}) has unknown impact because CWE/advisory metadata is missing "
                "or unsupported. No credential or tool reach is asserted without evidence.")
    return f"{kev}Vulnerability ({cwe_str}) with {category} impact."


# ── Device-side encoding (shared with ops/csrc) ─────────────────────────────
# u8 impact codes, ordered most→least severe so min() = worst case.
IMPACT_CODE = {
    IMPACT_CODE_EXECUTION: 0,
    IMPACT_CREDENTIAL_ACCESS: 1,
    IMPACT_FILE_ACCESS: 2,
    IMPACT_SSRF: 3,
    IMPACT_INJECTION: 4,
    IMPACT_DATA_LEAK: 5,
    IMPACT_AVAILABILITY: 6,
    IMPACT_CLIENT_SIDE: 7,
    IMPACT_UNKNOWN: 8,
}
IMPACT_FROM_CODE = {v: k for k, v in IMPACT_CODE.items()}
# Bit 0: reaches all creds; bit 1: reaches db creds only; bit 2: reaches all
# tools; bit 3: reaches db tools only.
IMPACT_REACH_MASK = {
    IMPACT_CODE.get(cat): (
        (1 if cat in _FULL_CRED_REACH else 0)
        | (2 if cat == IMPACT_INJECTION else 0)
        | (4 if cat in (IMPACT_CODE_EXECUTION, IMPACT_CREDENTIAL_ACCESS, IMPACT_FILE_ACCESS, IMPACT_SSRF, IMPACT_DATA_LEAK) else 0)
        | (8 if cat == IMPACT_INJECTION else 0)
    )
    for cat in IMPACT_CODE
}
