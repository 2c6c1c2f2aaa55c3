"""Non-human identity (NHI) discovery + agent identity lifecycle.

Reference surface: src/agent_bom/identity/{okta_nhi,entra_nhi}.py (discovery),
src/agent_bom/api/agent_identity_store.py (lifecycle).  Re-designed here:
discovery runs against injectable clients or exported inventory files (this
environment has no egress), lifecycle is a SQLite/in-memory store with
hash-only token storage.
"""

from agentbom_amd.identity.lifecycle import (
    AgentIdentity,
    AgentIdentityStore,
    AgentJITGrant,
    generate_token,
    hash_token,
)
from agentbom_amd.identity.nhi import (
    DiscoveredNonHumanIdentity,
    NHIDiscoveryResult,
    NHIDiscoveryStatus,
    discover_entra_nhis,
    discover_okta_nhis,
)

__all__ = [
    "AgentIdentity",
    "AgentIdentityStore",
    "AgentJITGrant",
    "DiscoveredNonHumanIdentity",
    "NHIDiscoveryResult",
    "NHIDiscoveryStatus",
    "discover_entra_nhis",
    "discover_okta_nhis",
    "generate_token",
    "hash_token",
]
