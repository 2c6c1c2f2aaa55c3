"""Graph enums — the closed node/edge vocabulary (graph contract).

Reference: src/agent_bom/graph/types.py (42 EntityTypes, 14 semantic
layers, 49 RelationshipTypes, 4 NodeStatus, 6 layouts) — cited whole by
docs/graph_contract.json; these values are the persisted wire contract.

``REL_CODE``/``ENTITY_CODE`` add the compact u8 encodings the GPU CSR
engine carries per edge/node (traversal classes become 32-bit masks).
"""

from __future__ import annotations

from enum import Enum


class EntityType(str, Enum):
    AGENT = "agent"
    SERVER = "server"
    PACKAGE = "package"
    TOOL = "tool"
    TOOL_CALL = "tool_call"
    MODEL = "model"
    DATASET = "dataset"
    FRAMEWORK = "framework"
    CONTAINER = "container"
    CLOUD_RESOURCE = "cloud_resource"
    RESOURCE = "resource"
    SOURCE_FILE = "source_file"
    CODE_MODULE = "code_module"
    CONFIG_FILE = "config_file"
    EXTERNAL_IMPORT = "external_import"
    CI_JOB = "ci_job"
    DIRECTORY = "directory"
    VULNERABILITY = "vulnerability"
    MISCONFIGURATION = "misconfiguration"
    CREDENTIAL = "credential"
    CREDENTIAL_REF = "credential_ref"
    ORG = "org"
    ACCOUNT = "account"
    USER = "user"
    GROUP = "group"
    ROLE = "role"
    POLICY = "policy"
    SERVICE_ACCOUNT = "service_account"
    SERVICE_PRINCIPAL = "service_principal"
    FEDERATED_IDENTITY = "federated_identity"
    MANAGED_IDENTITY = "managed_identity"
    ACCESS_GRANT = "access_grant"
    ACCESS_POLICY = "access_policy"
    BLUEPRINT = "blueprint"
    DRIFT_INCIDENT = "drift_incident"
    DATA_STORE = "data_store"
    API_GATEWAY = "api_gateway"
    APPLICATION = "application"
    PROVIDER = "provider"
    ENVIRONMENT = "environment"
    FLEET = "fleet"
    CLUSTER = "cluster"


class GraphSemanticLayer(str, Enum):
    USER = "user"
    IDENTITY = "identity"
    APP = "app"
    API_GATEWAY = "api_gateway"
    ORCHESTRATION = "orchestration"
    MCP_SERVER = "mcp_server"
    TOOL = "tool"
    PACKAGE = "package"
    RUNTIME_EVIDENCE = "runtime_evidence"
    ASSET = "asset"
    INFRA = "infra"
    FINDING = "finding"
    CODE = "code"
    CI = "ci"


class RelationshipType(str, Enum):
    # static inventory
    HOSTS = "hosts"
    USES = "uses"
    USES_FRAMEWORK = "uses_framework"
    DEPENDS_ON = "depends_on"
    PROVIDES_TOOL = "provides_tool"
    EXPOSES_CRED = "exposes_cred"
    REACHES_TOOL = "reaches_tool"
    SERVES_MODEL = "serves_model"
    CONTAINS = "contains"
    IMPORTS = "imports"
    DEFINES = "defines"
    RUNS = "runs"
    CONFIGURES = "configures"
    OBSERVES = "observes"
    # vulnerability
    AFFECTS = "affects"
    VULNERABLE_TO = "vulnerable_to"
    EXPLOITABLE_VIA = "exploitable_via"
    REMEDIATES = "remediates"
    TRIGGERS = "triggers"
    # lateral movement (computed)
    SHARES_SERVER = "shares_server"
    SHARES_CRED = "shares_cred"
    LATERAL_PATH = "lateral_path"
    # ownership & governance
    MANAGES = "manages"
    OWNS = "owns"
    PART_OF = "part_of"
    MEMBER_OF = "member_of"
    ASSUMES = "assumes"
    TRUSTS = "trusts"
    ATTACHED = "attached"
    INHERITS = "inherits"
    CAN_ACCESS = "can_access"
    CROSS_ACCOUNT_TRUST = "cross_account_trust"
    # agent-identity governance
    AUTHENTICATES_AS = "authenticates_as"
    SCOPED_TO = "scoped_to"
    GOVERNS = "governs"
    EXHIBITS_DRIFT = "exhibits_drift"
    # cloud-CNAPP
    EXPOSED_TO = "exposed_to"
    STORES = "stores"
    HAS_PERMISSION = "has_permission"
    PROTECTS = "protects"
    # runtime events
    ACTED_AS = "acted_as"
    INVOKED = "invoked"
    CALLED = "called"
    USED_CREDENTIAL = "used_credential"
    ACCESSED = "accessed"
    DELEGATED_TO = "delegated_to"
    # cross-environment correlation
    CORRELATES_WITH = "correlates_with"
    POSSIBLY_CORRELATES_WITH = "possibly_correlates_with"
    # ASPM
    BELONGS_TO = "belongs_to"


class NodeStatus(str, Enum):
    ACTIVE = "active"
    INACTIVE = "inactive"
    VULNERABLE = "vulnerable"
    REMEDIATED = "remediated"


class GraphLayout(str, Enum):
    DAGRE = "dagre"
    FORCE = "force"
    RADIAL = "radial"
    SANKEY = "sankey"
    HIERARCHICAL = "hierarchical"
    GRID = "grid"


# ── Compact device-side encodings ──────────────────────────────────────────
# Edge types travel as u8 in the GPU CSR; traversal classes are 32-bit masks.
# Only relationship classes that participate in GPU traversals get codes
# 0-31 (mask-addressable); the rest map to 32+ (never traversed on device).
_GPU_RELS = [
    RelationshipType.USES, RelationshipType.CONTAINS, RelationshipType.EXPOSES_CRED,
    RelationshipType.PROVIDES_TOOL, RelationshipType.DEPENDS_ON,
    RelationshipType.VULNERABLE_TO, RelationshipType.AFFECTS,
    RelationshipType.EXPLOITABLE_VIA, RelationshipType.SHARES_SERVER,
    RelationshipType.SHARES_CRED, RelationshipType.HOSTS, RelationshipType.USES_FRAMEWORK,
    RelationshipType.REACHES_TOOL, RelationshipType.PART_OF, RelationshipType.CAN_ACCESS,
    RelationshipType.EXPOSED_TO, RelationshipType.STORES, RelationshipType.HAS_PERMISSION,
    RelationshipType.DELEGATED_TO, RelationshipType.LATERAL_PATH,
    RelationshipType.BELONGS_TO, RelationshipType.MEMBER_OF, RelationshipType.ASSUMES,
    RelationshipType.TRUSTS, RelationshipType.ATTACHED,
]
REL_CODE: dict[RelationshipType, int] = {r: i for i, r in enumerate(_GPU_RELS)}
for _r in RelationshipType:
    REL_CODE.setdefault(_r, 32 + len([k for k in REL_CODE if REL_CODE[k] >= 32]))

ENTITY_CODE: dict[EntityType, int] = {e: i for i, e in enumerate(EntityType)}


def rel_mask(*rels: RelationshipType) -> int:
    """32-bit allowed-mask over GPU-traversable relationship codes."""
    mask = 0
    for r in rels:
        code = REL_CODE[r]
        if code < 32:
            mask |= 1 << code
    return mask


# The dependency-reach traversal class (reference graph/dependency_reach.py:109
# follows USES / DEPENDS_ON / CONTAINS / PROVIDES_TOOL).
DEPENDENCY_REACH_MASK = rel_mask(
    RelationshipType.USES, RelationshipType.DEPENDS_ON,
    RelationshipType.CONTAINS, RelationshipType.PROVIDES_TOOL,
)
