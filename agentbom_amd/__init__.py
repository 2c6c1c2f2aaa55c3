"""agentbom_amd — MI355X-native AI-BOM scanner and blast-radius graph engine.

A from-scratch rebuild of the capabilities of msaad00/agent-bom (reference:
/root/reference) designed GPU-first for AMD Instinct MI355X (gfx950):

- estate graph (package -> vulnerability -> MCP server -> tool/credential ->
  agent) held in CSR form in HBM3E, hash-partitioned over up to 8 GPUs with
  RCCL collectives over xGMI;
- bulk OSV/GHSA version-range matching, blast-radius BFS/reachability,
  exposure-path scoring and rollup reductions as hand-written CDNA4 HIP
  kernels (``agentbom_amd/ops/csrc``);
- the CLI / REST / MCP surfaces and JSON/SARIF/SBOM output schemas of the
  reference (``agent-bom agents``, ``agent-bom serve``, ...).

Layout:
  models/    core data model (reference: src/agent_bom/models.py, finding.py)
  utils/     config, canonical ids, version comparators + GPU key encoding
  scan/      discovery, parsers, match orchestration (reference: scanners/)
  db/        local advisory store + columnar GPU advisory arena
  graph/     unified estate graph: CPU reference container + GPU CSR engine
  ops/       HIP kernels (gfx950) + ctypes host bindings + CPU references
  parallel/  multi-GPU partitioning, RCCL frontier exchange
  output/    JSON / SARIF / CycloneDX / SPDX / console / graph exports
  cli/       click command-line surface
  api/       FastAPI control plane
  mcp/       MCP server (stdio JSON-RPC) tool surface
"""

__version__ = "0.1.0"
