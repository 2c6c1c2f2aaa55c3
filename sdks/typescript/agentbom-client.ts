/**
 * agent-bom control-plane TypeScript SDK.
 *
 * Reference: sdks/typescript — a thin typed client over the /v1 REST
 * surface; mirrors sdks/python/agentbom_client.py.
 */

export interface ScanSummary {
  total_agents: number;
  total_mcp_servers: number;
  total_packages: number;
  total_vulnerabilities: number;
  critical_findings: number;
  total_findings: number;
}

export interface ScanJob {
  job_id: string;
  status: "pending" | "running" | "done" | "failed";
}

export interface DeployVerdict {
  verdict: "allow" | "warn" | "block";
  max_risk: number;
  has_malicious: boolean;
  has_kev: boolean;
}

export class AgentBomError extends Error {
  constructor(public statusCode: number, public detail: string) {
    super(`HTTP ${statusCode}: ${detail}`);
  }
}

export interface ClientOptions {
  baseUrl?: string;
  apiKey?: string;
  fetchImpl?: typeof fetch;
}

export class AgentBomClient {
  private baseUrl: string;
  private headers: Record<string, string>;
  private fetchImpl: typeof fetch;

  constructor(opts: ClientOptions = {}) {
    this.baseUrl = (opts.baseUrl ?? "http://127.0.0.1:8000").replace(/\/$/, "");
    this.headers = { "content-type": "application/json" };
    if (opts.apiKey) this.headers["x-api-key"] = opts.apiKey;
    this.fetchImpl = opts.fetchImpl ?? fetch;
  }

  private async req<T>(method: string, path: string, body?: unknown): Promise<T> {
    const resp = await this.fetchImpl(`${this.baseUrl}${path}`, {
      method,
      headers: this.headers,
      body: body === undefined ? undefined : JSON.stringify(body),
    });
    if (!resp.ok) {
      let detail = resp.statusText;
      try {
        const data = (await resp.json()) as { detail?: string };
        detail = data.detail ?? detail;
      } catch {
        /* non-JSON error body */
      }
      throw new AgentBomError(resp.status, detail);
    }
    return (await resp.json()) as T;
  }

  health(): Promise<{ status: string; version: string }> {
    return this.req("GET", "/healthz");
  }

  scan(opts: { inventory?: object; demo?: boolean; blastRadiusDepth?: number } = {}): Promise<ScanJob> {
    return this.req("POST", "/v1/scan", {
      inventory: opts.inventory ?? null,
      demo: opts.demo ?? false,
      blast_radius_depth: opts.blastRadiusDepth ?? 1,
    });
  }

  scanJob(jobId: string): Promise<Record<string, unknown>> {
    return this.req("GET", `/v1/scan/${jobId}`);
  }

  scanReport(jobId: string): Promise<Record<string, unknown>> {
    return this.req("GET", `/v1/scan/${jobId}/report`);
  }

  findings(severity?: string, limit = 100): Promise<Record<string, unknown>> {
    const q = new URLSearchParams({ limit: String(limit) });
    if (severity) q.set("severity", severity);
    return this.req("GET", `/v1/findings?${q}`);
  }

  graphSearch(q: string, entityType?: string): Promise<Record<string, unknown>> {
    const params = new URLSearchParams({ q });
    if (entityType) params.set("entity_type", entityType);
    return this.req("GET", `/v1/graph/search?${params}`);
  }

  graphPaths(limit = 25): Promise<Record<string, unknown>> {
    return this.req("GET", `/v1/graph/paths?limit=${limit}`);
  }

  exposurePaths(limit = 50): Promise<Record<string, unknown>> {
    return this.req("GET", `/v1/graph/exposure-paths?limit=${limit}`);
  }

  graphQuery(start: string, maxDepth = 3, maxNodes = 500): Promise<Record<string, unknown>> {
    return this.req("POST", "/v1/graph/query", {
      start, max_depth: maxDepth, max_nodes: maxNodes,
    });
  }

  rollup(): Promise<Record<string, unknown>> {
    return this.req("GET", "/v1/graph/rollup");
  }

  shouldIDeploy(): Promise<DeployVerdict> {
    return this.req("GET", "/v1/graph/should-i-deploy");
  }

  heartbeat(memberId: string, stats: Record<string, unknown> = {}): Promise<Record<string, unknown>> {
    return this.req("POST", "/v1/fleet/heartbeat", {
      member_id: memberId, hostname: memberId, ...stats,
    });
  }

  complianceReport(framework: string): Promise<Record<string, unknown>> {
    return this.req("GET", `/v1/compliance/${framework}/report`);
  }
}
