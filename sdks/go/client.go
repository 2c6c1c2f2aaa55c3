// Package agentbom is the Go control-plane client for the agent-bom API.
//
// Reference parity: sdks/go/client.go in the upstream project — a thin,
// dependency-free HTTP client over the /v1 surface (scans, findings,
// graph queries, fleet, compliance).  Mirrors sdks/python/agentbom_client.py
// and sdks/typescript/agentbom-client.ts method-for-method.
//
// Usage:
//
//	c := agentbom.New("http://127.0.0.1:8000", os.Getenv("AGENT_BOM_API_KEY"))
//	job, err := c.Scan(nil, true)           // demo scan
//	report, err := c.ScanReport(job["job_id"].(string))
package agentbom

import (
	"bytes"
	"encoding/json"
	"fmt"
	"io"
	"net/http"
	"net/url"
	"strconv"
	"time"
)

// Error is a non-2xx API response.
type Error struct {
	StatusCode int
	Detail     string
}

func (e *Error) Error() string {
	return fmt.Sprintf("agent-bom API %d: %s", e.StatusCode, e.Detail)
}

// Client talks to one agent-bom control plane.
type Client struct {
	BaseURL string
	APIKey  string
	HTTP    *http.Client
}

// New returns a client; apiKey may be empty for unauthenticated deployments.
func New(baseURL, apiKey string) *Client {
	return &Client{
		BaseURL: baseURL,
		APIKey:  apiKey,
		HTTP:    &http.Client{Timeout: 30 * time.Second},
	}
}

type Obj = map[string]interface{}

func (c *Client) req(method, path string, query url.Values, body interface{}) (Obj, error) {
	u := c.BaseURL + path
	if len(query) > 0 {
		u += "?" + query.Encode()
	}
	var rd io.Reader
	if body != nil {
		b, err := json.Marshal(body)
		if err != nil {
			return nil, err
		}
		rd = bytes.NewReader(b)
	}
	req, err := http.NewRequest(method, u, rd)
	if err != nil {
		return nil, err
	}
	if body != nil {
		req.Header.Set("Content-Type", "application/json")
	}
	if c.APIKey != "" {
		req.Header.Set("X-API-Key", c.APIKey)
	}
	resp, err := c.HTTP.Do(req)
	if err != nil {
		return nil, err
	}
	defer resp.Body.Close()
	raw, err := io.ReadAll(io.LimitReader(resp.Body, 64<<20))
	if err != nil {
		return nil, err
	}
	if resp.StatusCode < 200 || resp.StatusCode >= 300 {
		return nil, &Error{StatusCode: resp.StatusCode, Detail: string(raw)}
	}
	var out Obj
	if len(raw) > 0 {
		if err := json.Unmarshal(raw, &out); err != nil {
			return nil, err
		}
	}
	return out, nil
}

// Health returns /healthz.
func (c *Client) Health() (Obj, error) { return c.req("GET", "/healthz", nil, nil) }

// Scan starts a scan job; inventory may be nil with demo=true.
func (c *Client) Scan(inventory Obj, demo bool) (Obj, error) {
	body := Obj{"demo": demo}
	if inventory != nil {
		body["inventory"] = inventory
	}
	return c.req("POST", "/v1/scan", nil, body)
}

// ScanJob returns job status for a scan started with Scan.
func (c *Client) ScanJob(jobID string) (Obj, error) {
	return c.req("GET", "/v1/scan/"+url.PathEscape(jobID), nil, nil)
}

// ScanReport returns the finished report document for a job.
func (c *Client) ScanReport(jobID string) (Obj, error) {
	return c.req("GET", "/v1/scan/"+url.PathEscape(jobID)+"/report", nil, nil)
}

// WaitForScan polls a job until it leaves pending/running or the deadline hits.
func (c *Client) WaitForScan(jobID string, timeout time.Duration) (Obj, error) {
	deadline := time.Now().Add(timeout)
	for {
		job, err := c.ScanJob(jobID)
		if err != nil {
			return nil, err
		}
		state, _ := job["status"].(string)
		if state != "pending" && state != "running" {
			return job, nil
		}
		if time.Now().After(deadline) {
			return job, fmt.Errorf("scan %s still %s after %s", jobID, state, timeout)
		}
		time.Sleep(500 * time.Millisecond)
	}
}

// Findings lists unified findings, optionally filtered by severity.
func (c *Client) Findings(severity string, limit int) (Obj, error) {
	q := url.Values{"limit": {strconv.Itoa(limit)}}
	if severity != "" {
		q.Set("severity", severity)
	}
	return c.req("GET", "/v1/findings", q, nil)
}

// FindingsDelta returns the new/changed/resolved stream since the watermark.
func (c *Client) FindingsDelta() (Obj, error) {
	return c.req("GET", "/v1/findings/delta", nil, nil)
}

// Graph returns the unified graph (bounded by limit).
func (c *Client) Graph(limit int) (Obj, error) {
	return c.req("GET", "/v1/graph", url.Values{"limit": {strconv.Itoa(limit)}}, nil)
}

// GraphSearch full-text searches nodes.
func (c *Client) GraphSearch(q, entityType string) (Obj, error) {
	v := url.Values{"q": {q}}
	if entityType != "" {
		v.Set("entity_type", entityType)
	}
	return c.req("GET", "/v1/graph/search", v, nil)
}

// GraphPaths returns ranked attack paths.
func (c *Client) GraphPaths(limit int) (Obj, error) {
	return c.req("GET", "/v1/graph/paths",
		url.Values{"limit": {strconv.Itoa(limit)}}, nil)
}

// ExposurePaths returns per-finding exposure paths.
func (c *Client) ExposurePaths(limit int) (Obj, error) {
	return c.req("GET", "/v1/graph/exposure-paths",
		url.Values{"limit": {strconv.Itoa(limit)}}, nil)
}

// GraphQuery runs a bounded traversal from a start node.
func (c *Client) GraphQuery(start string, maxDepth, maxNodes int) (Obj, error) {
	return c.req("POST", "/v1/graph/query", nil, Obj{
		"start": start, "max_depth": maxDepth, "max_nodes": maxNodes})
}

// Impact returns the bounded impact set of one node.
func (c *Client) Impact(nodeID string, maxHops int) (Obj, error) {
	return c.req("GET", "/v1/graph/impact/"+url.PathEscape(nodeID),
		url.Values{"max_hops": {strconv.Itoa(maxHops)}}, nil)
}

// Rollup returns the estate rollup view.
func (c *Client) Rollup() (Obj, error) { return c.req("GET", "/v1/graph/rollup", nil, nil) }

// ShouldIDeploy returns the deploy-gate verdict.
func (c *Client) ShouldIDeploy() (Obj, error) {
	return c.req("GET", "/v1/graph/should-i-deploy", nil, nil)
}

// EvidenceManifest returns the graph-evidence manifest.
func (c *Client) EvidenceManifest() (Obj, error) {
	return c.req("GET", "/v1/graph/evidence-manifest", nil, nil)
}

// Heartbeat reports one fleet member's liveness.
func (c *Client) Heartbeat(memberID string, stats Obj) (Obj, error) {
	body := Obj{"member_id": memberID}
	for k, v := range stats {
		body[k] = v
	}
	return c.req("POST", "/v1/fleet/heartbeat", nil, body)
}

// Fleet lists fleet members.
func (c *Client) Fleet() (Obj, error) { return c.req("GET", "/v1/fleet", nil, nil) }

// CreateSchedule registers a recurring scan schedule.
func (c *Client) CreateSchedule(scheduleID string, intervalSeconds float64) (Obj, error) {
	return c.req("POST", "/v1/schedules", nil, Obj{
		"schedule_id": scheduleID, "interval_s": intervalSeconds})
}

// ComplianceReport returns posture for one framework (e.g. "nist-ai-rmf").
func (c *Client) ComplianceReport(framework string) (Obj, error) {
	return c.req("GET", "/v1/compliance/"+url.PathEscape(framework)+"/report", nil, nil)
}

// Posture returns the self-hardening posture grade.
func (c *Client) Posture() (Obj, error) { return c.req("GET", "/v1/posture", nil, nil) }
