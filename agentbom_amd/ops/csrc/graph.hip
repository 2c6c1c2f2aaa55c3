// Estate-graph kernels: level-synchronous multi-source BFS over CSR in HBM,
// batched bounded blast-radius queries, and the risk-score formula.
//
// Replaces the reference's Python adjacency-dict walks
// (reference: src/agent_bom/graph/dependency_reach.py:109-198 per-agent BFS,
// src/agent_bom/graph/container.py:613-683 BFS/impact_of/traverse_subgraph)
// with frontier kernels over a CSR whose 10M-100M nodes stay resident in
// 288 GB of HBM3E.
//
// BFS design (CDNA4):
// - one thread per frontier vertex, grid-stride; dist claims via atomicCAS
//   from UNVISITED so each vertex is pushed exactly once (level-synchronous
//   => first touch is the minimal hop).
// - edge-type filtering via a 32-bit allowed-mask over per-edge type bytes
//   (the reference traverses only USES/DEPENDS_ON/CONTAINS/PROVIDES_TOOL
//   classes for dependency reach — types are a closed enum).
// - high-degree frontier vertices (skewed estates: ~1% of agents fan out
//   18-32x) are handled wave-cooperatively: degree >= 64 vertices are
//   expanded by whole waves from a secondary queue filled in pass 1.
//
// The host-side level loop lives in abom_api.hip (C++, one sync per level).

#include "abom_common.h"

namespace abom {

// One atomic per wave: lanes accumulate locally, reduce, lane 0 adds.
__device__ __forceinline__ void wave_add_degree(unsigned int* dst, unsigned v) {
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if ((threadIdx.x & 63) == 0 && v) atomicAdd(dst, v);
}

// Pass over the frontier: expand vertices with degree < WAVE_DEG inline;
// defer heavy vertices to the heavy queue.
__global__ void bfs_expand_kernel(
    const uint64_t* __restrict__ row_off,   // [N+1]
    const uint32_t* __restrict__ col,       // [E]
    const uint8_t* __restrict__ etype,      // [E] or nullptr
    uint32_t allowed_mask,
    const uint32_t* __restrict__ frontier,
    long long frontier_size,
    uint32_t* __restrict__ dist,
    uint32_t next_level,
    uint32_t* __restrict__ next_frontier,
    unsigned int* __restrict__ next_count,
    uint32_t* __restrict__ heavy_queue,
    unsigned int* __restrict__ heavy_count,
    unsigned int* __restrict__ next_degree_sum,
    long long capacity) {
    constexpr uint64_t WAVE_DEG = 64;
    const long long stride = (long long)gridDim.x * blockDim.x;
    unsigned my_deg = 0;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < frontier_size;
         i += stride) {
        const uint32_t u = frontier[i];
        const uint64_t beg = row_off[u];
        const uint64_t end = row_off[u + 1];
        if (end - beg >= WAVE_DEG) {
            heavy_queue[atomicAdd(heavy_count, 1u)] = u;
            continue;
        }
        for (uint64_t e = beg; e < end; ++e) {
            if (etype && !((allowed_mask >> etype[e]) & 1u)) continue;
            const uint32_t v = col[e];
            if (dist[v] == ABOM_UNVISITED &&
                atomicCAS(&dist[v], ABOM_UNVISITED, next_level) == ABOM_UNVISITED) {
                const unsigned idx = atomicAdd(next_count, 1u);
                if ((long long)idx < capacity) next_frontier[idx] = v;
                my_deg += (unsigned)(row_off[v + 1] - row_off[v]);
            }
        }
    }
    wave_add_degree(next_degree_sum, my_deg);
}

// Wave-cooperative expansion of heavy vertices: one wave (64 lanes) walks one
// vertex's adjacency with coalesced col[] reads.
__global__ void bfs_expand_heavy_kernel(
    const uint64_t* __restrict__ row_off,
    const uint32_t* __restrict__ col,
    const uint8_t* __restrict__ etype,
    uint32_t allowed_mask,
    const uint32_t* __restrict__ heavy_queue,
    const unsigned int* __restrict__ heavy_size_ptr,
    uint32_t* __restrict__ dist,
    uint32_t next_level,
    uint32_t* __restrict__ next_frontier,
    unsigned int* __restrict__ next_count,
    unsigned int* __restrict__ next_degree_sum,
    long long capacity) {
    const unsigned heavy_size = *heavy_size_ptr;
    const int lane = threadIdx.x & 63;
    const long long wave = ((long long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const long long nwaves = ((long long)gridDim.x * blockDim.x) >> 6;
    unsigned my_deg = 0;
    for (long long q = wave; q < heavy_size; q += nwaves) {
        const uint32_t u = heavy_queue[q];
        const uint64_t beg = row_off[u];
        const uint64_t end = row_off[u + 1];
        for (uint64_t e = beg + lane; e < end; e += 64) {
            if (etype && !((allowed_mask >> etype[e]) & 1u)) continue;
            const uint32_t v = col[e];
            if (dist[v] == ABOM_UNVISITED &&
                atomicCAS(&dist[v], ABOM_UNVISITED, next_level) == ABOM_UNVISITED) {
                const unsigned idx = atomicAdd(next_count, 1u);
                if ((long long)idx < capacity) next_frontier[idx] = v;
                my_deg += (unsigned)(row_off[v + 1] - row_off[v]);
            }
        }
    }
    wave_add_degree(next_degree_sum, my_deg);
}

// Edge-centric expansion for DENSE frontiers: one thread per edge, fully
// coalesced src/col streams.  Used when the frontier would touch a large
// share of all edges (the agent->server->everything levels of estate
// graphs), where per-vertex serial neighbor loops are latency-bound.
// Requires the per-edge source array (edge_src, same order as col).
// build_frontier == 0: dist-driven mode for subsequent edge-centric levels —
// claims are only counted (one wave-reduced atomic per wave), no frontier
// append, no degree accumulation.  Removes the single-counter atomic storm
// on claim-heavy levels (~10M returning atomicAdds on one word otherwise).
__global__ void bfs_expand_edges_kernel(
    const uint32_t* __restrict__ edge_src,   // [E]
    const uint32_t* __restrict__ col,        // [E]
    const uint8_t* __restrict__ etype,       // [E] or nullptr
    uint32_t allowed_mask,
    long long num_edges,
    const uint64_t* __restrict__ row_off,
    uint32_t* __restrict__ dist,
    uint32_t cur_level,                      // frontier's level (claimed at cur_level)
    uint32_t* __restrict__ next_frontier,
    unsigned int* __restrict__ next_count,
    unsigned int* __restrict__ next_degree_sum,
    int build_frontier,
    long long capacity) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    unsigned my_deg = 0;
    unsigned my_claims = 0;
    for (long long e = (long long)blockIdx.x * blockDim.x + threadIdx.x; e < num_edges;
         e += stride) {
        if (etype && !((allowed_mask >> etype[e]) & 1u)) continue;
        if (dist[edge_src[e]] != cur_level) continue;
        const uint32_t v = col[e];
        if (build_frontier) {
            if (dist[v] == ABOM_UNVISITED &&
                atomicCAS(&dist[v], ABOM_UNVISITED, cur_level + 1) == ABOM_UNVISITED) {
                const unsigned idx = atomicAdd(next_count, 1u);
                if ((long long)idx < capacity) next_frontier[idx] = v;
                my_deg += (unsigned)(row_off[v + 1] - row_off[v]);
            }
        } else if (dist[v] == ABOM_UNVISITED) {
            // dist-driven mode: every racer writes the SAME value, so a plain
            // store is exact (level-synchronous); the claim count may double-
            // count racers but is only used for termination (>0) and the
            // hand-back threshold — both tolerant of overcounting.
            dist[v] = cur_level + 1;
            ++my_claims;
        }
    }
    wave_add_degree(next_degree_sum, my_deg);
    if (!build_frontier) wave_add_degree(next_count, my_claims);
}

// Bitmap-mode dense expansion.  The random dist[] probes in the edge pass
// are 64B-granule bound (two probes x E edges x a cache line each); the
// same membership tests against 1-bit-per-node bitmaps keep the whole
// working set (3 x N/8 bytes ~ 4 MB at 11M nodes) resident in L2/LLC:
//   cur_bits      nodes claimed at cur_level (the implicit frontier)
//   visited_bits  any claimed node
//   next_bits     nodes claimed this level (becomes cur next level)
// atomicOr claims are exact (first setter wins), so the claim count is not
// overcounted; dist[] is written once per claim for the output contract.
__global__ void build_bits_kernel(
    const uint32_t* __restrict__ dist, long long num_nodes, uint32_t cur_level,
    uint32_t* __restrict__ cur_bits, uint32_t* __restrict__ visited_bits) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    const long long words = (num_nodes + 31) >> 5;
    for (long long w = (long long)blockIdx.x * blockDim.x + threadIdx.x; w < words;
         w += stride) {
        uint32_t cur = 0, vis = 0;
        const long long base = w << 5;
        const int lanes = (int)((num_nodes - base) < 32 ? (num_nodes - base) : 32);
        for (int b = 0; b < lanes; ++b) {
            const uint32_t d = dist[base + b];
            if (d != ABOM_UNVISITED) vis |= (1u << b);
            if (d == cur_level) cur |= (1u << b);
        }
        cur_bits[w] = cur;
        visited_bits[w] = vis;
    }
}

__global__ void bfs_expand_edges_bits_kernel(
    const uint32_t* __restrict__ edge_src,   // [E]
    const uint32_t* __restrict__ col,        // [E]
    const uint8_t* __restrict__ etype,       // [E] or nullptr
    uint32_t allowed_mask,
    long long num_edges,
    uint32_t* __restrict__ dist,
    const uint32_t* __restrict__ cur_bits,
    uint32_t* __restrict__ visited_bits,
    uint32_t* __restrict__ next_bits,
    uint32_t cur_level,
    unsigned int* __restrict__ next_count) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    unsigned my_claims = 0;
    for (long long e = (long long)blockIdx.x * blockDim.x + threadIdx.x; e < num_edges;
         e += stride) {
        if (etype && !((allowed_mask >> etype[e]) & 1u)) continue;
        const uint32_t u = edge_src[e];
        if (!((cur_bits[u >> 5] >> (u & 31)) & 1u)) continue;
        const uint32_t v = col[e];
        const uint32_t bit = 1u << (v & 31);
        if ((visited_bits[v >> 5] & bit)) continue;
        const uint32_t old = atomicOr(&visited_bits[v >> 5], bit);
        if (!(old & bit)) {  // exact first claim
            dist[v] = cur_level + 1;
            atomicOr(&next_bits[v >> 5], bit);
            ++my_claims;
        }
    }
    wave_add_degree(next_count, my_claims);
}

// Direction-optimized bottom-up step: every UNVISITED vertex probes its
// REVERSE edges for a parent in the current frontier and claims itself.
// dist[v] reads/writes are coalesced (v is the loop index), the per-vertex
// probe loop EARLY-EXITS on the first frontier parent, and no frontier is
// materialized — the dense-level complement of the edge-centric push pass.
__global__ void bfs_bottom_up_kernel(
    const uint64_t* __restrict__ rev_off,
    const uint32_t* __restrict__ rev_col,
    const uint8_t* __restrict__ rev_et,   // type of edge (parent -> v)
    uint32_t allowed_mask,
    long long num_nodes,
    uint32_t* __restrict__ dist,
    uint32_t cur_level,
    unsigned int* __restrict__ next_count) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    unsigned my_claims = 0;
    for (long long v = (long long)blockIdx.x * blockDim.x + threadIdx.x; v < num_nodes;
         v += stride) {
        if (dist[v] != ABOM_UNVISITED) continue;
        const uint64_t beg = rev_off[v];
        const uint64_t end = rev_off[v + 1];
        for (uint64_t e = beg; e < end; ++e) {
            if (rev_et && !((allowed_mask >> rev_et[e]) & 1u)) continue;
            if (dist[rev_col[e]] == cur_level) {
                dist[v] = cur_level + 1;
                ++my_claims;
                break;
            }
        }
    }
    wave_add_degree(next_count, my_claims);
}

// Rebuild a frontier from dist (nodes claimed at `level`): used when
// dist-driven dense mode hands back to vertex-frontier mode after the
// claim rate drops.  Claim counts are small here, so appends are cheap.
__global__ void collect_frontier_kernel(
    const uint32_t* __restrict__ dist, long long num_nodes, uint32_t level,
    const uint64_t* __restrict__ row_off,
    uint32_t* __restrict__ frontier, unsigned int* __restrict__ count,
    unsigned int* __restrict__ degree_sum, long long capacity) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    unsigned my_deg = 0;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < num_nodes;
         i += stride) {
        if (dist[i] == level) {
            const unsigned idx = atomicAdd(count, 1u);
            if ((long long)idx < capacity) frontier[idx] = (uint32_t)i;
            my_deg += (unsigned)(row_off[i + 1] - row_off[i]);
        }
    }
    wave_add_degree(degree_sum, my_deg);
}

__global__ void init_dist_kernel(uint32_t* __restrict__ dist, long long n, uint32_t value) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        dist[i] = value;
}

__global__ void seed_sources_kernel(
    const uint32_t* __restrict__ sources, long long n_sources,
    const uint64_t* __restrict__ row_off,
    uint32_t* __restrict__ dist, uint32_t* __restrict__ frontier,
    unsigned int* __restrict__ degree_sum) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    unsigned my_deg = 0;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n_sources;
         i += stride) {
        const uint32_t s = sources[i];
        dist[s] = 0;
        frontier[i] = s;
        my_deg += (unsigned)(row_off[s + 1] - row_off[s]);
    }
    wave_add_degree(degree_sum, my_deg);
}

// ── Batched bounded blast-radius query (impact_of / traverse_subgraph) ─────
//
// One BLOCK per query: a small BFS bounded by max_hops (<=4 default, matching
// container.impact_of) and max_nodes per query.  The visited set is an
// open-addressed hash table in LDS (8K slots), the per-level frontier also
// lives in LDS.  Collected (node, hop) pairs stream to the query's output
// slab.  Used for p50 blast-radius query latency and the /v1/graph read path.

constexpr int IQ_HASH = 8192;       // LDS hash slots (32 KB)
constexpr int IQ_FRONTIER = 2048;   // per-level frontier cap in LDS

__global__ void impact_query_kernel(
    const uint64_t* __restrict__ row_off,
    const uint32_t* __restrict__ col,
    const uint8_t* __restrict__ etype,
    uint32_t allowed_mask,
    const uint32_t* __restrict__ query_sources,   // [Q]
    int num_queries,
    int max_hops,
    int max_nodes_per_query,                      // slab stride
    uint32_t* __restrict__ out_nodes,             // [Q * stride]
    uint8_t* __restrict__ out_hops,               // [Q * stride]
    uint32_t* __restrict__ out_counts,            // [Q] (clamped to stride)
    uint8_t* __restrict__ out_truncated) {        // [Q]
    __shared__ uint32_t h_table[IQ_HASH];
    __shared__ uint32_t fr[2][IQ_FRONTIER];
    __shared__ unsigned fr_size[2];
    __shared__ unsigned out_cursor;
    __shared__ unsigned truncated;

    for (int q = blockIdx.x; q < num_queries; q += gridDim.x) {
        // reset LDS state
        for (int i = threadIdx.x; i < IQ_HASH; i += blockDim.x) h_table[i] = ABOM_UNVISITED;
        if (threadIdx.x == 0) {
            fr_size[0] = 1;
            fr_size[1] = 0;
            out_cursor = 0;
            truncated = 0;
            fr[0][0] = query_sources[q];
            // seed visited
            uint32_t s = query_sources[q];
            uint32_t h = (s * 2654435761u) & (IQ_HASH - 1);
            h_table[h] = s;
        }
        __syncthreads();

        uint32_t* slab_nodes = out_nodes + (long long)q * max_nodes_per_query;
        uint8_t* slab_hops = out_hops + (long long)q * max_nodes_per_query;
        if (threadIdx.x == 0 && max_nodes_per_query > 0) {
            slab_nodes[0] = fr[0][0];
            slab_hops[0] = 0;
            out_cursor = 1;
        }
        __syncthreads();

        int cur = 0;
        for (int hop = 1; hop <= max_hops; ++hop) {
            const int nxt = cur ^ 1;
            const unsigned fsz = fr_size[cur];
            if (fsz == 0) break;
            __syncthreads();
            // Threads stride the (frontier x neighbor) work by frontier entry.
            for (unsigned i = threadIdx.x; i < fsz; i += blockDim.x) {
                const uint32_t u = fr[cur][i];
                const uint64_t beg = row_off[u];
                const uint64_t end = row_off[u + 1];
                for (uint64_t e = beg; e < end; ++e) {
                    if (etype && !((allowed_mask >> etype[e]) & 1u)) continue;
                    const uint32_t v = col[e];
                    // LDS open-addressing insert; full table => truncate.
                    uint32_t h = (v * 2654435761u) & (IQ_HASH - 1);
                    bool inserted = false, seen = false;
                    for (int probe = 0; probe < 64; ++probe) {
                        uint32_t prev = atomicCAS(&h_table[h], ABOM_UNVISITED, v);
                        if (prev == ABOM_UNVISITED) { inserted = true; break; }
                        if (prev == v) { seen = true; break; }
                        h = (h + 1) & (IQ_HASH - 1);
                    }
                    if (seen) continue;
                    if (!inserted) { truncated = 1; continue; }
                    const unsigned oi = atomicAdd(&out_cursor, 1u);
                    if ((int)oi < max_nodes_per_query) {
                        slab_nodes[oi] = v;
                        slab_hops[oi] = (uint8_t)hop;
                    } else {
                        truncated = 1;
                    }
                    const unsigned fi = atomicAdd(&fr_size[nxt], 1u);
                    if ((int)fi < IQ_FRONTIER) fr[nxt][fi] = v;
                    else truncated = 1;
                }
            }
            __syncthreads();
            if (threadIdx.x == 0) {
                if (fr_size[nxt] > IQ_FRONTIER) fr_size[nxt] = IQ_FRONTIER;
                fr_size[cur] = 0;
            }
            cur = nxt;
            __syncthreads();
        }
        if (threadIdx.x == 0) {
            out_counts[q] = out_cursor < (unsigned)max_nodes_per_query ? out_cursor
                                                                       : (unsigned)max_nodes_per_query;
            out_truncated[q] = (uint8_t)truncated;
        }
        __syncthreads();
    }
}

// ── Risk-score kernel ──────────────────────────────────────────────────────
// Implements models/blast.py risk_score_from_counts exactly (f32): the CPU
// formula is the specification; parity is asserted in tests/test_ops_gpu.py.

struct RiskWeights {
    float base_critical, base_high, base_medium, base_low;
    float agent_w, agent_cap, cred_w, cred_cap, tool_w, tool_cap;
    float ai_boost, kev_boost, epss_boost, epss_threshold;
    float sc_t1, sc_b1, sc_t2, sc_b2, sc_t3, sc_b3;
    float reach_boost, unreach_penalty;
};

__device__ __forceinline__ float score_one(
    uint8_t sev, uint32_t n_agents, uint32_t n_creds, uint32_t n_tools,
    uint8_t f, float epss, float scorecard, int8_t reach, const RiskWeights& w) {
    if (f & 4u) return 0.0f;  // suppressed / VEX
    float base = 0.0f;
    switch (sev) {
        case 5: base = w.base_critical; break;
        case 4: base = w.base_high; break;
        case 3: base = w.base_medium; break;
        case 2: base = w.base_low; break;
        default: base = 0.0f;
    }
    const float af = fminf((float)n_agents * w.agent_w, w.agent_cap);
    const float cf = fminf((float)n_creds * w.cred_w, w.cred_cap);
    const float tf = fminf((float)n_tools * w.tool_w, w.tool_cap);
    const int ai_signals = (int)(f & 1u) + (n_creds > 0) + (n_tools > 0);
    const float ai = ai_signals >= 2 ? w.ai_boost : 0.0f;
    const float kev = (f & 2u) ? w.kev_boost : 0.0f;
    const float ep = (epss >= w.epss_threshold) ? w.epss_boost : 0.0f;
    float sc = 0.0f;
    if (scorecard >= 0.0f) {
        if (scorecard < w.sc_t1) sc = w.sc_b1;
        else if (scorecard < w.sc_t2) sc = w.sc_b2;
        else if (scorecard < w.sc_t3) sc = w.sc_b3;
    }
    float ra = 0.0f;
    if (reach == 1) ra = w.reach_boost;
    else if (reach == 0) ra = -w.unreach_penalty;
    return fmaxf(0.0f, fminf(base + af + cf + tf + ai + kev + ep + sc + ra, 10.0f));
}

// Fused per-finding gather + score: replaces the ~14-op torch chain
// (arena gathers, CWE-impact LUT classification, count selection,
// reachability compare, score) with one pass.
__global__ void score_gather_kernel(
    const long long* __restrict__ win_idx,    // [n] -> arena window rows
    const long long* __restrict__ pkg_nodes,  // [n] -> global node ids
    const long long* __restrict__ pos,        // [n] -> rows into counts2d
    const uint8_t* __restrict__ a_sev,        // [W]
    const uint8_t* __restrict__ a_kev,        // [W]
    const float* __restrict__ a_epss,         // [W]
    const uint8_t* __restrict__ a_impact,     // [W]
    const uint8_t* __restrict__ cred_lut,     // [9]
    const uint8_t* __restrict__ tool_lut,     // [9]
    const int* __restrict__ counts2d,         // [U*6]
    const uint32_t* __restrict__ dist,        // [N]
    float* __restrict__ out_scores,
    int* __restrict__ out_agents,
    int* __restrict__ out_creds,
    int* __restrict__ out_tools,
    long long n, RiskWeights w) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        const long long wrow = win_idx[i];
        const long long crow = pos[i] * 6;
        const uint8_t impact = a_impact[wrow];
        const uint8_t ccls = cred_lut[impact];
        const uint8_t tcls = tool_lut[impact];
        const int n_creds = ccls == 2 ? counts2d[crow + 2]
                          : ccls == 1 ? counts2d[crow + 3] : 0;
        const int n_tools = tcls == 2 ? counts2d[crow + 4]
                          : tcls == 1 ? counts2d[crow + 5] : 0;
        const int n_agents = counts2d[crow + 1];
        const uint8_t flags = (uint8_t)(a_kev[wrow] ? 2u : 0u);
        const int8_t reach = dist[pkg_nodes[i]] != ABOM_UNVISITED ? 1 : 0;
        out_scores[i] = score_one(a_sev[wrow], (uint32_t)n_agents,
                                  (uint32_t)n_creds, (uint32_t)n_tools, flags,
                                  a_epss[wrow], -1.0f, reach, w);
        out_agents[i] = n_agents;
        out_creds[i] = n_creds;
        out_tools[i] = n_tools;
    }
}

__global__ void risk_score_kernel(
    const uint8_t* __restrict__ severity,     // SEVERITY_CODE (5=crit..2=low)
    const uint32_t* __restrict__ n_agents,
    const uint32_t* __restrict__ n_creds,
    const uint32_t* __restrict__ n_tools,
    const uint8_t* __restrict__ flags,        // bit0 ai_ctx, bit1 kev, bit2 suppressed
    const float* __restrict__ epss,           // <0 => none
    const float* __restrict__ scorecard,      // <0 => none
    const int8_t* __restrict__ reach,         // -1 unknown / 0 unreachable / 1 reachable
    float* __restrict__ out,
    long long n, RiskWeights w) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        out[i] = score_one(severity[i], n_agents[i], n_creds[i], n_tools[i],
                           flags[i], epss[i], scorecard[i], reach[i], w);
    }
}

// ── Segmented reductions for rollup ────────────────────────────────────────
// Per-finding severity scatter into per-container histograms (6 severity
// buckets) — container.rollup's descendant aggregates.
__global__ void severity_histogram_kernel(
    const uint32_t* __restrict__ owner,       // [F] container index per finding
    const uint8_t* __restrict__ severity,     // [F] SEVERITY_CODE
    unsigned int* __restrict__ hist,          // [C * 6]
    long long n) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        atomicAdd(&hist[(long long)owner[i] * 6 + severity[i]], 1u);
    }
}

}  // namespace abom

// ── C ABI ──────────────────────────────────────────────────────────────────

extern "C" int abom_bfs_init(void* dist, long long n, void* stream) {
    // ABOM_UNVISITED is 0xFFFFFFFF — a byte-uniform pattern, so the DMA
    // memset path beats a grid-stride store kernel (and frees the CUs for
    // the match kernel running concurrently on the side stream).
    return (int)hipMemsetAsync(dist, 0xFF, (size_t)n * sizeof(uint32_t),
                               (hipStream_t)stream);
}

extern "C" int abom_bfs_seed(const void* sources, long long n_sources, const void* row_off,
                             void* dist, void* frontier, void* degree_sum, void* stream) {
    const int block = 256;
    hipLaunchKernelGGL(abom::seed_sources_kernel, dim3(abom::grid_for(n_sources, block)),
                       dim3(block), 0, (hipStream_t)stream, (const uint32_t*)sources, n_sources,
                       (const uint64_t*)row_off, (uint32_t*)dist, (uint32_t*)frontier,
                       (unsigned int*)degree_sum);
    return (int)hipGetLastError();
}

extern "C" int abom_bfs_expand(
    const void* row_off, const void* col, const void* etype, unsigned int allowed_mask,
    const void* frontier, long long frontier_size, void* dist, unsigned int next_level,
    void* next_frontier, void* next_count, void* heavy_queue, void* heavy_count,
    void* next_degree_sum, long long capacity, void* stream) {
    const int block = 256;
    hipLaunchKernelGGL(abom::bfs_expand_kernel, dim3(abom::grid_for(frontier_size, block)),
                       dim3(block), 0, (hipStream_t)stream, (const uint64_t*)row_off,
                       (const uint32_t*)col, (const uint8_t*)etype, allowed_mask,
                       (const uint32_t*)frontier, frontier_size, (uint32_t*)dist, next_level,
                       (uint32_t*)next_frontier, (unsigned int*)next_count,
                       (uint32_t*)heavy_queue, (unsigned int*)heavy_count,
                       (unsigned int*)next_degree_sum, capacity);
    return (int)hipGetLastError();
}

extern "C" int abom_bfs_expand_heavy(
    const void* row_off, const void* col, const void* etype, unsigned int allowed_mask,
    const void* heavy_queue, const void* heavy_size_ptr, void* dist, unsigned int next_level,
    void* next_frontier, void* next_count, void* next_degree_sum, long long capacity,
    void* stream) {
    const int block = 256;  // 4 waves per block; kernel grid-strides by wave
    hipLaunchKernelGGL(abom::bfs_expand_heavy_kernel, dim3(2048), dim3(block), 0,
                       (hipStream_t)stream, (const uint64_t*)row_off, (const uint32_t*)col,
                       (const uint8_t*)etype, allowed_mask, (const uint32_t*)heavy_queue,
                       (const unsigned int*)heavy_size_ptr, (uint32_t*)dist, next_level,
                       (uint32_t*)next_frontier, (unsigned int*)next_count,
                       (unsigned int*)next_degree_sum, capacity);
    return (int)hipGetLastError();
}

extern "C" int abom_bfs_expand_edges(
    const void* edge_src, const void* col, const void* etype, unsigned int allowed_mask,
    long long num_edges, const void* row_off, void* dist, unsigned int cur_level,
    void* next_frontier, void* next_count, void* next_degree_sum, int build_frontier,
    long long capacity, void* stream) {
    const int block = 256;
    hipLaunchKernelGGL(abom::bfs_expand_edges_kernel, dim3(abom::grid_for(num_edges, block)),
                       dim3(block), 0, (hipStream_t)stream, (const uint32_t*)edge_src,
                       (const uint32_t*)col, (const uint8_t*)etype, allowed_mask, num_edges,
                       (const uint64_t*)row_off, (uint32_t*)dist, cur_level,
                       (uint32_t*)next_frontier, (unsigned int*)next_count,
                       (unsigned int*)next_degree_sum, build_frontier, capacity);
    return (int)hipGetLastError();
}

// Full multi-source BFS: host-side level loop, one device->host count read
// per level.  dist must be u32[N]; frontier_a/b u32[N]; heavy_queue u32[N];
// counters = device u32[2] (next_count, heavy_count).  ``edge_src`` (the
// per-edge source array, col-aligned) may be null; when present, levels
// whose frontier would touch > ~1/8 of all edges switch to the edge-centric
// kernel (coalesced streams) instead of per-vertex expansion.
// Returns negative hip error or the number of levels run.
extern "C" int abom_bfs_run(
    const void* row_off, const void* col, const void* etype, unsigned int allowed_mask,
    const void* sources, long long n_sources, void* dist, long long num_nodes,
    void* frontier_a, void* frontier_b, void* heavy_queue, void* counters,
    int max_levels, const void* edge_src, long long num_edges, double avg_degree,
    void* bits, const void* rev_off, const void* rev_col, const void* rev_et,
    void* stream) {
    // counters layout: [0] next_count, [1] heavy_count, [2] frontier degree sum
    // bits (optional): u32[3 * ceil(N/32)] — cur/next/visited bitmaps for the
    // L2-resident dense mode; nullptr falls back to dist-probe dense mode.
    hipStream_t s = (hipStream_t)stream;
    unsigned int* ctr = (unsigned int*)counters;
    const long long bit_words = (num_nodes + 31) >> 5;
    uint32_t* cur_bits = (uint32_t*)bits;
    uint32_t* next_bits = cur_bits ? cur_bits + bit_words : nullptr;
    uint32_t* visited_bits = cur_bits ? cur_bits + 2 * bit_words : nullptr;
    bool bits_ready = false;
    int rc = abom_bfs_init(dist, num_nodes, stream);
    if (rc) return -rc;
    ABOM_CHECK(hipMemsetAsync(ctr, 0, 3 * sizeof(unsigned int), s));
    rc = abom_bfs_seed(sources, n_sources, row_off, dist, frontier_a, ctr + 2, stream);
    if (rc) return -rc;
    unsigned int host_vals[2] = {0, 0};
    ABOM_CHECK(hipMemcpyAsync(&host_vals[1], ctr + 2, sizeof(unsigned int),
                              hipMemcpyDeviceToHost, s));
    ABOM_CHECK(hipStreamSynchronize(s));
    unsigned int frontier_degree = host_vals[1];

    uint32_t* cur = (uint32_t*)frontier_a;
    uint32_t* nxt = (uint32_t*)frontier_b;
    long long frontier_size = n_sources;
    int level = 0;
    bool stay_dense = false;
    while (frontier_size != 0 && level < max_levels) {
        ++level;
        ABOM_CHECK(hipMemsetAsync(ctr, 0, 3 * sizeof(unsigned int), s));
        // Dense frontier (its edges are a big share of ALL edges): the
        // one-thread-per-edge pass with coalesced src/col streams beats
        // per-vertex serial neighbor loops.  Once dense mode fires, stay
        // dist-driven: subsequent edge-centric levels never materialize a
        // frontier (build_frontier=0), claims are wave-counted only.
        const bool dense = stay_dense ||
                           (edge_src != nullptr && num_edges > 0 &&
                            (double)frontier_degree > (double)num_edges / 8.0);
        if (dense) {
            stay_dense = true;
            if (rev_off && rev_col) {
                hipLaunchKernelGGL(abom::bfs_bottom_up_kernel,
                                   dim3(abom::grid_for(num_nodes, 256)), dim3(256), 0, s,
                                   (const uint64_t*)rev_off, (const uint32_t*)rev_col,
                                   (const uint8_t*)rev_et, allowed_mask, num_nodes,
                                   (uint32_t*)dist, (unsigned int)(level - 1), ctr);
                rc = (int)hipGetLastError();
                if (rc) return -rc;
            } else if (cur_bits) {
                if (!bits_ready) {
                    hipLaunchKernelGGL(abom::build_bits_kernel,
                                       dim3(abom::grid_for((num_nodes + 31) >> 5, 256)),
                                       dim3(256), 0, s,
                                       (const uint32_t*)dist, num_nodes,
                                       (unsigned int)(level - 1), cur_bits, visited_bits);
                    rc = (int)hipGetLastError();
                    if (rc) return -rc;
                    bits_ready = true;
                }
                ABOM_CHECK(hipMemsetAsync(next_bits, 0,
                                          bit_words * sizeof(uint32_t), s));
                hipLaunchKernelGGL(abom::bfs_expand_edges_bits_kernel,
                                   dim3(abom::grid_for(num_edges, 256)), dim3(256), 0, s,
                                   (const uint32_t*)edge_src, (const uint32_t*)col,
                                   (const uint8_t*)etype, allowed_mask, num_edges,
                                   (uint32_t*)dist, cur_bits, visited_bits, next_bits,
                                   (unsigned int)(level - 1), ctr);
                rc = (int)hipGetLastError();
                if (rc) return -rc;
                uint32_t* tb = cur_bits; cur_bits = next_bits; next_bits = tb;
            } else {
                rc = abom_bfs_expand_edges(edge_src, col, etype, allowed_mask, num_edges,
                                           row_off, dist, (unsigned int)(level - 1), nxt,
                                           ctr, ctr + 2, /*build_frontier=*/0,
                                           num_nodes, stream);
                if (rc) return -rc;
            }
        } else if (frontier_size < 0) {
            // dense mode handed back: rebuild the frontier from dist
            hipLaunchKernelGGL(abom::collect_frontier_kernel,
                               dim3(abom::grid_for(num_nodes, 256)), dim3(256), 0, s,
                               (const uint32_t*)dist, num_nodes,
                               (unsigned int)(level - 1), (const uint64_t*)row_off,
                               cur, ctr + 1, ctr + 2, num_nodes);
            rc = (int)hipGetLastError();
            if (rc) return -rc;
            unsigned int rebuilt[2] = {0, 0};
            ABOM_CHECK(hipMemcpyAsync(&rebuilt[0], ctr + 1, sizeof(unsigned int),
                                      hipMemcpyDeviceToHost, s));
            ABOM_CHECK(hipStreamSynchronize(s));
            frontier_size = rebuilt[0];
            // reset all three counters: the rebuilt frontier's degree sum is
            // the CURRENT level's, while the dense decision needs the NEXT
            // frontier's (accumulated by the expand below)
            ABOM_CHECK(hipMemsetAsync(ctr, 0, 3 * sizeof(unsigned int), s));
            rc = abom_bfs_expand(row_off, col, etype, allowed_mask, cur, frontier_size, dist,
                                 (unsigned int)level, nxt, ctr, heavy_queue, ctr + 1, ctr + 2,
                                 num_nodes, stream);
            if (rc) return -rc;
            rc = abom_bfs_expand_heavy(row_off, col, etype, allowed_mask, heavy_queue, ctr + 1,
                                       dist, (unsigned int)level, nxt, ctr, ctr + 2,
                                       num_nodes, stream);
            if (rc) return -rc;
        } else {
            rc = abom_bfs_expand(row_off, col, etype, allowed_mask, cur, frontier_size, dist,
                                 (unsigned int)level, nxt, ctr, heavy_queue, ctr + 1, ctr + 2,
                                 num_nodes, stream);
            if (rc) return -rc;
            rc = abom_bfs_expand_heavy(row_off, col, etype, allowed_mask, heavy_queue, ctr + 1,
                                       dist, (unsigned int)level, nxt, ctr, ctr + 2,
                                       num_nodes, stream);
            if (rc) return -rc;
        }
        unsigned int host_pair[3] = {0, 0, 0};
        ABOM_CHECK(hipMemcpyAsync(host_pair, ctr, 3 * sizeof(unsigned int),
                                  hipMemcpyDeviceToHost, s));
        ABOM_CHECK(hipStreamSynchronize(s));
        frontier_size = host_pair[0];
        frontier_degree = host_pair[2];
        if (stay_dense && frontier_size > 0 &&
            (double)frontier_size < (double)num_edges / 4096.0) {
            // dense claims have trickled out: hand back to vertex mode next
            // level (frontier rebuilt from dist — signalled by negative size)
            stay_dense = false;
            frontier_size = -1;
            bits_ready = false;  // re-entry rebuilds bitmaps from dist
        }
        uint32_t* t = cur; cur = nxt; nxt = t;
    }
    return level;
}

extern "C" int abom_impact_query(
    const void* row_off, const void* col, const void* etype, unsigned int allowed_mask,
    const void* query_sources, int num_queries, int max_hops, int max_nodes_per_query,
    void* out_nodes, void* out_hops, void* out_counts, void* out_truncated, void* stream) {
    const int block = 256;
    const int grid = num_queries < 2048 ? (num_queries > 0 ? num_queries : 1) : 2048;
    hipLaunchKernelGGL(abom::impact_query_kernel, dim3(grid), dim3(block), 0,
                       (hipStream_t)stream, (const uint64_t*)row_off, (const uint32_t*)col,
                       (const uint8_t*)etype, allowed_mask, (const uint32_t*)query_sources,
                       num_queries, max_hops, max_nodes_per_query, (uint32_t*)out_nodes,
                       (uint8_t*)out_hops, (uint32_t*)out_counts, (uint8_t*)out_truncated);
    return (int)hipGetLastError();
}

extern "C" int abom_risk_score(
    const void* severity, const void* n_agents, const void* n_creds, const void* n_tools,
    const void* flags, const void* epss, const void* scorecard, const void* reach,
    void* out, long long n, const float* weights22, void* stream) {
    abom::RiskWeights w;
    const float* p = weights22;
    w.base_critical = p[0]; w.base_high = p[1]; w.base_medium = p[2]; w.base_low = p[3];
    w.agent_w = p[4]; w.agent_cap = p[5]; w.cred_w = p[6]; w.cred_cap = p[7];
    w.tool_w = p[8]; w.tool_cap = p[9]; w.ai_boost = p[10]; w.kev_boost = p[11];
    w.epss_boost = p[12]; w.epss_threshold = p[13]; w.sc_t1 = p[14]; w.sc_b1 = p[15];
    w.sc_t2 = p[16]; w.sc_b2 = p[17]; w.sc_t3 = p[18]; w.sc_b3 = p[19];
    w.reach_boost = p[20]; w.unreach_penalty = p[21];
    const int block = 256;
    hipLaunchKernelGGL(abom::risk_score_kernel, dim3(abom::grid_for(n, block)), dim3(block), 0,
                       (hipStream_t)stream, (const uint8_t*)severity, (const uint32_t*)n_agents,
                       (const uint32_t*)n_creds, (const uint32_t*)n_tools, (const uint8_t*)flags,
                       (const float*)epss, (const float*)scorecard, (const int8_t*)reach,
                       (float*)out, n, w);
    return (int)hipGetLastError();
}

extern "C" int abom_score_gather(
    const void* win_idx, const void* pkg_nodes, const void* pos,
    const void* a_sev, const void* a_kev, const void* a_epss, const void* a_impact,
    const void* cred_lut, const void* tool_lut, const void* counts2d,
    const void* dist, void* out_scores, void* out_agents, void* out_creds,
    void* out_tools, long long n, const float* weights22, void* stream) {
    abom::RiskWeights w;
    const float* p = weights22;
    w.base_critical = p[0]; w.base_high = p[1]; w.base_medium = p[2]; w.base_low = p[3];
    w.agent_w = p[4]; w.agent_cap = p[5]; w.cred_w = p[6]; w.cred_cap = p[7];
    w.tool_w = p[8]; w.tool_cap = p[9]; w.ai_boost = p[10]; w.kev_boost = p[11];
    w.epss_boost = p[12]; w.epss_threshold = p[13]; w.sc_t1 = p[14]; w.sc_b1 = p[15];
    w.sc_t2 = p[16]; w.sc_b2 = p[17]; w.sc_t3 = p[18]; w.sc_b3 = p[19];
    w.reach_boost = p[20]; w.unreach_penalty = p[21];
    const int block = 256;
    hipLaunchKernelGGL(abom::score_gather_kernel, dim3(abom::grid_for(n, block)),
                       dim3(block), 0, (hipStream_t)stream,
                       (const long long*)win_idx, (const long long*)pkg_nodes,
                       (const long long*)pos, (const uint8_t*)a_sev,
                       (const uint8_t*)a_kev, (const float*)a_epss,
                       (const uint8_t*)a_impact, (const uint8_t*)cred_lut,
                       (const uint8_t*)tool_lut, (const int*)counts2d,
                       (const uint32_t*)dist, (float*)out_scores,
                       (int*)out_agents, (int*)out_creds, (int*)out_tools, n, w);
    return (int)hipGetLastError();
}

extern "C" int abom_severity_histogram(
    const void* owner, const void* severity, void* hist, long long n, void* stream) {
    const int block = 256;
    hipLaunchKernelGGL(abom::severity_histogram_kernel, dim3(abom::grid_for(n, block)),
                       dim3(block), 0, (hipStream_t)stream, (const uint32_t*)owner,
                       (const uint8_t*)severity, (unsigned int*)hist, n);
    return (int)hipGetLastError();
}
