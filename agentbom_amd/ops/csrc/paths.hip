// Attack / exposure path enumeration: hop-bounded max-score path DP.
//
// Replaces the reference's Python DFS walk (agent-bom
// src/agent_bom/graph/attack_path_fusion.py:194-377 compute_fused_attack_paths
// + graph/path_ranking.py:62) with a CDNA4 label-correcting relaxation that
// scales to the HBM-resident estate CSR:
//
//   label(v, d, g) = best score of any d-hop path entry->v whose gate state
//   is g (g=1 once the path used a vuln-class edge or touched a gated node),
//   with the winning edge recorded for exact path reconstruction.
//
// One kernel launch per hop, edge-centric (dense coalesced src/col/etype
// streams — the same shape as the BFS dense mode that measured fastest on
// this estate in round 1).  Labels are packed u64:
//     (ordered_f32(score) << 32) | winner_edge_index
// so ONE device-scope atomicMax both ranks by score AND breaks ties
// deterministically by edge index — the result is order-independent, which
// gives run-to-run deterministic paths (graph contract: determinism given
// the same inventory).  Scores are non-negative by construction (all boosts
// >= 0), so packed==0 doubles as the invalid label.
//
// Per-path scores are exact (computed along the path, never from partial
// atomics), so float reproducibility holds regardless of relaxation order.

#include "abom_common.h"

namespace {

__device__ __forceinline__ uint32_t ordered_f32(float f) {
    // non-negative floats: setting the sign bit preserves order vs u32
    uint32_t u = __float_as_uint(f);
    return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

__device__ __forceinline__ float unordered_f32(uint32_t u) {
    uint32_t raw = (u & 0x80000000u) ? (u & 0x7FFFFFFFu) : ~u;
    return __uint_as_float(raw);
}

__device__ __forceinline__ uint64_t pack_label(float score, uint32_t edge) {
    return ((uint64_t)ordered_f32(score) << 32) | edge;
}

// One relaxation hop: read labels at hop d-1 (cur), atomicMax into hop d
// (nxt).  Layout: labels[node*2 + gate_state].
__global__ void path_relax_kernel(
    const int32_t* __restrict__ edge_src,   // col-aligned, [E]
    const int32_t* __restrict__ col,        // [E]
    const uint8_t* __restrict__ etype,      // [E]
    const float* __restrict__ edge_weight,  // [E] or nullptr
    const unsigned long long* __restrict__ cur,  // [N*2]
    unsigned long long* __restrict__ nxt,        // [N*2] pre-zeroed
    const float* __restrict__ node_boost,        // [N]
    const float* __restrict__ etype_boost,       // [256]
    const uint8_t* __restrict__ etype_trav,      // [256] 0/1
    const uint8_t* __restrict__ etype_gate,      // [256] 0/1
    const uint8_t* __restrict__ node_gate,       // [N] 0/1 (or nullptr)
    long long E) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long e = (long long)blockIdx.x * blockDim.x + threadIdx.x; e < E; e += stride) {
        uint8_t et = etype[e];
        if (!etype_trav[et]) continue;
        int32_t u = edge_src[e];
        unsigned long long lu0 = cur[(long long)u * 2 + 0];
        unsigned long long lu1 = cur[(long long)u * 2 + 1];
        if (!(lu0 | lu1)) continue;
        int32_t v = col[e];
        float step = etype_boost[et] + node_boost[v];
        if (edge_weight) step += edge_weight[e] * 0.3f;
        bool gate = etype_gate[et] || (node_gate && node_gate[v]);
        if (lu0) {
            // predecessor-avoidance: never relax straight back to the node
            // this label came from (kills 2-cycles, the dominant non-simple
            // walk class on lateral agent<->server edges)
            uint32_t pe = (uint32_t)lu0;
            if (pe == 0xFFFFFFFFu || edge_src[pe] != v) {
                float s = unordered_f32((uint32_t)(lu0 >> 32)) + step;
                unsigned long long cand = pack_label(s, (uint32_t)e);
                atomicMax(&nxt[(long long)v * 2 + (gate ? 1 : 0)], cand);
            }
        }
        if (lu1) {
            uint32_t pe = (uint32_t)lu1;
            if (pe == 0xFFFFFFFFu || edge_src[pe] != v) {
                float s = unordered_f32((uint32_t)(lu1 >> 32)) + step;
                atomicMax(&nxt[(long long)v * 2 + 1], pack_label(s, (uint32_t)e));
            }
        }
    }
}

}  // namespace

extern "C" int abom_path_relax(
    const void* edge_src, const void* col, const void* etype,
    const void* edge_weight,  // nullable
    const void* cur, void* nxt,
    const void* node_boost, const void* etype_boost,
    const void* etype_trav, const void* etype_gate,
    const void* node_gate,  // nullable
    long long num_edges, void* stream) {
    int block = 256;
    int grid = abom::grid_for(num_edges, block);
    hipLaunchKernelGGL(path_relax_kernel, dim3(grid), dim3(block), 0,
                       (hipStream_t)stream,
                       (const int32_t*)edge_src, (const int32_t*)col,
                       (const uint8_t*)etype, (const float*)edge_weight,
                       (const unsigned long long*)cur, (unsigned long long*)nxt,
                       (const float*)node_boost, (const float*)etype_boost,
                       (const uint8_t*)etype_trav, (const uint8_t*)etype_gate,
                       (const uint8_t*)node_gate, num_edges);
    return (int)hipGetLastError();
}
