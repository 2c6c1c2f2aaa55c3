// Fused blast-radius reach counting on gfx950.
//
// Replaces the torch sort/unique/bincount join pipeline (gpu_engine.py
// blast_counts — ~20 kernel launches per step) with ONE wave-cooperative
// kernel: each 64-lane wave owns one finding package and walks
// package -> servers (reverse CONTAINS) -> {agents (reverse USES),
// creds (forward EXPOSES_CRED), tools (forward PROVIDES_TOOL)} collecting
// candidates into per-wave LDS arrays, then distinct-counts them in LDS
// (O(m^2/64) first-occurrence scan — m is capped, LDS reads are 2 cycles).
//
// Zipf-head packages that overflow the LDS caps are flagged; the host
// resolves just those with the sort-based path (hybrid, exact either way).
// CDNA4 notes: one workgroup = 4 waves = 4 packages; LDS slice per wave =
// (SRV_CAP + AG_CAP + CR_CAP + TL_CAP) * 4 B = 4.5 KiB -> 18 KiB/block,
// sized so LDS is not the occupancy limiter (see cap comment below).

#include "abom_common.h"

namespace abom {

// Caps sized for occupancy: 4.1 KiB LDS per wave -> 16.5 KiB per 4-wave
// block -> 9 blocks/CU (full 32-wave occupancy; the previous 8.5 KiB/wave
// layout measured 11.1/32).  Zipf-head packages beyond a cap overflow to
// the exact sort-based join, so caps trade fallback rate for occupancy.
constexpr int SRV_CAP = 64;
constexpr int AG_CAP = 256;
constexpr int CR_CAP = 256;
constexpr int TL_CAP = 512;
constexpr int WAVES_PER_BLOCK = 4;

__device__ __forceinline__ int wave_reduce_add(int v) {
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    return v;
}

// Collect neighbors of `node` with edge type `want` into lds[cap]; returns
// new count or -1 on overflow (count accumulated across calls).
__device__ __forceinline__ int collect(
    const uint64_t* __restrict__ row_off, const uint32_t* __restrict__ col,
    const uint8_t* __restrict__ etype, uint32_t node, uint8_t want,
    uint32_t* lds, int count, int cap, int lane, bool* overflow) {
    const uint64_t beg = row_off[node];
    const uint64_t end = row_off[node + 1];
    for (uint64_t base = beg; base < end; base += 64) {
        const uint64_t e = base + lane;
        const bool valid = e < end && etype[e] == want;
        const unsigned long long mask = __ballot(valid);
        const int my_rank = __popcll(mask & ((1ull << lane) - 1));
        const int n_new = __popcll(mask);
        if (valid) {
            const int idx = count + my_rank;
            if (idx < cap) lds[idx] = col[e];
        }
        count += n_new;
        if (count > cap) { *overflow = true; return cap; }
    }
    return count;
}

// One pass over `node`'s FORWARD edge row appending creds and tools into
// their LDS arrays simultaneously — the fwd row of a server is dominated by
// CONTAINS->package edges, so scanning it once instead of once per type
// saves the bulk of this kernel's edge traffic.
__device__ __forceinline__ void collect_fwd2(
    const uint64_t* __restrict__ row_off, const uint32_t* __restrict__ col,
    const uint8_t* __restrict__ etype, uint32_t node,
    uint8_t want_a, uint32_t* lds_a, int* count_a, int cap_a,
    uint8_t want_b, uint32_t* lds_b, int* count_b, int cap_b,
    int lane, bool* overflow) {
    const uint64_t beg = row_off[node];
    const uint64_t end = row_off[node + 1];
    for (uint64_t base = beg; base < end; base += 64) {
        const uint64_t e = base + lane;
        const bool in = e < end;
        const uint8_t et = in ? etype[e] : (uint8_t)0xFF;
        const uint32_t v = in ? col[e] : 0u;
        const bool va = in && et == want_a;
        const bool vb = in && et == want_b;
        const unsigned long long ma = __ballot(va);
        const unsigned long long mb = __ballot(vb);
        if (va) {
            const int idx = *count_a + __popcll(ma & ((1ull << lane) - 1));
            if (idx < cap_a) lds_a[idx] = v;
        }
        if (vb) {
            const int idx = *count_b + __popcll(mb & ((1ull << lane) - 1));
            if (idx < cap_b) lds_b[idx] = v;
        }
        *count_a += __popcll(ma);
        *count_b += __popcll(mb);
        if (*count_a > cap_a || *count_b > cap_b) { *overflow = true; return; }
    }
}

// Distinct count of lds[0..m): lane-strided first-occurrence scan.
// Also counts distinct elements whose db_flag[node] is set.
__device__ __forceinline__ void distinct_count(
    const uint32_t* lds, int m, const uint8_t* __restrict__ db_flag,
    int lane, int* out_all, int* out_db) {
    int cnt = 0, cnt_db = 0;
    for (int i = lane; i < m; i += 64) {
        const uint32_t v = lds[i];
        bool first = true;
        for (int j = 0; j < i; ++j) {
            if (lds[j] == v) { first = false; break; }
        }
        if (first) {
            ++cnt;
            if (db_flag && db_flag[v]) ++cnt_db;
        }
    }
    *out_all = wave_reduce_add(cnt);
    *out_db = wave_reduce_add(cnt_db);
}

__global__ void __launch_bounds__(256) blast_count_kernel(
    const uint32_t* __restrict__ pkgs,          // [n] package node ids
    long long n,
    const uint64_t* __restrict__ rev_off, const uint32_t* __restrict__ rev_col,
    const uint8_t* __restrict__ rev_et,
    const uint64_t* __restrict__ fwd_off, const uint32_t* __restrict__ fwd_col,
    const uint8_t* __restrict__ fwd_et,
    uint8_t et_contains, uint8_t et_uses, uint8_t et_cred, uint8_t et_tool,
    const uint8_t* __restrict__ node_is_db_cred,  // [N]
    const uint8_t* __restrict__ node_is_db_tool,  // [N]
    uint32_t* __restrict__ out_counts,            // [n*6]
    uint8_t* __restrict__ out_overflow) {         // [n]
    __shared__ uint32_t lds_srv[WAVES_PER_BLOCK][SRV_CAP];
    __shared__ uint32_t lds_ag[WAVES_PER_BLOCK][AG_CAP];
    __shared__ uint32_t lds_cr[WAVES_PER_BLOCK][CR_CAP];
    __shared__ uint32_t lds_tl[WAVES_PER_BLOCK][TL_CAP];

    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const long long wave_global = (long long)blockIdx.x * WAVES_PER_BLOCK + wid;
    const long long n_waves = (long long)gridDim.x * WAVES_PER_BLOCK;

    for (long long p = wave_global; p < n; p += n_waves) {
        const uint32_t pkg = pkgs[p];
        bool overflow = false;

        int n_srv = collect(rev_off, rev_col, rev_et, pkg, et_contains,
                            lds_srv[wid], 0, SRV_CAP, lane, &overflow);
        int n_ag = 0, n_cr = 0, n_tl = 0;
        if (!overflow) {
            for (int s = 0; s < n_srv && !overflow; ++s) {
                const uint32_t srv = lds_srv[wid][s];
                n_ag = collect(rev_off, rev_col, rev_et, srv, et_uses,
                               lds_ag[wid], n_ag, AG_CAP, lane, &overflow);
                if (overflow) break;
                collect_fwd2(fwd_off, fwd_col, fwd_et, srv,
                             et_cred, lds_cr[wid], &n_cr, CR_CAP,
                             et_tool, lds_tl[wid], &n_tl, TL_CAP,
                             lane, &overflow);
            }
        }

        if (overflow) {
            if (lane == 0) {
                out_overflow[p] = 1;
                for (int k = 0; k < 6; ++k) out_counts[p * 6 + k] = 0;
            }
            continue;
        }

        int srv_all, srv_db, ag_all, ag_db, cr_all, cr_db, tl_all, tl_db;
        distinct_count(lds_srv[wid], n_srv, nullptr, lane, &srv_all, &srv_db);
        distinct_count(lds_ag[wid], n_ag, nullptr, lane, &ag_all, &ag_db);
        distinct_count(lds_cr[wid], n_cr, node_is_db_cred, lane, &cr_all, &cr_db);
        distinct_count(lds_tl[wid], n_tl, node_is_db_tool, lane, &tl_all, &tl_db);

        if (lane == 0) {
            out_overflow[p] = 0;
            out_counts[p * 6 + 0] = (uint32_t)srv_all;
            out_counts[p * 6 + 1] = (uint32_t)ag_all;
            out_counts[p * 6 + 2] = (uint32_t)cr_all;
            out_counts[p * 6 + 3] = (uint32_t)cr_db;
            out_counts[p * 6 + 4] = (uint32_t)tl_all;
            out_counts[p * 6 + 5] = (uint32_t)tl_db;
        }
    }
}

}  // namespace abom

extern "C" int abom_blast_counts(
    const void* pkgs, long long n,
    const void* rev_off, const void* rev_col, const void* rev_et,
    const void* fwd_off, const void* fwd_col, const void* fwd_et,
    int et_contains, int et_uses, int et_cred, int et_tool,
    const void* node_is_db_cred, const void* node_is_db_tool,
    void* out_counts, void* out_overflow, void* stream) {
    const int block = 256;
    long long blocks = (n + abom::WAVES_PER_BLOCK - 1) / abom::WAVES_PER_BLOCK;
    if (blocks > 2048) blocks = 2048;
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(abom::blast_count_kernel, dim3((int)blocks), dim3(block), 0,
                       (hipStream_t)stream,
                       (const uint32_t*)pkgs, n,
                       (const uint64_t*)rev_off, (const uint32_t*)rev_col,
                       (const uint8_t*)rev_et,
                       (const uint64_t*)fwd_off, (const uint32_t*)fwd_col,
                       (const uint8_t*)fwd_et,
                       (uint8_t)et_contains, (uint8_t)et_uses, (uint8_t)et_cred,
                       (uint8_t)et_tool,
                       (const uint8_t*)node_is_db_cred, (const uint8_t*)node_is_db_tool,
                       (uint32_t*)out_counts, (uint8_t*)out_overflow);
    return (int)hipGetLastError();
}
