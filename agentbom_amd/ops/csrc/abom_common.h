// Common device helpers for the agent-bom MI355X engine (gfx950 / CDNA4).
//
// Design notes:
// - wave64: all wave-level idioms use 64-lane masks.
// - Memory-bound kernels follow the CDNA4 guidelines: grid-stride loops
//   capped at ~2048 blocks, coalesced SoA access, one atomic per wave where
//   the compiler can aggregate (hipcc coalesces per-lane atomicAdd(p,1)).
// - u128 keys are compared as (hi, lo) unsigned pairs; see
//   agentbom_amd/utils/version_keys.py for the encoding contract.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define ABOM_UNVISITED 0xFFFFFFFFu

namespace abom {

__device__ __forceinline__ bool key_lt(uint64_t ahi, uint64_t alo, uint64_t bhi, uint64_t blo) {
    return (ahi < bhi) | ((ahi == bhi) & (alo < blo));
}
__device__ __forceinline__ bool key_ge(uint64_t ahi, uint64_t alo, uint64_t bhi, uint64_t blo) {
    return !key_lt(ahi, alo, bhi, blo);
}
__device__ __forceinline__ bool key_gt(uint64_t ahi, uint64_t alo, uint64_t bhi, uint64_t blo) {
    return key_lt(bhi, blo, ahi, alo);
}

// Window flags (shared contract with agentbom_amd/db/arena.py)
constexpr uint8_t WF_HAS_INTRO = 1u << 0;
constexpr uint8_t WF_HAS_FIXED = 1u << 1;
constexpr uint8_t WF_HAS_LAST = 1u << 2;
constexpr uint8_t WF_CPU_FALLBACK = 1u << 3;  // unencodable bound: host resolves
constexpr uint8_t WF_UNFIXED_SUPPRESSED = 1u << 4;  // distro unfixed, suppressed by default

// Package flags
constexpr uint8_t PF_ENCODABLE = 1u << 0;

inline int grid_for(long long work, int block) {
    long long blocks = (work + block - 1) / block;
    if (blocks > 2048) blocks = 2048;  // grid-stride the rest (256 CUs x 8)
    if (blocks < 1) blocks = 1;
    return (int)blocks;
}

}  // namespace abom

#define ABOM_CHECK(expr)                                                     \
    do {                                                                     \
        hipError_t _e = (expr);                                              \
        if (_e != hipSuccess) return (int)_e;                                \
    } while (0)
