// rank.hip — quantized-score finding rank as a stable 10-bit counting sort.
//
// Replaces the int32 radix sort in rank_order (graph/gpu_engine.py): scores
// quantize to 1001 buckets (0.01 steps, the serialization granularity), so
// the rank "sort" is really a 1024-bin stable counting sort:
//
//   1. rank_hist:    per-block bin histograms (wave-ballot equality groups —
//                    ten __ballot()s split the 64 lanes into same-bin groups,
//                    one LDS add per group leader, no atomics);
//   2. rank_scan:    exclusive scan, bin-major then block-minor, giving each
//                    (block, bin) its global output base;
//   3. rank_scatter: replay the histogram walk; each element's position is
//                    base[block][bin] + running[bin] + rank-within-wave-group,
//                    which reproduces (bucket asc, index asc) EXACTLY —
//                    bit-equal to the torch key sort it replaces.
//
// wave64 throughout: one wave per block, lockstep LDS (no atomics, no
// __syncthreads in the inner loop).  Four small launches replace the
// multi-kernel rocprim onesweep pipeline.
#include "abom_common.h"

namespace {

constexpr int RANK_BINS = 1024;   // 1001 used; power-of-two for the ballots
constexpr int RANK_CHUNK = 256;   // elements per block, wave does 4 rounds

__device__ __forceinline__ int score_bin(float s) {
    // match rank_order: q = round(score*100).clamp(0,1000); bin = 1000 - q.
    // rintf = round-half-to-even, same as torch .round().
    float q = rintf(s * 100.0f);
    q = fminf(fmaxf(q, 0.0f), 1000.0f);
    return 1000 - (int)q;
}

// lanes holding the same 10-bit bin, restricted to `act`
__device__ __forceinline__ uint64_t eq_group(int bin, uint64_t act) {
    uint64_t m = act;
    #pragma unroll
    for (int b = 0; b < 10; ++b) {
        uint64_t bb = __ballot((bin >> b) & 1);
        m &= ((bin >> b) & 1) ? bb : ~bb;
    }
    return m;
}

__global__ void rank_hist_kernel(const float* __restrict__ scores,
                                 long long n,
                                 int* __restrict__ hist /* [nblocks][1024] */) {
    __shared__ int lhist[RANK_BINS];
    const int lane = threadIdx.x;  // blockDim.x == 64
    for (int i = lane; i < RANK_BINS; i += 64) lhist[i] = 0;
    __syncthreads();
    const long long base = (long long)blockIdx.x * RANK_CHUNK;
    for (int r = 0; r < RANK_CHUNK; r += 64) {
        const long long idx = base + r + lane;
        const bool active = idx < n;
        const int bin = active ? score_bin(scores[idx]) : 0;
        const uint64_t act = __ballot(active);
        const uint64_t eq = eq_group(bin, act);
        if (active && (eq & ((1ull << lane) - 1)) == 0)  // group leader
            lhist[bin] += __popcll(eq);
    }
    __syncthreads();
    int* gh = hist + (long long)blockIdx.x * RANK_BINS;
    for (int i = lane; i < RANK_BINS; i += 64) gh[i] = lhist[i];
}

// Exclusive scan across blocks for each bin; bin totals out.
// One block (one wave) per bin: 64-lane tiles with a shfl_up scan ladder
// and a carried running total — the serial single-thread version of this
// walk measured ~0.1 ms at 1280 blocks and dominated the whole rank.
__global__ void rank_scan_blocks_kernel(int* __restrict__ hist,
                                        int nblocks,
                                        int* __restrict__ bin_total) {
    const int bin = blockIdx.x;
    const int lane = threadIdx.x;  // blockDim.x == 64
    int carry = 0;
    for (int t = 0; t < nblocks; t += 64) {
        const int idx = t + lane;
        const int v = (idx < nblocks)
            ? hist[(long long)idx * RANK_BINS + bin] : 0;
        int x = v;  // wave inclusive scan
        #pragma unroll
        for (int off = 1; off < 64; off <<= 1) {
            const int y = __shfl_up(x, off);
            if (lane >= off) x += y;
        }
        if (idx < nblocks)
            hist[(long long)idx * RANK_BINS + bin] = x - v + carry;
        carry += __shfl(x, 63);  // tile total, broadcast to all lanes
    }
    if (lane == 0) bin_total[bin] = carry;
}

// Exclusive scan of the 1024 bin totals (single wave, LDS ladder).
__global__ void rank_scan_bins_kernel(int* __restrict__ bin_total,
                                      int* __restrict__ bin_base) {
    __shared__ int vals[RANK_BINS];
    const int lane = threadIdx.x;
    for (int i = lane; i < RANK_BINS; i += 64) vals[i] = bin_total[i];
    __syncthreads();
    if (lane == 0) {  // 1024 serial adds on one lane — trivial vs launch cost
        int acc = 0;
        for (int i = 0; i < RANK_BINS; ++i) {
            const int v = vals[i];
            vals[i] = acc;
            acc += v;
        }
    }
    __syncthreads();
    for (int i = lane; i < RANK_BINS; i += 64) bin_base[i] = vals[i];
}

__global__ void rank_scatter_kernel(const float* __restrict__ scores,
                                    long long n,
                                    const int* __restrict__ hist,
                                    const int* __restrict__ bin_base,
                                    long long* __restrict__ order) {
    __shared__ int run[RANK_BINS];
    const int lane = threadIdx.x;
    for (int i = lane; i < RANK_BINS; i += 64) run[i] = 0;
    __syncthreads();
    const long long base = (long long)blockIdx.x * RANK_CHUNK;
    const int* gb = hist + (long long)blockIdx.x * RANK_BINS;
    for (int r = 0; r < RANK_CHUNK; r += 64) {
        const long long idx = base + r + lane;
        const bool active = idx < n;
        const int bin = active ? score_bin(scores[idx]) : 0;
        const uint64_t act = __ballot(active);
        const uint64_t eq = eq_group(bin, act);
        if (active) {
            const uint64_t lower = eq & ((1ull << lane) - 1);
            const long long pos =
                (long long)bin_base[bin] + gb[bin] + run[bin] + __popcll(lower);
            order[pos] = idx;
            if (lower == 0)  // leader bumps the block-local running count
                run[bin] += __popcll(eq);
        }
    }
}

}  // namespace

extern "C" int abom_rank_order(const void* scores, long long n,
                               void* hist, void* bin_scratch, int nblocks,
                               void* order, void* stream) {
    if (n <= 0) return 0;
    hipStream_t s = (hipStream_t)stream;
    int* bin_total = (int*)bin_scratch;
    int* bin_base = bin_total + RANK_BINS;
    hipLaunchKernelGGL(rank_hist_kernel, dim3(nblocks), dim3(64), 0, s,
                       (const float*)scores, n, (int*)hist);
    hipLaunchKernelGGL(rank_scan_blocks_kernel, dim3(RANK_BINS), dim3(64), 0, s,
                       (int*)hist, nblocks, bin_total);
    hipLaunchKernelGGL(rank_scan_bins_kernel, dim3(1), dim3(64), 0, s,
                       bin_total, bin_base);
    hipLaunchKernelGGL(rank_scatter_kernel, dim3(nblocks), dim3(64), 0, s,
                       (const float*)scores, n, (const int*)hist,
                       (const int*)bin_base, (long long*)order);
    return (int)hipGetLastError();
}
