/* t9_sort_msb9.hip — 9-bit level-2 digit for the two-level MSB pipeline.
 * This is the PRODUCTION DEFAULT for 2^22 <= n < T9_L3_MIN
 * (pass2_bits() in t9_sort_msb.hip defaults to 9; T9_PASS2_BITS=8 is the
 * byte-digit fallback): 256 x 512 = 131072
 * (b7, 9-bit) sub-buckets of ~n/131072 elements, so the level-3 LDS sort
 * runs the 1024-element / 256-thread variant at ~5 blocks per CU — the
 * measured limiter of the 8-bit flow was the LDS sort's barrier parking
 * at 2 blocks per CU. Level-2 digit = key bits 47..55; level 3 sorts the
 * low 47 bits (6 passes, bit 47 re-covered harmlessly).
 *
 * Self-contained variants of hist/seg-scan/scatter over 512 digits with
 * u16 count rows; digit-of-slot is recomputed from the staged key
 * instead of stored. Shares the pass-1 machinery and fallbacks with
 * t9_sort_msb.hip via its exported helpers.
 */

#include "t9_common.h"
#include "t9_rank_scatter.h"

#include <cstdlib>
#include <vector>

#define T9_MSB_TILE 8192
#define NDIG9 512

__device__ inline u32 bucket_of9(const u32* s_abase, u32 pos) {
    u32 lo = 0, hi = 255;
    while (lo < hi) {
        u32 mid = (lo + hi + 1) >> 1;
        if (s_abase[mid] <= pos) lo = mid; else hi = mid - 1;
    }
    return lo;
}

/* segmented per-block histogram over the padded pass-1 output, 512 bins */
__global__ __launch_bounds__(256) void k_hist_seg9(
    const u64* __restrict__ keys, const u32* __restrict__ abase,
    const u32* __restrict__ bucket_n, u32* __restrict__ hist) {
    __shared__ u32 s_cnt[NDIG9];
    __shared__ u32 s_abase[257];
    const u32 tid = threadIdx.x;
    s_abase[tid] = abase[tid];
    if (tid == 0) s_abase[256] = abase[256];
    s_cnt[tid] = 0;
    s_cnt[tid + 256] = 0;
    __syncthreads();
    /* u64 tile base before narrowing (grid is B2max-sized; see the same
     * guard in t9_sort_msb.hip k_hist_seg — ADVICE r01, medium) */
    const u64 tbase64 = (u64)blockIdx.x * T9_MSB_TILE;
    if (tbase64 < s_abase[256]) {
        const u32 tbase = (u32)tbase64;
        const u32 b = bucket_of9(s_abase, tbase);
        const u32 off = tbase - s_abase[b];
        const u32 tn = (bucket_n[b] > off)
                           ? ((bucket_n[b] - off < T9_MSB_TILE)
                                  ? bucket_n[b] - off
                                  : T9_MSB_TILE)
                           : 0;
        const u32 lane = tid & 63;
        for (u32 i0 = 0; i0 < (u32)T9_MSB_TILE; i0 += 256) {
            const u32 i = i0 + tid;
            const bool valid = i < tn;
            const u32 d =
                valid ? (u32)(keys[tbase + i] >> 47) & 511u : 0;
            t9_hist_ballot_add<9>(s_cnt, d, valid, lane);
        }
    }
    __syncthreads();
    hist[(u64)blockIdx.x * NDIG9 + tid] = s_cnt[tid];
    hist[(u64)blockIdx.x * NDIG9 + tid + 256] = s_cnt[tid + 256];
}

/* per-bucket segmented column scan over 512 digits (thread t owns digits
 * t and t+256) -> final offsets in hist, sub index arrays */
__global__ __launch_bounds__(256) void k_seg_scan9(
    u32* __restrict__ hist, const u32* __restrict__ abase,
    const u32* __restrict__ bucket_n, const u32* __restrict__ true_base,
    u32* __restrict__ sub_start, u32* __restrict__ sub_n) {
    const u32 b = blockIdx.x;
    const u32 tid = threadIdx.x;
    const u32 r0 = abase[b] / T9_MSB_TILE;
    const u32 rows = (bucket_n[b] + T9_MSB_TILE - 1) / T9_MSB_TILE;
    u32 tot0 = 0, tot1 = 0;
    for (u32 r = 0; r < rows; ++r) {
        tot0 += hist[(u64)(r0 + r) * NDIG9 + tid];
        tot1 += hist[(u64)(r0 + r) * NDIG9 + tid + 256];
    }
    __shared__ u32 sh[NDIG9];
    sh[tid] = tot0;
    sh[tid + 256] = tot1;
    __syncthreads();
    t9_scan_onewave<NDIG9>(sh, tid);
    __syncthreads();
    const u32 gstart0 = true_base[b] + sh[tid];
    const u32 gstart1 = true_base[b] + sh[tid + 256];
    sub_start[(u64)b * NDIG9 + tid] = gstart0;
    sub_start[(u64)b * NDIG9 + tid + 256] = gstart1;
    sub_n[(u64)b * NDIG9 + tid] = tot0;
    sub_n[(u64)b * NDIG9 + tid + 256] = tot1;
    u32 run0 = gstart0, run1 = gstart1;
    for (u32 r = 0; r < rows; ++r) {
        u32 v0 = hist[(u64)(r0 + r) * NDIG9 + tid];
        u32 v1 = hist[(u64)(r0 + r) * NDIG9 + tid + 256];
        hist[(u64)(r0 + r) * NDIG9 + tid] = run0;
        hist[(u64)(r0 + r) * NDIG9 + tid + 256] = run1;
        run0 += v0;
        run1 += v1;
    }
}

/* segmented 1024-thread scatter over 512 digits; u16 count/offset rows,
 * digit recomputed from the staged key in the write phase */
template <bool HAS_VAL>
__global__ __launch_bounds__(1024, 4) void k_scatter_seg9(
    const u64* __restrict__ in_keys, const u32* __restrict__ in_vals,
    const u32* __restrict__ abase, const u32* __restrict__ bucket_n,
    u64* __restrict__ out_keys, u32* __restrict__ out_vals,
    const u32* __restrict__ offs) {
    constexpr int TILE = T9_MSB_TILE;
    constexpr int NW = 16;
    constexpr int SUB = TILE / NW;
    constexpr int GROUPS = SUB / 64;
    __shared__ u64 s_okeys[TILE];
    __shared__ u32 s_ovals[HAS_VAL ? TILE : 1];
    __shared__ u16 s_rank[TILE];
    __shared__ u16 s_wcnt[NW * NDIG9];
    __shared__ u16 s_woff[NW * NDIG9];
    __shared__ u32 s_start[NDIG9];
    __shared__ u32 s_goff[NDIG9];
    __shared__ u32 s_abase[257];

    const u32 tid = threadIdx.x, wave = tid >> 6, lane = tid & 63;
    if (tid < 257) s_abase[tid] = abase[tid];
    __syncthreads();
    /* u64 guard before narrowing — see k_hist_seg9 */
    const u64 tbase64 = (u64)blockIdx.x * TILE;
    if (tbase64 >= s_abase[256]) return;
    const u32 tbase = (u32)tbase64;
    const u32 b = bucket_of9(s_abase, tbase);
    const u32 off_in_bucket = tbase - s_abase[b];
    const u32 bn = bucket_n[b];
    const u32 tn = (bn > off_in_bucket)
                       ? ((bn - off_in_bucket < (u32)TILE)
                              ? bn - off_in_bucket
                              : (u32)TILE)
                       : 0;

    if (tid < NDIG9)
        s_goff[tid] = offs[(u64)blockIdx.x * NDIG9 + tid];
    for (u32 t = lane; t < NDIG9; t += 64) s_wcnt[wave * NDIG9 + t] = 0;

    const u32 wbase = wave * SUB;
    for (int g = 0; g < GROUPS; ++g) {
        const u32 i = wbase + g * 64 + lane;
        const bool valid = i < tn;
        u32 d = 0;
        if (valid) d = (u32)(in_keys[tbase + i] >> 47) & 511u;
        u64 m = __ballot(valid);
        for (int bit = 0; bit < 9; ++bit) {
            u64 bb = __ballot((d >> bit) & 1u);
            m &= ((d >> bit) & 1u) ? bb : ~bb;
        }
        const u32 wr = (u32)__popcll(m & ((1ull << lane) - 1ull));
        const u32 before = valid ? (u32)s_wcnt[wave * NDIG9 + d] : 0;
        if (valid) {
            s_rank[i] = (u16)(before + wr);
            if (wr == 0)
                s_wcnt[wave * NDIG9 + d] =
                    (u16)(before + (u32)__popcll(m));
        }
    }
    __syncthreads();

    if (tid < NDIG9) {
        u32 run = 0;
        for (int w = 0; w < NW; ++w) {
            s_woff[w * NDIG9 + tid] = (u16)run;
            run += s_wcnt[w * NDIG9 + tid];
        }
        s_start[tid] = run;
    }
    __syncthreads();
    t9_scan_onewave<NDIG9>(s_start, tid);
    __syncthreads();

    for (int g = 0; g < GROUPS; ++g) {
        const u32 i = wbase + g * 64 + lane;
        if (i < tn) {
            const u64 k = in_keys[tbase + i];
            const u32 d = (u32)(k >> 47) & 511u;
            const u32 pos = s_start[d] + (u32)s_woff[wave * NDIG9 + d] +
                            s_rank[i];
            s_okeys[pos] = k;
            if (HAS_VAL) s_ovals[pos] = in_vals[tbase + i];
        }
    }
    __syncthreads();

    constexpr int CHUNKS = TILE / 1024;
    for (int c = 0; c < CHUNKS; ++c) {
        const u32 j = c * 1024 + tid;
        if (j < tn) {
            const u32 d = (u32)(s_okeys[j] >> 47) & 511u;
            const u64 gpos = (u64)s_goff[d] + (j - s_start[d]);
            out_keys[gpos] = s_okeys[j];
            if (HAS_VAL) out_vals[gpos] = s_ovals[j];
        }
    }
}

/* C wrappers so the host flow in t9_sort_msb.hip can launch these
 * without cross-TU template plumbing */
extern "C" {

void t9i_launch_hist_seg9(u32 grid, void* stream, const u64* keys,
                          const u32* abase, const u32* bucket_n,
                          u32* hist) {
    hipLaunchKernelGGL(k_hist_seg9, dim3(grid), dim3(256), 0,
                       (hipStream_t)stream, keys, abase, bucket_n, hist);
}

void t9i_launch_seg_scan9(u32 grid, void* stream, u32* hist,
                          const u32* abase, const u32* bucket_n,
                          const u32* true_base, u32* sub_start,
                          u32* sub_n) {
    hipLaunchKernelGGL(k_seg_scan9, dim3(grid), dim3(256), 0,
                       (hipStream_t)stream, hist, abase, bucket_n,
                       true_base, sub_start, sub_n);
}

void t9i_launch_scatter_seg9(u32 grid, int has_val, void* stream,
                             const u64* in_keys, const u32* in_vals,
                             const u32* abase, const u32* bucket_n,
                             u64* out_keys, u32* out_vals,
                             const u32* offs) {
    if (has_val)
        hipLaunchKernelGGL((k_scatter_seg9<true>), dim3(grid), dim3(1024),
                           0, (hipStream_t)stream, in_keys, in_vals, abase,
                           bucket_n, out_keys, out_vals, offs);
    else
        hipLaunchKernelGGL((k_scatter_seg9<false>), dim3(grid), dim3(1024),
                           0, (hipStream_t)stream, in_keys, in_vals, abase,
                           bucket_n, out_keys, out_vals, offs);
}

} /* extern "C" */
