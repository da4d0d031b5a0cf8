/* t9_sort_msb.hip — two-level MSB radix sort for (u64 key, u32 idx) pairs.
 *
 * Replaces the 8-pass LSD pipeline for large n with:
 *   pass 1: stable scatter by key byte 7 into 256 buckets whose starts are
 *           TILE-aligned (padding), so every pass-2 tile lies in exactly
 *           one bucket;
 *   pass 2: stable scatter by key byte 6, segmented per bucket (a
 *           per-bucket column scan computes the (bucket, digit) bases),
 *           landing at final compact positions grouped by the top 16 bits;
 *   level 3: each of the 65536 (b7, b6) sub-buckets (~n/65536 pairs) is
 *           sorted IN LDS over the remaining 48 bits (6 ballot-ranked
 *           passes, no HBM traffic beyond one read + one write).
 *
 * HBM traffic ≈ 2 scatter passes + 1 LDS-sort read/write ≈ 3×24 B/pair vs
 * the LSD pipeline's 8×24 B. Every stage is stable, so the composition is
 * stable (the tie-fix and the parity tests rely on that).
 *
 * Skew handling: sub-buckets larger than T9_SUBMAX fall back — few: each
 * is re-sorted in place by the (stable) LSD path on its range; many
 * (pathological, e.g. all-equal top bytes): one full LSD sort of the
 * current (stable-permuted) array. Sub-buckets whose remaining 48 bits are
 * all equal are skipped entirely (already in stable order), so the
 * all-equal-key case degrades to ~2 passes, not 8.
 */

#include "t9_common.h"
#include "t9_rank_scatter.h"
#include "t9_bitonic.h"

#include <cstdlib>
#include <vector>

#define T9_MSB_TILE 8192       /* pass tile (matches scatter v7 geometry) */
#define T9_SUBMAX 4096         /* max sub-bucket for the LDS sort */
#define T9_MSB_MIN (1u << 22)  /* below this, plain LSD */

/* kernels defined in t9_sort.hip (non-template, external linkage) */
__global__ void k_colsum(const u32*, u64, u32*);
__global__ void k_chunkscan(u32*, u64, u32*);
__global__ void k_finaloffs(u32*, u64, const u32*, const u32*);
/* exclusive u32 scan + total (t9_group.hip) */
__global__ void k_grp_scan(u32*, u64, u64*);

/* 9-bit level-2 experiment (t9_sort_msb9.hip, T9_PASS2_BITS=9) */
extern "C" void t9i_launch_hist_seg9(u32, void*, const u64*, const u32*,
                                     const u32*, u32*);
extern "C" void t9i_launch_seg_scan9(u32, void*, u32*, const u32*,
                                     const u32*, const u32*, u32*, u32*);
extern "C" void t9i_launch_scatter_seg9(u32, int, void*, const u64*,
                                        const u32*, const u32*, const u32*,
                                        u64*, u32*, const u32*);
constexpr u64 NSUB9 = 256ull * 512;

/* level-2 digit width: 9 bits by default (measured -1.5% end to end:
 * lds_sort 3.00 -> 2.66 ms at the 10 GiB bench), T9_PASS2_BITS=8
 * restores the byte digit */
static int pass2_bits() {
    const char* e = getenv("T9_PASS2_BITS");
    return (e && atoi(e) == 8) ? 8 : 9;
}

#define T9_L3_MIN 240000000ull   /* 3 MSB levels above this n */

extern "C" int t9i_sort_pairs_lsd(t9_context*, u64*, u32*, u64, void*,
                                  void*);
extern "C" u64 t9i_sort_pairs_lsd_workspace(u64 n);
extern "C" int t9i_sort_keys_lsd(t9_context*, u64*, u64, void*, void*);

/* ------------------------------------------------------------------ */

/* plain per-block histogram (tile = T9_MSB_TILE). RW = 0: digit from the
 * packed u64 key; RW != 0: keys argument is the record byte array
 * (RW u32 words per record), digit = big-endian key byte (7 - shift/8). */
template <int RW>
__global__ __launch_bounds__(256) void k_hist_msb(
    const u64* __restrict__ keys, u64 n, u32 shift, u32* __restrict__ hist) {
    __shared__ u32 s_cnt[T9_RADIX];
    const u32 tid = threadIdx.x;
    const u64 base = (u64)blockIdx.x * T9_MSB_TILE;
    const u32 tn =
        (u32)((n - base < (u64)T9_MSB_TILE) ? (n - base) : (u64)T9_MSB_TILE);
    s_cnt[tid] = 0;
    __syncthreads();
    const u8* rec8 = (const u8*)keys;
    const u32 lane = tid & 63;
    for (u32 i0 = 0; i0 < (u32)T9_MSB_TILE; i0 += 256) {
        const u32 i = i0 + tid;
        const bool valid = i < tn;
        u32 d = 0;
        if (valid) {
            if (RW)
                d = rec8[(base + i) * (u64)RW * 4 + (7 - shift / 8)];
            else
                d = (u32)(keys[base + i] >> shift) & 255u;
        }
        t9_hist_ballot_add<8>(s_cnt, d, valid, lane);
    }
    __syncthreads();
    hist[(u64)blockIdx.x * T9_RADIX + tid] = s_cnt[tid];
}

/* aligned bucket bases from the pass-1 exclusive digit bases.
 * true_base = copy of the exclusive scan (compact positions);
 * abase[d] = TILE-aligned start of bucket d in the padded layout,
 * abase[256] = padded total; bucket_n[d] = count. One block. */
__global__ __launch_bounds__(256) void k_align_bases(
    const u32* __restrict__ digit_base, u64 n, u32* __restrict__ true_base,
    u32* __restrict__ abase, u32* __restrict__ bucket_n) {
    const u32 tid = threadIdx.x;
    __shared__ u32 s[T9_RADIX];
    const u32 b0 = digit_base[tid];
    const u32 b1 = (tid == 255) ? (u32)n : digit_base[tid + 1];
    const u32 cnt = b1 - b0;
    true_base[tid] = b0;
    bucket_n[tid] = cnt;
    const u32 padded =
        (cnt + T9_MSB_TILE - 1) / T9_MSB_TILE * T9_MSB_TILE;
    s[tid] = padded;
    __syncthreads();
    for (int off = 1; off < T9_RADIX; off <<= 1) {
        u32 y = (tid >= (u32)off) ? s[tid - off] : 0;
        __syncthreads();
        s[tid] += y;
        __syncthreads();
    }
    abase[tid] = s[tid] - padded;
    if (tid == 255) abase[256] = s[255];
}

/* find the bucket containing padded position `pos` (abase in LDS) */
__device__ inline u32 bucket_of(const u32* s_abase, u32 pos) {
    u32 lo = 0, hi = 255;
    while (lo < hi) {
        u32 mid = (lo + hi + 1) >> 1;
        if (s_abase[mid] <= pos) lo = mid; else hi = mid - 1;
    }
    return lo;
}

/* segmented per-block histogram over the padded pass-1 output */
__global__ __launch_bounds__(256) void k_hist_seg(
    const u64* __restrict__ keys, const u32* __restrict__ abase,
    const u32* __restrict__ bucket_n, u32 shift, u32* __restrict__ hist) {
    __shared__ u32 s_cnt[T9_RADIX];
    __shared__ u32 s_abase[257];
    const u32 tid = threadIdx.x;
    s_abase[tid] = abase[tid];
    if (tid == 0) s_abase[256] = abase[256];
    s_cnt[tid] = 0;
    __syncthreads();
    /* u64 tile base: the pass-2 grid is sized B2max (which includes the
     * 3-level row allowance), so blockIdx.x * TILE can exceed 2^32 for
     * large n — a wrapped u32 would pass the < abase[256] guard and
     * re-scatter a stale tile. */
    const u64 tbase64 = (u64)blockIdx.x * T9_MSB_TILE;
    if (tbase64 < s_abase[256]) {
        const u32 tbase = (u32)tbase64;
        const u32 b = bucket_of(s_abase, tbase);
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
            const u32 d = valid
                              ? (u32)(keys[tbase + i] >> shift) & 255u
                              : 0;
            t9_hist_ballot_add<8>(s_cnt, d, valid, lane);
        }
    }
    __syncthreads();
    hist[(u64)blockIdx.x * T9_RADIX + tid] = s_cnt[tid];
}

/* per-bucket segmented column scan: block b scans its bucket's hist rows,
 * producing final global offsets (compact positions, from true_base) in
 * place, plus sub_start/sub_n for the 256 (b, b6) sub-buckets. */
__global__ __launch_bounds__(256) void k_seg_scan(
    u32* __restrict__ hist, const u32* __restrict__ abase,
    const u32* __restrict__ bucket_n, const u32* __restrict__ true_base,
    u32* __restrict__ sub_start, u32* __restrict__ sub_n) {
    const u32 b = blockIdx.x;
    const u32 tid = threadIdx.x;
    const u32 r0 = abase[b] / T9_MSB_TILE;
    const u32 rows = (bucket_n[b] + T9_MSB_TILE - 1) / T9_MSB_TILE;
    u32 total = 0;
    for (u32 r = 0; r < rows; ++r)
        total += hist[(u64)(r0 + r) * T9_RADIX + tid];
    /* exclusive scan of totals across digits */
    __shared__ u32 s[T9_RADIX];
    s[tid] = total;
    __syncthreads();
    for (int off = 1; off < T9_RADIX; off <<= 1) {
        u32 y = (tid >= (u32)off) ? s[tid - off] : 0;
        __syncthreads();
        s[tid] += y;
        __syncthreads();
    }
    const u32 excl = s[tid] - total;
    const u32 gstart = true_base[b] + excl;
    sub_start[(u64)b * T9_RADIX + tid] = gstart;
    sub_n[(u64)b * T9_RADIX + tid] = total;
    u32 run = gstart;
    for (u32 r = 0; r < rows; ++r) {
        u32 v = hist[(u64)(r0 + r) * T9_RADIX + tid];
        hist[(u64)(r0 + r) * T9_RADIX + tid] = run;
        run += v;
    }
}

/* segmented 1024-thread wave-autonomous stable scatter (pass 2): like
 * k_scatter_wave512 but the tile's element range comes from the padded
 * bucket layout. */
template <bool HAS_VAL>
__global__ __launch_bounds__(1024, 4) void k_scatter_seg(
    const u64* __restrict__ in_keys, const u32* __restrict__ in_vals,
    const u32* __restrict__ abase, const u32* __restrict__ bucket_n,
    u64* __restrict__ out_keys, u32* __restrict__ out_vals,
    const u32* __restrict__ offs, u32 shift) {
    constexpr int TILE = T9_MSB_TILE;
    constexpr int NW = 16;
    constexpr int SUB = TILE / NW;
    constexpr int GROUPS = SUB / 64;
    __shared__ u64 s_okeys[TILE];
    __shared__ u32 s_ovals[HAS_VAL ? TILE : 1];
    __shared__ u16 s_rank[TILE];
    __shared__ u8 s_digof[TILE];
    __shared__ u32 s_wcnt[NW * T9_RADIX];
    __shared__ u32 s_woff[NW * T9_RADIX];
    __shared__ u32 s_start[T9_RADIX];
    __shared__ u32 s_goff[T9_RADIX];
    __shared__ u32 s_abase[257];

    const u32 tid = threadIdx.x, wave = tid >> 6, lane = tid & 63;
    if (tid < 257) s_abase[tid] = abase[tid];
    __syncthreads();
    /* u64 guard before narrowing — see k_hist_seg */
    const u64 tbase64 = (u64)blockIdx.x * TILE;
    if (tbase64 >= s_abase[256]) return;
    const u32 tbase = (u32)tbase64;
    const u32 b = bucket_of(s_abase, tbase);
    const u32 off_in_bucket = tbase - s_abase[b];
    const u32 bn = bucket_n[b];
    const u32 tn = (bn > off_in_bucket)
                       ? ((bn - off_in_bucket < (u32)TILE)
                              ? bn - off_in_bucket
                              : (u32)TILE)
                       : 0;

    if (tid < T9_RADIX)
        s_goff[tid] = offs[(u64)blockIdx.x * T9_RADIX + tid];
    for (u32 t = lane; t < T9_RADIX; t += 64) s_wcnt[wave * T9_RADIX + t] = 0;

    const u32 wbase = wave * SUB;
    for (int g = 0; g < GROUPS; ++g) {
        const u32 i = wbase + g * 64 + lane;
        const bool valid = i < tn;
        u32 d = 0;
        if (valid) d = (u32)(in_keys[tbase + i] >> shift) & 255u;
        u64 m = __ballot(valid);
        for (int bit = 0; bit < 8; ++bit) {
            u64 bb = __ballot((d >> bit) & 1u);
            m &= ((d >> bit) & 1u) ? bb : ~bb;
        }
        const u32 wr = (u32)__popcll(m & ((1ull << lane) - 1ull));
        const u32 before = valid ? s_wcnt[wave * T9_RADIX + d] : 0;
        if (valid) {
            s_rank[i] = (u16)(before + wr);
            if (wr == 0)
                s_wcnt[wave * T9_RADIX + d] = before + (u32)__popcll(m);
        }
    }
    __syncthreads();

    if (tid < T9_RADIX) {
        u32 run = 0;
        for (int w = 0; w < NW; ++w) {
            s_woff[w * T9_RADIX + tid] = run;   /* wave-relative for now */
            run += s_wcnt[w * T9_RADIX + tid];
        }
        s_start[tid] = run;
    }
    __syncthreads();
    t9_scan256_onewave(s_start, tid);   /* s_start: totals -> exclusive */
    __syncthreads();
    if (tid < T9_RADIX) {
        const u32 excl = s_start[tid];
        for (int w = 0; w < NW; ++w) s_woff[w * T9_RADIX + tid] += excl;
    }
    __syncthreads();

    for (int g = 0; g < GROUPS; ++g) {
        const u32 i = wbase + g * 64 + lane;
        if (i < tn) {
            const u64 k = in_keys[tbase + i];
            const u32 d = (u32)(k >> shift) & 255u;
            const u32 pos = s_woff[wave * T9_RADIX + d] + s_rank[i];
            s_okeys[pos] = k;
            if (HAS_VAL) s_ovals[pos] = in_vals[tbase + i];
            s_digof[pos] = (u8)d;
        }
    }
    __syncthreads();

    constexpr int CHUNKS = TILE / 1024;
    for (int c = 0; c < CHUNKS; ++c) {
        const u32 j = c * 1024 + tid;
        if (j < tn) {
            const u32 d = s_digof[j];
            const u64 gpos = (u64)s_goff[d] + (j - s_start[d]);
            out_keys[gpos] = s_okeys[j];
            if (HAS_VAL) out_vals[gpos] = s_ovals[j];
        }
    }
}

/* nb[i] = blocks needed by sub-bucket i at T9_MSB_TILE granularity
 * (input to the exclusive scan that builds the pass-3 block map) */
__global__ __launch_bounds__(256) void k_nb_of(const u32* __restrict__ sub_n,
                                               u32 nsub,
                                               u32* __restrict__ nb) {
    const u32 i = blockIdx.x * 256 + threadIdx.x;
    if (i < nsub) nb[i] = (sub_n[i] + T9_MSB_TILE - 1) / T9_MSB_TILE;
}

/* find the segment owning block b via the exclusive blocks-prefix bp */
__device__ inline u32 seg_of_block(const u32* __restrict__ bp, u32 nsub,
                                   u32 b) {
    u32 lo = 0, hi = nsub - 1;
    while (lo < hi) {
        u32 mid = (lo + hi + 1) >> 1;
        if (bp[mid] <= b) lo = mid; else hi = mid - 1;
    }
    return lo;
}

/* pass-3 histogram: tiles mapped to (b7,b6) sub-buckets via bp */
__global__ __launch_bounds__(256) void k_hist_seg3(
    const u64* __restrict__ keys, const u32* __restrict__ bp, u32 nsub,
    const u64* __restrict__ d_total, const u32* __restrict__ sub_start,
    const u32* __restrict__ sub_n, u32 shift, u32* __restrict__ hist) {
    __shared__ u32 s_cnt[T9_RADIX];
    const u32 tid = threadIdx.x;
    s_cnt[tid] = 0;
    __syncthreads();
    const u32 b = blockIdx.x;
    if (b < (u32)*d_total) {
        const u32 s = seg_of_block(bp, nsub, b);
        const u32 toff = (b - bp[s]) * T9_MSB_TILE;
        const u32 base = sub_start[s] + toff;
        const u32 rem = sub_n[s] - toff;
        const u32 tn = (rem < (u32)T9_MSB_TILE) ? rem : (u32)T9_MSB_TILE;
        const u32 lane = tid & 63;
        for (u32 i0 = 0; i0 < (u32)T9_MSB_TILE; i0 += 256) {
            const u32 i = i0 + tid;
            const bool valid = i < tn;
            const u32 d =
                valid ? (u32)(keys[(u64)base + i] >> shift) & 255u : 0;
            t9_hist_ballot_add<8>(s_cnt, d, valid, lane);
        }
    }
    __syncthreads();
    hist[(u64)blockIdx.x * T9_RADIX + tid] = s_cnt[tid];
}

/* pass-3 per-sub2 segmented scan -> final offsets + sub3 index */
__global__ __launch_bounds__(256) void k_seg_scan3(
    u32* __restrict__ hist, const u32* __restrict__ bp, u32 nsub,
    const u32* __restrict__ sub_start, const u32* __restrict__ sub_n,
    u32* __restrict__ sub3_start, u32* __restrict__ sub3_n) {
    const u32 s = blockIdx.x;
    const u32 tid = threadIdx.x;
    const u32 r0 = bp[s];
    const u32 rows = (sub_n[s] + T9_MSB_TILE - 1) / T9_MSB_TILE;
    u32 total = 0;
    for (u32 r = 0; r < rows; ++r)
        total += hist[(u64)(r0 + r) * T9_RADIX + tid];
    __shared__ u32 sh[T9_RADIX];
    sh[tid] = total;
    __syncthreads();
    for (int off = 1; off < T9_RADIX; off <<= 1) {
        u32 y = (tid >= (u32)off) ? sh[tid - off] : 0;
        __syncthreads();
        sh[tid] += y;
        __syncthreads();
    }
    const u32 excl = sh[tid] - total;
    const u32 gstart = sub_start[s] + excl;
    sub3_start[(u64)s * T9_RADIX + tid] = gstart;
    sub3_n[(u64)s * T9_RADIX + tid] = total;
    u32 run = gstart;
    for (u32 r = 0; r < rows; ++r) {
        u32 v = hist[(u64)(r0 + r) * T9_RADIX + tid];
        hist[(u64)(r0 + r) * T9_RADIX + tid] = run;
        run += v;
    }
}

/* pass-3 scatter (1024 threads, wave-autonomous, bp-mapped tiles) */
template <bool HAS_VAL>
__global__ __launch_bounds__(1024, 4) void k_scatter_seg3(
    const u64* __restrict__ in_keys, const u32* __restrict__ in_vals,
    const u32* __restrict__ bp, u32 nsub, const u64* __restrict__ d_total,
    const u32* __restrict__ sub_start, const u32* __restrict__ sub_n,
    u64* __restrict__ out_keys, u32* __restrict__ out_vals,
    const u32* __restrict__ offs, u32 shift) {
    constexpr int TILE = T9_MSB_TILE;
    constexpr int NW = 16;
    constexpr int SUB = TILE / NW;
    constexpr int GROUPS = SUB / 64;
    __shared__ u64 s_okeys[TILE];
    __shared__ u32 s_ovals[HAS_VAL ? TILE : 1];
    __shared__ u16 s_rank[TILE];
    __shared__ u8 s_digof[TILE];
    __shared__ u32 s_wcnt[NW * T9_RADIX];
    __shared__ u32 s_woff[NW * T9_RADIX];
    __shared__ u32 s_start[T9_RADIX];
    __shared__ u32 s_goff[T9_RADIX];
    __shared__ u32 s_range[2];   /* base, tn */

    const u32 tid = threadIdx.x, wave = tid >> 6, lane = tid & 63;
    const u32 b = blockIdx.x;
    if (b >= (u32)*d_total) return;
    if (tid == 0) {
        const u32 s = seg_of_block(bp, nsub, b);
        const u32 toff = (b - bp[s]) * TILE;
        s_range[0] = sub_start[s] + toff;
        const u32 rem = sub_n[s] - toff;
        s_range[1] = (rem < (u32)TILE) ? rem : (u32)TILE;
    }
    if (tid < T9_RADIX)
        s_goff[tid] = offs[(u64)b * T9_RADIX + tid];
    for (u32 t = lane; t < T9_RADIX; t += 64) s_wcnt[wave * T9_RADIX + t] = 0;
    __syncthreads();
    const u32 base = s_range[0], tn = s_range[1];

    const u32 wbase = wave * SUB;
    for (int g = 0; g < GROUPS; ++g) {
        const u32 i = wbase + g * 64 + lane;
        const bool valid = i < tn;
        u32 d = 0;
        if (valid) d = (u32)(in_keys[(u64)base + i] >> shift) & 255u;
        u64 m = __ballot(valid);
        for (int bit = 0; bit < 8; ++bit) {
            u64 bb = __ballot((d >> bit) & 1u);
            m &= ((d >> bit) & 1u) ? bb : ~bb;
        }
        const u32 wr = (u32)__popcll(m & ((1ull << lane) - 1ull));
        const u32 before = valid ? s_wcnt[wave * T9_RADIX + d] : 0;
        if (valid) {
            s_rank[i] = (u16)(before + wr);
            if (wr == 0)
                s_wcnt[wave * T9_RADIX + d] = before + (u32)__popcll(m);
        }
    }
    __syncthreads();

    if (tid < T9_RADIX) {
        u32 run = 0;
        for (int w = 0; w < NW; ++w) {
            s_woff[w * T9_RADIX + tid] = run;
            run += s_wcnt[w * T9_RADIX + tid];
        }
        s_start[tid] = run;
    }
    __syncthreads();
    t9_scan256_onewave(s_start, tid);
    __syncthreads();
    if (tid < T9_RADIX) {
        const u32 excl = s_start[tid];
        for (int w = 0; w < NW; ++w) s_woff[w * T9_RADIX + tid] += excl;
    }
    __syncthreads();

    for (int g = 0; g < GROUPS; ++g) {
        const u32 i = wbase + g * 64 + lane;
        if (i < tn) {
            const u64 k = in_keys[(u64)base + i];
            const u32 d = (u32)(k >> shift) & 255u;
            const u32 pos = s_woff[wave * T9_RADIX + d] + s_rank[i];
            s_okeys[pos] = k;
            if (HAS_VAL) s_ovals[pos] = in_vals[(u64)base + i];
            s_digof[pos] = (u8)d;
        }
    }
    __syncthreads();

    constexpr int CHUNKS = TILE / 1024;
    for (int c = 0; c < CHUNKS; ++c) {
        const u32 j = c * 1024 + tid;
        if (j < tn) {
            const u32 d = s_digof[j];
            const u64 gpos = (u64)s_goff[d] + (j - s_start[d]);
            out_keys[gpos] = s_okeys[j];
            if (HAS_VAL) out_vals[gpos] = s_ovals[j];
        }
    }
}

/* Greedy span packing: pack consecutive (b7,b6) sub-buckets into spans
 * of <= SPANMAX elements. A span may cover several sub-buckets because
 * their top-16 key bits are distinct and ordered, so sorting the span by
 * the FULL 64-bit key both preserves the bucket grouping and sorts within
 * — at the cost of 8 LDS passes instead of 6. One 256-thread block;
 * thread t packs the 256 sub-buckets of top-byte bucket t (buckets are
 * independent, so spans never cross a b7 boundary). A sub-bucket larger
 * than SPANMAX becomes its own (oversize) span, handled by the ranged-LSD
 * fallback. */
__global__ __launch_bounds__(256) void k_span_pack(
    const u32* __restrict__ sub_start, const u32* __restrict__ sub_n,
    u32 spanmax, u32* __restrict__ span_count,
    u32* __restrict__ span_start, u32* __restrict__ span_len) {
    const u32 t = blockIdx.x * 256 + threadIdx.x;
    /* walk 1: count this walker's spans; one atomicAdd claims the range
     * (a per-span add on one counter serializes at 3-level span counts) */
    u32 nspans = 0, cur_len = 0;
    for (u32 j = 0; j < 256; ++j) {
        const u32 ns = sub_n[t * 256 + j];
        if (ns == 0) continue;
        if (cur_len == 0 || cur_len + ns > spanmax) {
            if (cur_len) ++nspans;
            cur_len = ns;
        }
        else {
            cur_len += ns;
        }
    }
    if (cur_len) ++nspans;
    if (nspans == 0) return;
    u32 slot = atomicAdd(span_count, nspans);
    /* walk 2: emit */
    u32 cur_start = 0;
    cur_len = 0;
    for (u32 j = 0; j < 256; ++j) {
        const u32 i = t * 256 + j;
        const u32 ns = sub_n[i];
        if (ns == 0) continue;
        const u32 st = sub_start[i];
        if (cur_len == 0) {
            cur_start = st;
            cur_len = ns;
        }
        else if (cur_len + ns <= spanmax) {
            cur_len += ns;
        }
        else {
            span_start[slot] = cur_start;
            span_len[slot] = cur_len;
            ++slot;
            cur_start = st;
            cur_len = ns;
        }
    }
    span_start[slot] = cur_start;
    span_len[slot] = cur_len;
}

/* differ[0] = 1 if any keys[i] != keys[0] over the range — guards the
 * oversize-sub-bucket LSD fallback: an all-equal oversize sub-bucket
 * (the degenerate-key adversarial inputs) is already in stable order
 * and must NOT pay 8 LSD passes (measured ~10 ms on the all-identical
 * 10 GiB tie bench before this check). */
__global__ __launch_bounds__(256) void k_range_differ(
    const u64* __restrict__ keys, u64 n, u32* __restrict__ differ) {
    __shared__ u32 s;
    if (threadIdx.x == 0) s = 0;
    __syncthreads();
    const u64 v0 = keys[0];
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < n; i += stride)
        if (keys[i] != v0) { s = 1; break; }
    __syncthreads();
    if (threadIdx.x == 0 && s) atomicExch(differ, 1u);
}

/* host: does the key range hold >1 distinct value? (scratch = any device
 * u32) */
static bool range_differs(const u64* d_keys, u64 cnt, u32* d_scratch,
                          hipStream_t s) {
    if (cnt < 2) return false;
    if (hipMemsetAsync(d_scratch, 0, 4, s) != hipSuccess) return true;
    u64 want = t9_ceil_div(cnt, 256);
    hipLaunchKernelGGL(k_range_differ,
                       dim3((u32)((want < 4096) ? want : 4096)), dim3(256),
                       0, s, d_keys, cnt, d_scratch);
    u32 f = 1;
    if (hipMemcpyAsync(&f, d_scratch, 4, hipMemcpyDeviceToHost, s) !=
        hipSuccess)
        return true;
    if (hipStreamSynchronize(s) != hipSuccess) return true;
    return f != 0;
}

/* sub-bucket stats: info[0] = max size (atomicMax), info[1] = count of
 * sub-buckets above hardmax, list = their indices */
__global__ __launch_bounds__(256) void k_subinfo(
    const u32* __restrict__ sub_n, u32 nsub, u32 hardmax, u32 list_cap,
    u32* __restrict__ info, u32* __restrict__ list) {
    __shared__ u32 s_max;
    if (threadIdx.x == 0) s_max = 0;
    __syncthreads();
    const u32 i = blockIdx.x * 256 + threadIdx.x;
    if (i < nsub) {
        const u32 v = sub_n[i];
        atomicMax(&s_max, v);   /* block-local; one global atomic below */
        if (v > hardmax) {
            u32 pos = atomicAdd(&info[1], 1u);
            if (pos < list_cap) list[pos] = i;
        }
    }
    __syncthreads();
    if (threadIdx.x == 0 && s_max) atomicMax(&info[0], s_max);
}

/* wave16 blocksort is the default for sub-buckets <= 1024 (measured
 * 2.44 vs 2.64 ms radix on the bench workload); T9_LDS_WAVE16=0
 * restores the 6-pass radix LDS sort. */
static bool t9i_lds_wave16() {
    static const bool on = [] {
        const char* e = getenv("T9_LDS_WAVE16");
        return !(e && e[0] == '0');
    }();
    return on;
}

/* largest sub-bucket size routed to wave16 (tiers 1024/2048/4096);
 * measured tier by tier before raising the default */
static u32 t9i_wave16_max() {
    static const u32 v = [] {
        const char* e = getenv("T9_WAVE16_MAX");
        return e ? (u32)atoi(e) : 1024u;
    }();
    return v;
}

/* span wave16 is default-on (68.3 vs 69.5 ms at 2^30): spans are packed
 * nearly full so the fixed-size network wastes no padding work.
 * T9_SPAN_WAVE16=0 restores the 6-pass radix span sort. */
static bool t9i_span_wave16() {
    static const bool on = [] {
        const char* e = getenv("T9_SPAN_WAVE16");
        return !(e && e[0] == '0');
    }();
    return on;
}

template <bool HAS_VAL>
static void t9i_launch_wave16_sub(u32 grid, u32 maxsub, hipStream_t s,
                                  u64* d_keys, u32* d_vals,
                                  const u32* sub_start, const u32* sub_n) {
    if (maxsub <= 1024)
        hipLaunchKernelGGL((k_wave16_sort_sub<1024, 64, HAS_VAL>),
                           dim3(grid), dim3(64), 0, s, d_keys, d_vals,
                           sub_start, sub_n);
    else if (maxsub <= 2048)
        hipLaunchKernelGGL((k_wave16_sort_sub<2048, 128, HAS_VAL>),
                           dim3(grid), dim3(128), 0, s, d_keys, d_vals,
                           sub_start, sub_n);
    else
        hipLaunchKernelGGL((k_wave16_sort_sub<4096, 256, HAS_VAL>),
                           dim3(grid), dim3(256), 0, s, d_keys, d_vals,
                           sub_start, sub_n);
}

static bool t9i_lds_bitonic() {
    static const bool on = [] {
        const char* e = getenv("T9_LDS_BITONIC");
        return e && e[0] == '1';
    }();
    return on;
}

template <bool HAS_VAL>
static void t9i_launch_bitonic_sub(u32 grid, u32 maxsub, hipStream_t s,
                                   u64* d_keys, u32* d_vals,
                                   const u32* sub_start, const u32* sub_n) {
    if (maxsub <= 1024)
        hipLaunchKernelGGL((k_bitonic_sort_sub<1024, 256, HAS_VAL>),
                           dim3(grid), dim3(256), 0, s, d_keys, d_vals,
                           sub_start, sub_n);
    else if (maxsub <= 2048)
        hipLaunchKernelGGL((k_bitonic_sort_sub<2048, 512, HAS_VAL>),
                           dim3(grid), dim3(512), 0, s, d_keys, d_vals,
                           sub_start, sub_n);
    else
        hipLaunchKernelGGL((k_bitonic_sort_sub<4096, 1024, HAS_VAL>),
                           dim3(grid), dim3(1024), 0, s, d_keys, d_vals,
                           sub_start, sub_n);
}

/* level 3: sort one sub-bucket (<= SUBMAX pairs) in LDS over the low
 * 48 key bits — 6 stable ballot-ranked passes, then write back.
 * <2048,512>: 70 KB LDS, 2 blocks/CU (cross-block overlap hides the
 * pass barriers); <4096,1024>: 137 KB, 1 block/CU — chosen by the host
 * from the measured max sub-bucket size. Sub-buckets with all-equal
 * low-48 bits are skipped (already in stable order, in place). */
template <int SUBMAX, int BLOCK, bool HAS_VAL>
__global__ __launch_bounds__(BLOCK, 4) void k_lds_sort_sub(
    u64* __restrict__ keys, u32* __restrict__ vals,
    const u32* __restrict__ sub_start, const u32* __restrict__ sub_n) {
    constexpr int NW = BLOCK / 64;
    constexpr int SUBQ = SUBMAX / NW;        /* 256 */
    constexpr int GROUPS = SUBQ / 64;        /* 4 */
    __shared__ u64 s_k[2][SUBMAX];
    __shared__ u32 s_v[2][HAS_VAL ? SUBMAX : 1];
    __shared__ u16 s_rank[SUBMAX];
    __shared__ u32 s_wcnt[NW * T9_RADIX];
    __shared__ u32 s_woff[NW * T9_RADIX];
    __shared__ u32 s_start[T9_RADIX];
    __shared__ u32 s_differ;

    const u32 sb = blockIdx.x;
    const u32 ns = sub_n[sb];
    if (ns <= 1 || ns > (u32)SUBMAX) return;
    const u32 gbase = sub_start[sb];
    const u32 tid = threadIdx.x, wave = tid >> 6, lane = tid & 63;

    if (tid == 0) s_differ = 0;
    __syncthreads();
    /* load + equal-check over the low 48 bits */
    const u64 mask48 = 0x0000FFFFFFFFFFFFull;
    u64 k0ref = keys[gbase] & mask48;
    for (u32 i = tid; i < ns; i += BLOCK) {
        u64 k = keys[gbase + i];
        s_k[0][i] = k;
        if (HAS_VAL) s_v[0][i] = vals[gbase + i];
        if ((k & mask48) != k0ref) s_differ = 1;
    }
    __syncthreads();
    if (!s_differ) return;   /* stable order already — nothing to do */

    int cur = 0;
    for (int pass = 0; pass < 6; ++pass) {
        const u32 shift = pass * 8;
        for (u32 t = lane; t < T9_RADIX; t += 64)
            s_wcnt[wave * T9_RADIX + t] = 0;
        __syncthreads();
        const u32 wbase = wave * SUBQ;
        for (int g = 0; g < GROUPS; ++g) {
            const u32 i = wbase + g * 64 + lane;
            const bool valid = i < ns;
            u32 d = 0;
            if (valid) d = (u32)(s_k[cur][i] >> shift) & 255u;
            u64 m = __ballot(valid);
            for (int bit = 0; bit < 8; ++bit) {
                u64 bb = __ballot((d >> bit) & 1u);
                m &= ((d >> bit) & 1u) ? bb : ~bb;
            }
            const u32 wr = (u32)__popcll(m & ((1ull << lane) - 1ull));
            const u32 before = valid ? s_wcnt[wave * T9_RADIX + d] : 0;
            if (valid) {
                s_rank[i] = (u16)(before + wr);
                if (wr == 0)
                    s_wcnt[wave * T9_RADIX + d] =
                        before + (u32)__popcll(m);
            }
        }
        __syncthreads();
        if (tid < T9_RADIX) {
            u32 run = 0;
            for (int w = 0; w < NW; ++w) {
                s_woff[w * T9_RADIX + tid] = run;   /* wave-relative for now */
                run += s_wcnt[w * T9_RADIX + tid];
            }
            s_start[tid] = run;
        }
        __syncthreads();
        t9_scan256_onewave(s_start, tid);   /* s_start: totals -> exclusive */
        __syncthreads();
        if (tid < T9_RADIX) {
            const u32 excl = s_start[tid];
            for (int w = 0; w < NW; ++w) s_woff[w * T9_RADIX + tid] += excl;
        }
        __syncthreads();
        for (int g = 0; g < GROUPS; ++g) {
            const u32 i = wbase + g * 64 + lane;
            if (i < ns) {
                const u64 k = s_k[cur][i];
                const u32 d = (u32)(k >> shift) & 255u;
                const u32 pos = s_woff[wave * T9_RADIX + d] + s_rank[i];
                s_k[cur ^ 1][pos] = k;
                if (HAS_VAL) s_v[cur ^ 1][pos] = s_v[cur][i];
            }
        }
        __syncthreads();
        cur ^= 1;
    }
    for (u32 i = tid; i < ns; i += BLOCK) {
        keys[gbase + i] = s_k[cur][i];
        if (HAS_VAL) vals[gbase + i] = s_v[cur][i];
    }
}

/* ------------------------------------------------------------------ *
 * host
 * ------------------------------------------------------------------ */

/* span LDS sort: up to SPANMAX pairs, 8 stable ballot-ranked passes over
 * the FULL 64-bit key (spans cover several top-16 groups). in == out is
 * allowed (in-place); reads go to LDS before any write-back. */
template <int SPANMAX, int BLOCK, bool HAS_VAL, int PASSES>
__global__ __launch_bounds__(BLOCK, 4) void k_lds_sort_span(
    const u64* __restrict__ keys_in, const u32* __restrict__ vals_in,
    u64* __restrict__ keys_out, u32* __restrict__ vals_out,
    const u32* __restrict__ span_start, const u32* __restrict__ span_len) {
    constexpr int NW = BLOCK / 64;
    constexpr int SUBQ = SPANMAX / NW;
    constexpr int GROUPS = SUBQ / 64;
    __shared__ u64 s_k[2][SPANMAX];
    __shared__ u32 s_v[2][HAS_VAL ? SPANMAX : 1];
    __shared__ u16 s_rank[SPANMAX];
    __shared__ u32 s_wcnt[NW * T9_RADIX];
    __shared__ u32 s_woff[NW * T9_RADIX];
    __shared__ u32 s_start[T9_RADIX];
    __shared__ u32 s_differ;

    const u32 sb = blockIdx.x;
    const u32 ns = span_len[sb];
    if (ns == 0) return;
    const u32 gbase = span_start[sb];
    const u32 tid = threadIdx.x, wave = tid >> 6, lane = tid & 63;

    if (ns > (u32)SPANMAX) {
        /* oversize span (single huge sub-bucket): copy through when
         * out-of-place; the ranged-LSD fallback re-sorts it afterwards */
        if (keys_out != keys_in)
            for (u32 i = tid; i < ns; i += BLOCK) {
                keys_out[gbase + i] = keys_in[gbase + i];
                if (HAS_VAL) vals_out[gbase + i] = vals_in[gbase + i];
            }
        return;
    }

    if (tid == 0) s_differ = 0;
    __syncthreads();
    const u64 k0ref = keys_in[gbase];
    for (u32 i = tid; i < ns; i += BLOCK) {
        u64 k = keys_in[gbase + i];
        s_k[0][i] = k;
        if (HAS_VAL) s_v[0][i] = vals_in[gbase + i];
        if (k != k0ref) s_differ = 1;
    }
    __syncthreads();
    if (!s_differ) {
        if (keys_out != keys_in)
            for (u32 i = tid; i < ns; i += BLOCK) {
                keys_out[gbase + i] = s_k[0][i];
                if (HAS_VAL) vals_out[gbase + i] = s_v[0][i];
            }
        return;
    }

    int cur = 0;
    for (int pass = 0; pass < PASSES; ++pass) {
        const u32 shift = pass * 8;
        for (u32 t = lane; t < T9_RADIX; t += 64)
            s_wcnt[wave * T9_RADIX + t] = 0;
        __syncthreads();
        const u32 wbase = wave * SUBQ;
        for (int g = 0; g < GROUPS; ++g) {
            const u32 i = wbase + g * 64 + lane;
            const bool valid = i < ns;
            u32 d = 0;
            if (valid) d = (u32)(s_k[cur][i] >> shift) & 255u;
            u64 m = __ballot(valid);
            for (int bit = 0; bit < 8; ++bit) {
                u64 bb = __ballot((d >> bit) & 1u);
                m &= ((d >> bit) & 1u) ? bb : ~bb;
            }
            const u32 wr = (u32)__popcll(m & ((1ull << lane) - 1ull));
            const u32 before = valid ? s_wcnt[wave * T9_RADIX + d] : 0;
            if (valid) {
                s_rank[i] = (u16)(before + wr);
                if (wr == 0)
                    s_wcnt[wave * T9_RADIX + d] = before + (u32)__popcll(m);
            }
        }
        __syncthreads();
        if (tid < T9_RADIX) {
            u32 run = 0;
            for (int w = 0; w < NW; ++w) {
                s_woff[w * T9_RADIX + tid] = run;
                run += s_wcnt[w * T9_RADIX + tid];
            }
            s_start[tid] = run;
        }
        __syncthreads();
        t9_scan256_onewave(s_start, tid);
        __syncthreads();
        if (tid < T9_RADIX) {
            const u32 excl = s_start[tid];
            for (int w = 0; w < NW; ++w) s_woff[w * T9_RADIX + tid] += excl;
        }
        __syncthreads();
        for (int g = 0; g < GROUPS; ++g) {
            const u32 i = wbase + g * 64 + lane;
            if (i < ns) {
                const u64 k = s_k[cur][i];
                const u32 d = (u32)(k >> shift) & 255u;
                const u32 pos = s_woff[wave * T9_RADIX + d] + s_rank[i];
                s_k[cur ^ 1][pos] = k;
                if (HAS_VAL) s_v[cur ^ 1][pos] = s_v[cur][i];
            }
        }
        __syncthreads();
        cur ^= 1;
    }
    for (u32 i = tid; i < ns; i += BLOCK) {
        keys_out[gbase + i] = s_k[cur][i];
        if (HAS_VAL) vals_out[gbase + i] = s_v[cur][i];
    }
}

namespace {
struct MsbWs {
    u64* alt_k;
    u32* alt_v;
    u32* hist;
    u32* chunkpart;
    u32* digit_base;
    u32* true_base;
    u32* abase;
    u32* bucket_n;
    u32* sub_start;
    u32* sub_n;
    u32* ovr;        /* [0] = max or count, [1] = count, [2..] = list */
    u32* span;
    /* 3-level extras (allocated only for n >= T9_L3_MIN) */
    u32* bp;         /* NSUB blocks-prefix */
    u64* d_total;
    u32* sub3_start;
    u32* sub3_n;
    u32* span3;      /* [0]=count, starts, lens */
    u64 B1, B2max;
};

constexpr u64 NSUB3 = 256ull * 256 * 256;

constexpr u32 NSUB = 256 * 256;

MsbWs carve_msb(char* p, u64 n) {
    MsbWs w;
    const u64 npad = n + 256ull * T9_MSB_TILE;
    w.B1 = t9_ceil_div(n, T9_MSB_TILE);
    w.B2max = t9_ceil_div(n, T9_MSB_TILE) + 256;
    {
        const char* l3c = getenv("T9_MSB_LEVELS");
        if (n >= T9_L3_MIN || (l3c && atoi(l3c) == 3))
            w.B2max = t9_ceil_div(n, T9_MSB_TILE) + NSUB;
        if (pass2_bits() == 9) w.B2max *= 2;
    }
    w.alt_k = (u64*)p;
    p += t9_align256(npad * 8);
    w.alt_v = (u32*)p;
    p += t9_align256(npad * 4);
    w.hist = (u32*)p;
    p += t9_align256(w.B2max * T9_RADIX * 4);
    w.chunkpart = (u32*)p;
    p += t9_align256(t9_ceil_div(w.B2max, T9_SCAN_CHUNK) * T9_RADIX * 4);
    w.digit_base = (u32*)p;
    p += t9_align256(T9_RADIX * 4);
    w.true_base = (u32*)p;
    p += t9_align256(T9_RADIX * 4);
    w.abase = (u32*)p;
    p += t9_align256(257 * 4);
    w.bucket_n = (u32*)p;
    p += t9_align256(T9_RADIX * 4);
    w.sub_start = (u32*)p;
    p += t9_align256(NSUB9 * 4);
    w.sub_n = (u32*)p;
    p += t9_align256(NSUB9 * 4);
    w.ovr = (u32*)p;
    p += t9_align256((u64)(NSUB + 2) * 4);
    w.span = (u32*)p;   /* [0]=count, [1..CAP]=start, [1+CAP..]=len */
    p += t9_align256((u64)(2 * (NSUB + 64) + 1) * 4);
    const char* l3 = getenv("T9_MSB_LEVELS");
    if (n >= T9_L3_MIN || (l3 && atoi(l3) == 3)) {
        w.bp = (u32*)p;
        p += t9_align256((u64)(NSUB + 1) * 4);
        w.d_total = (u64*)p;
        p += 256;
        w.sub3_start = (u32*)p;
        p += t9_align256(NSUB3 * 4);
        w.sub3_n = (u32*)p;
        p += t9_align256(NSUB3 * 4);
        w.span3 = (u32*)p;
        p += t9_align256((2 * (NSUB3 + 64) + 1) * 4);
    }
    return w;
}

u64 msb_ws_bytes(u64 n) {
    const u64 npad = n + 256ull * T9_MSB_TILE;
    const char* l3e = getenv("T9_MSB_LEVELS");
    u64 B2max = t9_ceil_div(n, T9_MSB_TILE) + 256;
    if (n >= T9_L3_MIN || (l3e && atoi(l3e) == 3))
        B2max = t9_ceil_div(n, T9_MSB_TILE) + NSUB;   /* pass-3 rows */
    if (pass2_bits() == 9) B2max *= 2;   /* 512-bin hist rows */
    u64 l3 = 0;
    if (n >= T9_L3_MIN || (l3e && atoi(l3e) == 3))
        l3 = t9_align256((u64)(NSUB + 1) * 4) + 256 +
             2 * t9_align256(NSUB3 * 4) +
             t9_align256((2 * (NSUB3 + 64) + 1) * 4);
    return l3 + t9_align256(npad * 8) + t9_align256(npad * 4) +
           t9_align256(B2max * T9_RADIX * 4) +
           t9_align256(t9_ceil_div(B2max, T9_SCAN_CHUNK) * T9_RADIX * 4) +
           4 * t9_align256(T9_RADIX * 4) + t9_align256(257 * 4) +
           2 * t9_align256(NSUB9 * 4) +
           t9_align256((u64)(NSUB + 2) * 4) +
           t9_align256((u64)(2 * (NSUB + 64) + 1) * 4);
}
} // namespace

extern "C" u64 t9i_sort_pairs_msb_workspace(u64 n) { return msb_ws_bytes(n); }

template <bool HAS_VAL, int RW = 0>
static int sort_msb_impl(t9_context* ctx, const u64* pass1_src,
                         u64* d_keys, u32* d_vals, u64 n,
                         void* d_workspace, void* stream,
                         bool pre_hist = false) {
    hipStream_t s = (hipStream_t)stream;
    MsbWs w = carve_msb((char*)d_workspace, n);

    /* ---- pass 1: byte 7 into TILE-aligned buckets (input -> alt) ---- */
    {
        const u64 B = w.B1;
        const u64 Bc = t9_ceil_div(B, T9_SCAN_CHUNK);
        /* pre_hist: the caller already filled w.hist for shift 56 (the
         * fused extract pass, t9i_extract_hist) */
        if (!pre_hist)
            T9_PERF_WRAP(s, "hist_pairs",
                         hipLaunchKernelGGL((k_hist_msb<RW>), dim3((u32)B),
                                            dim3(256), 0, s, pass1_src, n,
                                            56, w.hist));
        hipLaunchKernelGGL(k_colsum, dim3((u32)Bc), dim3(256), 0, s, w.hist,
                           B, w.chunkpart);
        hipLaunchKernelGGL(k_chunkscan, dim3(1), dim3(256), 0, s,
                           w.chunkpart, Bc, w.digit_base);
        hipLaunchKernelGGL(k_align_bases, dim3(1), dim3(256), 0, s,
                           w.digit_base, n, w.true_base, w.abase,
                           w.bucket_n);
        hipLaunchKernelGGL(k_finaloffs, dim3((u32)Bc), dim3(256), 0, s,
                           w.hist, B, w.chunkpart, w.abase);
        T9_PERF_WRAP(
            s, "pair_scatter",
            hipLaunchKernelGGL(
                (k_scatter_wave512<T9_MSB_TILE, 1024, true, HAS_VAL, RW>),
                dim3((u32)B), dim3(1024), 0, s, pass1_src, d_vals, w.alt_k,
                w.alt_v, w.hist, n, 56));
        T9_LAUNCH_CHECK();
    }

    /* ---- optional 9-bit level 2 (experiment, T9_PASS2_BITS=9) ---- */
    {
        const char* l3x = getenv("T9_MSB_LEVELS");
        const bool blocked3 = (n >= T9_L3_MIN) ||
                              (l3x && atoi(l3x) == 3 && n >= (1ull << 14));
        if (pass2_bits() == 9 && !blocked3) {
            const u64 B2 = t9_ceil_div(n, T9_MSB_TILE) + 256;
            T9_PERF_WRAP(s, "hist_pairs",
                         t9i_launch_hist_seg9((u32)B2, stream, w.alt_k,
                                              w.abase, w.bucket_n,
                                              w.hist));
            t9i_launch_seg_scan9(256, stream, w.hist, w.abase, w.bucket_n,
                                 w.true_base, w.sub_start, w.sub_n);
            T9_PERF_WRAP(s, "pair_scatter",
                         t9i_launch_scatter_seg9((u32)B2, HAS_VAL, stream,
                                                 w.alt_k, w.alt_v, w.abase,
                                                 w.bucket_n, d_keys,
                                                 d_vals, w.hist));
            T9_LAUNCH_CHECK();
            HIP_TRY(hipMemsetAsync(w.ovr, 0, 8, s));
            hipLaunchKernelGGL(k_subinfo, dim3((u32)(NSUB9 / 256)),
                               dim3(256), 0, s, w.sub_n, (u32)NSUB9,
                               T9_SUBMAX, NSUB, w.ovr, w.ovr + 2);
            u32 info[2] = { 0, 0 };
            HIP_TRY(hipMemcpyAsync(info, w.ovr, 8, hipMemcpyDeviceToHost,
                                   s));
            HIP_TRY(hipStreamSynchronize(s));
            const u32 maxsub = info[0], novr9 = info[1];
            T9_PERF_WRAP(
                s, "lds_sort",
                if (t9i_lds_wave16() && maxsub <= t9i_wave16_max())
                    t9i_launch_wave16_sub<HAS_VAL>((u32)NSUB9, maxsub, s,
                                                   d_keys, d_vals,
                                                   w.sub_start, w.sub_n);
                else if (t9i_lds_bitonic())
                    t9i_launch_bitonic_sub<HAS_VAL>((u32)NSUB9, maxsub, s,
                                                    d_keys, d_vals,
                                                    w.sub_start, w.sub_n);
                else if (maxsub <= 1024)
                    hipLaunchKernelGGL(
                        (k_lds_sort_sub<1024, 256, HAS_VAL>), dim3(NSUB9),
                        dim3(256), 0, s, d_keys, d_vals, w.sub_start,
                        w.sub_n);
                else if (maxsub <= 2048)
                    hipLaunchKernelGGL(
                        (k_lds_sort_sub<2048, 512, HAS_VAL>), dim3(NSUB9),
                        dim3(512), 0, s, d_keys, d_vals, w.sub_start,
                        w.sub_n);
                else
                    hipLaunchKernelGGL(
                        (k_lds_sort_sub<T9_SUBMAX, 1024, HAS_VAL>),
                        dim3(NSUB9), dim3(1024), 0, s, d_keys, d_vals,
                        w.sub_start, w.sub_n));
            T9_LAUNCH_CHECK();
            if (novr9 == 0) return T9_OK;
            if (novr9 > 64)
                return HAS_VAL ? t9i_sort_pairs_lsd(ctx, d_keys, d_vals, n,
                                                    d_workspace, stream)
                               : t9i_sort_keys_lsd(ctx, d_keys, n,
                                                   d_workspace, stream);
            std::vector<u32> list9(novr9), st9(novr9), cn9(novr9);
            HIP_TRY(hipMemcpy(list9.data(), w.ovr + 2, novr9 * 4,
                              hipMemcpyDeviceToHost));
            for (u32 i = 0; i < novr9; ++i) {
                HIP_TRY(hipMemcpy(&st9[i], w.sub_start + list9[i], 4,
                                  hipMemcpyDeviceToHost));
                HIP_TRY(hipMemcpy(&cn9[i], w.sub_n + list9[i], 4,
                                  hipMemcpyDeviceToHost));
            }
            for (u32 i = 0; i < novr9; ++i) {
                if (!range_differs(d_keys + st9[i], cn9[i], w.ovr, s))
                    continue;   /* all-equal: stable order already */
                int rc = HAS_VAL
                             ? t9i_sort_pairs_lsd(ctx, d_keys + st9[i],
                                                  d_vals + st9[i], cn9[i],
                                                  d_workspace, stream)
                             : t9i_sort_keys_lsd(ctx, d_keys + st9[i],
                                                 cn9[i], d_workspace,
                                                 stream);
                if (rc) return rc;
            }
            return T9_OK;
        }
    }

    /* ---- pass 2: byte 6, segmented per bucket (alt -> input, compact
     * final positions by the top 16 bits) ---- */
    {
        const u64 B2 = w.B2max;   /* blocks beyond the padded end exit */
        T9_PERF_WRAP(s, "hist_pairs",
                     hipLaunchKernelGGL(k_hist_seg, dim3((u32)B2),
                                        dim3(256), 0, s, w.alt_k, w.abase,
                                        w.bucket_n, 48, w.hist));
        hipLaunchKernelGGL(k_seg_scan, dim3(256), dim3(256), 0, s, w.hist,
                           w.abase, w.bucket_n, w.true_base, w.sub_start,
                           w.sub_n);
        T9_PERF_WRAP(
            s, "pair_scatter",
            hipLaunchKernelGGL((k_scatter_seg<HAS_VAL>), dim3((u32)B2),
                               dim3(1024), 0, s, w.alt_k, w.alt_v, w.abase,
                               w.bucket_n, d_keys, d_vals, w.hist, 48));
        T9_LAUNCH_CHECK();
    }

    const char* l3e = getenv("T9_MSB_LEVELS");
    const bool use3 = (n >= T9_L3_MIN) ||
                      (l3e && atoi(l3e) == 3 && n >= (1ull << 14));
    if (use3) {
        /* ---- pass 3: byte 5, segmented per (b7,b6) sub-bucket
         * (d_keys -> alt), then level 4: span-packed LDS sort over the
         * (b7,b6,b5) sub-buckets (low 48 bits; spans stay inside one
         * (b7,b6) bucket), alt -> d_keys ---- */
        hipLaunchKernelGGL(k_nb_of, dim3(NSUB / 256), dim3(256), 0, s,
                           w.sub_n, NSUB, w.bp);
        hipLaunchKernelGGL(k_grp_scan, dim3(1), dim3(256), 0, s, w.bp,
                           (u64)NSUB, w.d_total);
        const u64 B3 = t9_ceil_div(n, T9_MSB_TILE) + NSUB;
        T9_PERF_WRAP(s, "hist_pairs",
                     hipLaunchKernelGGL(k_hist_seg3, dim3((u32)B3),
                                        dim3(256), 0, s, d_keys, w.bp,
                                        NSUB, w.d_total, w.sub_start,
                                        w.sub_n, 40, w.hist));
        hipLaunchKernelGGL(k_seg_scan3, dim3(NSUB), dim3(256), 0, s,
                           w.hist, w.bp, NSUB, w.sub_start, w.sub_n,
                           w.sub3_start, w.sub3_n);
        T9_PERF_WRAP(
            s, "pair_scatter",
            hipLaunchKernelGGL((k_scatter_seg3<HAS_VAL>), dim3((u32)B3),
                               dim3(1024), 0, s, d_keys, d_vals, w.bp,
                               NSUB, w.d_total, w.sub_start, w.sub_n,
                               w.alt_k, w.alt_v, w.hist, 40));
        T9_LAUNCH_CHECK();

        /* max (b7,b6,b5) group size decides the span geometry: tiny
         * groups (the common big-n case) pack into 2048-element spans at
         * 2 blocks/CU — same lever that won on the level-3 sub sort */
        HIP_TRY(hipMemsetAsync(w.ovr, 0, 8, s));
        hipLaunchKernelGGL(k_subinfo, dim3((u32)(NSUB3 / 256)), dim3(256),
                           0, s, w.sub3_n, (u32)NSUB3, T9_SUBMAX, NSUB,
                           w.ovr, w.ovr + 2);
        u32 mx[1] = { 0 };
        HIP_TRY(hipMemcpyAsync(mx, w.ovr, 4, hipMemcpyDeviceToHost, s));
        HIP_TRY(hipStreamSynchronize(s));
        const u32 spanmax = (mx[0] <= 2048) ? 2048 : T9_SUBMAX;

        const u64 CAP3 = NSUB3 + 64;
        u32* s3_count = w.span3;
        u32* s3_start = w.span3 + 1;
        u32* s3_len = w.span3 + 1 + CAP3;
        HIP_TRY(hipMemsetAsync(w.span3, 0, (2 * CAP3 + 1) * 4, s));
        hipLaunchKernelGGL(k_span_pack, dim3(NSUB / 256), dim3(256), 0, s,
                           w.sub3_start, w.sub3_n, spanmax, s3_count,
                           s3_start, s3_len);
        HIP_TRY(hipMemsetAsync(w.ovr, 0, 8, s));
        hipLaunchKernelGGL(k_subinfo, dim3((u32)(CAP3 / 256 + 1)),
                           dim3(256), 0, s, s3_len, (u32)CAP3, spanmax,
                           NSUB, w.ovr, w.ovr + 2);
        u32 hdr[2] = { 0, 0 };
        u32 nspan = 0;
        HIP_TRY(hipMemcpyAsync(&nspan, s3_count, 4,
                               hipMemcpyDeviceToHost, s));
        HIP_TRY(hipMemcpyAsync(hdr, w.ovr, 8, hipMemcpyDeviceToHost, s));
        HIP_TRY(hipStreamSynchronize(s));
        const u32 novr3 = hdr[1];
        T9_PERF_WRAP(
            s, "lds_sort",
            if (t9i_span_wave16() && spanmax == 2048)
                hipLaunchKernelGGL(
                    (k_wave16_sort_span<2048, 128, HAS_VAL>),
                    dim3(nspan ? nspan : 1), dim3(128), 0, s, w.alt_k,
                    w.alt_v, d_keys, d_vals, s3_start, s3_len);
            else if (t9i_span_wave16())
                hipLaunchKernelGGL(
                    (k_wave16_sort_span<4096, 256, HAS_VAL>),
                    dim3(nspan ? nspan : 1), dim3(256), 0, s, w.alt_k,
                    w.alt_v, d_keys, d_vals, s3_start, s3_len);
            else if (spanmax == 2048)
                hipLaunchKernelGGL(
                    (k_lds_sort_span<2048, 512, HAS_VAL, 6>),
                    dim3(nspan ? nspan : 1), dim3(512), 0, s, w.alt_k,
                    w.alt_v, d_keys, d_vals, s3_start, s3_len);
            else
                hipLaunchKernelGGL(
                    (k_lds_sort_span<T9_SUBMAX, 1024, HAS_VAL, 6>),
                    dim3(nspan ? nspan : 1), dim3(1024), 0, s, w.alt_k,
                    w.alt_v, d_keys, d_vals, s3_start, s3_len));
        T9_LAUNCH_CHECK();
        if (novr3 == 0) return T9_OK;
        if (novr3 > 64)
            return HAS_VAL ? t9i_sort_pairs_lsd(ctx, d_keys, d_vals, n,
                                                d_workspace, stream)
                           : t9i_sort_keys_lsd(ctx, d_keys, n, d_workspace,
                                               stream);
        HIP_TRY(hipStreamSynchronize(s));
        std::vector<u32> list3(novr3);
        HIP_TRY(hipMemcpy(list3.data(), w.ovr + 2, novr3 * 4,
                          hipMemcpyDeviceToHost));
        std::vector<u32> st3(novr3), cn3(novr3);
        for (u32 i = 0; i < novr3; ++i) {
            HIP_TRY(hipMemcpy(&st3[i], s3_start + list3[i], 4,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(&cn3[i], s3_len + list3[i], 4,
                              hipMemcpyDeviceToHost));
        }
        for (u32 i = 0; i < novr3; ++i) {
            if (!range_differs(d_keys + st3[i], cn3[i], w.ovr, s))
                continue;   /* all-equal: stable order already */
            int rc = HAS_VAL
                         ? t9i_sort_pairs_lsd(ctx, d_keys + st3[i],
                                              d_vals + st3[i], cn3[i],
                                              d_workspace, stream)
                         : t9i_sort_keys_lsd(ctx, d_keys + st3[i], cn3[i],
                                             d_workspace, stream);
            if (rc) return rc;
        }
        return T9_OK;
    }

    /* ---- level 3: in-LDS sort; spans pack consecutive sub-buckets to
     * full 4096-element blocks sorted by the complete 64-bit key
     * (T9_LDS_SPAN=0 falls back to one block per sub-bucket) ---- */
    /* measured: span mode 4.6 ms vs per-sub 3.0 ms at the 10 GiB bench
     * (8 full-key passes + 1 block/CU outweigh the 2.5x block-count
     * reduction) — kept opt-in for skew shapes where packing fills
     * blocks better */
    const char* se = getenv("T9_LDS_SPAN");
    const bool use_span = se && atoi(se) == 1;
    const u32* sel_start;
    const u32* sel_n;
    u32 novr = 0;
    if (use_span) {
        const u32 CAP = NSUB + 64;
        u32* span_count = w.span;
        u32* span_start = w.span + 1;
        u32* span_len = w.span + 1 + CAP;
        HIP_TRY(hipMemsetAsync(w.span, 0, (u64)(2 * CAP + 1) * 4, s));
        hipLaunchKernelGGL(k_span_pack, dim3(1), dim3(256), 0, s,
                           w.sub_start, w.sub_n, T9_SUBMAX, span_count,
                           span_start, span_len);
        HIP_TRY(hipMemsetAsync(w.ovr, 0, 8, s));
        hipLaunchKernelGGL(k_subinfo, dim3(CAP / 256 + 1), dim3(256), 0, s,
                           span_len, CAP, T9_SUBMAX, NSUB, w.ovr,
                           w.ovr + 2);
        u32 info[2] = { 0, 0 };
        HIP_TRY(hipMemcpyAsync(info, w.ovr, 8, hipMemcpyDeviceToHost, s));
        HIP_TRY(hipStreamSynchronize(s));
        novr = info[1];
        T9_PERF_WRAP(
            s, "lds_sort",
            hipLaunchKernelGGL(
                (k_lds_sort_span<T9_SUBMAX, 1024, HAS_VAL, 7>),
                dim3(CAP), dim3(1024), 0, s, d_keys, d_vals, d_keys,
                d_vals, span_start, span_len));
        T9_LAUNCH_CHECK();
        sel_start = span_start;
        sel_n = span_len;
    }
    else {
        HIP_TRY(hipMemsetAsync(w.ovr, 0, 8, s));
        hipLaunchKernelGGL(k_subinfo, dim3(NSUB / 256), dim3(256), 0, s,
                           w.sub_n, NSUB, T9_SUBMAX, NSUB, w.ovr,
                           w.ovr + 2);
        u32 info[2] = { 0, 0 };
        HIP_TRY(hipMemcpyAsync(info, w.ovr, 8, hipMemcpyDeviceToHost, s));
        HIP_TRY(hipStreamSynchronize(s));
        const u32 maxsub = info[0];
        novr = info[1];
        T9_PERF_WRAP(
            s, "lds_sort",
            if (t9i_lds_wave16() && maxsub <= t9i_wave16_max())
                t9i_launch_wave16_sub<HAS_VAL>((u32)NSUB, maxsub, s,
                                               d_keys, d_vals,
                                               w.sub_start, w.sub_n);
            else if (t9i_lds_bitonic())
                t9i_launch_bitonic_sub<HAS_VAL>((u32)NSUB, maxsub, s,
                                                d_keys, d_vals,
                                                w.sub_start, w.sub_n);
            else if (maxsub <= 2048)
                hipLaunchKernelGGL((k_lds_sort_sub<2048, 512, HAS_VAL>),
                                   dim3(NSUB), dim3(512), 0, s, d_keys,
                                   d_vals, w.sub_start, w.sub_n);
            else
                hipLaunchKernelGGL(
                    (k_lds_sort_sub<T9_SUBMAX, 1024, HAS_VAL>), dim3(NSUB),
                    dim3(1024), 0, s, d_keys, d_vals, w.sub_start,
                    w.sub_n));
        T9_LAUNCH_CHECK();
        sel_start = w.sub_start;
        sel_n = w.sub_n;
    }
    if (novr == 0) return T9_OK;

    if (novr > 64) {
        /* heavy skew: one full stable LSD sort of the current (already
         * stable-permuted) array — correct, at LSD speed */
        return HAS_VAL ? t9i_sort_pairs_lsd(ctx, d_keys, d_vals, n,
                                            d_workspace, stream)
                       : t9i_sort_keys_lsd(ctx, d_keys, n, d_workspace,
                                           stream);
    }
    /* few oversize sub-buckets: stable LSD on each range (shares the
     * workspace, so copy the metadata out first) */
    HIP_TRY(hipStreamSynchronize(s));
    std::vector<u32> list(novr);
    HIP_TRY(hipMemcpy(list.data(), w.ovr + 2, novr * 4,
                      hipMemcpyDeviceToHost));
    std::vector<u32> starts(novr), counts(novr);
    for (u32 i = 0; i < novr; ++i) {
        HIP_TRY(hipMemcpy(&starts[i], sel_start + list[i], 4,
                          hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(&counts[i], sel_n + list[i], 4,
                          hipMemcpyDeviceToHost));
    }
    for (u32 i = 0; i < novr; ++i) {
        if (!range_differs(d_keys + starts[i], counts[i], w.ovr, s))
            continue;   /* all-equal: stable order already */
        int rc = HAS_VAL
                     ? t9i_sort_pairs_lsd(ctx, d_keys + starts[i],
                                          d_vals + starts[i], counts[i],
                                          d_workspace, stream)
                     : t9i_sort_keys_lsd(ctx, d_keys + starts[i],
                                         counts[i], d_workspace, stream);
        if (rc) return rc;
    }
    return T9_OK;
}

extern "C" int t9i_sort_pairs_msb(t9_context* ctx, u64* d_keys,
                                  u32* d_vals, u64 n, void* d_workspace,
                                  void* stream) {
    return sort_msb_impl<true>(ctx, d_keys, d_keys, d_vals, n, d_workspace,
                               stream);
}

/* pre-hist variant: the caller (fused extract, t9_records.hip) already
 * wrote the pass-1 byte-7 histogram rows into the workspace slot
 * returned by t9i_msb_pass1_hist */
extern "C" int t9i_sort_pairs_msb_ph(t9_context* ctx, u64* d_keys,
                                     u32* d_vals, u64 n,
                                     void* d_workspace, void* stream) {
    return sort_msb_impl<true>(ctx, d_keys, d_keys, d_vals, n, d_workspace,
                               stream, true);
}

extern "C" u32* t9i_msb_pass1_hist(void* d_workspace, u64 n) {
    MsbWs w = carve_msb((char*)d_workspace, n);
    return w.hist;
}

extern "C" int t9i_sort_keys_msb(t9_context* ctx, u64* d_keys, u64 n,
                                 void* d_workspace, void* stream) {
    return sort_msb_impl<false>(ctx, d_keys, d_keys, nullptr, n,
                                d_workspace, stream);
}

/* fused-extract entry: records in, sorted (key, idx) pairs out — MSB
 * pass 1 reads the record bytes directly (digit = key byte, payload =
 * record index), so the separate extract pass and its packed-key round
 * trip disappear. rec_size/4 must be 25 (100 B) or 32 (128 B). */
extern "C" int t9i_sort_recs_msb(t9_context* ctx, const u8* d_recs,
                                 u32 rec_size, u64* d_keys, u32* d_vals,
                                 u64 n, void* d_workspace, void* stream) {
    const u32 rw = rec_size / 4;
    if (rw == 25)
        return sort_msb_impl<true, 25>(ctx, (const u64*)d_recs, d_keys,
                                       d_vals, n, d_workspace, stream);
    if (rw == 32)
        return sort_msb_impl<true, 32>(ctx, (const u64*)d_recs, d_keys,
                                       d_vals, n, d_workspace, stream);
    return T9_EINVAL;
}
