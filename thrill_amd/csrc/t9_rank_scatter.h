/* t9_rank_scatter.h — shared wave-autonomous stable scatter template
 * (included by t9_sort.hip and t9_sort_msb.hip). See t9_sort.hip header
 * comment for the design narrative. */
#pragma once
#include "t9_common.h"

/* exclusive scan of NDIG LDS entries by wave 0 alone (generalized form
 * below; the 256 wrapper keeps existing call sites) */
template <int NDIG>
__device__ inline void t9_scan_onewave(u32* s_vals, u32 tid) {
    if (tid < 64) {
        const u32 lane = tid;
        u32 carry = 0;
        for (int c = 0; c < NDIG / 64; ++c) {
            const u32 v0 = s_vals[c * 64 + lane];
            u32 v = v0;
            for (int off = 1; off < 64; off <<= 1) {
                u32 y = __shfl_up(v, off);
                if (lane >= (u32)off) v += y;
            }
            const u32 total = __shfl(v, 63);
            s_vals[c * 64 + lane] = (v - v0) + carry;
            carry += total;
        }
    }
}

/* exclusive scan of 256 LDS entries by wave 0 alone (4 shfl-scanned
 * 64-lane chunks with a running carry) — replaces the 256-thread
 * Hillis-Steele scan and its 16 block barriers with one. Caller barriers
 * before (values visible to wave 0) and after. */
__device__ inline void t9_scan256_onewave(u32* s_vals, u32 tid) {
    if (tid < 64) {
        const u32 lane = tid;
        u32 carry = 0;
        for (int c = 0; c < 4; ++c) {
            const u32 v0 = s_vals[c * 64 + lane];
            u32 v = v0;
            for (int off = 1; off < 64; off <<= 1) {
                u32 y = __shfl_up(v, off);
                if (lane >= (u32)off) v += y;
            }
            const u32 total = __shfl(v, 63);
            s_vals[c * 64 + lane] = (v - v0) + carry;
            carry += total;
        }
    }
}

/* ballot-combined LDS histogram add: lanes holding the same digit are
 * grouped by BITS ballots; the group leader issues ONE atomicAdd of the
 * group size. Same cost as a plain atomicAdd on uniform digits
 * (memory-bound either way) but immune to the degenerate all-equal-digit
 * case, where 64-way same-address LDS atomics serialize (measured
 * 16x on the segmented hist during the all-identical-records tie
 * bench). */
template <int BITS>
__device__ inline void t9_hist_ballot_add(u32* __restrict__ s_cnt, u32 d,
                                          bool valid, u32 lane) {
    u64 m = __ballot(valid);
    for (int b = 0; b < BITS; ++b) {
        const u64 bb = __ballot((d >> b) & 1u);
        m &= ((d >> b) & 1u) ? bb : ~bb;
    }
    if (valid && (u32)__popcll(m & ((1ull << lane) - 1ull)) == 0)
        atomicAdd(&s_cnt[d], (u32)__popcll(m));
}

/* 512-thread (8-wave) wave-autonomous scatter, always global re-read:
 * doubles waves/SIMD to 4 at the same LDS footprint — the PMC-measured
 * limiter of the 256-thread version was 81% SQ_WAIT_ANY at 2 waves/SIMD.
 * Digit recomputed from the L1/L2-resident re-read instead of an LDS
 * cache. */
/* REC_WORDS != 0: "fused extract" mode — in_keys is the packed record
 * array (REC_WORDS u32 words per record); the big-endian u64 key prefix
 * is built from the record bytes and the payload is the record index
 * (iota), eliminating the separate extract pass and the packed key-array
 * round trip for MSB pass 1. */
template <int TILE, int BLOCK, bool HAS_KEY, bool HAS_VAL,
          int REC_WORDS = 0>
__global__ __launch_bounds__(BLOCK, 4) void k_scatter_wave512(
    const u64* __restrict__ in_keys, const u32* __restrict__ in_vals,
    u64* __restrict__ out_keys, u32* __restrict__ out_vals,
    const u32* __restrict__ offs, u64 n, u32 shift) {
    constexpr int NW = BLOCK / 64;
    constexpr int SUB = TILE / NW;
    constexpr int GROUPS = SUB / 64;
    const u32* rec32 = (const u32*)in_keys;
    __shared__ u64 s_okeys[HAS_KEY ? TILE : 1];
    __shared__ u32 s_ovals[HAS_VAL ? TILE : 1];
    __shared__ u16 s_rank[TILE];
    __shared__ u8 s_digof[TILE];
    __shared__ u32 s_wcnt[NW * T9_RADIX];
    __shared__ u32 s_woff[NW * T9_RADIX];
    __shared__ u32 s_start[T9_RADIX];
    __shared__ u32 s_goff[T9_RADIX];

    const u32 tid = threadIdx.x, wave = tid >> 6, lane = tid & 63;
    const u64 base = (u64)blockIdx.x * TILE;
    const u32 tn = (u32)((n - base < (u64)TILE) ? (n - base) : (u64)TILE);

    if (tid < T9_RADIX)
        s_goff[tid] = offs[(u64)blockIdx.x * T9_RADIX + tid];
    for (u32 t = lane; t < T9_RADIX; t += 64) s_wcnt[wave * T9_RADIX + t] = 0;

    const u32 wbase = wave * SUB;
    for (int g = 0; g < GROUPS; ++g) {
        const u32 i = wbase + g * 64 + lane;
        const bool valid = i < tn;
        u32 d = 0;
        if (valid) {
            if (REC_WORDS)
                d = ((const u8*)rec32)[(base + i) * (u64)REC_WORDS * 4 +
                                       (7 - shift / 8)];
            else
                d = (u32)(in_keys[base + i] >> shift) & 255u;
        }
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

    /* combine (threads 0..255 own digit tid) */
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
            u64 k;
            if (REC_WORDS) {
                const u64 w0 = (u64)rec32[(base + i) * REC_WORDS];
                const u64 w1 = (u64)rec32[(base + i) * REC_WORDS + 1];
                k = ((u64)__builtin_bswap32((u32)w0) << 32) |
                    __builtin_bswap32((u32)w1);
            }
            else {
                k = HAS_KEY ? in_keys[base + i] : 0;
            }
            const u32 d = (u32)(k >> shift) & 255u;
            const u32 pos = s_woff[wave * T9_RADIX + d] + s_rank[i];
            if (HAS_KEY) s_okeys[pos] = k;
            if (HAS_VAL)
                s_ovals[pos] =
                    REC_WORDS ? (u32)(base + i) : in_vals[base + i];
            s_digof[pos] = (u8)d;
        }
    }
    __syncthreads();

    constexpr int CHUNKS = TILE / BLOCK;
    for (int c = 0; c < CHUNKS; ++c) {
        const u32 j = c * BLOCK + tid;
        if (j < tn) {
            const u32 d = s_digof[j];
            const u64 gpos = (u64)s_goff[d] + (j - s_start[d]);
            if (HAS_KEY) out_keys[gpos] = s_okeys[j];
            if (HAS_VAL) out_vals[gpos] = s_ovals[j];
        }
    }
}

