/* t9_bitonic.h — register/shfl bitonic alternative to k_lds_sort_sub
 * (included by t9_sort_msb.hip; selected with T9_LDS_BITONIC=1).
 *
 * Motivation (profiles/r01_pmc_wavecycle_decomposition.txt): the 6-pass
 * LDS counting sort is 58% wave-parked on its ~38 block barriers with
 * only 4% LDS-conflict stall, so the fix is structural: a bitonic
 * network runs 52 of its 55 compare-exchange stages entirely in
 * registers/crosslane (no barrier, no LDS), leaving only the j>=256
 * stages (3/6/10 for SUBMAX 1024/2048/4096) to stage through LDS.
 *
 * Stability: bitonic networks are not stable, so the network sorts the
 * composite c = (key & mask48) << 12 | pos (pos = load index < 4096).
 * Composites are unique, ties in low-48 resolve to input order — the
 * exact order the stable radix passes produce. The high 16 key bits are
 * shared by every element of a sub-bucket (byte-7 + level-2 digit), so
 * the key is reconstructed as shared_high | (c >> 12) and need not be
 * carried through the network.
 *
 * Element mapping: e = tid*4 + r (4 registers per thread, wave w owns
 * the contiguous run [w*256, (w+1)*256)). CE distance j=1,2: register
 * pair inside the lane; j=4..128: __shfl_xor at lane distance j/4;
 * j>=256: LDS exchange at wave distance j/256.
 * Padding: slots >= ns carry c = ~0 (sinks to the end, never written
 * back); pad-vs-pad CEs may duplicate pad payloads, which is harmless
 * because the composite multiset above ns stays all-max. */
#pragma once
#include "t9_common.h"

/* one compare-exchange against a crosslane partner: both sides compute;
 * take the partner's element iff (mine > partner) == (I am the lower
 * position in an ascending pair). Composites are unique so c == oc
 * cannot occur between live elements. */
__device__ inline void t9_ce_xor(u64& c, u32& v, u32 lanemask, bool lower,
                                 bool asc) {
    const u64 oc = __shfl_xor(c, lanemask);
    const u32 ov = __shfl_xor(v, lanemask);
    if ((c > oc) == (lower == asc)) { c = oc; v = ov; }
}

template <int SUBMAX, int BLOCK, bool HAS_VAL>
__global__ __launch_bounds__(BLOCK, 2) void k_bitonic_sort_sub(
    u64* __restrict__ keys, u32* __restrict__ vals,
    const u32* __restrict__ sub_start, const u32* __restrict__ sub_n) {
    constexpr int E = SUBMAX / BLOCK;        /* 4 */
    static_assert(E == 4, "mapping assumes 4 elements per thread");
    __shared__ u64 s_c[SUBMAX];
    __shared__ u32 s_v[HAS_VAL ? SUBMAX : 1];
    __shared__ u32 s_differ;

    const u32 sb = blockIdx.x;
    const u32 ns = sub_n[sb];
    if (ns <= 1 || ns > (u32)SUBMAX) return;
    const u32 gbase = sub_start[sb];
    const u32 tid = threadIdx.x, lane = tid & 63;

    if (tid == 0) s_differ = 0;
    __syncthreads();

    const u64 mask48 = 0x0000FFFFFFFFFFFFull;
    const u64 high16 = keys[gbase] & ~mask48;
    const u64 ref48 = keys[gbase] & mask48;

    /* load: 4 contiguous elements per thread -> composite registers */
    u64 c[E];
    u32 v[E];
    u32 differ = 0;
    const u32 e0 = tid * E;
#pragma unroll
    for (int r = 0; r < E; ++r) {
        const u32 i = e0 + r;
        if (i < ns) {
            const u64 k = keys[gbase + i];
            c[r] = ((k & mask48) << 12) | (u64)i;
            if (HAS_VAL) v[r] = vals[gbase + i];
            differ |= ((k & mask48) != ref48);
        } else {
            c[r] = ~0ull;
            if (HAS_VAL) v[r] = 0;
        }
    }
    if (differ) s_differ = 1;
    __syncthreads();
    if (!s_differ) return;   /* all-equal low-48: stable order is in place */

    /* bitonic network over SUBMAX elements — fully unrolled (lk/lj are
     * compile-time constants, so lane masks become immediates and the
     * per-stage branch selection disappears) */
    constexpr int LOG = (SUBMAX == 1024) ? 10 : (SUBMAX == 2048) ? 11 : 12;
#pragma unroll
    for (int lk = 1; lk <= LOG; ++lk) {
        const u32 k = 1u << lk;
#pragma unroll
        for (int lj = lk - 1; lj >= 0; --lj) {
            const u32 j = 1u << lj;
            if (j >= 256) {
                /* cross-wave: stage through LDS */
                __syncthreads();   /* WAR: prior reads of s_c complete */
#pragma unroll
                for (int r = 0; r < E; ++r) {
                    s_c[e0 + r] = c[r];
                    if (HAS_VAL) s_v[e0 + r] = v[r];
                }
                __syncthreads();
#pragma unroll
                for (int r = 0; r < E; ++r) {
                    const u32 e = e0 + r;
                    const u32 p = e ^ j;
                    const u64 oc = s_c[p];
                    const bool lower = (e & j) == 0;
                    const bool asc = (e & k) == 0;
                    if ((c[r] > oc) == (lower == asc)) {
                        c[r] = oc;
                        if (HAS_VAL) v[r] = s_v[p];
                    }
                }
            } else if (j >= 4) {
                /* crosslane within the wave: lane distance j/4 */
                const u32 lm = j >> 2;
#pragma unroll
                for (int r = 0; r < E; ++r) {
                    const u32 e = e0 + r;
                    t9_ce_xor(c[r], v[r], lm, (lane & lm) == 0,
                              (e & k) == 0);
                }
            } else {
                /* register pair inside the lane: r ^ j (j = 1 or 2) */
#pragma unroll
                for (int r = 0; r < E; ++r) {
                    const int q = r ^ (int)j;
                    if (q > r) {
                        const bool asc = ((e0 + r) & k) == 0;
                        if ((c[r] > c[q]) == asc) {
                            u64 tc = c[r]; c[r] = c[q]; c[q] = tc;
                            if (HAS_VAL) {
                                u32 tv = v[r]; v[r] = v[q]; v[q] = tv;
                            }
                        }
                    }
                }
            }
        }
    }

    /* write back: reconstruct keys from composites; element e0+r of the
     * sorted order lives in register r of thread tid */
#pragma unroll
    for (int r = 0; r < E; ++r) {
        const u32 i = e0 + r;
        if (i < ns) {
            keys[gbase + i] = high16 | (c[r] >> 12);
            if (HAS_VAL) vals[gbase + i] = v[r];
        }
    }
}

/* ------------------------------------------------------------------ *
 * wave16 blocksort (T9_LDS_WAVE16=1): single-wave block, 16 elements
 * per lane. The in-lane sort is a pure-VALU bitonic network over
 * registers (zero LDS traffic where the radix sort spends 4 of its 6
 * passes), then 6 merge rounds (16->1024) via merge-path splits into
 * LDS: ~2 LDS writes + ~3 LDS reads per element per round. With a
 * 64-thread block every __syncthreads() compiles to a wave-level
 * scheduling barrier (no s_barrier cost). Same stable-composite scheme
 * as k_bitonic_sort_sub above.
 * ------------------------------------------------------------------ */
template <int NSORT, int BLOCK, bool HAS_VAL>
__global__ __launch_bounds__(BLOCK, 2) void k_wave16_sort_sub(
    u64* __restrict__ keys, u32* __restrict__ vals,
    const u32* __restrict__ sub_start, const u32* __restrict__ sub_n) {
    constexpr int E = NSORT / BLOCK;
    static_assert(E == 16, "wave16 mapping: 16 elements per lane");
    constexpr int ROUNDS = (NSORT == 1024) ? 6 : (NSORT == 2048) ? 7 : 8;
    __shared__ u64 lc[NSORT];
    __shared__ u32 s_differ;

    const u32 sb = blockIdx.x;
    const u32 ns = sub_n[sb];
    if (ns <= 1 || ns > (u32)NSORT) return;
    const u32 gbase = sub_start[sb];
    const u32 lane = threadIdx.x;   /* block-level thread id */
    const u32 e0 = lane * E;

    if (lane == 0) s_differ = 0;
    __syncthreads();

    const u64 mask48 = 0x0000FFFFFFFFFFFFull;
    const u64 high16 = keys[gbase] & ~mask48;
    const u64 ref48 = keys[gbase] & mask48;

    /* the composite embeds the ORIGINAL position (low 12 bits), so the
     * value array never travels through the sort at all — it is
     * re-gathered from global at writeback (r2: removing the u32 lv LDS
     * array and the v[] registers raises occupancy, the measured
     * limiter of this barrier-parked kernel) */
    u64 c[E];
    u32 differ = 0;
#pragma unroll
    for (int r = 0; r < E; ++r) {
        const u32 i = e0 + r;
        if (i < ns) {
            const u64 k = keys[gbase + i];
            c[r] = ((k & mask48) << 12) | (u64)i;
            differ |= ((k & mask48) != ref48);
        } else {
            c[r] = ~0ull;
        }
    }
    if (differ) s_differ = 1;
    __syncthreads();
    if (!s_differ) return;

    /* in-register bitonic sort of the lane's 16 elements (pure VALU) */
#pragma unroll
    for (int lk = 1; lk <= 4; ++lk) {
        const int k = 1 << lk;
#pragma unroll
        for (int lj = lk - 1; lj >= 0; --lj) {
            const int j = 1 << lj;
#pragma unroll
            for (int r = 0; r < E; ++r) {
                const int q = r ^ j;
                if (q > r) {
                    const bool asc = (r & k) == 0;
                    if ((c[r] > c[q]) == asc) {
                        u64 tc = c[r]; c[r] = c[q]; c[q] = tc;
                    }
                }
            }
        }
    }

    /* 6 merge rounds: sorted runs of L pairwise -> 2L via merge-path.
     * Composites are unique, so `<=` against the B run is a stable merge
     * (A-run elements precede equal... equality cannot occur). */
#pragma unroll
    for (int lm = 0; lm < ROUNDS; ++lm) {
        const u32 L = (u32)E << lm;
        __syncthreads();
#pragma unroll
        for (int r = 0; r < E; ++r) lc[e0 + r] = c[r];
        __syncthreads();
        const u32 pairbase = e0 & ~(2 * L - 1);
        const u32 d = e0 - pairbase;          /* my first output diagonal */
        const u64* A = lc + pairbase;
        const u64* B = lc + pairbase + L;
        u32 lo = (d > L) ? d - L : 0;
        u32 hi = (d < L) ? d : L;
        while (lo < hi) {
            const u32 m = (lo + hi) >> 1;
            if (A[m] <= B[d - m - 1]) lo = m + 1;
            else hi = m;
        }
        u32 i = lo, j = d - lo;
#pragma unroll
        for (int r = 0; r < E; ++r) {
            const bool ta = (j >= L) || (i < L && A[i] <= B[j]);
            c[r] = (ta ? A[i] : B[j]);
            if (ta) ++i; else ++j;
        }
    }

    /* writeback: keys from the composite; values re-gathered from their
     * embedded original positions. All reads complete before any lane
     * writes (barrier), since the sort is in place. */
    u32 v[HAS_VAL ? E : 1];
    if (HAS_VAL) {
#pragma unroll
        for (int r = 0; r < E; ++r) {
            const u32 i = e0 + r;
            if (i < ns) v[r] = vals[gbase + (u32)(c[r] & 0xFFFu)];
        }
    }
    __syncthreads();
#pragma unroll
    for (int r = 0; r < E; ++r) {
        const u32 i = e0 + r;
        if (i < ns) {
            keys[gbase + i] = high16 | (c[r] >> 12);
            if (HAS_VAL) vals[gbase + i] = v[r];
        }
    }
}

/* wave16 span sort (3-level path, PASSES=6 spans): same blocksort as
 * k_wave16_sort_sub, but out-of-place over span descriptors. Valid for
 * the 6-pass span case only: spans pack (b7,b6,b5) groups that share
 * their top 16 key bits, so the 60-bit stable composite applies.
 * Oversize spans copy through (the ranged-LSD fallback re-sorts them),
 * mirroring k_lds_sort_span. in == out is safe: a block barrier
 * separates every global read from the first global write. */
template <int NSORT, int BLOCK, bool HAS_VAL>
__global__ __launch_bounds__(BLOCK, 2) void k_wave16_sort_span(
    const u64* __restrict__ keys_in, const u32* __restrict__ vals_in,
    u64* __restrict__ keys_out, u32* __restrict__ vals_out,
    const u32* __restrict__ span_start, const u32* __restrict__ span_len) {
    constexpr int E = NSORT / BLOCK;
    static_assert(E == 16, "wave16 mapping: 16 elements per lane");
    constexpr int ROUNDS = (NSORT == 1024) ? 6 : (NSORT == 2048) ? 7 : 8;
    __shared__ u64 lc[NSORT];
    __shared__ u32 s_differ;

    const u32 sb = blockIdx.x;
    const u32 ns = span_len[sb];
    if (ns == 0) return;
    const u32 gbase = span_start[sb];
    const u32 lane = threadIdx.x;
    const u32 e0 = lane * E;

    if (ns > (u32)NSORT) {
        if (keys_out != keys_in)
            for (u32 i = lane; i < ns; i += BLOCK) {
                keys_out[gbase + i] = keys_in[gbase + i];
                if (HAS_VAL) vals_out[gbase + i] = vals_in[gbase + i];
            }
        return;
    }

    if (lane == 0) s_differ = 0;
    __syncthreads();

    const u64 mask48 = 0x0000FFFFFFFFFFFFull;
    const u64 high16 = keys_in[gbase] & ~mask48;
    const u64 ref48 = keys_in[gbase] & mask48;

    /* values never travel through the sort (positions embedded in the
     * composite; re-gathered at writeback — see k_wave16_sort_sub) */
    u64 c[E];
    u32 differ = 0;
#pragma unroll
    for (int r = 0; r < E; ++r) {
        const u32 i = e0 + r;
        if (i < ns) {
            const u64 k = keys_in[gbase + i];
            c[r] = ((k & mask48) << 12) | (u64)i;
            differ |= ((k & mask48) != ref48);
        } else {
            c[r] = ~0ull;
        }
    }
    if (differ) s_differ = 1;
    __syncthreads();
    if (!s_differ) {
        if (keys_out != keys_in) {
#pragma unroll
            for (int r = 0; r < E; ++r) {
                const u32 i = e0 + r;
                if (i < ns) {
                    keys_out[gbase + i] = high16 | ((c[r] >> 12) & mask48);
                    if (HAS_VAL) vals_out[gbase + i] = vals_in[gbase + i];
                }
            }
        }
        return;
    }

#pragma unroll
    for (int lk = 1; lk <= 4; ++lk) {
        const int k = 1 << lk;
#pragma unroll
        for (int lj = lk - 1; lj >= 0; --lj) {
            const int j = 1 << lj;
#pragma unroll
            for (int r = 0; r < E; ++r) {
                const int q = r ^ j;
                if (q > r) {
                    const bool asc = (r & k) == 0;
                    if ((c[r] > c[q]) == asc) {
                        u64 tc = c[r]; c[r] = c[q]; c[q] = tc;
                    }
                }
            }
        }
    }

#pragma unroll
    for (int lm = 0; lm < ROUNDS; ++lm) {
        const u32 L = (u32)E << lm;
        __syncthreads();
#pragma unroll
        for (int r = 0; r < E; ++r) lc[e0 + r] = c[r];
        __syncthreads();
        const u32 pairbase = e0 & ~(2 * L - 1);
        const u32 d = e0 - pairbase;
        const u64* A = lc + pairbase;
        const u64* B = lc + pairbase + L;
        u32 lo = (d > L) ? d - L : 0;
        u32 hi = (d < L) ? d : L;
        while (lo < hi) {
            const u32 m = (lo + hi) >> 1;
            if (A[m] <= B[d - m - 1]) lo = m + 1;
            else hi = m;
        }
        u32 i = lo, j = d - lo;
#pragma unroll
        for (int r = 0; r < E; ++r) {
            const bool ta = (j >= L) || (i < L && A[i] <= B[j]);
            c[r] = (ta ? A[i] : B[j]);
            if (ta) ++i; else ++j;
        }
    }

    /* gather values from their embedded original positions, then write
     * (barrier keeps all reads ahead of the in-place writes) */
    u32 v[HAS_VAL ? E : 1];
    if (HAS_VAL) {
#pragma unroll
        for (int r = 0; r < E; ++r) {
            const u32 i = e0 + r;
            if (i < ns) v[r] = vals_in[gbase + (u32)(c[r] & 0xFFFu)];
        }
    }
    __syncthreads();
#pragma unroll
    for (int r = 0; r < E; ++r) {
        const u32 i = e0 + r;
        if (i < ns) {
            keys_out[gbase + i] = high16 | (c[r] >> 12);
            if (HAS_VAL) vals_out[gbase + i] = v[r];
        }
    }
}
