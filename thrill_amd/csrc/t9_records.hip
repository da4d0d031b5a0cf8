/* t9_records.hip — 100-byte record path: device-side generation (layout
 * restated from examples/terasort/terasort.cpp:31-118, key bytes from a
 * seeded counter-based splitmix64 identical to oracle/t9_oracle.cpp),
 * key-prefix extraction, payload gather and the end-to-end record sort
 * (SURVEY.md §7 step 4). Zipf token sampling for the word_count input
 * (thrill/common/zipf_distribution.hpp:55-120 mass function by inverse CDF).
 */

#include "t9_common.h"
#include "t9_rank_scatter.h"

#include <cstdlib>
#include <cstring>
#include <vector>
#include <algorithm>

extern "C" int t9i_count_tied(const u64* d_keys, u64 n, u32* d_ntied,
                              hipStream_t s);
extern "C" u64 t9_sort_pairs_workspace(u64 n);
extern "C" int t9_sort_pairs_u64_u32(t9_context*, u64*, u32*, u64, void*,
                                     void*);
extern "C" int t9i_sort_recs_msb(t9_context*, const u8*, u32, u64*, u32*,
                                 u64, void*, void*);
extern "C" u64 t9_partition_idx_workspace(u64 n);
extern "C" int t9_partition_idx(t9_context*, const u32*, u64, u32, u32*,
                                u64*, void*, void*);
extern "C" int t9i_sort_pairs_msb_ph(t9_context*, u64*, u32*, u64, void*,
                                     void*);
extern "C" u32* t9i_msb_pass1_hist(void* d_workspace, u64 n);

/* ------------------------------------------------------------------ */

__global__ __launch_bounds__(256) void k_gen_u64(u64* __restrict__ out,
                                                 u64 index0, u64 n,
                                                 u64 seed) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < n; i += stride)
        out[i] = t9_splitmix64_at(seed, index0 + i);
}

/* One record per thread, staged through LDS so global writes are
 * word-coalesced. Value layout: terasort.cpp:73-111. */
__global__ __launch_bounds__(256) void k_gen_records(u8* __restrict__ out,
                                                     u64 index0, u64 n,
                                                     u64 seed) {
    __shared__ u8 buf[256 * 100];
    const u32 tid = threadIdx.x;
    const u64 blk0 = (u64)blockIdx.x * 256;
    const u64 r = blk0 + tid;
    const u32 nrec = (u32)((n - blk0 < 256) ? (n - blk0) : 256);
    if (r < n) {
        const u64 rec = index0 + r;
        u8* o = buf + tid * 100;
        u64 k0 = t9_splitmix64_at(seed, 2 * rec);
        u64 k1 = t9_splitmix64_at(seed, 2 * rec + 1);
        for (int j = 0; j < 8; ++j) o[j] = (u8)(k0 >> (56 - 8 * j));
        o[8] = (u8)(k1 >> 56);
        o[9] = (u8)(k1 >> 48);
        u8* v = o + 10;
        const char* hexd = "0123456789ABCDEF";
        *v++ = 0x00;
        *v++ = 0x11;
        for (int j = 0; j != 16; ++j) *v++ = hexd[(rec >> (4 * j)) & 0x0F];
        for (int j = 0; j != 16; ++j) *v++ = '0';
        *v++ = 0x88; *v++ = 0x99; *v++ = 0xAA; *v++ = 0xBB;
        for (int j = 0; j < 12; ++j) {
            u8 f = hexd[((20 + rec) >> (4 * j)) & 0x0F];
            *v++ = f; *v++ = f; *v++ = f; *v++ = f;
        }
        *v++ = 0xCC; *v++ = 0xDD; *v++ = 0xEE; *v++ = 0xFF;
    }
    __syncthreads();
    const u32 nb = nrec * 100;
    u32* gout = (u32*)(out + blk0 * 100);
    const u32* gin = (const u32*)buf;
    for (u32 w = tid; w * 4 < nb; w += 256) gout[w] = gin[w];
}

/* u64 key prefix at key_off + record index iota. BE (default): the key
 * is 8 lexicographic bytes (TeraSort, sorts like memcmp); LE: a native
 * uint64_t field (BASELINE config 5's struct{u64 key; u8 payload[120]}),
 * sorted numerically. */
template <bool LE>
__global__ __launch_bounds__(256) void k_extract_key64(
    const u8* __restrict__ recs, u64 n, u32 rec_words, u32 key_off,
    u64* __restrict__ keys, u32* __restrict__ idx) {
    const u64 stride = (u64)gridDim.x * 256;
    const u32* r32 = (const u32*)recs;
    const u64* r64 = (const u64*)recs;
    const bool aligned8 = LE && (rec_words % 2 == 0) && (key_off % 8 == 0);
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
        u64 k;
        if (aligned8) {
            k = r64[(u64)i * (rec_words / 2) + key_off / 8];
        }
        else {
            u64 w = (u64)i * rec_words + key_off / 4;
            u32 w0 = r32[w], w1 = r32[w + 1];
            k = LE ? (((u64)w1 << 32) | w0)
                   : (((u64)__builtin_bswap32(w0) << 32) |
                      __builtin_bswap32(w1));
        }
        keys[i] = k;
        idx[i] = (u32)i;
    }
}

/* out[i] = recs[idx[i]], word-wise: consecutive threads write consecutive
 * output words; a wave's 64 word-reads touch only ~3 source records
 * (contiguous lines), which measured faster than 16-byte-per-lane
 * variants that spread each read instruction over ~10 random records.
 * RW is the compile-time record width in words (0 = runtime): the
 * word->record division then compiles to a multiply-shift instead of a
 * ~30-instruction emulated u64 division per word. */
template <int RW, bool NT>
__global__ __launch_bounds__(256) void k_gather_records(
    const u8* __restrict__ recs, const u32* __restrict__ idx, u64 n,
    u32 rec_words_rt, u8* __restrict__ out) {
    const u32 rec_words = RW ? (u32)RW : rec_words_rt;
    const u64 total_words = n * rec_words;
    const u64 stride = (u64)gridDim.x * 256;
    const u32* rin = (const u32*)recs;
    u32* rout = (u32*)out;
    u64 g = (u64)blockIdx.x * 256 + threadIdx.x;
    /* 4 independent loads in flight per iteration */
    for (; g + 3 * stride < total_words; g += 4 * stride) {
        u32 v[4];
        for (int j = 0; j < 4; ++j) {
            u64 gj = g + (u64)j * stride;
            u64 rec = gj / rec_words;
            const u32* src =
                &rin[(u64)idx[rec] * rec_words + (u32)(gj - rec * rec_words)];
            v[j] = NT ? __builtin_nontemporal_load(src) : *src;
        }
        for (int j = 0; j < 4; ++j) rout[g + (u64)j * stride] = v[j];
    }
    for (; g < total_words; g += stride) {
        u64 rec = g / rec_words;
        rout[g] = rin[(u64)idx[rec] * rec_words +
                      (u32)(g - rec * rec_words)];
    }
}

/* block-contiguous variant: each block owns one contiguous output span,
 * iterated 256 words at a time — the idx window a block touches is a
 * compact ~SPAN/RW range (L1-resident). */
template <int RW, bool NT>
__global__ __launch_bounds__(256) void k_gather_records_span(
    const u8* __restrict__ recs, const u32* __restrict__ idx, u64 n,
    u32 rec_words_rt, u8* __restrict__ out) {
    const u32 rec_words = RW ? (u32)RW : rec_words_rt;
    const u64 total_words = n * rec_words;
    const u64 span = (total_words + gridDim.x - 1) / gridDim.x;
    const u64 s0 = (u64)blockIdx.x * span;
    const u64 s1 = (s0 + span < total_words) ? s0 + span : total_words;
    const u32* rin = (const u32*)recs;
    u32* rout = (u32*)out;
    u64 g = s0 + threadIdx.x;
    for (; g + 3 * 256 < s1; g += 4 * 256) {
        u32 v[4];
        for (int j = 0; j < 4; ++j) {
            u64 gj = g + (u64)j * 256;
            u64 rec = gj / rec_words;
            const u32* src =
                &rin[(u64)idx[rec] * rec_words + (u32)(gj - rec * rec_words)];
            v[j] = NT ? __builtin_nontemporal_load(src) : *src;
        }
        for (int j = 0; j < 4; ++j) rout[g + (u64)j * 256] = v[j];
    }
    for (; g < s1; g += 256) {
        u64 rec = g / rec_words;
        rout[g] = rin[(u64)idx[rec] * rec_words +
                      (u32)(g - rec * rec_words)];
    }
}

/* ---- on-device tie sort (equal-u64-prefix runs) ------------------- *
 * After the stable radix sort of (prefix, idx) pairs, records whose
 * 8-byte prefix collides must be re-ordered by the remaining bytes (the
 * reference compares whole items, api/sort.hpp:480,487-501). Round 1 did
 * this with one synchronous hipMemcpy per tied record (a perf cliff on
 * adversarial duplicate-heavy inputs — VERDICT r01 item 2). Now fully on
 * device: compact the tied positions, then LSD over the tail — for each
 * 8-byte big-endian chunk of the tail from LAST to FIRST, stably sort
 * the (chunk, record-idx) pairs with the existing radix pipeline,
 * finishing with one stable pass on the u64 prefix to regroup the runs.
 * Chunks that are identical across ALL tied records are skipped (a
 * one-word reduction), so the all-equal-records input costs extraction
 * passes only. */

/* bucket[i] = 0 if keys[i] participates in an equal-prefix run */
__global__ __launch_bounds__(256) void k_tie_flags(
    const u64* __restrict__ keys, u64 n, u32* __restrict__ bucket) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
        const u64 k = keys[i];
        const bool tied = (i > 0 && keys[i - 1] == k) ||
                          (i + 1 < n && keys[i + 1] == k);
        bucket[i] = tied ? 0u : 1u;
    }
}

/* out[j] = big-endian u64 of record bytes [off, off+8) (zero-padded past
 * rec_size) for record tidx[j]; LE=true loads a native u64 (the keyle
 * prefix pass). */
template <bool LE>
__global__ __launch_bounds__(256) void k_tie_chunk(
    const u8* __restrict__ recs, const u32* __restrict__ tidx, u64 m,
    u32 rec_size, u32 off, u64* __restrict__ out) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 j = (u64)blockIdx.x * 256 + threadIdx.x; j < m; j += stride) {
        const u8* r = recs + (u64)tidx[j] * rec_size + off;
        const u32 avail = rec_size - off;
        u64 k = 0;
        if (LE) {
            /* off==0; record base is 4-aligned (rec_size % 4 == 0) */
            const u32* r32 = (const u32*)r;
            k = ((u64)r32[1] << 32) | r32[0];
        }
        else if (avail >= 8 && (off % 4) == 0) {
            const u32* r32 = (const u32*)r;
            k = ((u64)__builtin_bswap32(r32[0]) << 32) |
                __builtin_bswap32(r32[1]);
        }
        else {
            for (u32 t = 0; t < 8 && t < avail; ++t)
                k |= (u64)r[t] << (56 - 8 * t);
        }
        out[j] = k;
    }
}

/* One-pass prescan over the tied set: flags[0] = 1 if the u64 prefix
 * differs anywhere (>= 2 runs -> the regroup pass is needed); flags[1+c]
 * = 1 if tail chunk c differs anywhere. Order-invariant (compares
 * against record tidx[0]'s bytes), so it runs ONCE before the sort
 * loop; chunks that never differ are skipped entirely — the
 * all-identical-records adversarial case costs this single read of the
 * tied records instead of one extract+check round per chunk. */
__global__ __launch_bounds__(256) void k_tie_prescan(
    const u8* __restrict__ recs, const u32* __restrict__ tidx, u64 m,
    u32 rec_size, u32 nc, u32* __restrict__ flags) {
    __shared__ u32 s_flags[64];
    const u32 tid = threadIdx.x;
    const u32 rw = rec_size / 4;
    for (u32 c = tid; c <= nc && c < 64; c += 256) s_flags[c] = 0;
    __syncthreads();
    const u32* rin = (const u32*)recs;
    const u32* ref = rin + (u64)tidx[0] * rw;
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 j = (u64)blockIdx.x * 256 + tid; j < m; j += stride) {
        const u32* r = rin + (u64)tidx[j] * rw;
        /* word-granular diffs (4x fewer loads than bytes); word w covers
         * bytes [4w, 4w+4): prefix = words 0..1, tail chunk c = words
         * 2+2c..3+2c */
        if ((r[0] ^ ref[0]) | (r[1] ^ ref[1])) s_flags[0] = 1;
        for (u32 w = 2; w < rw; ++w)
            if (r[w] ^ ref[w]) s_flags[1 + (w - 2) / 2] = 1;
    }
    __syncthreads();
    for (u32 c = tid; c <= nc && c < 64; c += 256)
        if (s_flags[c]) atomicExch(&flags[c], 1u);
}

/* sequential prescan for the m == n case (every record tied — the
 * adversarial all-equal benches): word g of the packed record array is
 * compared directly against the reference record's word g % rw, fully
 * coalesced (no index indirection; the set of records scanned is the
 * same, and the flags are order-invariant). */
__global__ __launch_bounds__(256) void k_tie_prescan_seq(
    const u8* __restrict__ recs, const u32* __restrict__ tidx, u64 m,
    u32 rec_size, u32 nc, u32* __restrict__ flags) {
    __shared__ u32 s_flags[64];
    __shared__ u32 s_ref[160];
    const u32 tid = threadIdx.x;
    const u32 rw = rec_size / 4;
    for (u32 c = tid; c <= nc && c < 64; c += 256) s_flags[c] = 0;
    const u32* rin = (const u32*)recs;
    if (tid < rw && rw <= 160)
        s_ref[tid] = rin[(u64)tidx[0] * rw + tid];
    __syncthreads();
    const u64 total = m * rw;
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 g = (u64)blockIdx.x * 256 + tid; g < total; g += stride) {
        const u32 w = (u32)(g % rw);
        if (rin[g] ^ s_ref[w])
            s_flags[w < 2 ? 0 : 1 + (w - 2) / 2] = 1;
    }
    __syncthreads();
    for (u32 c = tid; c <= nc && c < 64; c += 256)
        if (s_flags[c]) atomicExch(&flags[c], 1u);
}

/* ---- MSD tie strategy (group-id sorts) --------------------------- *
 * LSD over the tail is oblivious: it sorts EVERY differing chunk even
 * when the first one already separates everything (random tails — 13
 * full sorts of m = n on the shared-prefix adversarial input). The MSD
 * strategy processes chunks most-significant first and STOPS as soon as
 * no two elements still tie: per level, stably sort by the chunk, then
 * stably sort by the current group id (= run-start position), which
 * yields exactly the segmented order (group asc, chunk asc, stable);
 * then recompute groups as runs of equal (group, chunk) and count the
 * remaining tied elements. Random tails exit after ONE level (two pair
 * sorts); inputs whose ties persist past 2 levels fall back to the
 * bounded LSD loop. Group ids sort as (u64)grp << 32 so the MSB radix
 * pipeline sees their entropy in its top bytes. */

/* v[j] = j if a group boundary starts at j (j==0, group changed, or
 * chunk value changed), else 0; d_ties += #non-boundary elements.
 * grp == NULL: boundaries from ck alone (the init pass on the prefix). */
__global__ __launch_bounds__(256) void k_tie_bound_vals(
    const u32* __restrict__ grp, const u64* __restrict__ ck, u64 m,
    u32* __restrict__ v, u32* __restrict__ ties) {
    __shared__ u32 s_t;
    if (threadIdx.x == 0) s_t = 0;
    __syncthreads();
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 j = (u64)blockIdx.x * 256 + threadIdx.x; j < m; j += stride) {
        bool b = (j == 0) || (ck[j] != ck[j - 1]) ||
                 (grp && grp[j] != grp[j - 1]);
        v[j] = b ? (u32)j : 0;
        if (!b) atomicAdd(&s_t, 1u);
    }
    __syncthreads();
    if (threadIdx.x == 0 && s_t) atomicAdd(ties, s_t);
}

#define T9_MAXSCAN_TILE 2048

/* per-tile inclusive max-scan of v (in place); partials[b] = tile max */
__global__ __launch_bounds__(256) void k_maxscan_local(
    u32* __restrict__ v, u64 m, u32* __restrict__ partials) {
    __shared__ u32 s[256];
    const u32 tid = threadIdx.x;
    const u64 base = (u64)blockIdx.x * T9_MAXSCAN_TILE;
    u32 carry = 0;
    for (u32 c = 0; c < T9_MAXSCAN_TILE / 256; ++c) {
        const u64 j = base + c * 256 + tid;
        u32 x = (j < m) ? v[j] : 0;
        s[tid] = x;
        __syncthreads();
        for (int off = 1; off < 256; off <<= 1) {
            u32 y = (tid >= (u32)off) ? s[tid - off] : 0;
            __syncthreads();
            if (y > s[tid]) s[tid] = y;
            __syncthreads();
        }
        const u32 val = (s[tid] > carry) ? s[tid] : carry;
        if (j < m) v[j] = val;
        const u32 cmax = (s[255] > carry) ? s[255] : carry;
        __syncthreads();
        carry = cmax;
    }
    if (tid == 0) partials[blockIdx.x] = carry;
}

/* single block: EXCLUSIVE max-scan over B partials, in place */
__global__ __launch_bounds__(256) void k_maxscan_part(
    u32* __restrict__ p, u64 B) {
    __shared__ u32 s[256];
    const u32 tid = threadIdx.x;
    u32 carry = 0;
    for (u64 c0 = 0; c0 < B; c0 += 256) {
        const u64 b = c0 + tid;
        u32 x = (b < B) ? p[b] : 0;
        s[tid] = x;
        __syncthreads();
        for (int off = 1; off < 256; off <<= 1) {
            u32 y = (tid >= (u32)off) ? s[tid - off] : 0;
            __syncthreads();
            if (y > s[tid]) s[tid] = y;
            __syncthreads();
        }
        /* exclusive: shift by one (carry for lane 0) */
        const u32 excl = (tid == 0) ? carry
                                    : ((s[tid - 1] > carry) ? s[tid - 1]
                                                            : carry);
        if (b < B) p[b] = excl;
        const u32 cmax = (s[255] > carry) ? s[255] : carry;
        __syncthreads();
        carry = cmax;
    }
}

/* v[j] = max(v[j], partials[tile]) — v becomes the new group-id array */
__global__ __launch_bounds__(256) void k_maxscan_fix(
    u32* __restrict__ v, u64 m, const u32* __restrict__ partials) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 j = (u64)blockIdx.x * 256 + threadIdx.x; j < m; j += stride) {
        const u32 pmax = partials[j / T9_MAXSCAN_TILE];
        if (pmax > v[j]) v[j] = pmax;
    }
}

/* grpkey[x] = (u64)grp[x] << 32 (entropy into the MSB pipeline's bytes) */
__global__ __launch_bounds__(256) void k_grp_keys(
    const u32* __restrict__ grp, u64 m, u64* __restrict__ out) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 j = (u64)blockIdx.x * 256 + threadIdx.x; j < m; j += stride)
        out[j] = (u64)grp[j] << 32;
}

__global__ __launch_bounds__(256) void k_iota32(u32* __restrict__ v,
                                                u64 m) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 j = (u64)blockIdx.x * 256 + threadIdx.x; j < m; j += stride)
        v[j] = (u32)j;
}

/* fused permutation applies for the MSD level (one read of the
 * permutation drives all arrays, replacing per-array gather launches):
 * after the chunk sort:  a_out[x] = a_in[p[x]], b_out[x] = b_in[p[x]]
 * after the group sort:  + the chunk values c as a third array */
__global__ __launch_bounds__(256) void k_msd_apply2(
    const u32* __restrict__ p, u64 m, const u32* __restrict__ a_in,
    u32* __restrict__ a_out, const u32* __restrict__ b_in,
    u32* __restrict__ b_out) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 x = (u64)blockIdx.x * 256 + threadIdx.x; x < m; x += stride) {
        const u32 j = p[x];
        a_out[x] = a_in[j];
        b_out[x] = b_in[j];
    }
}

__global__ __launch_bounds__(256) void k_msd_apply3(
    const u32* __restrict__ p, u64 m, const u32* __restrict__ a_in,
    u32* __restrict__ a_out, const u32* __restrict__ b_in,
    u32* __restrict__ b_out, const u64* __restrict__ c_in,
    u64* __restrict__ c_out) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 x = (u64)blockIdx.x * 256 + threadIdx.x; x < m; x += stride) {
        const u32 j = p[x];
        a_out[x] = a_in[j];
        b_out[x] = b_in[j];
        c_out[x] = c_in[j];
    }
}

/* d_idx[perm[j]] = tidx[j] — write the re-ordered record indices back
 * into the (ascending) tied positions */
__global__ __launch_bounds__(256) void k_tie_scatter(
    u32* __restrict__ d_idx, const u32* __restrict__ perm,
    const u32* __restrict__ tidx, u64 m) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 j = (u64)blockIdx.x * 256 + threadIdx.x; j < m; j += stride)
        d_idx[perm[j]] = tidx[j];
}

__global__ __launch_bounds__(256) void k_zipf_tokens(
    u64* __restrict__ out, const double* __restrict__ cdf, u64 N, u64 index0,
    u64 n, u64 seed) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
        double u = (double)(t9_splitmix64_at(seed, index0 + i) >> 11) *
                   (1.0 / 9007199254740992.0);
        u64 lo = 0, hi = N - 1;
        while (lo < hi) {
            u64 mid = (lo + hi) / 2;
            if (cdf[mid] > u) hi = mid; else lo = mid + 1;
        }
        out[i] = lo + 1;
    }
}

/* Fused extract + MSB pass-1 histogram: one block per 8192-record tile
 * (the MSB pipeline's T9_MSB_TILE), 32 sub-tiles of 256 records staged
 * through LDS by coalesced word loads — every input line is read ONCE,
 * coalesced (the standalone extract's per-lane strided loads reach only
 * ~60% of streaming rate on 100-B records), and the tile's byte-7 digit
 * counts come out of the same pass, eliminating the separate pass-1
 * histogram read (k_hist_msb). Output layout identical to
 * k_extract_key64 + k_hist_msb<0> at shift 56. */
template <int RW, bool LE, bool DB = true>
__global__ __launch_bounds__(256) void k_extract_hist(
    const u8* __restrict__ recs, u64 n, u64* __restrict__ keys,
    u32* __restrict__ idx, u32* __restrict__ hist) {
    constexpr int SUBREC = 256;                /* records per LDS stage */
    constexpr int SUBW = SUBREC * RW;          /* words per stage */
    /* DB: double-buffered (one barrier per stage, 3 blocks/CU at 100-B
       records) vs single-buffered (two barriers, 6 blocks/CU) — the
       wave-cycle decomposition shows this kernel parked on the stage
       barrier, so occupancy vs barrier count is the tradeoff to measure
       (T9_EXTRACT_SB=1 selects single). */
    __shared__ u32 s_buf[DB ? 2 : 1][SUBW];
    __shared__ u32 s_cnt[256];
    const u32 tid = threadIdx.x;
    const u64 tile0 = (u64)blockIdx.x * 8192;
    const u32 tn = (u32)((n - tile0 < 8192) ? (n - tile0) : 8192);
    s_cnt[tid] = 0;
    const u32* rin = (const u32*)recs;

    auto load_stage = [&](u32 s0, int buf) {
        const u32 sn = (tn - s0 < SUBREC) ? tn - s0 : SUBREC;
        const u64 w0 = (tile0 + s0) * RW;
        const u32 wn = sn * RW;
        for (u32 w = tid; w < wn; w += 256) s_buf[buf][w] = rin[w0 + w];
    };
    if (DB) {
        load_stage(0, 0);
        __syncthreads();
    }
    int cur = 0;
    for (u32 s0 = 0; s0 < tn; s0 += SUBREC, cur ^= DB ? 1 : 0) {
        if (!DB) {
            load_stage(s0, 0);
            __syncthreads();
        }
        else if (s0 + SUBREC < tn) {
            load_stage(s0 + SUBREC, cur ^ 1);
        }
        const u32 sn = (tn - s0 < SUBREC) ? tn - s0 : SUBREC;
        const bool valid = tid < sn;
        u64 k = 0;
        if (valid) {
            if (LE)
                k = ((u64)s_buf[cur][tid * RW + 1] << 32) |
                    s_buf[cur][tid * RW];
            else
                k = ((u64)__builtin_bswap32(s_buf[cur][tid * RW]) << 32) |
                    __builtin_bswap32(s_buf[cur][tid * RW + 1]);
            const u64 gi = tile0 + s0 + tid;
            keys[gi] = k;
            idx[gi] = (u32)gi;
        }
        t9_hist_ballot_add<8>(s_cnt, (u32)(k >> 56), valid, tid & 63);
        __syncthreads();
    }
    hist[(u64)blockIdx.x * 256 + tid] = s_cnt[tid];
}

/* scatter-records experiment (gather-wall probe): read the input
 * SEQUENTIALLY (every line fetched once, fully used) and write each
 * word to its record's destination: out[inv[rec]]. Mirrors
 * k_gather_records with the random side moved from reads to writes —
 * measures whether CDNA4 partial-line writes (byte-enable sectors) beat
 * random-line read amplification. */
template <int RW>
__global__ __launch_bounds__(256) void k_scatter_records(
    const u8* __restrict__ recs, const u32* __restrict__ inv, u64 n,
    u8* __restrict__ out) {
    const u64 total_words = n * RW;
    const u64 stride = (u64)gridDim.x * 256;
    const u32* rin = (const u32*)recs;
    u32* rout = (u32*)out;
    for (u64 g = (u64)blockIdx.x * 256 + threadIdx.x; g < total_words;
         g += stride) {
        const u64 rec = g / RW;
        const u32 off = (u32)(g - rec * RW);
        rout[(u64)inv[rec] * RW + off] = rin[g];
    }
}

__global__ __launch_bounds__(256) void k_invert_perm(
    const u32* __restrict__ idx, u64 n, u32* __restrict__ inv) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < n; i += stride)
        inv[idx[i]] = (u32)i;
}

/* ------------------------------------------------------------------ *
 * host orchestration
 * ------------------------------------------------------------------ */

namespace {
u32 grid_for(u64 work) {
    u64 want = t9_ceil_div(work, 256);
    return (u32)((want < 4096) ? (want ? want : 1) : 4096);
}
} // namespace

extern "C" {

int t9_gen_u64(t9_context* ctx, u64* d_out, u64 index0, u64 n, u64 seed,
               void* stream) {
    (void)ctx;
    if (!d_out) return T9_EINVAL;
    if (n == 0) return T9_OK;
    hipLaunchKernelGGL(k_gen_u64, dim3(grid_for(n)), dim3(256), 0,
                       (hipStream_t)stream, d_out, index0, n, seed);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_gen_records(t9_context* ctx, u8* d_out, u64 index0, u64 n, u64 seed,
                   void* stream) {
    (void)ctx;
    if (!d_out) return T9_EINVAL;
    if (n == 0) return T9_OK;
    u64 blocks = t9_ceil_div(n, 256);
    if (blocks >= (1ull << 31)) return T9_EINVAL;
    hipLaunchKernelGGL(k_gen_records, dim3((u32)blocks), dim3(256), 0,
                       (hipStream_t)stream, d_out, index0, n, seed);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_extract_key64(t9_context* ctx, const u8* d_recs, u64 n, u32 rec_size,
                     u32 key_off, u64* d_keys, u32* d_idx, void* stream) {
    (void)ctx;
    if (!d_recs || !d_keys || !d_idx) return T9_EINVAL;
    if (rec_size % 4 || key_off % 4 || key_off + 8 > rec_size)
        return T9_EINVAL;
    if (n == 0) return T9_OK;
    const char* eg = getenv("T9_EXTRACT_GRID");
    u64 ewant = t9_ceil_div(n, 256);
    const u32 ecap = eg ? (u32)atoi(eg) : 16384;
    const dim3 egrid((u32)((ewant < ecap) ? (ewant ? ewant : 1) : ecap));
    T9_PERF_WRAP((hipStream_t)stream, "extract",
                 hipLaunchKernelGGL((k_extract_key64<false>), egrid,
                                    dim3(256), 0, (hipStream_t)stream,
                                    d_recs, n, rec_size / 4, key_off,
                                    d_keys, d_idx));
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_extract_key64_le(t9_context* ctx, const u8* d_recs, u64 n,
                        u32 rec_size, u32 key_off, u64* d_keys, u32* d_idx,
                        void* stream) {
    (void)ctx;
    if (!d_recs || !d_keys || !d_idx) return T9_EINVAL;
    if (rec_size % 4 || key_off % 4 || key_off + 8 > rec_size)
        return T9_EINVAL;
    if (n == 0) return T9_OK;
    u64 ewant = t9_ceil_div(n, 256);
    const dim3 egrid((u32)((ewant < 16384) ? (ewant ? ewant : 1) : 16384));
    T9_PERF_WRAP((hipStream_t)stream, "extract",
                 hipLaunchKernelGGL((k_extract_key64<true>), egrid,
                                    dim3(256), 0, (hipStream_t)stream,
                                    d_recs, n, rec_size / 4, key_off,
                                    d_keys, d_idx));
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_gather_records(t9_context* ctx, const u8* d_recs, const u32* d_idx,
                      u64 n, u32 rec_size, u8* d_out, void* stream) {
    (void)ctx;
    if (!d_recs || !d_idx || !d_out || rec_size % 4) return T9_EINVAL;
    if (n == 0) return T9_OK;
    const u32 rw = rec_size / 4;
    hipStream_t s = (hipStream_t)stream;
    const char* gv = getenv("T9_GATHER_VARIANT");
    const int var = gv ? atoi(gv) : 3;   /* 1 strided, 2 +nt, 3 span,
                                            4 span+nt, 5 scatter probe
                                            (sequential read + random
                                            write via inverse perm);
                                            measured best: span,
                                            grid 16384 */
    const char* gg = getenv("T9_GATHER_GRID");
    const u32 gcap = gg ? (u32)atoi(gg) : 16384;
    u64 want = t9_ceil_div(n * rw, 256);
    const dim3 grid((u32)((want < gcap) ? (want ? want : 1) : gcap));
    if (var == 5 && rw == 25) {
        /* scatter experiment: build the inverse permutation, then read
         * the input sequentially and write each record to its
         * destination (probe for the random-read-line amplification vs
         * partial-line-write cost tradeoff) */
        u32* d_inv = nullptr;
        HIP_TRY(hipMallocAsync((void**)&d_inv, n * 4, s));
        u64 iwant = t9_ceil_div(n, 256);
        hipLaunchKernelGGL(k_invert_perm,
                           dim3((u32)((iwant < 4096) ? iwant : 4096)),
                           dim3(256), 0, s, d_idx, n, d_inv);
        T9_PERF_WRAP(s, "gather",
                     hipLaunchKernelGGL((k_scatter_records<25>), grid,
                                        dim3(256), 0, s, d_recs, d_inv, n,
                                        d_out));
        HIP_TRY(hipFreeAsync(d_inv, s));
        T9_LAUNCH_CHECK();
        return T9_OK;
    }
    T9_PERF_WRAP(
        s, "gather",
        if (rw == 25 && var == 1)
            hipLaunchKernelGGL((k_gather_records<25, false>), grid,
                               dim3(256), 0, s, d_recs, d_idx, n, rw,
                               d_out);
        else if (rw == 25 && var == 2)
            hipLaunchKernelGGL((k_gather_records<25, true>), grid,
                               dim3(256), 0, s, d_recs, d_idx, n, rw,
                               d_out);
        else if (rw == 25 && var == 3)
            hipLaunchKernelGGL((k_gather_records_span<25, false>), grid,
                               dim3(256), 0, s, d_recs, d_idx, n, rw,
                               d_out);
        else if (rw == 25 && var == 4)
            hipLaunchKernelGGL((k_gather_records_span<25, true>), grid,
                               dim3(256), 0, s, d_recs, d_idx, n, rw,
                               d_out);
        else if (rw == 32)
            hipLaunchKernelGGL((k_gather_records<32, false>), grid,
                               dim3(256), 0, s, d_recs, d_idx, n, rw,
                               d_out);
        else if (rw == 2)
            hipLaunchKernelGGL((k_gather_records<2, false>), grid,
                               dim3(256), 0, s, d_recs, d_idx, n, rw,
                               d_out);
        else
            hipLaunchKernelGGL((k_gather_records<0, false>), grid,
                               dim3(256), 0, s, d_recs, d_idx, n, rw,
                               d_out));
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_zipf_tokens(t9_context* ctx, u64* d_out, const double* d_cdf, u64 N,
                   u64 index0, u64 n, u64 seed, void* stream) {
    (void)ctx;
    if (!d_out || !d_cdf || N == 0) return T9_EINVAL;
    if (n == 0) return T9_OK;
    hipLaunchKernelGGL(k_zipf_tokens, dim3(grid_for(n)), dim3(256), 0,
                       (hipStream_t)stream, d_out, d_cdf, N, index0, n,
                       seed);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_sort_records_keyle(t9_context* ctx, const u8* d_in, u8* d_out,
                          u64 n, u32 rec_size, void* d_workspace,
                          void* stream);

u64 t9_sort_records_workspace(u64 n, u32 rec_size) {
    (void)rec_size;
    if (n < 2) return 256;
    return t9_align256(n * 8) + t9_align256(n * 4) +
           t9_sort_pairs_workspace(n) + 256 /* ntied */;
}

/* Sort records by the acceptance total order (full-record lexicographic;
 * the u64 big-endian prefix of the key is the radix sort key, stability of
 * the radix pipeline keeps equal-prefix records in input order, and
 * equal-prefix runs are re-ordered ON DEVICE by the remaining bytes —
 * the segmented tail LSD above k_tie_flags. At the benchmark's
 * uniform-key sizes the expected number of colliding 8-byte prefixes is
 * < 1 (SURVEY.md §7 hard part (a)), so the tie machinery is normally
 * idle; adversarial duplicate-heavy inputs stay on device. */
static int sort_records_impl(t9_context* ctx, const u8* d_in, u8* d_out,
                             u64 n, u32 rec_size, u32 key_len,
                             void* d_workspace, void* stream, bool le);

int t9_sort_records(t9_context* ctx, const u8* d_in, u8* d_out, u64 n,
                    u32 rec_size, u32 key_len, void* d_workspace,
                    void* stream) {
    return sort_records_impl(ctx, d_in, d_out, n, rec_size, key_len,
                             d_workspace, stream, false);
}

/* config-5 variant: the record key is a native little-endian uint64_t at
 * offset 0, ordered numerically; payload ties by byte order. */
int t9_sort_records_keyle(t9_context* ctx, const u8* d_in, u8* d_out,
                          u64 n, u32 rec_size, void* d_workspace,
                          void* stream) {
    return sort_records_impl(ctx, d_in, d_out, n, rec_size, 8, d_workspace,
                             stream, true);
}

static int sort_records_impl(t9_context* ctx, const u8* d_in, u8* d_out,
                             u64 n, u32 rec_size, u32 key_len,
                             void* d_workspace, void* stream, bool le) {
    if (rec_size % 4 || key_len > rec_size || n >= (1ull << 32))
        return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    if (n == 0) return T9_OK;
    if (!d_in || !d_out || !d_workspace) return T9_EINVAL;
    /* d_in must be preserved (API contract) and d_out doubles as tie
     * scratch — aliasing would corrupt the input */
    if (d_in == d_out) return T9_EINVAL;
    if (n == 1) {
        HIP_TRY(hipMemcpyAsync(d_out, d_in, rec_size, hipMemcpyDeviceToDevice,
                               s));
        return T9_OK;
    }
    char* p = (char*)d_workspace;
    u64* d_keys = (u64*)p;
    p += t9_align256(n * 8);
    u32* d_idx = (u32*)p;
    p += t9_align256(n * 4);
    void* pair_ws = p;
    p += t9_sort_pairs_workspace(n);
    u32* d_ntied = (u32*)p;

    /* extract strategy ladder:
     * (1) DEFAULT for MSB-dispatch sizes: fused extract+hist — one
     *     LDS-staged coalesced pass over the records writes keys, the
     *     index iota AND the MSB pass-1 byte-7 histogram
     *     (k_extract_hist), so the standalone extract's strided loads
     *     and the pass-1 histogram read both disappear. T9_EXTRACT_HIST=0
     *     disables.
     * (2) the round-1 per-lane fused path (T9_FUSED_EXTRACT=1): measured
     *     SLOWER (byte reads at record stride, 17.3 vs 15.5 ms/10 GiB) —
     *     kept as a recorded negative result.
     * (3) plain extract + pair-sort dispatch (small n / LSD override). */
    const char* algo = getenv("T9_SORT_ALGO");
    bool msb_dispatch = n >= (1ull << 22);
    if (algo && strcmp(algo, "lsd") == 0) msb_dispatch = false;
    if (algo && strcmp(algo, "msb") == 0) msb_dispatch = n >= (1ull << 14);
    const char* ehe = getenv("T9_EXTRACT_HIST");
    const u32 rw = rec_size / 4;
    const bool eh = !(ehe && ehe[0] == '0') && msb_dispatch &&
                    (rw == 25 || rw == 32);
    const char* fe = getenv("T9_FUSED_EXTRACT");
    bool fused = fe && atoi(fe) && n >= (1ull << 14) &&
                 (rec_size == 100 || rec_size == 128);
    int rc;
    if (eh && !fused) {
        u32* hist = t9i_msb_pass1_hist(pair_ws, n);
        const u64 B = t9_ceil_div(n, 8192);
        /* single-buffer default: 6 blocks/CU measured 2.447 vs 2.56 ms
           (double-buffer, 3 blocks/CU) — occupancy beats the saved
           barrier; T9_EXTRACT_SB=0 restores double buffering */
        const char* sbe = getenv("T9_EXTRACT_SB");
        const bool sb = !(sbe && sbe[0] == '0');
        T9_PERF_WRAP(
            s, "extract",
            if (rw == 25 && !le && sb)
                hipLaunchKernelGGL((k_extract_hist<25, false, false>),
                                   dim3((u32)B), dim3(256), 0, s, d_in, n,
                                   d_keys, d_idx, hist);
            else if (rw == 25 && !le)
                hipLaunchKernelGGL((k_extract_hist<25, false>),
                                   dim3((u32)B), dim3(256), 0, s, d_in, n,
                                   d_keys, d_idx, hist);
            else if (rw == 25)
                hipLaunchKernelGGL((k_extract_hist<25, true>),
                                   dim3((u32)B), dim3(256), 0, s, d_in, n,
                                   d_keys, d_idx, hist);
            else if (!le)
                hipLaunchKernelGGL((k_extract_hist<32, false>),
                                   dim3((u32)B), dim3(256), 0, s, d_in, n,
                                   d_keys, d_idx, hist);
            else
                hipLaunchKernelGGL((k_extract_hist<32, true>),
                                   dim3((u32)B), dim3(256), 0, s, d_in, n,
                                   d_keys, d_idx, hist));
        T9_LAUNCH_CHECK();
        rc = t9i_sort_pairs_msb_ph(ctx, d_keys, d_idx, n, pair_ws,
                                   stream);
    }
    else if (fused && !le) {
        rc = t9i_sort_recs_msb(ctx, d_in, rec_size, d_keys, d_idx, n,
                               pair_ws, stream);
    }
    else {
        rc = le ? t9_extract_key64_le(ctx, d_in, n, rec_size, 0, d_keys,
                                      d_idx, stream)
                : t9_extract_key64(ctx, d_in, n, rec_size, 0, d_keys,
                                   d_idx, stream);
        if (rc) return rc;
        rc = t9_sort_pairs_u64_u32(ctx, d_keys, d_idx, n, pair_ws, stream);
    }
    if (rc) return rc;
    rc = t9i_count_tied(d_keys, n, d_ntied, s);
    if (rc) return rc;
    u32 ntied = 0;
    HIP_TRY(hipMemcpyAsync(&ntied, d_ntied, 4, hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));

    if (ntied && rec_size > 8) {
        /* on-device tie sort (kernels + plan above k_tie_flags). Scratch
         * lives in d_out, which is dead until the final gather (~21 B
         * per element vs rec_size B available); records narrower than
         * 24 B fall back to a temporary allocation. */
        const u64 need = 6 * t9_align256(n * 4) +
                         2 * t9_align256(n * 8) +
                         t9_align256(3 * 8) + 256 +
                         t9_align256((n / T9_MAXSCAN_TILE + 2) * 4) + 256 +
                         t9_partition_idx_workspace(n);
        void* tmp_alloc = nullptr;
        char* q;
        if ((u64)n * rec_size >= need) {
            q = (char*)d_out;
        }
        else {
            HIP_TRY(hipMalloc(&tmp_alloc, need));
            q = (char*)tmp_alloc;
        }
        u32* t_bucket = (u32*)q; q += t9_align256(n * 4);
        u32* t_perm = (u32*)q;   q += t9_align256(n * 4);
        u32* t_idx = (u32*)q;    q += t9_align256(n * 4);
        u64* t_chunk = (u64*)q;  q += t9_align256(n * 8);
        u64* t_offs = (u64*)q;   q += t9_align256(3 * 8);
        u32* t_flag = (u32*)q;   q += 256;
        /* MSD strategy state (t_bucket doubles as the iota/permutation
         * buffer once the partition is done) */
        u32* t_grp = (u32*)q;    q += t9_align256(n * 4);
        u32* t_grp2 = (u32*)q;   q += t9_align256(n * 4);
        u32* t_tmp4 = (u32*)q;   q += t9_align256(n * 4);
        u64* t_chunk2 = (u64*)q; q += t9_align256(n * 8);
        u32* t_part = (u32*)q;
        q += t9_align256((n / T9_MAXSCAN_TILE + 2) * 4);
        u32* t_ties = (u32*)q;   q += 256;
        void* t_pws = q;

        const u32 ngrid =
            (u32)((t9_ceil_div(n, 256) < 4096) ? t9_ceil_div(n, 256) : 4096);
        T9_PERF_WRAP(s, "tie_partition",
                     hipLaunchKernelGGL(k_tie_flags, dim3(ngrid),
                                        dim3(256), 0, s, d_keys, n,
                                        t_bucket));
        T9_LAUNCH_CHECK();
        rc = t9_partition_idx(ctx, t_bucket, n, 2, t_perm, t_offs, t_pws,
                              stream);
        u64 m = 0;
        if (!rc) {
            HIP_TRY(hipMemcpyAsync(&m, t_offs + 1, 8,
                                   hipMemcpyDeviceToHost, s));
            HIP_TRY(hipStreamSynchronize(s));
        }
        if (!rc && m >= 2) {
            rc = t9_gather_records(ctx, (const u8*)d_idx, t_perm, m, 4,
                                   (u8*)t_idx, stream);
            const u32 mgrid =
                (u32)((t9_ceil_div(m, 256) < 4096) ? t9_ceil_div(m, 256)
                                                   : 4096);
            /* prescan: which chunks differ at all (one read of the tied
             * records); then LSD over the differing tail chunks (last ->
             * first) and, if >= 2 runs, the prefix regroup pass */
            const u32 nc = (rec_size > 8)
                               ? (u32)t9_ceil_div(rec_size - 8, 8) : 0;
            u32 hflags[64];
            if (nc <= 63) {
                HIP_TRY(hipMemsetAsync(t_flag, 0, 64 * 4, s));
                T9_PERF_WRAP(
                    s, "tie_prescan",
                    if (m == n && rec_size / 4 <= 160)
                        /* every record tied: fully coalesced word scan */
                        hipLaunchKernelGGL(k_tie_prescan_seq, dim3(8192),
                                           dim3(256), 0, s, d_in, t_idx,
                                           m, rec_size, nc, t_flag);
                    else
                        hipLaunchKernelGGL(k_tie_prescan, dim3(mgrid),
                                           dim3(256), 0, s, d_in, t_idx,
                                           m, rec_size, nc, t_flag));
                T9_LAUNCH_CHECK();
                HIP_TRY(hipMemcpyAsync(hflags, t_flag, (nc + 1) * 4,
                                       hipMemcpyDeviceToHost, s));
                HIP_TRY(hipStreamSynchronize(s));
            }
            else {
                for (u32 c = 0; c < 64; ++c) hflags[c] = 1;
            }
            /* differing tail chunks, most-significant first */
            std::vector<u32> tailC;
            if (nc <= 63) {
                for (u32 c = 1; c <= nc; ++c)
                    if (hflags[c]) tailC.push_back(c - 1);
            }
            else {
                for (u32 c = 0; c < nc; ++c) tailC.push_back(c);
            }
            const char* me_ = getenv("T9_TIE_MSD");
            const bool use_msd = !(me_ && me_[0] == '0') &&
                                 tailC.size() >= 2;
            bool msd_done = false;
            if (use_msd && !rc) {
                /* init groups = runs of equal prefix (gather the tied
                 * prefixes, boundary where they change) */
                rc = t9_gather_records(ctx, (const u8*)d_keys, t_perm, m,
                                       8, (u8*)t_chunk, stream);
                const u32 B2k = (u32)t9_ceil_div(m, T9_MAXSCAN_TILE);
                auto regroup = [&](const u32* old_grp, const u64* ck,
                                   u32* out_grp, u32* h_ties) -> int {
                    if (hipMemsetAsync(t_ties, 0, 4, s) != hipSuccess)
                        return T9_EIO;
                    hipLaunchKernelGGL(k_tie_bound_vals, dim3(mgrid),
                                       dim3(256), 0, s, old_grp, ck, m,
                                       out_grp, t_ties);
                    hipLaunchKernelGGL(k_maxscan_local, dim3(B2k),
                                       dim3(256), 0, s, out_grp, m,
                                       t_part);
                    hipLaunchKernelGGL(k_maxscan_part, dim3(1), dim3(256),
                                       0, s, t_part, B2k);
                    hipLaunchKernelGGL(k_maxscan_fix, dim3(mgrid),
                                       dim3(256), 0, s, out_grp, m,
                                       t_part);
                    T9_LAUNCH_CHECK();
                    if (!h_ties) return T9_OK;
                    if (hipMemcpyAsync(h_ties, t_ties, 4,
                                       hipMemcpyDeviceToHost, s) !=
                            hipSuccess ||
                        hipStreamSynchronize(s) != hipSuccess)
                        return T9_EIO;
                    return T9_OK;
                };
                if (!rc) rc = regroup(nullptr, t_chunk, t_grp, nullptr);
                u32* jbuf = t_bucket;   /* partition is done; reuse */
                size_t ci = 0;
                for (; ci < tailC.size() && ci < 2 && !rc; ++ci) {
                    const u32 off = 8 + tailC[ci] * 8;
                    hipLaunchKernelGGL((k_tie_chunk<false>), dim3(mgrid),
                                       dim3(256), 0, s, d_in, t_idx, m,
                                       rec_size, off, t_chunk);
                    hipLaunchKernelGGL(k_iota32, dim3(mgrid), dim3(256),
                                       0, s, jbuf, m);
                    T9_LAUNCH_CHECK();
                    rc = t9_sort_pairs_u64_u32(ctx, t_chunk, jbuf, m,
                                               pair_ws, stream);
                    if (rc) break;
                    hipLaunchKernelGGL(k_msd_apply2, dim3(mgrid),
                                       dim3(256), 0, s, jbuf, m, t_idx,
                                       t_tmp4, t_grp, t_grp2);
                    hipLaunchKernelGGL(k_grp_keys, dim3(mgrid), dim3(256),
                                       0, s, t_grp2, m, t_chunk2);
                    hipLaunchKernelGGL(k_iota32, dim3(mgrid), dim3(256),
                                       0, s, jbuf, m);
                    T9_LAUNCH_CHECK();
                    rc = t9_sort_pairs_u64_u32(ctx, t_chunk2, jbuf, m,
                                               pair_ws, stream);
                    if (rc) break;
                    hipLaunchKernelGGL(k_msd_apply3, dim3(mgrid),
                                       dim3(256), 0, s, jbuf, m, t_tmp4,
                                       t_idx, t_grp2, t_grp, t_chunk,
                                       t_chunk2);
                    T9_LAUNCH_CHECK();
                    u32 ties = 0;
                    rc = regroup(t_grp, t_chunk2, t_grp2, &ties);
                    if (rc) break;
                    std::swap(t_grp, t_grp2);
                    if (ties == 0) {
                        msd_done = true;
                        ++ci;
                        break;
                    }
                }
                if (!rc && !msd_done && ci >= tailC.size())
                    msd_done = true;   /* all chunks processed: remaining
                                          ties are true duplicates,
                                          stable order is correct */
            }
            if (!msd_done) {
                /* LSD over the differing chunks (last -> first), then
                 * the prefix regroup pass — also the MSD fallback for
                 * tie patterns that persist past 2 levels (bounded) */
                for (u32 c = nc + 1; !rc && c-- > 0;) {
                    if (nc <= 63 && !hflags[c]) continue;
                    const u32 off = c ? 8 + (c - 1) * 8 : 0;
                    if (off == 0 && le)
                        hipLaunchKernelGGL((k_tie_chunk<true>),
                                           dim3(mgrid), dim3(256), 0, s,
                                           d_in, t_idx, m, rec_size, off,
                                           t_chunk);
                    else
                        hipLaunchKernelGGL((k_tie_chunk<false>),
                                           dim3(mgrid), dim3(256), 0, s,
                                           d_in, t_idx, m, rec_size, off,
                                           t_chunk);
                    T9_LAUNCH_CHECK();
                    rc = t9_sort_pairs_u64_u32(ctx, t_chunk, t_idx, m,
                                               pair_ws, stream);
                }
            }
            if (!rc) {
                hipLaunchKernelGGL(k_tie_scatter, dim3(mgrid), dim3(256),
                                   0, s, d_idx, t_perm, t_idx, m);
                T9_LAUNCH_CHECK();
            }
        }
        if (tmp_alloc) {
            HIP_TRY(hipStreamSynchronize(s));
            HIP_TRY(hipFree(tmp_alloc));
        }
        if (rc) return rc;
    }

    return t9_gather_records(ctx, d_in, d_idx, n, rec_size, d_out, stream);
}

} /* extern "C" */
