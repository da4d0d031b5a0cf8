/* t9_sort.hip — radix sort dispatch, LSD pipeline + splitter
 * classification kernels, gfx950.
 *
 * MI355X-native replacement for the reference's local sort path
 * (thrill/api/sort.hpp:665-786 SortAndWriteToFile/std::sort + PushData
 * loser-tree merge via core/multiway_merge.hpp:30-116): the whole per-GPU
 * partition is sorted in one radix pipeline, so the run/merge split of
 * the reference — an artifact of bounded RAM — vanishes in 288 GB HBM3E.
 * Large n dispatches to the two-level MSB pipeline (t9_sort_msb.hip);
 * this file holds the 8-pass LSD pipeline (small n + skew fallback), the
 * shared hist/scan kernels and the classification/partition kernels.
 *
 * Classification (k_classify) replaces TransmitItems' tree-descent +
 * EqualSampleGreaterIndex walk (thrill/api/sort.hpp:434-535, :424-426) by
 * the equivalent closed form: bucket = #{ j : (splitter_key[j],
 * splitter_idx[j]) < (key, gidx) lexicographically }. Equivalence is
 * cross-checked against the literal tree restatement in the oracle tests.
 *
 * Roofline: integer/byte work, HBM-bound; no MFMA (BASELINE.json
 * north_star). LSD moves hist 8 B + scatter 24 B per pair per pass over
 * 8 passes; the MSB pipeline replaces that with 2 passes + an LDS-only
 * level.
 */

#include "t9_common.h"
#include "t9_rank_scatter.h"

#include <algorithm>
#include <cstdlib>
#include <cstring>

/* ------------------------------------------------------------------ *
 * kernels
 * ------------------------------------------------------------------ */

template <int TILE, bool EXT_DIGIT>
__global__ __launch_bounds__(256) void k_hist(
    const u64* __restrict__ in_keys, const u32* __restrict__ ext_digit,
    u64 n, u32 shift, u32* __restrict__ hist) {
    __shared__ u32 s_cnt[T9_RADIX];
    const u32 tid = threadIdx.x;
    const u64 base = (u64)blockIdx.x * TILE;
    const u32 tn = (u32)((n - base < (u64)TILE) ? (n - base) : (u64)TILE);
    s_cnt[tid] = 0;
    __syncthreads();
    const u32 lane = tid & 63;
    for (u32 i0 = 0; i0 < (u32)TILE; i0 += 256) {
        const u32 i = i0 + tid;
        const bool valid = i < tn;
        u32 d = 0;
        if (valid)
            d = EXT_DIGIT ? ext_digit[base + i]
                          : ((u32)(in_keys[base + i] >> shift) & 255u);
        t9_hist_ballot_add<8>(s_cnt, d, valid, lane);
    }
    __syncthreads();
    hist[(u64)blockIdx.x * T9_RADIX + tid] = s_cnt[tid];
}

/* column (per-digit) partial sums over T9_SCAN_CHUNK hist rows */
__global__ __launch_bounds__(256) void k_colsum(
    const u32* __restrict__ hist, u64 B, u32* __restrict__ chunkpart) {
    const u32 tid = threadIdx.x;
    const u64 r0 = (u64)blockIdx.x * T9_SCAN_CHUNK;
    const u64 r1 = (r0 + T9_SCAN_CHUNK < B) ? r0 + T9_SCAN_CHUNK : B;
    u32 sum = 0;
    for (u64 r = r0; r < r1; ++r) sum += hist[r * T9_RADIX + tid];
    chunkpart[(u64)blockIdx.x * T9_RADIX + tid] = sum;
}

/* single block: exclusive scan of chunk partials per digit (in place) and
 * exclusive digit base offsets from the column totals */
__global__ __launch_bounds__(256) void k_chunkscan(
    u32* __restrict__ chunkpart, u64 Bc, u32* __restrict__ digit_base) {
    const u32 tid = threadIdx.x;
    u32 running = 0;
    for (u64 c = 0; c < Bc; ++c) {
        u32 v = chunkpart[c * T9_RADIX + tid];
        chunkpart[c * T9_RADIX + tid] = running;
        running += v;
    }
    __shared__ u32 s[T9_RADIX];
    s[tid] = running;
    __syncthreads();
    for (int off = 1; off < T9_RADIX; off <<= 1) {
        u32 y = (tid >= (u32)off) ? s[tid - off] : 0;
        __syncthreads();
        s[tid] += y;
        __syncthreads();
    }
    digit_base[tid] = s[tid] - running;   /* exclusive */
}

/* rewrite hist rows into final global exclusive offsets per (block, digit) */
__global__ __launch_bounds__(256) void k_finaloffs(
    u32* __restrict__ hist, u64 B, const u32* __restrict__ chunkpart,
    const u32* __restrict__ digit_base) {
    const u32 tid = threadIdx.x;
    const u64 r0 = (u64)blockIdx.x * T9_SCAN_CHUNK;
    const u64 r1 = (r0 + T9_SCAN_CHUNK < B) ? r0 + T9_SCAN_CHUNK : B;
    u32 run = chunkpart[(u64)blockIdx.x * T9_RADIX + tid] + digit_base[tid];
    for (u64 r = r0; r < r1; ++r) {
        u32 v = hist[r * T9_RADIX + tid];
        hist[r * T9_RADIX + tid] = run;
        run += v;
    }
}

/* Stable scatter of one radix pass. Each 256-thread block owns a TILE:
 * (1) load + LDS digit cache + per-digit counts, (2) LDS exclusive scan,
 * (3) stable intra-tile ranking — per wave a ballot over the 8 digit bits
 * yields the same-digit lane mask; rank = running count + earlier-wave
 * counts + popcount of lower same-digit lanes — and reorder into an LDS
 * staging tile, (4) digit-run coalesced global writes at the scanned
 * offsets.
 * STAGE_IN=false drops the input staging tiles and re-reads the inputs
 * from global in phase (3) — the block just read them, so the re-read is
 * an L1/L2 hit; the freed LDS doubles the tile at equal occupancy, which
 * doubles the average digit-run length and so the write coalescing. */
template <int TILE, bool HAS_KEY, bool HAS_VAL, bool EXT_DIGIT,
          bool IOTA_VAL, bool STAGE_IN>
__global__ __launch_bounds__(256, 2) void k_scatter(
    const u64* __restrict__ in_keys, const u32* __restrict__ in_vals,
    const u32* __restrict__ ext_digit, u64* __restrict__ out_keys,
    u32* __restrict__ out_vals, const u32* __restrict__ offs, u64 n,
    u32 shift) {
    constexpr int CHUNKS = TILE / 256;
    __shared__ u64 s_keys[(HAS_KEY && STAGE_IN) ? TILE : 1];
    __shared__ u32 s_vals[(HAS_VAL && STAGE_IN) ? TILE : 1];
    __shared__ u64 s_okeys[HAS_KEY ? TILE : 1];
    __shared__ u32 s_ovals[HAS_VAL ? TILE : 1];
    __shared__ u8 s_dig[TILE];
    __shared__ u8 s_digof[TILE];
    __shared__ u32 s_cnt[T9_RADIX];
    __shared__ u32 s_start[T9_RADIX];
    __shared__ u32 s_run[T9_RADIX];
    __shared__ u32 s_wavecnt[4 * T9_RADIX];
    __shared__ u32 s_goff[T9_RADIX];

    const u32 tid = threadIdx.x;
    const u64 base = (u64)blockIdx.x * TILE;
    const u32 tn = (u32)((n - base < (u64)TILE) ? (n - base) : (u64)TILE);
    const u32 wave = tid >> 6, lane = tid & 63;

    s_cnt[tid] = 0;
    s_run[tid] = 0;
    s_goff[tid] = offs[(u64)blockIdx.x * T9_RADIX + tid];
    __syncthreads();

    for (int c = 0; c < CHUNKS; ++c) {
        u32 i = c * 256 + tid;
        if (i < tn) {
            u64 k = 0;
            if (HAS_KEY) {
                k = in_keys[base + i];
                if (STAGE_IN) s_keys[i] = k;
            }
            u32 d = EXT_DIGIT ? ext_digit[base + i]
                              : ((u32)(k >> shift) & 255u);
            if (HAS_VAL && STAGE_IN)
                s_vals[i] = IOTA_VAL ? (u32)(base + i) : in_vals[base + i];
            s_dig[i] = (u8)d;
            atomicAdd(&s_cnt[d], 1u);
        }
    }
    __syncthreads();

    /* exclusive scan of s_cnt into s_start (Hillis-Steele, in place) */
    {
        u32 x = s_cnt[tid];
        s_start[tid] = x;
        __syncthreads();
        for (int off = 1; off < T9_RADIX; off <<= 1) {
            u32 y = (tid >= (u32)off) ? s_start[tid - off] : 0;
            __syncthreads();
            s_start[tid] += y;
            __syncthreads();
        }
        u32 incl = s_start[tid];
        __syncthreads();
        s_start[tid] = incl - x;
    }
    __syncthreads();

    for (int c = 0; c < CHUNKS; ++c) {
        u32 i = c * 256 + tid;
        bool valid = i < tn;
        u32 d = valid ? (u32)s_dig[i] : 0u;
        for (int w = 0; w < 4; ++w) s_wavecnt[w * T9_RADIX + tid] = 0;
        __syncthreads();
        u64 vmask = __ballot(valid);
        u64 m = vmask;
        for (int bit = 0; bit < 8; ++bit) {
            u64 bb = __ballot((d >> bit) & 1u);
            m &= ((d >> bit) & 1u) ? bb : ~bb;
        }
        u32 wave_rank = (u32)__popcll(m & ((1ull << lane) - 1ull));
        if (valid && wave_rank == 0)
            s_wavecnt[wave * T9_RADIX + d] = (u32)__popcll(m);
        __syncthreads();
        if (valid) {
            u32 before = 0;
            for (u32 w = 0; w < wave; ++w)
                before += s_wavecnt[w * T9_RADIX + d];
            u32 pos = s_start[d] + s_run[d] + before + wave_rank;
            if (HAS_KEY)
                s_okeys[pos] = STAGE_IN ? s_keys[i] : in_keys[base + i];
            if (HAS_VAL)
                s_ovals[pos] = STAGE_IN ? s_vals[i]
                               : (IOTA_VAL ? (u32)(base + i)
                                           : in_vals[base + i]);
            s_digof[pos] = (u8)d;
        }
        __syncthreads();
        u32 tot = 0;
        for (int w = 0; w < 4; ++w) tot += s_wavecnt[w * T9_RADIX + tid];
        __syncthreads();
        s_run[tid] += tot;
        __syncthreads();
    }

    for (int c = 0; c < CHUNKS; ++c) {
        u32 j = c * 256 + tid;
        if (j < tn) {
            u32 d = s_digof[j];
            u64 gpos = (u64)s_goff[d] + (j - s_start[d]);
            if (HAS_KEY) out_keys[gpos] = s_okeys[j];
            if (HAS_VAL) out_vals[gpos] = s_ovals[j];
        }
    }
}

/* Wave-autonomous stable scatter: each of the 4 waves owns a contiguous
 * quarter of the tile and ranks it with NO block barriers — ballot over
 * the digit bits gives the same-digit lane mask, a wave-private LDS
 * counter row carries the running per-digit count (wave-ordered LDS, no
 * races). One barrier, a cross-wave combine (exclusive digit scan +
 * per-wave bases), one barrier, then the LDS reorder + digit-run
 * coalesced global writes. Replaces the chunk-serialized k_scatter rank
 * loop (4 barriers x TILE/256 chunks) which measured latency-bound. */
template <int TILE, bool HAS_KEY, bool HAS_VAL, bool STAGE_IN>
__global__ __launch_bounds__(256, 2) void k_scatter_wave(
    const u64* __restrict__ in_keys, const u32* __restrict__ in_vals,
    u64* __restrict__ out_keys, u32* __restrict__ out_vals,
    const u32* __restrict__ offs, u64 n, u32 shift) {
    constexpr int NW = 4;
    constexpr int SUB = TILE / NW;
    constexpr int GROUPS = SUB / 64;
    __shared__ u64 s_keys[(HAS_KEY && STAGE_IN) ? TILE : 1];
    __shared__ u32 s_vals[(HAS_VAL && STAGE_IN) ? TILE : 1];
    __shared__ u64 s_okeys[HAS_KEY ? TILE : 1];
    __shared__ u32 s_ovals[HAS_VAL ? TILE : 1];
    __shared__ u8 s_dig[TILE];
    __shared__ u16 s_rank[TILE];
    __shared__ u8 s_digof[TILE];
    __shared__ u32 s_wcnt[NW * T9_RADIX];
    __shared__ u32 s_woff[NW * T9_RADIX];
    __shared__ u32 s_start[T9_RADIX];
    __shared__ u32 s_goff[T9_RADIX];

    const u32 tid = threadIdx.x, wave = tid >> 6, lane = tid & 63;
    const u64 base = (u64)blockIdx.x * TILE;
    const u32 tn = (u32)((n - base < (u64)TILE) ? (n - base) : (u64)TILE);

    s_goff[tid] = offs[(u64)blockIdx.x * T9_RADIX + tid];
    for (u32 t = lane; t < T9_RADIX; t += 64) s_wcnt[wave * T9_RADIX + t] = 0;
    /* no barrier needed: each wave touches only its own s_wcnt row until
     * the combine barrier below */

    const u32 wbase = wave * SUB;
    for (int g = 0; g < GROUPS; ++g) {
        const u32 i = wbase + g * 64 + lane;
        const bool valid = i < tn;
        u64 k = 0;
        u32 d = 0;
        if (valid) {
            k = in_keys[base + i];
            d = (u32)(k >> shift) & 255u;
            if (STAGE_IN && HAS_KEY) s_keys[i] = k;
            if (STAGE_IN && HAS_VAL) s_vals[i] = in_vals[base + i];
            s_dig[i] = (u8)d;
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
                s_wcnt[wave * T9_RADIX + d] =
                    before + (u32)__popcll(m);
        }
    }
    __syncthreads();

    /* combine: thread tid owns digit tid */
    const u32 c0 = s_wcnt[0 * T9_RADIX + tid];
    const u32 c1 = s_wcnt[1 * T9_RADIX + tid];
    const u32 c2 = s_wcnt[2 * T9_RADIX + tid];
    const u32 c3 = s_wcnt[3 * T9_RADIX + tid];
    const u32 total = c0 + c1 + c2 + c3;
    /* exclusive scan of totals across digits (Hillis-Steele in s_start) */
    s_start[tid] = total;
    __syncthreads();
    for (int off = 1; off < T9_RADIX; off <<= 1) {
        u32 y = (tid >= (u32)off) ? s_start[tid - off] : 0;
        __syncthreads();
        s_start[tid] += y;
        __syncthreads();
    }
    const u32 excl = s_start[tid] - total;
    __syncthreads();
    s_start[tid] = excl;
    s_woff[0 * T9_RADIX + tid] = excl;
    s_woff[1 * T9_RADIX + tid] = excl + c0;
    s_woff[2 * T9_RADIX + tid] = excl + c0 + c1;
    s_woff[3 * T9_RADIX + tid] = excl + c0 + c1 + c2;
    __syncthreads();

    for (int g = 0; g < GROUPS; ++g) {
        const u32 i = wbase + g * 64 + lane;
        if (i < tn) {
            const u32 d = s_dig[i];
            const u32 pos = s_woff[wave * T9_RADIX + d] + s_rank[i];
            if (HAS_KEY)
                s_okeys[pos] = STAGE_IN ? s_keys[i] : in_keys[base + i];
            if (HAS_VAL)
                s_ovals[pos] = STAGE_IN ? s_vals[i] : in_vals[base + i];
            s_digof[pos] = (u8)d;
        }
    }
    __syncthreads();

    constexpr int CHUNKS = TILE / 256;
    for (int c = 0; c < CHUNKS; ++c) {
        const u32 j = c * 256 + tid;
        if (j < tn) {
            const u32 d = s_digof[j];
            const u64 gpos = (u64)s_goff[d] + (j - s_start[d]);
            if (HAS_KEY) out_keys[gpos] = s_okeys[j];
            if (HAS_VAL) out_vals[gpos] = s_ovals[j];
        }
    }
}

#include "t9_rank_scatter.h"

/* bucket offsets (u64, p+1 entries) from the digit base array */
__global__ __launch_bounds__(512) void k_bucket_offsets(
    const u32* __restrict__ digit_base, u32 p, u64 n,
    u64* __restrict__ offsets) {
    u32 i = threadIdx.x;
    if (i < p) offsets[i] = digit_base[i];
    if (i == p) offsets[p] = n;
}

/* classification: closed form of TransmitItems (api/sort.hpp:434-535) */
__global__ __launch_bounds__(256) void k_classify(
    const u64* __restrict__ keys, u64 n, u64 gidx0,
    const u64* __restrict__ spl_k, const u64* __restrict__ spl_i, u32 p,
    u32* __restrict__ bucket, u64* __restrict__ counts) {
    __shared__ u64 sk[T9_RADIX], si[T9_RADIX];
    __shared__ u32 scnt[T9_RADIX];
    const u32 tid = threadIdx.x;
    if (tid < p - 1) {
        sk[tid] = spl_k[tid];
        si[tid] = spl_i[tid];
    }
    scnt[tid] = 0;
    __syncthreads();
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + tid; i < n; i += stride) {
        u64 k = keys[i], g = gidx0 + i;
        u32 b = 0;
        for (u32 j = 0; j < p - 1; ++j)
            if (sk[j] < k || (sk[j] == k && si[j] < g)) b = j + 1;
        bucket[i] = b;
        atomicAdd(&scnt[b], 1u);
    }
    __syncthreads();
    if (tid < p && scnt[tid])
        atomicAdd((unsigned long long*)&counts[tid],
                  (unsigned long long)scnt[tid]);
}

/* Record classification under the acceptance total order: primary compare
 * on the precomputed big-endian u64 key prefix, byte fallback over the
 * remaining record bytes only on prefix ties (the reference classifies
 * with its comparator on whole items — api/sort.hpp:480 compare_function_ —
 * which under the acceptance order is full-record lexicographic), then the
 * splitter-index tiebreak (:487-501). */
__global__ __launch_bounds__(256) void k_classify_rec(
    const u8* __restrict__ recs, const u64* __restrict__ k64, u64 n,
    u64 gidx0, const u8* __restrict__ spl_recs,
    const u64* __restrict__ spl_k64, const u64* __restrict__ spl_idx, u32 p,
    u32 rec_size, u32* __restrict__ bucket, u64* __restrict__ counts) {
    __shared__ u64 sk[T9_RADIX], si[T9_RADIX];
    __shared__ u32 scnt[T9_RADIX];
    const u32 tid = threadIdx.x;
    if (tid < p - 1) {
        sk[tid] = spl_k64[tid];
        si[tid] = spl_idx[tid];
    }
    scnt[tid] = 0;
    __syncthreads();
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + tid; i < n; i += stride) {
        const u64 k = k64[i], g = gidx0 + i;
        u32 b = 0;
        for (u32 j = 0; j < p - 1; ++j) {
            bool less;
            if (sk[j] != k) {
                less = sk[j] < k;
            }
            else {
                const u8* a = spl_recs + (u64)j * rec_size + 8;
                const u8* r = recs + i * (u64)rec_size + 8;
                int c = 0;
                for (u32 t = 8; t < rec_size && c == 0; ++t) {
                    u8 x = *a++, y = *r++;
                    c = (x > y) - (x < y);
                }
                less = c < 0 || (c == 0 && si[j] < g);
            }
            if (less) b = j + 1;
        }
        bucket[i] = b;
        atomicAdd(&scnt[b], 1u);
    }
    __syncthreads();
    if (tid < p && scnt[tid])
        atomicAdd((unsigned long long*)&counts[tid],
                  (unsigned long long)scnt[tid]);
}

/* count elements whose u64 key equals their left neighbour's (tie probe) */
__global__ __launch_bounds__(256) void k_count_tied(
    const u64* __restrict__ keys, u64 n, u32* __restrict__ ntied) {
    __shared__ u32 s;
    if (threadIdx.x == 0) s = 0;
    __syncthreads();
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < n; i += stride)
        if (i > 0 && keys[i] == keys[i - 1]) atomicAdd(&s, 1u);
    __syncthreads();
    if (threadIdx.x == 0 && s) atomicAdd(ntied, s);
}

/* ------------------------------------------------------------------ *
 * host orchestration
 * ------------------------------------------------------------------ */

extern "C" u64 t9i_sort_pairs_msb_workspace(u64 n);

namespace {

struct ScanWs {
    u32* hist;
    u32* chunkpart;
    u32* digit_base;
    u64 B, Bc;
};

u64 scan_ws_bytes(u64 B) {
    u64 Bc = t9_ceil_div(B, T9_SCAN_CHUNK);
    return t9_align256(B * T9_RADIX * 4) + t9_align256(Bc * T9_RADIX * 4) +
           t9_align256(T9_RADIX * 4);
}

/* carve buffers sized for B_cap rows but use B_used (<= B_cap) rows */
ScanWs carve_scan_ws(char*& p, u64 B_cap, u64 B_used) {
    ScanWs w;
    w.B = B_used;
    w.Bc = t9_ceil_div(B_used, T9_SCAN_CHUNK);
    w.hist = (u32*)p;
    p += t9_align256(B_cap * T9_RADIX * 4);
    w.chunkpart = (u32*)p;
    p += t9_align256(t9_ceil_div(B_cap, T9_SCAN_CHUNK) * T9_RADIX * 4);
    w.digit_base = (u32*)p;
    p += t9_align256(T9_RADIX * 4);
    return w;
}

int env_variant(const char* name, int dflt, int maxv) {
    const char* e = getenv(name);
    if (!e) return dflt;
    int v = atoi(e);
    return (v >= 1 && v <= maxv) ? v : dflt;
}

int run_scan(const ScanWs& w, hipStream_t s) {
    hipLaunchKernelGGL(k_colsum, dim3((u32)w.Bc), dim3(256), 0, s, w.hist,
                       w.B, w.chunkpart);
    hipLaunchKernelGGL(k_chunkscan, dim3(1), dim3(256), 0, s, w.chunkpart,
                       w.Bc, w.digit_base);
    hipLaunchKernelGGL(k_finaloffs, dim3((u32)w.Bc), dim3(256), 0, s,
                       w.hist, w.B, w.chunkpart, w.digit_base);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

} // namespace

extern "C" {

u64 t9i_sort_keys_lsd_workspace(u64 n) {
    if (n < 2) return 256;
    u64 B = t9_ceil_div(n, T9_KEYS_TILE);
    return t9_align256(n * 8) + scan_ws_bytes(B);
}

/* MSB keys implementation (t9_sort_msb.hip) */
int t9i_sort_keys_msb(t9_context*, u64*, u64, void*, void*);
int t9i_sort_keys_lsd(t9_context*, u64*, u64, void*, void*);

u64 t9_sort_u64_workspace(u64 n) {
    if (n < 2) return 256;
    u64 a = t9i_sort_keys_lsd_workspace(n);
    u64 b = t9i_sort_pairs_msb_workspace(n);
    return a > b ? a : b;
}

/* algorithm dispatch, as for pairs */
int t9_sort_u64(t9_context* ctx, u64* d_keys, u64 n, void* d_workspace,
                void* stream) {
    if (n < 2) return T9_OK;
    if (!d_keys || !d_workspace || n >= (1ull << 32)) return T9_EINVAL;
    const char* e = getenv("T9_SORT_ALGO");
    bool use_msb = n >= (1ull << 22);
    if (e && strcmp(e, "lsd") == 0) use_msb = false;
    if (e && strcmp(e, "msb") == 0) use_msb = n >= (1ull << 14);
    if (use_msb)
        return t9i_sort_keys_msb(ctx, d_keys, n, d_workspace, stream);
    return t9i_sort_keys_lsd(ctx, d_keys, n, d_workspace, stream);
}

int t9i_sort_keys_lsd(t9_context* ctx, u64* d_keys, u64 n,
                      void* d_workspace, void* stream) {
    (void)ctx;
    if (n < 2) return T9_OK;
    if (!d_keys || !d_workspace || n >= (1ull << 32)) return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    char* p = (char*)d_workspace;
    u64* alt = (u64*)p;
    p += t9_align256(n * 8);
    const int var = env_variant("T9_KEYS_SCATTER", 5, 5);
    const u64 tile = (var == 2 || var == 5) ? 2 * T9_KEYS_TILE
                                            : T9_KEYS_TILE;
    const u64 B = t9_ceil_div(n, tile);
    ScanWs w = carve_scan_ws(p, t9_ceil_div(n, (u64)T9_KEYS_TILE), B);

    u64* bufA = d_keys;
    u64* bufB = alt;
    for (int pass = 0; pass < 8; ++pass) {
        u32 shift = pass * 8;
        T9_PERF_WRAP(
            s, "hist_keys",
            if (var == 2 || var == 5)
                hipLaunchKernelGGL((k_hist<2 * T9_KEYS_TILE, false>),
                                   dim3((u32)B), dim3(256), 0, s, bufA,
                                   nullptr, n, shift, w.hist);
            else
                hipLaunchKernelGGL((k_hist<T9_KEYS_TILE, false>),
                                   dim3((u32)B), dim3(256), 0, s, bufA,
                                   nullptr, n, shift, w.hist));
        int rc = run_scan(w, s);
        if (rc) return rc;
        T9_PERF_WRAP(
            s, "keys_scatter",
            if (var == 1)
                hipLaunchKernelGGL(
                    (k_scatter<T9_KEYS_TILE, true, false, false, false,
                               true>),
                    dim3((u32)B), dim3(256), 0, s, bufA, nullptr, nullptr,
                    bufB, nullptr, w.hist, n, shift);
            else if (var == 2)
                hipLaunchKernelGGL(
                    (k_scatter<2 * T9_KEYS_TILE, true, false, false, false,
                               false>),
                    dim3((u32)B), dim3(256), 0, s, bufA, nullptr, nullptr,
                    bufB, nullptr, w.hist, n, shift);
            else if (var == 3)
                hipLaunchKernelGGL(
                    (k_scatter_wave<T9_KEYS_TILE, true, false, false>),
                    dim3((u32)B), dim3(256), 0, s, bufA, nullptr, bufB,
                    nullptr, w.hist, n, shift);
            else if (var == 5)
                hipLaunchKernelGGL(
                    (k_scatter_wave512<2 * T9_KEYS_TILE, 1024, true,
                                       false>),
                    dim3((u32)B), dim3(1024), 0, s, bufA, nullptr, bufB,
                    nullptr, w.hist, n, shift);
            else
                hipLaunchKernelGGL(
                    (k_scatter_wave512<T9_KEYS_TILE, 512, true, false>),
                    dim3((u32)B), dim3(512), 0, s, bufA, nullptr, bufB,
                    nullptr, w.hist, n, shift));
        T9_LAUNCH_CHECK();
        std::swap(bufA, bufB);
    }
    /* 8 passes: result is back in d_keys */
    return T9_OK;
}

u64 t9i_sort_pairs_lsd_workspace(u64 n) {
    if (n < 2) return 256;
    u64 B = t9_ceil_div(n, T9_PAIRS_TILE);
    return t9_align256(n * 8) + t9_align256(n * 4) + scan_ws_bytes(B);
}

/* MSB implementation (t9_sort_msb.hip) */
u64 t9i_sort_pairs_msb_workspace(u64 n);
int t9i_sort_pairs_msb(t9_context*, u64*, u32*, u64, void*, void*);
/* LSD body, defined below */
int t9i_sort_pairs_lsd(t9_context*, u64*, u32*, u64, void*, void*);

u64 t9_sort_pairs_workspace(u64 n) {
    if (n < 2) return 256;
    u64 a = t9i_sort_pairs_lsd_workspace(n);
    u64 b = t9i_sort_pairs_msb_workspace(n);
    return a > b ? a : b;
}

/* algorithm dispatch: two-level MSB + LDS sub-sort for large n (see
 * t9_sort_msb.hip), 8-pass LSD otherwise; T9_SORT_ALGO=lsd|msb
 * overrides. */
int t9_sort_pairs_u64_u32(t9_context* ctx, u64* d_keys, u32* d_vals, u64 n,
                          void* d_workspace, void* stream) {
    if (n < 2) return T9_OK;
    if (!d_keys || !d_vals || !d_workspace || n >= (1ull << 32))
        return T9_EINVAL;
    const char* e = getenv("T9_SORT_ALGO");
    bool use_msb = n >= (1ull << 22);
    if (e && strcmp(e, "lsd") == 0) use_msb = false;
    if (e && strcmp(e, "msb") == 0) use_msb = n >= (1ull << 14);
    if (use_msb)
        return t9i_sort_pairs_msb(ctx, d_keys, d_vals, n, d_workspace,
                                  stream);
    return t9i_sort_pairs_lsd(ctx, d_keys, d_vals, n, d_workspace, stream);
}

/* the 8-pass LSD pipeline (also the fallback for skewed MSB inputs) */
int t9i_sort_pairs_lsd(t9_context* ctx, u64* d_keys, u32* d_vals, u64 n,
                       void* d_workspace, void* stream) {
    (void)ctx;
    if (n < 2) return T9_OK;
    if (!d_keys || !d_vals || !d_workspace || n >= (1ull << 32))
        return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    char* p = (char*)d_workspace;
    u64* alt_k = (u64*)p;
    p += t9_align256(n * 8);
    u32* alt_v = (u32*)p;
    p += t9_align256(n * 4);
    const int var = env_variant("T9_PAIR_SCATTER", 7, 7);
    const u64 tile = (var == 1 || var == 4) ? T9_PAIRS_TILE
                   : ((var == 3 || var == 7) ? 4 * T9_PAIRS_TILE
                                             : 2 * T9_PAIRS_TILE);
    const u64 B = t9_ceil_div(n, tile);
    ScanWs w = carve_scan_ws(p, t9_ceil_div(n, (u64)T9_PAIRS_TILE), B);

    u64* kA = d_keys;
    u64* kB = alt_k;
    u32* vA = d_vals;
    u32* vB = alt_v;
    for (int pass = 0; pass < 8; ++pass) {
        u32 shift = pass * 8;
        T9_PERF_WRAP(
            s, "hist_pairs",
            if (var == 1 || var == 4)
                hipLaunchKernelGGL((k_hist<T9_PAIRS_TILE, false>),
                                   dim3((u32)B), dim3(256), 0, s, kA,
                                   nullptr, n, shift, w.hist);
            else if (var == 3 || var == 7)
                hipLaunchKernelGGL((k_hist<4 * T9_PAIRS_TILE, false>),
                                   dim3((u32)B), dim3(256), 0, s, kA,
                                   nullptr, n, shift, w.hist);
            else
                hipLaunchKernelGGL((k_hist<2 * T9_PAIRS_TILE, false>),
                                   dim3((u32)B), dim3(256), 0, s, kA,
                                   nullptr, n, shift, w.hist));
        int rc = run_scan(w, s);
        if (rc) return rc;
        T9_PERF_WRAP(
            s, "pair_scatter",
            if (var == 1)
                hipLaunchKernelGGL(
                    (k_scatter<T9_PAIRS_TILE, true, true, false, false,
                               true>),
                    dim3((u32)B), dim3(256), 0, s, kA, vA, nullptr, kB, vB,
                    w.hist, n, shift);
            else if (var == 2)
                hipLaunchKernelGGL(
                    (k_scatter<2 * T9_PAIRS_TILE, true, true, false, false,
                               false>),
                    dim3((u32)B), dim3(256), 0, s, kA, vA, nullptr, kB, vB,
                    w.hist, n, shift);
            else if (var == 3)
                hipLaunchKernelGGL(
                    (k_scatter<4 * T9_PAIRS_TILE, true, true, false, false,
                               false>),
                    dim3((u32)B), dim3(256), 0, s, kA, vA, nullptr, kB, vB,
                    w.hist, n, shift);
            else if (var == 4)
                hipLaunchKernelGGL(
                    (k_scatter_wave<T9_PAIRS_TILE, true, true, true>),
                    dim3((u32)B), dim3(256), 0, s, kA, vA, kB, vB, w.hist,
                    n, shift);
            else if (var == 5)
                hipLaunchKernelGGL(
                    (k_scatter_wave<2 * T9_PAIRS_TILE, true, true, false>),
                    dim3((u32)B), dim3(256), 0, s, kA, vA, kB, vB, w.hist,
                    n, shift);
            else if (var == 7)
                hipLaunchKernelGGL(
                    (k_scatter_wave512<4 * T9_PAIRS_TILE, 1024, true,
                                       true>),
                    dim3((u32)B), dim3(1024), 0, s, kA, vA, kB, vB, w.hist,
                    n, shift);
            else
                hipLaunchKernelGGL(
                    (k_scatter_wave512<2 * T9_PAIRS_TILE, 512, true, true>),
                    dim3((u32)B), dim3(512), 0, s, kA, vA, kB, vB, w.hist,
                    n, shift));
        T9_LAUNCH_CHECK();
        std::swap(kA, kB);
        std::swap(vA, vB);
    }
    return T9_OK;
}

int t9_classify_u64(t9_context* ctx, const u64* d_keys, u64 n, u64 gidx0,
                    const u64* d_spl_keys, const u64* d_spl_idx, u32 p,
                    u32* d_bucket, u64* d_counts, void* stream) {
    (void)ctx;
    if (!d_counts || p < 1 || p > 256) return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_counts, 0, p * 8, s));
    if (n == 0) return T9_OK;
    if (!d_keys || !d_spl_keys || !d_spl_idx || !d_bucket) return T9_EINVAL;
    u64 want = t9_ceil_div(n, 256);
    u32 grid = (u32)((want < 2048) ? want : 2048);
    hipLaunchKernelGGL(k_classify, dim3(grid), dim3(256), 0, s, d_keys, n,
                       gidx0, d_spl_keys, d_spl_idx, p, d_bucket, d_counts);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_classify_rec(t9_context* ctx, const u8* d_recs, const u64* d_k64,
                    u64 n, u64 gidx0, const u8* d_spl_recs,
                    const u64* d_spl_k64, const u64* d_spl_idx, u32 p,
                    u32 rec_size, u32* d_bucket, u64* d_counts,
                    void* stream) {
    (void)ctx;
    if (!d_counts || p < 1 || p > 256 || rec_size < 8) return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_counts, 0, p * 8, s));
    if (n == 0) return T9_OK;
    if (!d_recs || !d_k64 || !d_spl_recs || !d_spl_k64 || !d_spl_idx ||
        !d_bucket)
        return T9_EINVAL;
    u64 want = t9_ceil_div(n, 256);
    u32 grid = (u32)((want < 2048) ? want : 2048);
    hipLaunchKernelGGL(k_classify_rec, dim3(grid), dim3(256), 0, s, d_recs,
                       d_k64, n, gidx0, d_spl_recs, d_spl_k64, d_spl_idx, p,
                       rec_size, d_bucket, d_counts);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

u64 t9_partition_idx_workspace(u64 n) {
    if (n == 0) return 256;
    u64 B = t9_ceil_div(n, T9_PAIRS_TILE);
    return scan_ws_bytes(B);
}

int t9_partition_idx(t9_context* ctx, const u32* d_bucket, u64 n, u32 p,
                     u32* d_perm, u64* d_offsets, void* d_workspace,
                     void* stream) {
    (void)ctx;
    if (!d_offsets || p < 1 || p > 256 || n >= (1ull << 32))
        return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    if (n == 0) {
        HIP_TRY(hipMemsetAsync(d_offsets, 0, (p + 1) * 8, s));
        return T9_OK;
    }
    if (!d_bucket || !d_perm || !d_workspace) return T9_EINVAL;
    char* cp = (char*)d_workspace;
    const u64 B = t9_ceil_div(n, T9_PAIRS_TILE);
    ScanWs w = carve_scan_ws(cp, B, B);
    hipLaunchKernelGGL((k_hist<T9_PAIRS_TILE, true>), dim3((u32)B),
                       dim3(256), 0, s, nullptr, d_bucket, n, 0, w.hist);
    int rc = run_scan(w, s);
    if (rc) return rc;
    hipLaunchKernelGGL(
        (k_scatter<T9_PAIRS_TILE, false, true, true, true, true>),
        dim3((u32)B), dim3(256), 0, s, nullptr, nullptr, d_bucket, nullptr,
        d_perm, w.hist, n, 0);
    hipLaunchKernelGGL(k_bucket_offsets, dim3(1), dim3(512), 0, s,
                       w.digit_base, p, n, d_offsets);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

/* internal: used by t9_sort_records (t9_records.hip) */
int t9i_count_tied(const u64* d_keys, u64 n, u32* d_ntied, hipStream_t s) {
    HIP_TRY(hipMemsetAsync(d_ntied, 0, 4, s));
    if (n < 2) return T9_OK;
    u64 want = t9_ceil_div(n, 256);
    u32 grid = (u32)((want < 2048) ? want : 2048);
    hipLaunchKernelGGL(k_count_tied, dim3(grid), dim3(256), 0, s, d_keys, n,
                       d_ntied);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

} /* extern "C" */
