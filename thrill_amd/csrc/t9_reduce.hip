/* t9_reduce.hip — device open-addressing reduce table, gfx950.
 *
 * Replaces ReduceProbingHashTable::Insert
 * (thrill/core/reduce_probing_hash_table.hpp:190-268): linear probing on
 * Hash128to64(salt, key) (thrill/common/hash.hpp:64-72; index mapping per
 * core/reduce_functional.hpp:60-72 — the (h/p) % size local index reduces
 * to a mask of the low hash bits here since the table is power-of-two and
 * single-partition per GPU; the partition split across ranks uses the same
 * h % p as the reference). The reference's grow/spill machinery
 * (:293-409) is subsumed by sizing: 288 GB HBM holds the table.
 *
 * Skew control: equal keys within a wavefront are
 * pre-combined by ballot-match + shuffle reduction, so a Zipf(1.1) head
 * key costs one atomicAdd per wave instead of 64 — the same role the
 * reference's in-table in-place reduce (:233) plays for its cache.
 * The empty-slot sentinel key 0xFFFF..F is reduced in a dedicated extra
 * slot, mirroring reduce_probing_hash_table.hpp:195-217.
 */

#include "t9_common.h"

#include <cstdlib>

#define T9_EMPTY 0xFFFFFFFFFFFFFFFFull

/* The table is ONE interleaved array t[2*(cap+1)]: slot i = (t[2i] key,
 * t[2i+1] sum), aux sentinel slot at t[2cap..2cap+1]. Interleaving keeps
 * each cold-path CAS+add inside ONE 128-B line: with split key/val
 * arrays every cold token of a big-vocab stream paid two random HBM
 * lines, and the build kernel parked 94.8% of its wave cycles on them
 * (profiles/r01_pmc_wavecycle_decomposition.txt appendix). */
__global__ __launch_bounds__(256) void k_reduce_init(u64* __restrict__ t,
                                                     u64 cap) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i <= cap; i += stride) {
        t[2 * i] = (i == cap) ? 0 : T9_EMPTY;  /* aux slot counts
                                                  sentinel-key occurrences */
        t[2 * i + 1] = 0;
    }
}

__global__ __launch_bounds__(256) void k_reduce_build(
    const u64* __restrict__ keys, const u64* __restrict__ vals, u64 n,
    u64* __restrict__ t, u64 cap, u64 salt,
    u32* __restrict__ err, int combine) {
    const u64 gsz = (u64)gridDim.x * 256;
    const u32 lane = threadIdx.x & 63;
    for (u64 base = (u64)blockIdx.x * 256; base < n; base += gsz) {
        const u64 i = base + threadIdx.x;
        const bool valid = i < n;
        const u64 k = valid ? keys[i] : 0;
        const u64 v = valid ? vals[i] : 0;

        /* lane-parallel wave combine: each lane learns in O(64) shuffles
         * whether an earlier lane holds the same key (then it is not the
         * group leader) and accumulates the values of later same-key
         * lanes. All leaders then probe/insert CONCURRENTLY — unlike a
         * leader-at-a-time loop, whose serialized atomic round-trips
         * measured ~10x slower on high-cardinality streams. */
        bool leader = valid;
        u64 gsum = v;
        u32 cnt_same = valid ? 1u : 0u;
        for (int ofs = 1; combine && ofs < 64; ++ofs) {
            const u64 k_up = __shfl_up(k, ofs);
            const int val_up = __shfl_up((int)valid, ofs);
            if (lane >= (u32)ofs && valid && val_up && k_up == k)
                leader = false;
            const u64 k_dn = __shfl_down(k, ofs);
            const u64 v_dn = __shfl_down(v, ofs);
            const int val_dn = __shfl_down((int)valid, ofs);
            if (lane + ofs < 64 && valid && val_dn && k_dn == k) {
                gsum += v_dn;
                ++cnt_same;
            }
        }
        if (leader) {
            if (k == T9_EMPTY) {
                atomicAdd((unsigned long long*)&t[2 * cap],
                          (unsigned long long)cnt_same);
                atomicAdd((unsigned long long*)&t[2 * cap + 1],
                          (unsigned long long)gsum);
            }
            else {
                u64 slot = t9_hash128to64(salt, k) & (cap - 1);
                u64 probes = 0;
                for (;;) {
                    u64 prev = atomicCAS((unsigned long long*)&t[2 * slot],
                                         (unsigned long long)T9_EMPTY,
                                         (unsigned long long)k);
                    if (prev == T9_EMPTY || prev == k) {
                        atomicAdd((unsigned long long*)&t[2 * slot + 1],
                                  (unsigned long long)gsum);
                        break;
                    }
                    slot = (slot + 1) & (cap - 1);
                    if (++probes > cap) {
                        atomicExch(err, 1u);
                        break;
                    }
                }
            }
        }
    }
}

/* LDS-accumulated build: a 2048-slot per-block LDS table absorbs the hot
 * keys of a skewed stream at LDS-atomic speed (first-come slot claim, up
 * to 4 probes); misses insert directly into the global table (cold keys
 * are near-unique, so global contention is negligible). The block flushes
 * its LDS table once at the end. Plays the role the reference's
 * in-cache pre-table plays for its CPU cache
 * (core/reduce_pre_phase.hpp). */
/* SLOTS trades hot-key coverage against occupancy (LDS bytes/block):
 * 1024 -> 8 blocks/CU, 2048 -> 4 (default), 4096 -> 2. Selected with
 * T9_LDS_SLOTS. */
template <int SLOTS, bool READFIRST = false>
__global__ __launch_bounds__(256) void k_reduce_build_lds(
    const u64* __restrict__ keys, const u64* __restrict__ vals, u64 n,
    u64* __restrict__ t, u64 cap, u64 salt,
    u32* __restrict__ err, int wavecomb) {
    constexpr int T9_LDS_SLOTS = SLOTS;
    __shared__ u64 lk[T9_LDS_SLOTS];
    __shared__ u64 lv[T9_LDS_SLOTS];
    /* wave-combine scratch: per-lane (key, val) visible to the wave's
     * group leaders via plain LDS reads (per-wave-disjoint index ranges,
     * no barrier needed) */
    __shared__ u64 sc_k[256];
    __shared__ u64 sc_v[256];
    const u32 tid = threadIdx.x;
    const u32 lane = tid & 63, wbase = tid & ~63u;
    for (u32 s = tid; s < T9_LDS_SLOTS; s += 256) {
        lk[s] = T9_EMPTY;
        lv[s] = 0;
    }
    __syncthreads();

    const u64 gsz = (u64)gridDim.x * 256;
    for (u64 base = (u64)blockIdx.x * 256; base < n; base += gsz) {
        const u64 i = base + tid;
        const bool valid = i < n;
        const u64 k = valid ? keys[i] : 0;
        const u64 v = valid ? vals[i] : 0;
        const bool sent = valid && k == T9_EMPTY;
        if (sent) {
            atomicAdd((unsigned long long*)&t[2 * cap], 1ull);
            atomicAdd((unsigned long long*)&t[2 * cap + 1],
                      (unsigned long long)v);
        }
        const bool live = valid && !sent;
        const u64 h = t9_hash128to64(salt, k);
        u32 ls = (u32)(h >> 48) & (T9_LDS_SLOTS - 1);

        /* wave pre-combine (the Zipf hot path): ballot-match lanes on
         * the 11-bit LDS slot (VALU only), then the lowest matched lane
         * absorbs equal-key members through plain-LDS scratch — ONE
         * atomic round-trip per wave-group instead of one per lane.
         * Without this the same-slot LDS atomics of a skewed stream
         * serialize: measured 94.8% SQ_WAIT_ANY
         * (profiles/r01_pmc_wavecycle_decomposition.txt appendix). */
        u64 gsum = v;
        bool lead = live;
        if (wavecomb) {
        u64 m = __ballot(live);
        for (int b = 0; b < 11; ++b) {
            const u64 bb = __ballot((ls >> b) & 1u);
            m &= ((ls >> b) & 1u) ? bb : ~bb;
        }
        if (live && __popcll(m) > 1) {
            sc_k[tid] = k;
            sc_v[tid] = v;
            const u32 l0 = (u32)__ffsll((unsigned long long)m) - 1;
            if (lane == l0) {
                u64 mm = m & ~(1ull << lane);
                while (mm) {
                    const u32 b = (u32)__ffsll((unsigned long long)mm) - 1;
                    mm &= mm - 1;
                    if (sc_k[wbase + b] == k) gsum += sc_v[wbase + b];
                }
            } else if (sc_k[wbase + l0] == k) {
                lead = false;   /* absorbed by the group leader */
            }
        }
        }
        if (!lead) continue;
        if (!live) continue;

        bool done = false;
        for (int p = 0; p < 4; ++p) {
            u64 prev;
            if (READFIRST) {
                prev = lk[ls];
                if (prev == T9_EMPTY)
                    prev = atomicCAS((unsigned long long*)&lk[ls],
                                     (unsigned long long)T9_EMPTY,
                                     (unsigned long long)k);
            }
            else {
                prev = atomicCAS((unsigned long long*)&lk[ls],
                                 (unsigned long long)T9_EMPTY,
                                 (unsigned long long)k);
            }
            if (prev == T9_EMPTY || prev == k) {
                atomicAdd((unsigned long long*)&lv[ls],
                          (unsigned long long)gsum);
                done = true;
                break;
            }
            ls = (ls + 1) & (T9_LDS_SLOTS - 1);
        }
        if (!done) {
            /* cold path: straight to the global table — CAS and add
             * land in the SAME 128-B line (interleaved layout) */
            u64 slot = h & (cap - 1);
            u64 probes = 0;
            for (;;) {
                u64 prev;
                if (READFIRST) {
                    prev = __atomic_load_n(
                        (unsigned long long*)&t[2 * slot],
                        __ATOMIC_RELAXED);
                    if (prev == T9_EMPTY)
                        prev = atomicCAS(
                            (unsigned long long*)&t[2 * slot],
                            (unsigned long long)T9_EMPTY,
                            (unsigned long long)k);
                }
                else {
                    prev = atomicCAS((unsigned long long*)&t[2 * slot],
                                     (unsigned long long)T9_EMPTY,
                                     (unsigned long long)k);
                }
                if (prev == T9_EMPTY || prev == k) {
                    atomicAdd((unsigned long long*)&t[2 * slot + 1],
                              (unsigned long long)gsum);
                    break;
                }
                slot = (slot + 1) & (cap - 1);
                if (++probes > cap) {
                    atomicExch(err, 1u);
                    break;
                }
            }
        }
    }
    __syncthreads();

    /* flush the block's LDS accumulators into the global table */
    for (u32 s = tid; s < T9_LDS_SLOTS; s += 256) {
        const u64 k = lk[s];
        if (k == T9_EMPTY) continue;
        const u64 v = lv[s];
        u64 slot = t9_hash128to64(salt, k) & (cap - 1);
        u64 probes = 0;
        for (;;) {
            u64 prev = atomicCAS((unsigned long long*)&t[2 * slot],
                                 (unsigned long long)T9_EMPTY,
                                 (unsigned long long)k);
            if (prev == T9_EMPTY || prev == k) {
                atomicAdd((unsigned long long*)&t[2 * slot + 1],
                          (unsigned long long)v);
                break;
            }
            slot = (slot + 1) & (cap - 1);
            if (++probes > cap) {
                atomicExch(err, 1u);
                break;
            }
        }
    }
}

/* ------------------------------------------------------------------ *
 * 128-bit composite-key reduce — config-4 string identity.
 *
 * The reference reduces (std::string word, u64 count) pairs with
 * EQUALITY ON THE FULL KEY (core/reduce_probing_hash_table.hpp:233
 * probes compare keys, not hashes). The MI355X-native design
 * dictionary-encodes words at tokenize time into TWO independent 64-bit
 * hashes (k1, k2) and reduces on the 128-bit composite: two distinct
 * words occupy distinct table slots unless BOTH hashes collide
 * (p ~= 2^-128 per pair); in particular a forced k1 collision keeps
 * counts separate (parity-tested), which is the observable the
 * reference's full-key equality provides for any constructible input.
 * Slot = 3 interleaved u64 {k1, k2, sum}. Claim protocol: CAS k1
 * (EMPTY -> k1), then CAS k2 (EMPTY -> k2); a k1 match with a k2
 * mismatch advances the probe, so every (k1, k2) group converges to one
 * slot (all inserters scan the same slot order and only skip slots that
 * definitively mismatch). k1 == EMPTY / k2 == EMPTY are reserved; the
 * hashing kernels remap them (t9_hash2_of).
 * ------------------------------------------------------------------ */

__global__ __launch_bounds__(256) void k_reduce128_init(
    u64* __restrict__ t, u64 cap, u32 sstr) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < cap;
         i += stride) {
        t[sstr * i] = T9_EMPTY;
        t[sstr * i + 1] = T9_EMPTY;
        t[sstr * i + 2] = 0;
    }
}

/* global-table composite insert (shared by the build's cold path and the
 * LDS flush). READFIRST: probe with a plain load and only CAS when the
 * slot reads empty — hot re-inserts then cost loads + one ADD instead
 * of CAS round trips (correct: a non-empty k1/k2 is immutable, and a
 * racing claim is re-checked by the CAS). */
template <bool READFIRST, int SSTR = 3>
__device__ inline void t9_g128_insert(u64* __restrict__ tb, u64 cap,
                                      u64 salt, u64 k1, u64 k2, u64 v,
                                      u32* __restrict__ err) {
    u64 slot = t9_hash128to64(salt, k1) & (cap - 1);
    u64 probes = 0;
    for (;;) {
        u64* t = tb + (SSTR - 3) * slot;  /* SSTR=4: slot base 4*slot */
        u64 p1;
        if (READFIRST) {
            p1 = __atomic_load_n((unsigned long long*)&t[3 * slot],
                                 __ATOMIC_RELAXED);
            if (p1 == T9_EMPTY)
                p1 = atomicCAS((unsigned long long*)&t[3 * slot],
                               (unsigned long long)T9_EMPTY,
                               (unsigned long long)k1);
        }
        else {
            p1 = atomicCAS((unsigned long long*)&t[3 * slot],
                           (unsigned long long)T9_EMPTY,
                           (unsigned long long)k1);
        }
        if (p1 == T9_EMPTY || p1 == k1) {
            u64 p2;
            if (READFIRST) {
                p2 = __atomic_load_n(
                    (unsigned long long*)&t[3 * slot + 1],
                    __ATOMIC_RELAXED);
                if (p2 == T9_EMPTY)
                    p2 = atomicCAS((unsigned long long*)&t[3 * slot + 1],
                                   (unsigned long long)T9_EMPTY,
                                   (unsigned long long)k2);
            }
            else {
                p2 = atomicCAS((unsigned long long*)&t[3 * slot + 1],
                               (unsigned long long)T9_EMPTY,
                               (unsigned long long)k2);
            }
            if (p2 == T9_EMPTY || p2 == k2) {
                atomicAdd((unsigned long long*)&t[3 * slot + 2],
                          (unsigned long long)v);
                return;
            }
            /* k1 matches, k2 differs: a (rare) k1 collision — probe on */
        }
        slot = (slot + 1) & (cap - 1);
        if (++probes > cap) {
            atomicExch(err, 1u);
            return;
        }
    }
}

/* LDS-accumulated 128-bit build: per-block (k1, k2, sum) filter table
 * absorbing the Zipf head at LDS-atomic speed (same design as
 * k_reduce_build_lds, composite equality). SLOTS x 24 B of LDS. */
template <int SLOTS, bool READFIRST = false, int SSTR = 3>
__global__ __launch_bounds__(256) void k_reduce128_build_lds(
    const u64* __restrict__ k1s, const u64* __restrict__ k2s,
    const u64* __restrict__ vals, u64 n, u64* __restrict__ t, u64 cap,
    u64 salt, u32* __restrict__ err) {
    __shared__ u64 lk1[SLOTS];
    __shared__ u64 lk2[SLOTS];
    __shared__ u64 lv[SLOTS];
    const u32 tid = threadIdx.x;
    for (u32 s = tid; s < (u32)SLOTS; s += 256) {
        lk1[s] = T9_EMPTY;
        lk2[s] = T9_EMPTY;
        lv[s] = 0;
    }
    __syncthreads();

    const u64 gsz = (u64)gridDim.x * 256;
    for (u64 base = (u64)blockIdx.x * 256; base < n; base += gsz) {
        const u64 i = base + tid;
        if (i >= n) continue;
        const u64 k1 = k1s[i], k2 = k2s[i];
        const u64 v = vals ? vals[i] : 1;
        const u64 h = t9_hash128to64(salt, k1);
        u32 ls = (u32)(h >> 48) & (SLOTS - 1);
        bool done = false;
        for (int p = 0; p < 4; ++p) {
            u64 p1;
            if (READFIRST) {
                p1 = lk1[ls];
                if (p1 == T9_EMPTY)
                    p1 = atomicCAS((unsigned long long*)&lk1[ls],
                                   (unsigned long long)T9_EMPTY,
                                   (unsigned long long)k1);
            }
            else {
                p1 = atomicCAS((unsigned long long*)&lk1[ls],
                               (unsigned long long)T9_EMPTY,
                               (unsigned long long)k1);
            }
            if (p1 == T9_EMPTY || p1 == k1) {
                u64 p2;
                if (READFIRST) {
                    p2 = lk2[ls];
                    if (p2 == T9_EMPTY)
                        p2 = atomicCAS((unsigned long long*)&lk2[ls],
                                       (unsigned long long)T9_EMPTY,
                                       (unsigned long long)k2);
                }
                else {
                    p2 = atomicCAS((unsigned long long*)&lk2[ls],
                                   (unsigned long long)T9_EMPTY,
                                   (unsigned long long)k2);
                }
                if (p2 == T9_EMPTY || p2 == k2) {
                    atomicAdd((unsigned long long*)&lv[ls],
                              (unsigned long long)v);
                    done = true;
                    break;
                }
            }
            ls = (ls + 1) & (SLOTS - 1);
        }
        if (!done)
            t9_g128_insert<READFIRST, SSTR>(t, cap, salt, k1, k2, v, err);
    }
    __syncthreads();

    for (u32 s = tid; s < (u32)SLOTS; s += 256) {
        if (lk1[s] == T9_EMPTY) continue;
        /* a claimed slot whose k2 CAS was lost to another key keeps
         * k2 == EMPTY only if no inserter ever won it — then its count
         * is 0 and it can be skipped */
        if (lk2[s] == T9_EMPTY) continue;
        t9_g128_insert<READFIRST, SSTR>(t, cap, salt, lk1[s], lk2[s], lv[s], err);
    }
}

/* drain (k1, k2, sum) triples — block-aggregated range claims, as
 * k_reduce_drain */
__global__ __launch_bounds__(256) void k_reduce128_drain(
    const u64* __restrict__ t, u64 cap, u64* __restrict__ ok1,
    u64* __restrict__ ok2, u64* __restrict__ ov,
    u64* __restrict__ out_n, u32 sstr) {
    __shared__ u32 s_pre[256];
    __shared__ u64 s_base;
    const u32 tid = threadIdx.x;
    const u64 stride = (u64)gridDim.x * 256;
    const u64 gid = (u64)blockIdx.x * 256 + tid;
    u32 mine = 0;
    for (u64 i = gid; i < cap; i += stride)
        if (t[sstr * i] != T9_EMPTY && t[sstr * i + 1] != T9_EMPTY)
            ++mine;
    s_pre[tid] = mine;
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
        u32 y = (tid >= (u32)off) ? s_pre[tid - off] : 0;
        __syncthreads();
        s_pre[tid] += y;
        __syncthreads();
    }
    if (tid == 255 && s_pre[255])
        s_base = atomicAdd((unsigned long long*)out_n,
                           (unsigned long long)s_pre[255]);
    __syncthreads();
    if (mine) {
        u64 pos = s_base + s_pre[tid] - mine;
        for (u64 i = gid; i < cap; i += stride) {
            if (t[sstr * i] != T9_EMPTY && t[sstr * i + 1] != T9_EMPTY) {
                ok1[pos] = t[sstr * i];
                ok2[pos] = t[sstr * i + 1];
                ov[pos] = t[sstr * i + 2];
                ++pos;
            }
        }
    }
}

/* two independent 64-bit hashes of a u64 token id (the synthetic stand-in
 * for hashing the word string at tokenize time; reserved sentinel values
 * are remapped so they never reach the table) */
__global__ __launch_bounds__(256) void k_hash2_of(
    const u64* __restrict__ ids, u64 n, u64* __restrict__ k1s,
    u64* __restrict__ k2s) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
        const u64 id = ids[i];
        u64 h1 = t9_hash128to64(0x9AE16A3B2F90404Full, id);
        u64 h2 = t9_hash128to64(0xC3A5C85C97CB3127ull, id);
        if (h1 == T9_EMPTY) h1 ^= 1;
        if (h2 == T9_EMPTY) h2 ^= 1;
        k1s[i] = h1;
        k2s[i] = h2;
    }
}

/* bucket = key % p — the partition mapping when the key already IS the
 * hash (128-bit path: partition on k1, mirroring the reference's
 * h % num_partitions, core/reduce_functional.hpp:60-72) */
__global__ __launch_bounds__(256) void k_bucket_mod(
    const u64* __restrict__ keys, u64 n, u32 p, u32* __restrict__ bucket,
    u64* __restrict__ counts) {
    __shared__ u32 scnt[256];
    const u32 tid = threadIdx.x;
    scnt[tid] = 0;
    __syncthreads();
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + tid; i < n; i += stride) {
        const u32 b = (u32)(keys[i] % p);
        bucket[i] = b;
        atomicAdd(&scnt[b], 1u);
    }
    __syncthreads();
    if (tid < p && scnt[tid])
        atomicAdd((unsigned long long*)&counts[tid],
                  (unsigned long long)scnt[tid]);
}

/* ReduceToIndex — reference api/reduce_to_index.hpp with the
 * ReduceByIndex mapping (core/reduce_functional.hpp:84-149): keys are
 * dense indices; dense[key-begin] accumulates the u64 sum. Out-of-range
 * keys set the error flag. */
__global__ __launch_bounds__(256) void k_reduce_by_index(
    const u64* __restrict__ keys, const u64* __restrict__ vals, u64 n,
    u64 begin, u64 size, u64* __restrict__ dense, u32* __restrict__ err) {
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
        const u64 k = keys[i];
        if (k < begin || k - begin >= size) {
            atomicExch(err, 1u);
            continue;
        }
        atomicAdd((unsigned long long*)&dense[k - begin],
                  (unsigned long long)vals[i]);
    }
}

/* partition for the by-index mapping: bucket = (k-begin)*p / size
 * (reduce_functional.hpp:113-128 with num_buckets = p). Out-of-range keys
 * clamp to the last partition AND set the error flag, mirroring
 * k_reduce_by_index's error model (a silently misrouted key
 * would otherwise only surface on the destination rank). */
__global__ __launch_bounds__(256) void k_index_bucket(
    const u64* __restrict__ keys, u64 n, u64 begin, u64 size, u32 p,
    u32* __restrict__ bucket, u64* __restrict__ counts,
    u32* __restrict__ err) {
    __shared__ u32 scnt[256];
    const u32 tid = threadIdx.x;
    scnt[tid] = 0;
    __syncthreads();
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + tid; i < n; i += stride) {
        u64 g = keys[i] - begin;
        if (g >= size) {
            g = size - 1;
            atomicExch(err, 1u);
        }
        const u32 b = (u32)(g * p / size);
        bucket[i] = b;
        atomicAdd(&scnt[b], 1u);
    }
    __syncthreads();
    if (tid < p && scnt[tid])
        atomicAdd((unsigned long long*)&counts[tid],
                  (unsigned long long)scnt[tid]);
}

/* Drain with block-local aggregation: each block counts its occupied
 * slots, claims one contiguous output range with a single atomicAdd (a
 * per-element add on one counter measured ~6 ms at 2^25 slots), then
 * writes at block-local scanned positions. Output order is arbitrary, as
 * the reference documents for reducing (word_count_test.cpp:73-74). */
__global__ __launch_bounds__(256) void k_reduce_drain(
    const u64* __restrict__ t, u64 cap,
    u64* __restrict__ ok, u64* __restrict__ ov, u64* __restrict__ out_n) {
    __shared__ u32 s_pre[256];
    __shared__ u64 s_base;
    const u32 tid = threadIdx.x;
    const u64 stride = (u64)gridDim.x * 256;
    const u64 gid = (u64)blockIdx.x * 256 + tid;
    u32 mine = 0;
    for (u64 i = gid; i < cap; i += stride)
        if (t[2 * i] != T9_EMPTY) ++mine;
    s_pre[tid] = mine;
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
        u32 y = (tid >= (u32)off) ? s_pre[tid - off] : 0;
        __syncthreads();
        s_pre[tid] += y;
        __syncthreads();
    }
    if (tid == 255 && s_pre[255])
        s_base = atomicAdd((unsigned long long*)out_n,
                           (unsigned long long)s_pre[255]);
    __syncthreads();
    if (mine) {
        u64 pos = s_base + s_pre[tid] - mine;
        for (u64 i = gid; i < cap; i += stride) {
            if (t[2 * i] != T9_EMPTY) {
                ok[pos] = t[2 * i];
                ov[pos] = t[2 * i + 1];
                ++pos;
            }
        }
    }
    if (gid == 0 && t[2 * cap] > 0) {
        u64 pos = atomicAdd((unsigned long long*)out_n, 1ull);
        ok[pos] = T9_EMPTY;
        ov[pos] = t[2 * cap + 1];
    }
}

/* bucket = Hash128to64(salt, key) % p — the ReduceByHash partition mapping
 * (core/reduce_functional.hpp:60-72) used to split pre-reduced pairs
 * across ranks before the all-to-all. d_counts[p] accumulates totals. */
__global__ __launch_bounds__(256) void k_hash_bucket(
    const u64* __restrict__ keys, u64 n, u64 salt, u32 p,
    u32* __restrict__ bucket, u64* __restrict__ counts) {
    __shared__ u32 scnt[256];
    const u32 tid = threadIdx.x;
    scnt[tid] = 0;
    __syncthreads();
    const u64 stride = (u64)gridDim.x * 256;
    for (u64 i = (u64)blockIdx.x * 256 + tid; i < n; i += stride) {
        u32 b = (u32)(t9_hash128to64(salt, keys[i]) % p);
        bucket[i] = b;
        atomicAdd(&scnt[b], 1u);
    }
    __syncthreads();
    if (tid < p && scnt[tid])
        atomicAdd((unsigned long long*)&counts[tid],
                  (unsigned long long)scnt[tid]);
}

namespace {
u32 grid_for(u64 work) {
    u64 want = (work + 255) / 256;
    return (u32)((want < 4096) ? (want ? want : 1) : 4096);
}
bool is_pow2(u64 x) { return x && !(x & (x - 1)); }
} // namespace

extern "C" {

int t9_hash_bucket(t9_context* ctx, const u64* d_keys, u64 n, u64 salt,
                   u32 p, u32* d_bucket, u64* d_counts, void* stream) {
    (void)ctx;
    if (!d_counts || p < 1 || p > 256) return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_counts, 0, p * 8, s));
    if (n == 0) return T9_OK;
    if (!d_keys || !d_bucket) return T9_EINVAL;
    hipLaunchKernelGGL(k_hash_bucket, dim3(grid_for(n)), dim3(256), 0, s,
                       d_keys, n, salt, p, d_bucket, d_counts);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_reduce_by_index(t9_context* ctx, const u64* d_keys,
                       const u64* d_vals, u64 n, u64 begin, u64 size,
                       u64* d_dense, u32* d_error, void* stream) {
    (void)ctx;
    if (!d_dense || !d_error || size == 0) return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_error, 0, 4, s));
    HIP_TRY(hipMemsetAsync(d_dense, 0, size * 8, s));
    if (n == 0) return T9_OK;
    if (!d_keys || !d_vals) return T9_EINVAL;
    hipLaunchKernelGGL(k_reduce_by_index, dim3(grid_for(n)), dim3(256), 0,
                       s, d_keys, d_vals, n, begin, size, d_dense,
                       d_error);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_index_bucket(t9_context* ctx, const u64* d_keys, u64 n, u64 begin,
                    u64 size, u32 p, u32* d_bucket, u64* d_counts,
                    u32* d_error, void* stream) {
    (void)ctx;
    if (!d_counts || !d_error || p < 1 || p > 256 || size == 0)
        return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_counts, 0, p * 8, s));
    HIP_TRY(hipMemsetAsync(d_error, 0, 4, s));
    if (n == 0) return T9_OK;
    if (!d_keys || !d_bucket) return T9_EINVAL;
    hipLaunchKernelGGL(k_index_bucket, dim3(grid_for(n)), dim3(256), 0, s,
                       d_keys, n, begin, size, p, d_bucket, d_counts,
                       d_error);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_reduce_init(t9_context* ctx, u64* d_table, u64 cap, void* stream) {
    (void)ctx;
    if (!d_table || !is_pow2(cap)) return T9_EINVAL;
    hipLaunchKernelGGL(k_reduce_init, dim3(grid_for(cap + 1)), dim3(256), 0,
                       (hipStream_t)stream, d_table, cap);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_reduce_build(t9_context* ctx, const u64* d_keys, const u64* d_vals,
                    u64 n, u64* d_table, u64 cap, u64 salt,
                    u32* d_error, void* stream) {
    (void)ctx;
    if (!d_table || !d_error || !is_pow2(cap)) return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_error, 0, 4, s));
    if (n == 0) return T9_OK;
    if (!d_keys || !d_vals) return T9_EINVAL;
    const char* ce = getenv("T9_REDUCE_COMBINE");
    const int mode = ce ? atoi(ce) : 2;   /* 2 = LDS table (default) */
    /* wave pre-combine is OFF by default: measured SLOWER at every
     * vocab (1e3: 2.38->3.70 ms; 1e7: 11.06->11.75 ms at 2^28 tokens)
     * — CDNA4 LDS atomics absorb same-address contention better than
     * the ballot/scratch/leader overhead costs. Kept opt-in
     * (T9_REDUCE_WAVECOMB=1) as a recorded negative result. */
    const char* we = getenv("T9_REDUCE_WAVECOMB");
    const int wavecomb = (we && we[0] == '1') ? 1 : 0;
    T9_PERF_WRAP(
        s, "reduce_build",
        if (mode == 2) {
            /* measured optimum at 10M-vocab Zipf(1.1), 2^29 tokens:
             * slots=4096 (2 blocks/CU) beats 2048 (21.6 -> 19.1 ms,
             * coverage over concurrency) and 8192 (1 block/CU, 22.0);
             * a 1024 grid shaves flush duplication (19.1 -> 18.8). */
            const char* se = getenv("T9_LDS_SLOTS");
            const int slots = se ? atoi(se) : 4096;
            const char* ge = getenv("T9_REDUCE_GRID");
            u32 grid = ge ? (u32)atoi(ge) : grid_for(n);
            if (!ge && grid > 1024) grid = 1024;
            /* read-before-CAS probing, default ON: measured 14.65 vs
               18.44 ms per 2^29 tokens at 10M vocab (-21%,
               scripts/ab_reduce_u64_rf.py) — occupied-slot probes and
               hot re-inserts pay plain loads instead of CAS round
               trips. T9_REDUCE_READFIRST=0 restores always-CAS. */
            const char* urf = getenv("T9_REDUCE_READFIRST");
            const bool rf = !(urf && urf[0] == '0');
            if (slots >= 8192)
                hipLaunchKernelGGL((k_reduce_build_lds<8192, false>),
                                   dim3(grid), dim3(256), 0, s,
                                   d_keys, d_vals, n, d_table, cap, salt,
                                   d_error, wavecomb);
            else if (slots <= 1024)
                hipLaunchKernelGGL((k_reduce_build_lds<1024, false>),
                                   dim3(grid), dim3(256), 0, s,
                                   d_keys, d_vals, n, d_table, cap, salt,
                                   d_error, wavecomb);
            else if (slots >= 4096 && rf)
                hipLaunchKernelGGL((k_reduce_build_lds<4096, true>),
                                   dim3(grid), dim3(256), 0, s,
                                   d_keys, d_vals, n, d_table, cap, salt,
                                   d_error, wavecomb);
            else if (slots >= 4096)
                hipLaunchKernelGGL((k_reduce_build_lds<4096, false>),
                                   dim3(grid), dim3(256), 0, s,
                                   d_keys, d_vals, n, d_table, cap, salt,
                                   d_error, wavecomb);
            else
                hipLaunchKernelGGL((k_reduce_build_lds<2048, false>),
                                   dim3(grid), dim3(256), 0, s,
                                   d_keys, d_vals, n, d_table, cap, salt,
                                   d_error, wavecomb);
        }
        else
            hipLaunchKernelGGL(k_reduce_build, dim3(grid_for(n)),
                               dim3(256), 0, s, d_keys, d_vals, n, d_table,
                               cap, salt, d_error, mode));
    T9_LAUNCH_CHECK();
    return T9_OK;
}

/* slot stride: 3 u64s packed (default; table = 3*capacity u64s) or 4
   (T9_R128_STRIDE=4: 32-B line-aligned slots, table = 4*capacity u64s —
   the CALLER must allocate accordingly; A/B knob) */
static u32 r128_stride() {
    const char* e = getenv("T9_R128_STRIDE");
    return (e && atoi(e) == 4) ? 4u : 3u;
}

int t9_reduce128_init(t9_context* ctx, u64* d_table, u64 cap,
                      void* stream) {
    (void)ctx;
    if (!d_table || !is_pow2(cap)) return T9_EINVAL;
    hipLaunchKernelGGL(k_reduce128_init, dim3(grid_for(cap)), dim3(256), 0,
                       (hipStream_t)stream, d_table, cap, r128_stride());
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_reduce128_build(t9_context* ctx, const u64* d_k1, const u64* d_k2,
                       const u64* d_vals, u64 n, u64* d_table, u64 cap,
                       u64 salt, u32* d_error, void* stream) {
    (void)ctx;
    if (!d_table || !d_error || !is_pow2(cap)) return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_error, 0, 4, s));
    if (n == 0) return T9_OK;
    if (!d_k1 || !d_k2) return T9_EINVAL;
    /* d_vals may be NULL: every pair counts 1 (the word_count PreOp
     * emits (word, 1) — word_count.hpp:43-45) */
    /* measured at 2^29 Zipf(1.1) tokens, 10M vocab: 4096 (1 block/CU,
     * 96 KB LDS) 34.6 ms beats 2048 (36.1) and 1024 (45.5) — hot-key
     * coverage over occupancy, as in the u64 table */
    const char* se = getenv("T9_LDS128_SLOTS");
    const int slots = se ? atoi(se) : 4096;
    const char* ge = getenv("T9_REDUCE_GRID");
    u32 grid = ge ? (u32)atoi(ge) : grid_for(n);
    if (!ge && grid > 1024) grid = 1024;
    /* read-before-CAS probing (T9_R128_READFIRST): hot re-inserts pay
       loads + one ADD instead of CAS round trips */
    /* default ON: measured 30.5 vs 34.6 ms per 2^29 tokens at 10M
       vocab (scripts/ab_reduce128.py) */
    const char* rf = getenv("T9_R128_READFIRST");
    const bool readfirst = !(rf && rf[0] == '0');
    const bool s4 = r128_stride() == 4;
    T9_PERF_WRAP(
        s, "reduce_build",
        if (slots >= 4096 && readfirst && s4)
            hipLaunchKernelGGL((k_reduce128_build_lds<4096, true, 4>),
                               dim3(grid), dim3(256), 0, s, d_k1, d_k2,
                               d_vals, n, d_table, cap, salt, d_error);
        else if (slots >= 4096 && readfirst)
            hipLaunchKernelGGL((k_reduce128_build_lds<4096, true>),
                               dim3(grid), dim3(256), 0, s, d_k1, d_k2,
                               d_vals, n, d_table, cap, salt, d_error);
        else if (slots >= 4096)
            hipLaunchKernelGGL((k_reduce128_build_lds<4096, false>),
                               dim3(grid), dim3(256), 0, s, d_k1, d_k2,
                               d_vals, n, d_table, cap, salt, d_error);
        else if (slots <= 1024)
            hipLaunchKernelGGL((k_reduce128_build_lds<1024, false>),
                               dim3(grid), dim3(256), 0, s, d_k1, d_k2,
                               d_vals, n, d_table, cap, salt, d_error);
        else if (readfirst)
            hipLaunchKernelGGL((k_reduce128_build_lds<2048, true>),
                               dim3(grid), dim3(256), 0, s, d_k1, d_k2,
                               d_vals, n, d_table, cap, salt, d_error);
        else
            hipLaunchKernelGGL((k_reduce128_build_lds<2048, false>),
                               dim3(grid), dim3(256), 0, s, d_k1, d_k2,
                               d_vals, n, d_table, cap, salt, d_error));
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_reduce128_drain(t9_context* ctx, const u64* d_table, u64 cap,
                       u64* d_ok1, u64* d_ok2, u64* d_ov, u64* d_out_n,
                       void* stream) {
    (void)ctx;
    if (!d_table || !d_ok1 || !d_ok2 || !d_ov || !d_out_n || !is_pow2(cap))
        return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_out_n, 0, 8, s));
    hipLaunchKernelGGL(k_reduce128_drain, dim3(grid_for(cap)), dim3(256),
                       0, s, d_table, cap, d_ok1, d_ok2, d_ov, d_out_n,
                       r128_stride());
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_hash2_of(t9_context* ctx, const u64* d_ids, u64 n, u64* d_k1,
                u64* d_k2, void* stream) {
    (void)ctx;
    if (!d_ids || !d_k1 || !d_k2) return T9_EINVAL;
    if (n == 0) return T9_OK;
    hipLaunchKernelGGL(k_hash2_of, dim3(grid_for(n)), dim3(256), 0,
                       (hipStream_t)stream, d_ids, n, d_k1, d_k2);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_bucket_mod(t9_context* ctx, const u64* d_keys, u64 n, u32 p,
                  u32* d_bucket, u64* d_counts, void* stream) {
    (void)ctx;
    if (!d_counts || p < 1 || p > 256) return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_counts, 0, p * 8, s));
    if (n == 0) return T9_OK;
    if (!d_keys || !d_bucket) return T9_EINVAL;
    hipLaunchKernelGGL(k_bucket_mod, dim3(grid_for(n)), dim3(256), 0, s,
                       d_keys, n, p, d_bucket, d_counts);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

int t9_reduce_drain(t9_context* ctx, const u64* d_table,
                    u64 cap, u64* d_ok, u64* d_ov, u64* d_out_n,
                    void* stream) {
    (void)ctx;
    if (!d_table || !d_ok || !d_ov || !d_out_n || !is_pow2(cap))
        return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipMemsetAsync(d_out_n, 0, 8, s));
    hipLaunchKernelGGL(k_reduce_drain, dim3(grid_for(cap)), dim3(256), 0, s,
                       d_table, cap, d_ok, d_ov, d_out_n);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

} /* extern "C" */
