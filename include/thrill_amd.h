/* thrill_amd.h — C-ABI of libt9.so, the MI355X-native implementation of
 * Thrill's Sort / ReduceByKey DOp hot path (hand-written HIP/CDNA4 kernels,
 * gfx950). This is the drop-in seam of SURVEY.md §8b: the reference's
 * SortNode (thrill/api/sort.hpp:64-787) and ReduceNode
 * (thrill/api/reduce_by_key.hpp:64-239) become thin C++ hosts that marshal
 * item Files <-> device buffers and call these entry points; anything
 * (including a patched reference build) can call the shim directly. The
 * reference-side binding a maintainer would add is shown in INTEGRATION.md.
 *
 * Conventions: all functions are synchronous launches on the passed HIP
 * stream (void* == hipStream_t; NULL = default stream); they return 0 on
 * success or a negative errno-style value. The caller owns every buffer;
 * d_* pointers are device memory. No torch types anywhere.
 *
 * The GPU library contains the PRODUCT path only: it never falls back to
 * CPU. Without a GPU, t9_create fails; nothing here routes through the
 * oracle.
 */
#ifndef THRILL_AMD_H
#define THRILL_AMD_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Opaque context: device id, rank, world size, optional RCCL communicator.
 * Mirrors the per-worker api::Context the reference nodes consume
 * (thrill/api/context.hpp:243-245: my_rank / num_workers). */
typedef struct t9_context t9_context;

/* comm: an existing ncclComm_t (RCCL) or NULL for single-GPU use. */
int t9_create(t9_context** out, int device, int rank, int world, void* comm);
int t9_destroy(t9_context* ctx);
const char* t9_version(void);

/* RCCL communicator bootstrap (world > 1): rank 0 generates the unique id
 * (t9_comm_id_size() bytes, = sizeof(ncclUniqueId)), the caller moves it
 * to every rank over any host channel (the reference distributes its TCP
 * endpoints the same way, thrill/api/context.cpp:604-614), then every
 * rank calls t9_comm_init collectively. The context owns the resulting
 * communicator (t9_destroy frees it). Passing an external ncclComm_t via
 * t9_create(comm) remains supported; t9_comm_init on such a context is an
 * error. */
int t9_comm_id_size(void);
int t9_comm_id(void* out_id);
int t9_comm_init(t9_context* ctx, const void* id);

/* ------------------------------------------------------------------ *
 * Synthetic input generation (device-side, seeded; bit-identical to the
 * oracle's t9o_gen_* so CPU/GPU parity runs on identical bytes).
 * Record layout restates examples/terasort/terasort.cpp:31-118.
 * ------------------------------------------------------------------ */
int t9_gen_u64(t9_context* ctx, uint64_t* d_out, uint64_t index0, uint64_t n,
               uint64_t seed, void* stream);
int t9_gen_records(t9_context* ctx, uint8_t* d_out, uint64_t index0,
                   uint64_t n, uint64_t seed, void* stream);

/* ------------------------------------------------------------------ *
 * Local sort — replaces SortAndWriteToFile's std::sort run formation and
 * the loser-tree merge (thrill/api/sort.hpp:665-786,
 * thrill/core/multiway_merge.hpp:30-116): the whole per-GPU partition is
 * sorted in one LSD radix pipeline, so the merge stage vanishes
 * (SURVEY.md §8a row a5/a6).
 * ------------------------------------------------------------------ */

/* Workspace bytes for t9_sort_u64 on n keys. */
uint64_t t9_sort_u64_workspace(uint64_t n);
/* In-place ascending sort of n u64 keys (n < 2^32). */
int t9_sort_u64(t9_context* ctx, uint64_t* d_keys, uint64_t n,
                void* d_workspace, void* stream);

/* Workspace bytes for t9_sort_pairs_u64_u32 on n pairs. */
uint64_t t9_sort_pairs_workspace(uint64_t n);
/* In-place stable ascending sort of (key, payload) pairs by key. */
int t9_sort_pairs_u64_u32(t9_context* ctx, uint64_t* d_keys,
                          uint32_t* d_vals, uint64_t n, void* d_workspace,
                          void* stream);

/* Extract the big-endian u64 prefix of each record's key (bytes
 * key_off .. key_off+7) and the record index iota. rec_size % 4 == 0. */
int t9_extract_key64(t9_context* ctx, const uint8_t* d_recs, uint64_t n,
                     uint32_t rec_size, uint32_t key_off, uint64_t* d_keys,
                     uint32_t* d_idx, void* stream);

/* Little-endian variant: the key is a native uint64_t field (numeric
 * order) — BASELINE config 5's struct{u64 key; u8 payload[120]}. */
int t9_extract_key64_le(t9_context* ctx, const uint8_t* d_recs, uint64_t n,
                        uint32_t rec_size, uint32_t key_off,
                        uint64_t* d_keys, uint32_t* d_idx, void* stream);

/* out[i] = recs[idx[i]] for fixed-size records (rec_size % 4 == 0). */
int t9_gather_records(t9_context* ctx, const uint8_t* d_recs,
                      const uint32_t* d_idx, uint64_t n, uint32_t rec_size,
                      uint8_t* d_out, void* stream);

/* Workspace bytes for t9_sort_records. */
uint64_t t9_sort_records_workspace(uint64_t n, uint32_t rec_size);
/* Sort n fixed-size records by the acceptance total order: lexicographic
 * over the whole record (key prefix first — terasort.cpp:35-37 compares the
 * 10-byte key; ties beyond the radix-sorted u64 key prefix are resolved by
 * comparing the remaining bytes). d_in is preserved; d_out receives the
 * sorted sequence. key_len <= rec_size; key bytes start at offset 0.
 * Equal-u64-prefix runs are re-ordered ON DEVICE (segmented LSD over the
 * tail bytes using the stable pair sort as the primitive; identical
 * chunks are skipped). Synchronizes the stream internally (tie counts
 * and skip flags round-trip to the host). */
int t9_sort_records(t9_context* ctx, const uint8_t* d_in, uint8_t* d_out,
                    uint64_t n, uint32_t rec_size, uint32_t key_len,
                    void* d_workspace, void* stream);

/* config-5 variant: sort records whose key is a native little-endian
 * uint64_t at offset 0 (numeric order; payload-byte tiebreak). */
int t9_sort_records_keyle(t9_context* ctx, const uint8_t* d_in,
                          uint8_t* d_out, uint64_t n, uint32_t rec_size,
                          void* d_workspace, void* stream);

/* ------------------------------------------------------------------ *
 * Classification + partition — replaces TransmitItems' tree-descent loop
 * (thrill/api/sort.hpp:434-535). bucket(item i) = #{ j : (splitter_key[j],
 * splitter_idx[j]) < (key_i, gidx0+i) lexicographically }, which is exactly
 * the tree descent + EqualSampleGreaterIndex walk (sort.hpp:424-426,
 * 487-501) in closed form (proof: oracle/t9_oracle.cpp, cross-checked
 * against the literal tree restatement in tests).
 * ------------------------------------------------------------------ */

/* d_bucket[i] = bucket of key i; d_counts[p] (u64, zeroed by the call)
 * accumulates per-bucket totals. p <= 256. */
int t9_classify_u64(t9_context* ctx, const uint64_t* d_keys, uint64_t n,
                    uint64_t gidx0, const uint64_t* d_spl_keys,
                    const uint64_t* d_spl_idx, uint32_t p,
                    uint32_t* d_bucket, uint64_t* d_counts, void* stream);

/* Record classification under the acceptance total order: primary compare
 * on the precomputed big-endian u64 key prefix (d_k64, from
 * t9_extract_key64), byte fallback over the remaining record bytes on
 * prefix ties, then the splitter-index tiebreak. Splitters are whole
 * records (as in the reference, where splitters are sampled items —
 * api/sort.hpp:300,368-374). */
int t9_classify_rec(t9_context* ctx, const uint8_t* d_recs,
                    const uint64_t* d_k64, uint64_t n, uint64_t gidx0,
                    const uint8_t* d_spl_recs, const uint64_t* d_spl_k64,
                    const uint64_t* d_spl_idx, uint32_t p, uint32_t rec_size,
                    uint32_t* d_bucket, uint64_t* d_counts, void* stream);

/* Workspace bytes for t9_partition_idx. */
uint64_t t9_partition_idx_workspace(uint64_t n);
/* Stable counting-sort of the identity permutation by bucket id:
 * d_perm receives record indices grouped by bucket (bucket-major, original
 * order within a bucket — the scatter of SURVEY.md §8b t9_scatter).
 * d_offsets[p+1] (u64) receives the exclusive bucket start offsets. */
int t9_partition_idx(t9_context* ctx, const uint32_t* d_bucket, uint64_t n,
                     uint32_t p, uint32_t* d_perm, uint64_t* d_offsets,
                     void* d_workspace, void* stream);

/* ------------------------------------------------------------------ *
 * Shuffle — replaces the CatStream/MixStream + Multiplexer TCP exchange
 * (thrill/data/stream_sink.cpp:97-226, multiplexer.cpp:282-463) with an
 * RCCL all-to-all-v over xGMI (grouped ncclSend/ncclRecv), one rank per
 * GPU. counts are element counts per destination rank (host memory);
 * displacements are their exclusive prefix sums (caller-computed).
 * ------------------------------------------------------------------ */
int t9_alltoall(t9_context* ctx, const void* d_send,
                const uint64_t* send_counts, const uint64_t* send_displs,
                void* d_recv, const uint64_t* recv_counts,
                const uint64_t* recv_displs, uint64_t elem_size,
                void* stream);

/* ------------------------------------------------------------------ *
 * Reduce — replaces ReduceProbingHashTable::Insert
 * (thrill/core/reduce_probing_hash_table.hpp:190-268) with a device
 * open-addressing table: linear probing on Hash128to64(salt, key)
 * (thrill/common/hash.hpp:64-72; index mapping
 * core/reduce_functional.hpp:60-72), wave-level pre-combination of equal
 * keys (ballot match + shuffle reduce) before one atomic CAS/ADD per
 * distinct key per wave — the skew control for Zipf keys (SURVEY.md §7
 * step 6). The empty-slot sentinel key 0xFFFF..F is reduced in a dedicated
 * extra slot, mirroring reduce_probing_hash_table.hpp:195-217.
 * Table arrays have capacity+1 entries; capacity must be a power of two
 * >= 2x the number of distinct keys (no grow/spill: 288 GB HBM holds the
 * table — reference grow/spill machinery is subsumed by sizing).
 * ------------------------------------------------------------------ */
/* bucket = Hash128to64(salt, key) % p, the reference's ReduceByHash
 * partition mapping (core/reduce_functional.hpp:60-72; libstdc++
 * std::hash<u64> is the identity) — splits pre-reduced pairs across ranks.
 * d_counts[p] (u64, zeroed by the call) accumulates per-rank totals. */
int t9_hash_bucket(t9_context* ctx, const uint64_t* d_keys, uint64_t n,
                   uint64_t salt, uint32_t p, uint32_t* d_bucket,
                   uint64_t* d_counts, void* stream);

/* The reduce table is ONE interleaved u64 array of 2*(capacity+1)
 * elements: slot i = (d_table[2i] key, d_table[2i+1] sum), aux sentinel
 * slot at [2*capacity .. 2*capacity+1]. Interleaving keeps each probing
 * CAS + add inside one cache line (the build of a big-vocab stream is
 * random-line bound). capacity must be a power of two. */
int t9_reduce_init(t9_context* ctx, uint64_t* d_table, uint64_t capacity,
                   void* stream);
/* Accumulate n (key, value) pairs into the table; value reduce = u64 add.
 * d_error (device u32, zeroed by the call) is set nonzero if the table
 * overflows. */
int t9_reduce_build(t9_context* ctx, const uint64_t* d_keys,
                    const uint64_t* d_vals, uint64_t n,
                    uint64_t* d_table, uint64_t capacity, uint64_t salt,
                    uint32_t* d_error, void* stream);
/* Compact occupied slots to (key, value) arrays (unordered);
 * d_out_n (device u64) receives the count. */
int t9_reduce_drain(t9_context* ctx, const uint64_t* d_table,
                    uint64_t capacity, uint64_t* d_out_keys,
                    uint64_t* d_out_vals, uint64_t* d_out_n, void* stream);

/* ------------------------------------------------------------------ *
 * 128-bit composite-key reduce — config-4 string identity. The
 * reference reduces (std::string, u64) with equality on the FULL key
 * (core/reduce_probing_hash_table.hpp:233); here words are
 * dictionary-encoded at tokenize time into TWO independent 64-bit
 * hashes (k1, k2) and reduced on the 128-bit composite: distinct words
 * stay separate unless both hashes collide (p ~= 2^-128 per pair; a
 * forced k1 collision stays separate — parity-tested). Table slot = 3
 * interleaved u64 {k1, k2, sum}, 3*capacity array, capacity a power of
 * two. k1 == ~0 / k2 == ~0 are reserved sentinels (t9_hash2_of remaps
 * them). d_vals == NULL means every pair counts 1.
 * ------------------------------------------------------------------ */
int t9_reduce128_init(t9_context* ctx, uint64_t* d_table,
                      uint64_t capacity, void* stream);
int t9_reduce128_build(t9_context* ctx, const uint64_t* d_k1,
                       const uint64_t* d_k2, const uint64_t* d_vals,
                       uint64_t n, uint64_t* d_table, uint64_t capacity,
                       uint64_t salt, uint32_t* d_error, void* stream);
int t9_reduce128_drain(t9_context* ctx, const uint64_t* d_table,
                       uint64_t capacity, uint64_t* d_out_k1,
                       uint64_t* d_out_k2, uint64_t* d_out_vals,
                       uint64_t* d_out_n, void* stream);
/* two independent 64-bit hashes of u64 token ids (synthetic stand-in for
 * hashing the word bytes at tokenize time; remaps the reserved
 * sentinels) */
int t9_hash2_of(t9_context* ctx, const uint64_t* d_ids, uint64_t n,
                uint64_t* d_k1, uint64_t* d_k2, void* stream);
/* bucket = key % p — the partition mapping when the key already is the
 * hash (128-bit path partitions on k1, mirroring h % num_partitions,
 * core/reduce_functional.hpp:60-72) */
int t9_bucket_mod(t9_context* ctx, const uint64_t* d_keys, uint64_t n,
                  uint32_t p, uint32_t* d_bucket, uint64_t* d_counts,
                  void* stream);

/* ReduceToIndex — reference
 * api/reduce_to_index.hpp + core/reduce_by_index_post_phase.hpp with the
 * ReduceByIndex mapping (core/reduce_functional.hpp:84-149): keys are
 * dense indices in [begin, begin+size); d_dense (size u64, zeroed by the
 * call) accumulates the per-index sums; absent indices stay at the
 * neutral 0, as the reference's by-index post phase emits. d_error set
 * nonzero on out-of-range keys. */
int t9_reduce_by_index(t9_context* ctx, const uint64_t* d_keys,
                       const uint64_t* d_vals, uint64_t n, uint64_t begin,
                       uint64_t size, uint64_t* d_dense, uint32_t* d_error,
                       void* stream);

/* by-index partition: bucket = (key-begin)*p/size
 * (core/reduce_functional.hpp:113-128). Out-of-range keys clamp to the
 * last partition and set d_error (same error model as
 * t9_reduce_by_index). */
int t9_index_bucket(t9_context* ctx, const uint64_t* d_keys, uint64_t n,
                    uint64_t begin, uint64_t size, uint32_t p,
                    uint32_t* d_bucket, uint64_t* d_counts,
                    uint32_t* d_error, void* stream);

/* GroupByKey support (SURVEY.md §8f item 3 — thrill/api/group_by_key.hpp
 * is sort-based): the group index of a key-sorted array. d_unique[g] /
 * d_offsets[g] (ascending run starts) for g in [0, *d_count); d_count is
 * a device u64. Output capacity n always suffices. */
uint64_t t9_group_index_workspace(uint64_t n);
int t9_group_index(t9_context* ctx, const uint64_t* d_sorted_keys,
                   uint64_t n, uint64_t* d_unique, uint64_t* d_offsets,
                   uint64_t* d_count, void* d_workspace, void* stream);

/* Merge two byte-lexicographically sorted fixed-size record sequences
 * (the reference Merge's comparator restricted to the GPU-executable
 * byte order, api/merge.hpp:368-520); equal records from d_a precede
 * those from d_b. rec_size % 4 == 0. */
int t9_merge_records(t9_context* ctx, const uint8_t* d_a, uint64_t na,
                     const uint8_t* d_b, uint64_t nb, uint32_t rec_size,
                     uint8_t* d_out, void* stream);

/* Merge (SURVEY.md §8f item 4 — thrill/api/merge.hpp merges pre-sorted
 * DIAs): merge two sorted u64 sequences into d_out (na+nb); equal keys
 * from d_a precede those from d_b. One merge-path pass. */
int t9_merge_u64(t9_context* ctx, const uint64_t* d_a, uint64_t na,
                 const uint64_t* d_b, uint64_t nb, uint64_t* d_out,
                 void* stream);

/* Zipf(s, q, N) token sampling by inverse CDF (bit-identical to the
 * oracle's t9o_zipf_tokens given the same d_cdf table — the CDF itself is
 * computed once by the oracle/host and copied to the device). Restates
 * thrill/common/zipf_distribution.hpp:55-120's mass function. */
int t9_zipf_tokens(t9_context* ctx, uint64_t* d_out, const double* d_cdf,
                   uint64_t N, uint64_t index0, uint64_t n, uint64_t seed,
                   void* stream);

/* ------------------------------------------------------------------ *
 * Optional per-kernel-class HIP event timing (off by default) for the
 * bench harness's roofline measurement. Classes: "pair_scatter",
 * "keys_scatter", "hist_pairs", "hist_keys", "extract", "gather".
 * ------------------------------------------------------------------ */
int t9_perf_enable(int on);
int t9_perf_read(const char* kernel_class, double* total_ms,
                 uint64_t* launches);
int t9_perf_reset(void);

#ifdef __cplusplus
}
#endif

#endif /* THRILL_AMD_H */
