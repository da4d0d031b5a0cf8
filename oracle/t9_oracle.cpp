/* t9_oracle.cpp — CPU oracle for the thrill_amd hot path.
 *
 * TEST INFRASTRUCTURE ONLY. This library is the parity checker and the
 * bench.py `cpu_baseline` leg. Nothing under the product path (thrill_amd/,
 * libt9.so) may import, call, or link this; the product path must fail loudly
 * when the HIP extension is missing. Only tests/, __graft_entry__.smoke()
 * (as the checker) and bench.py's cpu_baseline leg may use it.
 *
 * This is a from-scratch CPU restatement of the algorithmic semantics of
 * Thrill's Sort / ReduceByKey hot path (reference: /root/reference, BSD-2,
 * NOT copied — restated from reading the code; citations are file:line into
 * the reference tree):
 *
 *   - splitter selection:      thrill/api/sort.hpp:337-378 (FindAndSendSplitters)
 *   - classification + tiebreak: thrill/api/sort.hpp:380-535 (TreeBuilder,
 *     TransmitItems, EqualSampleGreaterIndex :424-426, tie walk :487-501)
 *   - local sort:              thrill/api/sort.hpp:696-742 (SortAndWriteToFile,
 *     DefaultSortAlgorithm = std::sort :789-796)
 *   - reduce hash table:       thrill/core/reduce_probing_hash_table.hpp:190-268
 *     (Insert), :293-333 (GrowAndRehash); index fn
 *     thrill/core/reduce_functional.hpp:60-72 (ReduceByHash)
 *   - Hash128to64:             thrill/common/hash.hpp:64-72
 *   - record generator layout: examples/terasort/terasort.cpp:63-118
 *     (GenerateRecord; the reference seeds from std::random_device —
 *     non-reproducible — so the KEY byte stream here is replaced by a
 *     counter-based splitmix64 so CPU and GPU generate identical bytes;
 *     the VALUE layout is byte-identical to the reference's).
 *
 * Parity pinning: validated against the reference's own in-repo known-answer
 * tests, restated in tests/test_oracle_golden.py:
 *   - tests/examples/word_count_test.cpp:36-79  (bacon-ipsum 71-word table,
 *     with the fixture tests/inputs/wordcount.in — the one true in-repo KAT)
 *   - tests/api/sort_node_test.cpp:25-53,162-276 (known-integer identity,
 *     degenerate distributions)
 *   - tests/api/reduce_node_test.cpp:83-137     (exact modulo-key sums)
 *
 * Parity definition: Sort — byte-identical output under a
 * total-order comparator (key bytes, ties by the remaining record bytes ==
 * full-record lexicographic order); ReduceByKey — equality of the key-sorted
 * (key,value) multiset. u64 sums are exact under any association, so reduced
 * values are bit-exact regardless of reduction order.
 *
 * Build: g++ -O2 -shared -fPIC (see Makefile). No GPU, no HIP, no torch.
 */

#include <algorithm>
#include <cassert>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <tuple>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#include <parallel/algorithm>
#endif

extern "C" {

/* ------------------------------------------------------------------ */
/* Hash128to64 — thrill/common/hash.hpp:64-72 (Google cityhash, MIT). */
uint64_t t9o_hash128to64(uint64_t upper, uint64_t lower) {
    const uint64_t k = 0x9DDFEA08EB382D69ull;
    uint64_t a = (lower ^ upper) * k;
    a ^= (a >> 47);
    uint64_t b = (upper ^ a) * k;
    b ^= (b >> 47);
    b *= k;
    return b;
}

/* ReduceByHash partition mapping — thrill/core/reduce_functional.hpp:60-72.
 * hash = Hash128to64(salt, std::hash<u64>(k)); libstdc++ std::hash<uint64_t>
 * is the identity, so hash = Hash128to64(salt, k). partition = hash % p. */
uint32_t t9o_partition_of_u64(uint64_t key, uint64_t salt, uint32_t p) {
    return (uint32_t)(t9o_hash128to64(salt, key) % p);
}

/* ------------------------------------------------------------------ */
/* splitmix64 random access: value at counter `ctr` of the stream with the
 * given seed. Used for all synthetic inputs (seeds logged by callers). */
static inline uint64_t splitmix64_at(uint64_t seed, uint64_t ctr) {
    uint64_t z = seed + (ctr + 1) * 0x9E3779B97F4A7C15ull;
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    return z ^ (z >> 31);
}

uint64_t t9o_splitmix64_at(uint64_t seed, uint64_t ctr) {
    return splitmix64_at(seed, ctr);
}

/* Uniform u64 keys: keys[i] = splitmix64(seed, index0 + i). */
void t9o_gen_u64(uint64_t* out, uint64_t index0, uint64_t n, uint64_t seed) {
    for (uint64_t i = 0; i < n; ++i)
        out[i] = splitmix64_at(seed, index0 + i);
}

/* 100-byte TeraSort records — layout from examples/terasort/terasort.cpp:31-118
 * (struct Record{uint8_t key[10]; uint8_t value[90];}, GenerateRecord).
 * Key bytes: 10 bytes from splitmix64(seed, 2*rec) (big-endian: key[0] is the
 * most significant byte of k0, so the u64 built from key[0..7] big-endian
 * equals k0) and the top 2 bytes of splitmix64(seed, 2*rec+1).
 * Value bytes: byte-identical to the reference generator given `rec` = the
 * global record index (terasort.cpp:73-111). */
void t9o_gen_records(uint8_t* out, uint64_t index0, uint64_t n, uint64_t seed) {
    static const uint8_t hexdigits[16] = {
        '0', '1', '2', '3', '4', '5', '6', '7',
        '8', '9', 'A', 'B', 'C', 'D', 'E', 'F'
    };
    for (uint64_t i = 0; i < n; ++i) {
        uint64_t rec = index0 + i;
        uint8_t* r = out + i * 100;
        uint64_t k0 = splitmix64_at(seed, 2 * rec);
        uint64_t k1 = splitmix64_at(seed, 2 * rec + 1);
        for (int j = 0; j < 8; ++j)
            r[j] = (uint8_t)(k0 >> (56 - 8 * j));
        r[8] = (uint8_t)(k1 >> 56);
        r[9] = (uint8_t)(k1 >> 48);
        uint8_t* v = r + 10;
        *v++ = 0x00;                               /* terasort.cpp:76-77 */
        *v++ = 0x11;
        for (int j = 0; j != 16; ++j)              /* :84-86 hex of rec */
            *v++ = hexdigits[(rec >> (4 * j)) & 0x0F];
        for (int j = 0; j != 16; ++j)              /* :87-88 */
            *v++ = '0';
        *v++ = 0x88; *v++ = 0x99; *v++ = 0xAA; *v++ = 0xBB;  /* :91-94 */
        for (int j = 0; j < 12; ++j) {             /* :97-103 filler */
            uint8_t f = hexdigits[((20 + rec) >> (4 * j)) & 0x0F];
            *v++ = f; *v++ = f; *v++ = f; *v++ = f;
        }
        *v++ = 0xCC; *v++ = 0xDD; *v++ = 0xEE; *v++ = 0xFF;  /* :106-109 */
    }
}

/* ------------------------------------------------------------------ */
/* Sort, u64 keys, ascending. Restates the observable output of the
 * reference sample-sort (api/sort.hpp MainOp/PushData): under the total
 * order on u64 the globally sorted sequence is unique, and the reference's
 * splitter partition + per-bucket std::sort + rank-ordered concatenation
 * produces exactly it (classification is by (key, global index) — see
 * t9o_classify_u64 — so buckets are contiguous ranges of the sorted
 * multiset). */
void t9o_sort_u64(uint64_t* keys, uint64_t n) {
    std::sort(keys, keys + n);
}

/* Sort fixed-size records ascending by the acceptance total order:
 * lexicographic over key_len key bytes, ties broken by the remaining bytes
 * == memcmp over the whole record (the key is a prefix). The reference
 * comparator (terasort.cpp:35-37) compares only the key bytes; feeding this
 * total-order refinement to BOTH sides makes the output unique
 * (SURVEY.md §8c parity definition). */
/* OpenMP-parallel variant of the same semantics for the bench.py
 * cpu_baseline leg ("OpenMP across all cores, core count
 * reported"): per-chunk index sort + iterative pairwise merges, then one
 * permute pass. Returns the number of threads used. */
int t9o_sort_records_parallel(uint8_t* recs, uint64_t n, uint32_t rec_size);

void t9o_sort_records(uint8_t* recs, uint64_t n, uint32_t rec_size) {
    /* index sort + permute to avoid O(n) 100-byte swaps inside std::sort */
    std::vector<uint64_t> idx(n);
    for (uint64_t i = 0; i < n; ++i) idx[i] = i;
    std::sort(idx.begin(), idx.end(), [&](uint64_t a, uint64_t b) {
        return std::memcmp(recs + a * rec_size, recs + b * rec_size,
                           rec_size) < 0;
    });
    std::vector<uint8_t> tmp((size_t)n * rec_size);
    for (uint64_t i = 0; i < n; ++i)
        std::memcpy(tmp.data() + i * rec_size, recs + idx[i] * rec_size,
                    rec_size);
    std::memcpy(recs, tmp.data(), (size_t)n * rec_size);
}

int t9o_sort_records_parallel(uint8_t* recs, uint64_t n,
                              uint32_t rec_size) {
    int threads = 1;
#ifdef _OPENMP
    threads = omp_get_max_threads();
#endif
    auto cmp = [&](uint64_t x, uint64_t y) {
        return std::memcmp(recs + x * rec_size, recs + y * rec_size,
                           rec_size) < 0;
    };
    std::vector<uint64_t> idx(n);
    for (uint64_t i = 0; i < n; ++i) idx[i] = i;
#ifdef _OPENMP
    __gnu_parallel::sort(idx.begin(), idx.end(), cmp);
#else
    std::sort(idx.begin(), idx.end(), cmp);
#endif
    std::vector<uint8_t> out((size_t)n * rec_size);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (uint64_t i = 0; i < n; ++i)
        std::memcpy(out.data() + i * rec_size, recs + idx[i] * rec_size,
                    rec_size);
    std::memcpy(recs, out.data(), (size_t)n * rec_size);
    return threads;
}

/* ------------------------------------------------------------------ */
/* Splitter selection — api/sort.hpp:337-378 (FindAndSendSplitters).
 * Input: the gathered samples of all workers as (key, global index) pairs.
 * Sorts them by LessSampleIndex (api/sort.hpp:419-422: key order, ties by
 * index) and picks samples[(size_t)(i * size / p)] for i = 1..p-1.
 * Output: p-1 (splitter key, splitter index) pairs. */
void t9o_select_splitters_u64(const uint64_t* sample_keys,
                              const uint64_t* sample_idx,
                              uint64_t num_samples, uint32_t p,
                              uint64_t* out_keys, uint64_t* out_idx) {
    std::vector<std::pair<uint64_t, uint64_t> > s(num_samples);
    for (uint64_t i = 0; i < num_samples; ++i)
        s[i] = { sample_keys[i], sample_idx[i] };
    std::sort(s.begin(), s.end());   /* == LessSampleIndex for u64 keys */
    double splitting_size =
        (double)num_samples / (double)p;    /* api/sort.hpp:364-365 */
    for (uint32_t i = 1; i < p; ++i) {
        const auto& sp = s[(size_t)(i * splitting_size)];
        out_keys[i - 1] = sp.first;
        out_idx[i - 1] = sp.second;
    }
}

/* Classification — api/sort.hpp:380-535. Restated in closed form: the
 * reference runs each item down the splitter tree (TransmitItems :478-482)
 * and then walks back while EqualSampleGreaterIndex(splitter[b-1],
 * (item, gidx)) (:487-501), where EqualSampleGreaterIndex(a,b) :=
 * !cmp(a.key,b.key) && a.idx >= b.idx (:424-426). After the tree descent
 * b = #splitters with key < item (ties resolved right), and the walk moves
 * left past splitters with equal key and idx >= gidx. Net effect:
 *     bucket(item, gidx) = #{ j : (splitter_key[j], splitter_idx[j])
 *                                   <  (item_key, gidx)  lexicographically }
 * This closed form is what the GPU kernel implements; tests also run the
 * literal tree-descent restatement and assert the two agree element-wise. */
void t9o_classify_u64(const uint64_t* keys, uint64_t n, uint64_t gidx0,
                      const uint64_t* splitter_keys,
                      const uint64_t* splitter_idx, uint32_t p,
                      uint32_t* out_bucket) {
    for (uint64_t i = 0; i < n; ++i) {
        uint64_t k = keys[i], g = gidx0 + i;
        uint32_t b = 0;
        for (uint32_t j = 0; j < p - 1; ++j) {
            if (splitter_keys[j] < k ||
                (splitter_keys[j] == k && splitter_idx[j] < g))
                b = j + 1;
        }
        out_bucket[i] = b;
    }
}

/* Literal tree-descent restatement of api/sort.hpp:434-535 (TransmitItems),
 * for cross-checking the closed form above. k = 2^ceil(log2 p) buckets,
 * sentinel splitters replicate the last (:606-609); tree built by
 * TreeBuilder (:380-417). */
void t9o_classify_u64_tree(const uint64_t* keys, uint64_t n, uint64_t gidx0,
                           const uint64_t* splitter_keys,
                           const uint64_t* splitter_idx, uint32_t p,
                           uint32_t* out_bucket) {
    uint32_t log_k = 0;
    while ((1u << log_k) < p) ++log_k;
    uint32_t k = 1u << log_k;
    /* splitters padded with sentinels == last splitter (api/sort.hpp:606-609) */
    std::vector<std::pair<uint64_t, uint64_t> > spl(k - 1);
    for (uint32_t j = 0; j < k - 1; ++j) {
        uint32_t jj = j < p - 1 ? j : p - 2;
        spl[j] = { splitter_keys[jj], splitter_idx[jj] };
    }
    /* TreeBuilder (api/sort.hpp:393-417): recursive midpoint tree */
    std::vector<uint64_t> tree(k + 1, 0);
    struct Rec {
        std::vector<uint64_t>& tree;
        const std::pair<uint64_t, uint64_t>* samples;
        size_t ssplitter;
        void recurse(const std::pair<uint64_t, uint64_t>* lo,
                     const std::pair<uint64_t, uint64_t>* hi,
                     unsigned treeidx) {
            const auto* mid = lo + (hi - lo) / 2;
            tree[treeidx] = mid->first;
            if (2 * treeidx < ssplitter) {
                recurse(lo, mid, 2 * treeidx + 0);
                recurse(mid + 1, hi, 2 * treeidx + 1);
            }
        }
    } rec{ tree, spl.data(), k - 1 };
    if (k > 1) rec.recurse(spl.data(), spl.data() + (k - 1), 1);

    for (uint64_t i = 0; i < n; ++i) {
        uint64_t el = keys[i], g = gidx0 + i;
        size_t j = 1;
        for (uint32_t l = 0; l < log_k; ++l)
            j = 2 * j + (el < tree[j] ? 0 : 1);   /* :480 */
        size_t b = j - k;
        /* EqualSampleGreaterIndex walk (:487-501, :424-426) */
        while (b && !(spl[b - 1].first < el) && spl[b - 1].second >= g)
            --b;
        /* bucket ids above p-1 can only arise from sentinel padding; the
         * reference swaps the last real writer into slot k-1 (:457-461):
         * every sentinel bucket IS the last real bucket */
        out_bucket[i] = (uint32_t)(b >= p ? p - 1 : b);
    }
}

/* Record variant of the closed-form classification: comparator = key-byte
 * lexicographic order (terasort.cpp:35-37), ties by splitter global index. */
void t9o_classify_rec(const uint8_t* recs, uint64_t n, uint64_t gidx0,
                      uint32_t rec_size, uint32_t key_len,
                      const uint8_t* splitters, const uint64_t* splitter_idx,
                      uint32_t p, uint32_t* out_bucket) {
    for (uint64_t i = 0; i < n; ++i) {
        const uint8_t* r = recs + i * rec_size;
        uint64_t g = gidx0 + i;
        uint32_t b = 0;
        for (uint32_t j = 0; j < p - 1; ++j) {
            int c = std::memcmp(splitters + j * key_len, r, key_len);
            if (c < 0 || (c == 0 && splitter_idx[j] < g)) b = j + 1;
        }
        out_bucket[i] = b;
    }
}

/* ------------------------------------------------------------------ */
/* ReduceByKey, u64 keys / u64 values, reduce = addition.
 * Restates core/reduce_probing_hash_table.hpp:190-268 (linear probing in a
 * partition slice, in-place reduce on key match, GrowAndRehash :293-333 by
 * doubling) with the ReduceByHash index mapping
 * (core/reduce_functional.hpp:60-72): h = Hash128to64(salt, k);
 * local = (h / num_partitions) % size. Single partition here (the output —
 * the key-sorted (key,sum) multiset — is invariant to partitioning; the
 * partition split across ranks is t9o_partition_of_u64). Sentinel key 0 is
 * reduced in a dedicated extra slot (:195-217). No spill: everything fits.
 * Output sorted by key; returns the number of unique keys (< 0 impossible;
 * if out capacity is exceeded returns UINT64_MAX). */
uint64_t t9o_reduce_u64(const uint64_t* keys, const uint64_t* vals,
                        uint64_t n, uint64_t salt, uint64_t num_partitions,
                        uint64_t* out_keys, uint64_t* out_vals,
                        uint64_t cap) {
    size_t size = 512;                 /* initial_items_per_partition_,
                                          core/reduce_table.hpp:55 */
    std::vector<uint64_t> tk(size, 0), tv(size, 0);
    std::vector<uint8_t> used(size, 0);
    uint64_t sentinel_val = 0; bool sentinel_used = false;
    uint64_t items = 0;
    const double fill = 0.5;           /* limit_partition_fill_rate,
                                          core/reduce_table.hpp:45 */
    if (num_partitions == 0) num_partitions = 1;

    auto grow = [&]() {                /* GrowAndRehash :293-333, ×2 */
        size_t nsize = size * 2;
        std::vector<uint64_t> nk(nsize, 0), nv(nsize, 0);
        std::vector<uint8_t> nu(nsize, 0);
        for (size_t s = 0; s < size; ++s) {
            if (!used[s]) continue;
            uint64_t h = t9o_hash128to64(salt, tk[s]);
            size_t idx = (size_t)((h / num_partitions) % nsize);
            while (nu[idx]) idx = (idx + 1) % nsize;
            nu[idx] = 1; nk[idx] = tk[s]; nv[idx] = tv[s];
        }
        tk.swap(nk); tv.swap(nv); used.swap(nu); size = nsize;
    };

    for (uint64_t i = 0; i < n; ++i) {
        uint64_t k = keys[i], v = vals[i];
        if (k == 0) {                  /* sentinel key handling :195-217 */
            if (!sentinel_used) { sentinel_used = true; sentinel_val = v; ++items; }
            else sentinel_val += v;
            continue;
        }
        for (;;) {
            uint64_t h = t9o_hash128to64(salt, k);
            size_t begin = (size_t)((h / num_partitions) % size);
            size_t idx = begin;
            bool done = false;
            while (used[idx]) {        /* probing loop :229-248 */
                if (tk[idx] == k) { tv[idx] += v; done = true; break; }
                idx = (idx + 1) % size;
                if (idx == begin) break;   /* full: grow and retry :244-247 */
            }
            if (done) break;
            if (!used[idx]) {
                used[idx] = 1; tk[idx] = k; tv[idx] = v; ++items;
                while ((double)items >= fill * (double)size) grow();
                break;
            }
            grow();
        }
    }

    std::vector<std::pair<uint64_t, uint64_t> > out;
    out.reserve(items);
    if (sentinel_used) out.push_back({ 0, sentinel_val });
    for (size_t s = 0; s < size; ++s)
        if (used[s]) out.push_back({ tk[s], tv[s] });
    std::sort(out.begin(), out.end());
    if (out.size() > cap) return UINT64_MAX;
    for (size_t s = 0; s < out.size(); ++s) {
        out_keys[s] = out[s].first;
        out_vals[s] = out[s].second;
    }
    return out.size();
}

/* ------------------------------------------------------------------ */
/* 128-bit composite-key reduce — the reference reduces (string, u64)
 * with EQUALITY ON THE FULL KEY (core/reduce_probing_hash_table.hpp:233
 * probes compare keys); the MI355X path dictionary-encodes words into
 * two independent 64-bit hashes and reduces on the composite. This
 * restatement keeps the same probing-table semantics as t9o_reduce_u64
 * above (probe start from Hash128to64(salt, k1), equality on the
 * (k1, k2) pair — a k1 match with a k2 mismatch probes on, exactly the
 * device protocol, t9_reduce.hip t9_g128_insert). Output sorted by
 * (k1, k2). Returns the number of pairs, or UINT64_MAX on cap overflow. */
extern "C" uint64_t t9o_reduce128(const uint64_t* k1s, const uint64_t* k2s,
                                  const uint64_t* vals, uint64_t n,
                                  uint64_t salt, uint64_t* out_k1,
                                  uint64_t* out_k2, uint64_t* out_vals,
                                  uint64_t cap) {
    size_t size = 512;
    std::vector<uint64_t> t1(size, 0), t2(size, 0), tv(size, 0);
    std::vector<uint8_t> used(size, 0);
    uint64_t items = 0;
    const double fill = 0.5;

    auto grow = [&]() {
        size_t nsize = size * 2;
        std::vector<uint64_t> n1(nsize, 0), n2(nsize, 0), nv(nsize, 0);
        std::vector<uint8_t> nu(nsize, 0);
        for (size_t s2 = 0; s2 < size; ++s2) {
            if (!used[s2]) continue;
            size_t idx = (size_t)(t9o_hash128to64(salt, t1[s2]) % nsize);
            while (nu[idx]) idx = (idx + 1) % nsize;
            nu[idx] = 1; n1[idx] = t1[s2]; n2[idx] = t2[s2];
            nv[idx] = tv[s2];
        }
        t1.swap(n1); t2.swap(n2); tv.swap(nv); used.swap(nu);
        size = nsize;
    };

    for (uint64_t i = 0; i < n; ++i) {
        const uint64_t k1 = k1s[i], k2 = k2s[i];
        const uint64_t v = vals ? vals[i] : 1;
        for (;;) {
            size_t begin = (size_t)(t9o_hash128to64(salt, k1) % size);
            size_t idx = begin;
            bool done = false;
            while (used[idx]) {
                if (t1[idx] == k1 && t2[idx] == k2) {
                    tv[idx] += v; done = true; break;
                }
                idx = (idx + 1) % size;
                if (idx == begin) break;
            }
            if (done) break;
            if (!used[idx]) {
                used[idx] = 1; t1[idx] = k1; t2[idx] = k2; tv[idx] = v;
                ++items;
                while ((double)items >= fill * (double)size) grow();
                break;
            }
            grow();
        }
    }
    std::vector<std::tuple<uint64_t, uint64_t, uint64_t> > out;
    out.reserve(items);
    for (size_t s2 = 0; s2 < size; ++s2)
        if (used[s2])
            out.push_back(std::make_tuple(t1[s2], t2[s2], tv[s2]));
    std::sort(out.begin(), out.end());
    if (out.size() > cap) return UINT64_MAX;
    for (size_t s2 = 0; s2 < out.size(); ++s2) {
        out_k1[s2] = std::get<0>(out[s2]);
        out_k2[s2] = std::get<1>(out[s2]);
        out_vals[s2] = std::get<2>(out[s2]);
    }
    return out.size();
}

/* ------------------------------------------------------------------ */
/* ReduceToIndex — thrill/api/reduce_to_index.hpp with the ReduceByIndex
 * mapping (core/reduce_functional.hpp:84-149): keys are dense indices in
 * [begin, begin+size); the result is the dense value array with the sum
 * per index (absent indices keep the neutral value 0, as the reference's
 * by-index post phase emits neutral elements). Returns 0, or -1 if a key
 * is out of range. */
int t9o_reduce_by_index(const uint64_t* keys, const uint64_t* vals,
                        uint64_t n, uint64_t begin, uint64_t size,
                        uint64_t* dense) {
    for (uint64_t i = 0; i < size; ++i) dense[i] = 0;
    for (uint64_t i = 0; i < n; ++i) {
        if (keys[i] < begin || keys[i] - begin >= size) return -1;
        dense[keys[i] - begin] += vals[i];
    }
    return 0;
}

/* ------------------------------------------------------------------ */
/* Zipf(s, N) token sampling by inverse CDF. The reference's
 * common/zipf_distribution.hpp:55-120 draws from std::discrete_distribution
 * over weights 1/(k+q)^s; we restate the same mass function with an explicit
 * CDF table + a splitmix64 uniform, so CPU and GPU sample identical token
 * streams. Tokens are in [1, N] (zipf_distribution.hpp:94). */
void t9o_zipf_cdf(double* cdf, uint64_t N, double s, double q) {
    double acc = 0.0;
    for (uint64_t k = 1; k <= N; ++k) {
        acc += 1.0 / std::pow((double)k + q, s);
        cdf[k - 1] = acc;
    }
    for (uint64_t k = 0; k < N; ++k) cdf[k] /= acc;
}

void t9o_zipf_tokens(uint64_t* out, const double* cdf, uint64_t N,
                     uint64_t index0, uint64_t n, uint64_t seed) {
    for (uint64_t i = 0; i < n; ++i) {
        double u = (double)(splitmix64_at(seed, index0 + i) >> 11)
                   * (1.0 / 9007199254740992.0);   /* [0,1) with 53 bits */
        /* smallest k with cdf[k] > u  (upper_bound) */
        uint64_t lo = 0, hi = N - 1;
        while (lo < hi) {
            uint64_t mid = (lo + hi) / 2;
            if (cdf[mid] > u) hi = mid; else lo = mid + 1;
        }
        out[i] = lo + 1;
    }
}

} /* extern "C" */
