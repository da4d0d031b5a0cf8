/* t9/dia.hpp — C++ operator surface mirroring the reference's DIA API for
 * the Sort/ReduceByKey hot path, executing on the GPU through the C ABI of
 * thrill_amd.h. Names, argument meaning and error behaviour follow
 * thrill/api/dia.hpp (Sort :798-937, ReduceByKey via ReducePair
 * :241-463, Generate api/generate.hpp:37, Size api/size.hpp:28,
 * AllGather api/all_gather.hpp) restricted to the GPU-executable contract:
 * fixed-size POD items, byte-lexicographic (or u64) key order, u64 sum
 * reduction. There is NO CPU execution path here — without a GPU,
 * Context construction fails (the CPU restatement lives in oracle/ and is
 * test infrastructure).
 *
 * Execution model: eager per-op evaluation (each DOp runs when
 * constructed) on a single rank; the reference's lazy Stage/Execute
 * machinery (api/dia_base.cpp:381-443) is unnecessary for the two target
 * pipelines, whose DAGs are straight lines — see DESIGN.md. Multi-rank
 * execution goes through the python pipeline (thrill_amd/pipeline.py) or
 * the C ABI's own RCCL bootstrap (t9_comm_id/t9_comm_init).
 *
 * Round 2 additions, driven by the verbatim example ports (examples/):
 * host-side DIAs for non-POD items (std::string lines), FlatMap, and a
 * generic ReduceByKey for (string, u64-counter) pairs that
 * dictionary-encodes words and reduces on the GPU 128-bit composite
 * table (string identity — DESIGN.md "Config-4 string identity").
 */
#pragma once

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <memory>
#include <stdexcept>
#include <string>
#include <type_traits>
#include <unordered_map>
#include <utility>
#include <vector>

#include "../thrill_amd.h"

#include <fcntl.h>
#include <unistd.h>

namespace t9 {
namespace api {

namespace detail {

//! O_DIRECT bulk file write through a 16 MiB aligned bounce buffer —
//! the buffered+fsync path measured 1.8 GB/s against the box medium's
//! 8.2 GB/s O_DIRECT line rate (profiles/r02_io_bench.json notes); the
//! unaligned tail goes through a buffered descriptor. Falls back to
//! stdio wholesale if O_DIRECT open fails (filesystem-dependent).
inline bool write_file_direct(const char* name, const void* data,
                              size_t bytes) {
    const size_t CH = 16u << 20, ALIGN = 4096;
    int fd = ::open(name, O_WRONLY | O_CREAT | O_TRUNC | O_DIRECT, 0644);
    if (fd < 0) return false;
    void* buf = nullptr;
    if (posix_memalign(&buf, ALIGN, CH) != 0) {
        ::close(fd);
        return false;
    }
    const size_t aligned = bytes / ALIGN * ALIGN;
    size_t off = 0;
    bool ok = true;
    while (ok && off < aligned) {
        size_t len = std::min(CH, aligned - off);
        std::memcpy(buf, (const char*)data + off, len);
        ok = ::write(fd, buf, len) == (ssize_t)len;
        off += len;
    }
    ::fsync(fd);
    ::close(fd);
    std::free(buf);
    if (ok && aligned < bytes) {
        int fd2 = ::open(name, O_WRONLY);
        ok = fd2 >= 0 &&
             ::pwrite(fd2, (const char*)data + aligned, bytes - aligned,
                      (off_t)aligned) == (ssize_t)(bytes - aligned);
        if (fd2 >= 0) {
            ::fsync(fd2);
            ::close(fd2);
        }
    }
    return ok;
}

inline bool read_file_direct(const char* name, void* data, size_t bytes) {
    const size_t CH = 16u << 20, ALIGN = 4096;
    int fd = ::open(name, O_RDONLY | O_DIRECT);
    if (fd < 0) return false;
    void* buf = nullptr;
    if (posix_memalign(&buf, ALIGN, CH) != 0) {
        ::close(fd);
        return false;
    }
    const size_t aligned = bytes / ALIGN * ALIGN;
    size_t off = 0;
    bool ok = true;
    while (ok && off < aligned) {
        size_t len = std::min(CH, aligned - off);
        ok = ::read(fd, buf, len) == (ssize_t)len;
        if (ok) std::memcpy((char*)data + off, buf, len);
        off += len;
    }
    ::close(fd);
    std::free(buf);
    if (ok && aligned < bytes) {
        int fd2 = ::open(name, O_RDONLY);
        ok = fd2 >= 0 &&
             ::pread(fd2, (char*)data + aligned, bytes - aligned,
                     (off_t)aligned) == (ssize_t)(bytes - aligned);
        if (fd2 >= 0) ::close(fd2);
    }
    return ok;
}

//! device-buffer -> file, O_DIRECT, PINNED ping-pong staging: the
//! pageable AllGather D2H measured ~1.5-2 GB/s and throttled the whole
//! write; pinned transfers overlap the previous chunk's disk write.
//! Returns false to signal the caller's fallback.
inline bool write_device_direct(const char* name, const void* d_ptr,
                                size_t bytes, hipStream_t s) {
    const size_t CH = 64u << 20;   /* 64 MiB: fewer sync/chunk overheads */
    int fd = ::open(name, O_WRONLY | O_CREAT | O_TRUNC | O_DIRECT, 0644);
    if (fd < 0) return false;
    void* pin[2] = { nullptr, nullptr };
    void* dbuf = nullptr;   /* plain aligned bounce: O_DIRECT DMA from
                               GPU-registered pinned pages measured
                               ~3x slower than from normal memory */
    if (hipHostMalloc(&pin[0], CH, 0) != hipSuccess ||
        hipHostMalloc(&pin[1], CH, 0) != hipSuccess ||
        posix_memalign(&dbuf, 4096, CH) != 0) {
        if (pin[0]) (void)hipHostFree(pin[0]);
        if (pin[1]) (void)hipHostFree(pin[1]);
        ::close(fd);
        return false;
    }
    const size_t ALIGN = 4096;
    const size_t aligned = bytes / ALIGN * ALIGN;
    bool ok = true;
    size_t off = 0;
    int cur = 0;
    size_t len0 = std::min(CH, aligned);
    if (len0)
        ok = hipMemcpyAsync(pin[0], (const char*)d_ptr, len0,
                            hipMemcpyDeviceToHost, s) == hipSuccess;
    while (ok && off < aligned) {
        const size_t len = std::min(CH, aligned - off);
        ok = hipStreamSynchronize(s) == hipSuccess;
        const size_t noff = off + len;
        if (ok && noff < aligned) {
            const size_t nlen = std::min(CH, aligned - noff);
            ok = hipMemcpyAsync(pin[cur ^ 1], (const char*)d_ptr + noff,
                                nlen, hipMemcpyDeviceToHost,
                                s) == hipSuccess;
        }
        /* write straight from the pinned buffer: an A/B with a plain
         * aligned bounce measured WORSE on the write side (box-variable
         * storage; profiles/r02_io_bench.json notes) */
        ok = ok && ::write(fd, pin[cur], len) == (ssize_t)len;
        off = noff;
        cur ^= 1;
    }
    std::free(dbuf);
    (void)dbuf;
    ::fsync(fd);
    ::close(fd);
    if (ok && aligned < bytes) {
        ok = hipMemcpy(pin[0], (const char*)d_ptr + aligned,
                       bytes - aligned, hipMemcpyDeviceToHost) ==
             hipSuccess;
        int fd2 = ::open(name, O_WRONLY);
        ok = ok && fd2 >= 0 &&
             ::pwrite(fd2, pin[0], bytes - aligned, (off_t)aligned) ==
                 (ssize_t)(bytes - aligned);
        if (fd2 >= 0) {
            ::fsync(fd2);
            ::close(fd2);
        }
    }
    (void)hipHostFree(pin[0]);
    (void)hipHostFree(pin[1]);
    return ok;
}

//! file -> device buffer, O_DIRECT, pinned ping-pong (mirror of the
//! writer: disk read of chunk i overlaps the H2D copy of chunk i-1)
inline bool read_device_direct(const char* name, void* d_ptr,
                               size_t bytes, hipStream_t s) {
    const size_t CH = 64u << 20;
    int fd = ::open(name, O_RDONLY | O_DIRECT);
    if (fd < 0) return false;
    void* pin[2] = { nullptr, nullptr };
    if (hipHostMalloc(&pin[0], CH, 0) != hipSuccess ||
        hipHostMalloc(&pin[1], CH, 0) != hipSuccess) {
        if (pin[0]) (void)hipHostFree(pin[0]);
        ::close(fd);
        return false;
    }
    const size_t ALIGN = 4096;
    const size_t aligned = bytes / ALIGN * ALIGN;
    bool ok = true;
    size_t off = 0;
    int cur = 0;
    void* dbuf2 = nullptr;
    if (posix_memalign(&dbuf2, ALIGN, CH) != 0) {
        (void)hipHostFree(pin[0]);
        (void)hipHostFree(pin[1]);
        ::close(fd);
        return false;
    }
    while (ok && off < aligned) {
        const size_t len = std::min(CH, aligned - off);
        /* O_DIRECT into plain aligned memory, then host->pinned, then
         * async H2D; the disk read of the next chunk overlaps the H2D
         * (sync only before reusing the pinned buffer) */
        ok = ::read(fd, dbuf2, len) == (ssize_t)len;
        if (ok) {
            std::memcpy(pin[cur], dbuf2, len);
            ok = hipMemcpyAsync((char*)d_ptr + off, pin[cur], len,
                                hipMemcpyHostToDevice, s) == hipSuccess;
        }
        cur ^= 1;
        off += len;
        /* sync unconditionally: the tail path below reuses pin[0], and
         * the next iteration reuses the flipped buffer — either way the
         * in-flight H2D must drain first (an intermittent corruption of
         * the LAST chunk escaped when this sync was skipped on the
         * final iteration) */
        if (ok)
            ok = hipStreamSynchronize(s) == hipSuccess;
    }
    std::free(dbuf2);
    ::close(fd);
    if (ok && aligned < bytes) {
        int fd2 = ::open(name, O_RDONLY);
        ok = fd2 >= 0 &&
             ::pread(fd2, pin[0], bytes - aligned, (off_t)aligned) ==
                 (ssize_t)(bytes - aligned);
        if (fd2 >= 0) ::close(fd2);
        if (ok)
            ok = hipMemcpyAsync((char*)d_ptr + aligned, pin[0],
                                bytes - aligned, hipMemcpyHostToDevice,
                                s) == hipSuccess;
    }
    ok = ok && hipStreamSynchronize(s) == hipSuccess;
    (void)hipHostFree(pin[0]);
    (void)hipHostFree(pin[1]);
    return ok;
}

} // namespace detail

#define T9_DIA_TRY(expr)                                                  \
    do {                                                                  \
        int _rc = (expr);                                                 \
        if (_rc != 0)                                                     \
            throw std::runtime_error(std::string(#expr) +                 \
                                     " failed rc=" + std::to_string(_rc)); \
    } while (0)

#define T9_DIA_HIP(expr)                                                  \
    do {                                                                  \
        hipError_t _e = (expr);                                           \
        if (_e != hipSuccess)                                             \
            throw std::runtime_error(std::string(#expr) + " failed: " +   \
                                     hipGetErrorString(_e));              \
    } while (0)

//! (key, value) pair for ReducePair — a POD TableItem (the reference's
//! pair<Key,Value> TableItem concept, core/reduce_table.hpp; std::pair is
//! not trivially copyable, so the GPU surface uses this layout-compatible
//! struct)
struct KeyValue {
    uint64_t key;
    uint64_t value;
    bool operator==(const KeyValue& o) const {
        return key == o.key && value == o.value;
    }
};
static_assert(sizeof(KeyValue) == 16, "KeyValue packing");

//! byte-lexicographic ascending order over a POD's bytes — the acceptance
//! comparator (TeraSort Record::operator< is its key-prefix restriction,
//! examples/terasort/terasort.cpp:35-37)
template <typename T>
struct LexicographicLess {
    bool operator()(const T& a, const T& b) const {
        return std::memcmp(&a, &b, sizeof(T)) < 0;
    }
};

//! Per-worker context (mirrors thrill api::Context services the hot path
//! consumes: my_rank/num_workers — api/context.hpp:243-245).
class Context {
public:
    explicit Context(int device = 0, int rank = 0, int world = 1,
                     void* rccl_comm = nullptr)
        : rank_(rank), world_(world) {
        T9_DIA_TRY(t9_create(&ctx_, device, rank, world, rccl_comm));
        T9_DIA_HIP(hipStreamCreate(&stream_));
    }
    ~Context() {
        if (stream_) (void)hipStreamDestroy(stream_);
        if (ctx_) t9_destroy(ctx_);
    }
    Context(const Context&) = delete;
    Context& operator=(const Context&) = delete;

    size_t my_rank() const { return rank_; }
    size_t num_workers() const { return world_; }
    t9_context* native() const { return ctx_; }
    hipStream_t stream() const { return stream_; }

    //! reference api/context.hpp consume mode (memory hint): accepted and
    //! ignored — DIAs here are explicit device buffers
    void enable_consume() {}

    //! scalar net services the examples touch (ctx.net.Barrier(),
    //! terasort.cpp:204): single-process no-ops; multi-rank scalar
    //! collectives run on the host control plane (pipeline.py) or a
    //! caller-provided communicator
    struct Net {
        void Barrier() {}
    } net;

private:
    t9_context* ctx_ = nullptr;
    hipStream_t stream_ = nullptr;
    int rank_, world_;
};

//! device buffer with shared ownership (DIAs are cheap handles, as the
//! reference's CountingPtr-refcounted nodes are)
struct DeviceBuf {
    void* ptr = nullptr;
    size_t bytes = 0;
    explicit DeviceBuf(size_t b) : bytes(b) {
        if (b) T9_DIA_HIP(hipMalloc(&ptr, b));
    }
    ~DeviceBuf() {
        if (ptr) (void)hipFree(ptr);
    }
};

//! Items that live on the GPU: fixed-size POD, raw-copied
//! (data/serialization.hpp:35-48 contract). Non-POD items (std::string
//! lines/words — the word_count tokenizer input) are held host-side, as
//! in the reference where string handling is CPU code fused into PreOp
//! chains; the DOp that consumes them
//! (ReduceByKey) dictionary-encodes and moves fixed-width work to the
//! GPU.
template <typename ValueType>
class DIA {
    static constexpr bool kDevice =
        std::is_trivially_copyable<ValueType>::value;

public:
    DIA() = default;
    DIA(Context* ctx, std::shared_ptr<DeviceBuf> buf, size_t n)
        : ctx_(ctx), buf_(buf), n_(n) {}
    DIA(Context* ctx, std::shared_ptr<std::vector<ValueType> > host)
        : ctx_(ctx), hvec_(host), n_(host ? host->size() : 0) {}

    size_t Size() const {
        // reference: ActionNode + AllReduce (api/size.hpp:64-69); world=1
        // here, so the local count is the global count
        return n_;
    }

    std::vector<ValueType> AllGather() const {
        if (hvec_) return *hvec_;
        std::vector<ValueType> out(n_);
        if (n_)
            T9_DIA_HIP(hipMemcpy(out.data(), buf_->ptr,
                                 n_ * sizeof(ValueType),
                                 hipMemcpyDeviceToHost));
        return out;
    }

    //! Sort with the default ascending order (reference api/sort.hpp:811:
    //! std::less; for byte-key PODs the acceptance order is
    //! LexicographicLess). Only these two comparator types are executable
    //! on the GPU path; anything else fails to compile, by design
    //! (INTEGRATION.md "Error behaviour").
    DIA Sort() const { return SortImpl(); }

    //! SortStable (reference api/sort.hpp:900-937): the radix pipeline is
    //! stable at every level, so Sort already implements the stable
    //! contract (sort_node_test.cpp:291-404's "lower source index wins").
    DIA SortStable() const { return SortImpl(); }

    template <typename Compare>
    DIA SortStable(const Compare& c) const { return Sort(c); }

    template <typename Compare>
    DIA Sort(const Compare&) const {
        static_assert(
            std::is_same<Compare, LexicographicLess<ValueType> >::value ||
                (std::is_same<Compare, std::less<ValueType> >::value &&
                 std::is_same<ValueType, uint64_t>::value),
            "GPU Sort supports LexicographicLess<T> (byte order) or "
            "std::less<uint64_t>");
        return SortImpl();
    }

    template <typename F>
    auto Map(const F& f) const
        -> DIA<typename std::result_of<F(ValueType)>::type> {
        // host-side map (the reference fuses Maps into PreOp chains on the
        // CPU — dia.hpp:358-405; tokenizers etc. stay CPU per SURVEY §3b)
        using Out = typename std::result_of<F(ValueType)>::type;
        auto in = AllGather();
        std::vector<Out> out;
        out.reserve(in.size());
        for (auto& v : in) out.push_back(f(v));
        return FromVector(*ctx_, out);
    }

    //! FlatMap — reference api/dia.hpp:405 (template on the output type,
    //! user lambda receives (item, emit)): host code, exactly as the
    //! reference fuses it into the CPU PreOp chain; the DOp that follows
    //! runs on the GPU.
    template <typename Out, typename F>
    DIA<Out> FlatMap(const F& f) const {
        auto in = AllGather();
        auto out = std::make_shared<std::vector<Out> >();
        auto emit = [&](const Out& o) { out->push_back(o); };
        for (auto& v : in) f(v, emit);
        return FromVector(*ctx_, *out);
    }

    //! ReduceByKey — reference api/reduce_by_key.hpp:241-463: reduce items
    //! with equal key_extractor(item) using the associative+commutative
    //! reduce_function; output order arbitrary (word_count_test.cpp:73-74).
    //! Implemented for pair<std::string, integral> items (the word_count
    //! shape): words are dictionary-encoded into two independent 64-bit
    //! hashes and reduced on the GPU 128-bit composite table
    //! (t9_reduce128_*; string identity per DESIGN.md — forced single-hash
    //! collisions stay separate). The u64-sum hot path is used when the
    //! user's reduce_function is additive on the counter — verified by
    //! ALGEBRAIC PROBING (the functor is a pure function of POD-ish pairs:
    //! it is evaluated on sample values and must return the counter sum,
    //! the shape word_count.hpp:50-53 has); a non-additive functor falls
    //! back to GPU grouping + a host fold with the user's functor.
    template <typename KeyExtractor, typename ReduceFn>
    DIA<ValueType> ReduceByKey(const KeyExtractor& key_ex,
                               const ReduceFn& red) const;

    //! WriteBinary — reference api/write_binary.hpp: the items as packed
    //! binary (the POD raw-copy wire format, data/serialization.hpp:35-48),
    //! one file per worker: pathbase + zero-padded rank. The on-disk bytes
    //! are bit-compatible with the reference's files (terasort.cpp:184-200
    //! file mode).
    void WriteBinary(const std::string& pathbase) const {
        static_assert(std::is_trivially_copyable<ValueType>::value,
                      "WriteBinary needs fixed-size POD items");
        char name[512];
        std::snprintf(name, sizeof(name), "%s%010zu", pathbase.c_str(),
                      ctx_->my_rank());
        if (buf_ && n_) {
            // device-resident items: O_DIRECT + pinned ping-pong
            // (HBM -> disk without the pageable-copy throttle)
            if (detail::write_device_direct(name, buf_->ptr,
                                            n_ * sizeof(ValueType),
                                            ctx_->stream()))
                return;
        }
        auto host = AllGather();
        const size_t bytes = host.size() * sizeof(ValueType);
        if (detail::write_file_direct(name, host.data(), bytes))
            return;
        // buffered fallback (O_DIRECT unsupported on this filesystem)
        std::FILE* f = std::fopen(name, "wb");
        if (!f) throw std::runtime_error("WriteBinary: cannot open " +
                                         std::string(name));
        if (!host.empty() &&
            std::fwrite(host.data(), sizeof(ValueType), host.size(), f) !=
                host.size()) {
            std::fclose(f);
            throw std::runtime_error("WriteBinary: short write");
        }
        std::fclose(f);
    }

    Context& context() const { return *ctx_; }
    void* device_ptr() const { return buf_ ? buf_->ptr : nullptr; }

private:
    DIA SortImpl() const;

    template <typename T>
    friend DIA<T> FromVector(Context&, const std::vector<T>&);

    Context* ctx_ = nullptr;
    std::shared_ptr<DeviceBuf> buf_;
    std::shared_ptr<std::vector<ValueType> > hvec_;  // non-POD host items
    size_t n_ = 0;
};

template <typename T>
typename std::enable_if<std::is_trivially_copyable<T>::value,
                        DIA<T> >::type
FromVectorImpl(Context& ctx, const std::vector<T>& v) {
    auto buf = std::make_shared<DeviceBuf>(v.size() * sizeof(T));
    if (!v.empty())
        T9_DIA_HIP(hipMemcpy(buf->ptr, v.data(), v.size() * sizeof(T),
                             hipMemcpyHostToDevice));
    return DIA<T>(&ctx, buf, v.size());
}

template <typename T>
typename std::enable_if<!std::is_trivially_copyable<T>::value,
                        DIA<T> >::type
FromVectorImpl(Context& ctx, const std::vector<T>& v) {
    return DIA<T>(&ctx, std::make_shared<std::vector<T> >(v));
}

template <typename T>
DIA<T> FromVector(Context& ctx, const std::vector<T>& v) {
    return FromVectorImpl<T>(ctx, v);
}

//! Generate — reference api/generate.hpp:37: DIA of generator(i) for
//! i in [0, size). The generator runs on the host (it is user code);
//! the DOps that follow run on the GPU.
template <typename Generator>
auto Generate(Context& ctx, size_t size, Generator gen)
    -> DIA<decltype(gen(size_t(0)))> {
    // generator taken by value and callable non-const, as the
    // reference's GeneratorFunction is (api/generate.hpp:37; stateful
    // rngs like terasort's GenerateRecord mutate)
    using T = decltype(gen(size_t(0)));
    std::vector<T> v;
    v.reserve(size);
    for (size_t i = 0; i < size; ++i) v.push_back(gen(i));
    return FromVector(ctx, v);
}

template <typename ValueType>
DIA<ValueType> DIA<ValueType>::SortImpl() const {
    static_assert(std::is_trivially_copyable<ValueType>::value,
                  "GPU Sort needs fixed-size POD items "
                  "(data/serialization.hpp:35-48 raw-copy contract)");
    constexpr size_t R = sizeof(ValueType);
    static_assert(R % 4 == 0, "record size must be a multiple of 4");
    auto out = std::make_shared<DeviceBuf>(n_ * R);
    if (n_) {
        if (std::is_same<ValueType, uint64_t>::value) {
            // keys are numeric u64: in-place radix of a copy
            T9_DIA_HIP(hipMemcpyAsync(out->ptr, buf_->ptr, n_ * 8,
                                      hipMemcpyDeviceToDevice,
                                      ctx_->stream()));
            DeviceBuf ws(t9_sort_u64_workspace(n_));
            T9_DIA_TRY(t9_sort_u64(ctx_->native(), (uint64_t*)out->ptr, n_,
                                   ws.ptr, ctx_->stream()));
            T9_DIA_HIP(hipStreamSynchronize(ctx_->stream()));
        }
        else {
            DeviceBuf ws(t9_sort_records_workspace(n_, R));
            T9_DIA_TRY(t9_sort_records(ctx_->native(),
                                       (const uint8_t*)buf_->ptr,
                                       (uint8_t*)out->ptr, n_, R,
                                       /*key_len*/ R, ws.ptr,
                                       ctx_->stream()));
            T9_DIA_HIP(hipStreamSynchronize(ctx_->stream()));
        }
    }
    return DIA(ctx_, out, n_);
}

//! ReducePair for (u64 key, u64 value) pairs with u64-sum reduction — the
//! reference's ReducePair (api/reduce_by_key.hpp:393-463) restricted to
//! the GPU-executable contract; word_count's (hash(word), count) pairs are
//! exactly this shape. Output order is arbitrary, as the
//! reference documents for reducing (word_count_test.cpp:73-74).
inline DIA<KeyValue> ReducePair(const DIA<KeyValue>& input,
                                uint64_t salt = 0) {
    using KV = KeyValue;
    Context& ctx = input.context();
    size_t n = input.Size();
    // SoA split ON DEVICE: the pairs are interleaved u64s, i.e. 16-byte
    // records with a u64 field at offset 0 (the key) and offset 8 (the
    // value) — two t9_extract_key64_le passes split them without any
    // host round trip (round-1 split staged through host vectors).
    DeviceBuf dk(n * 8), dv(n * 8), didx(n * 4);
    if (n) {
        T9_DIA_TRY(t9_extract_key64_le(ctx.native(),
                                       (const uint8_t*)input.device_ptr(),
                                       n, 16, 0, (uint64_t*)dk.ptr,
                                       (uint32_t*)didx.ptr, ctx.stream()));
        T9_DIA_TRY(t9_extract_key64_le(ctx.native(),
                                       (const uint8_t*)input.device_ptr(),
                                       n, 16, 8, (uint64_t*)dv.ptr,
                                       (uint32_t*)didx.ptr, ctx.stream()));
    }
    uint64_t cap = 1024;
    while (cap < 2 * n + 2) cap <<= 1;   // no grow/spill: size for 2x
    DeviceBuf tbl(2 * (cap + 1) * 8);   // interleaved (key, sum) slots
    DeviceBuf ok((cap + 1) * 8), ov((cap + 1) * 8);
    DeviceBuf derr(4), dn(8);
    hipStream_t s = ctx.stream();
    T9_DIA_TRY(t9_reduce_init(ctx.native(), (uint64_t*)tbl.ptr, cap, s));
    T9_DIA_TRY(t9_reduce_build(ctx.native(), (const uint64_t*)dk.ptr,
                               (const uint64_t*)dv.ptr, n,
                               (uint64_t*)tbl.ptr, cap,
                               salt, (uint32_t*)derr.ptr, s));
    T9_DIA_TRY(t9_reduce_drain(ctx.native(), (const uint64_t*)tbl.ptr,
                               cap,
                               (uint64_t*)ok.ptr, (uint64_t*)ov.ptr,
                               (uint64_t*)dn.ptr, s));
    uint64_t m = 0, err = 0;
    uint32_t err32 = 0;
    T9_DIA_HIP(hipMemcpy(&m, dn.ptr, 8, hipMemcpyDeviceToHost));
    T9_DIA_HIP(hipMemcpy(&err32, derr.ptr, 4, hipMemcpyDeviceToHost));
    err = err32;
    if (err) throw std::runtime_error("ReducePair: table overflow");
    std::vector<uint64_t> rk(m), rv(m);
    if (m) {
        T9_DIA_HIP(hipMemcpy(rk.data(), ok.ptr, m * 8,
                             hipMemcpyDeviceToHost));
        T9_DIA_HIP(hipMemcpy(rv.data(), ov.ptr, m * 8,
                             hipMemcpyDeviceToHost));
    }
    std::vector<KV> out(m);
    for (size_t i = 0; i < m; ++i) out[i] = KV{ rk[i], rv[i] };
    return FromVector(ctx, out);
}

namespace detail {

//! the framework's string hash pair: fnv-1a under two bases, mixed by
//! Hash128to64 with two salts (DESIGN.md "config-4 string identity"; the
//! reference's std::hash is implementation-defined and affects placement
//! only). Sentinels remapped as t9_hash2_of does.
inline uint64_t fnv1a64(const char* s, size_t n, uint64_t basis) {
    uint64_t h = basis;
    for (size_t i = 0; i < n; ++i)
        h = (h ^ (uint8_t)s[i]) * 0x100000001B3ull;
    return h;
}

inline uint64_t mix128to64(uint64_t upper, uint64_t lower) {
    const uint64_t k = 0x9DDFEA08EB382D69ull;
    uint64_t a = (lower ^ upper) * k;
    a ^= (a >> 47);
    uint64_t b = (upper ^ a) * k;
    b ^= (b >> 47);
    b *= k;
    return b;
}

inline void string_hash2(const std::string& w, uint64_t* h1,
                         uint64_t* h2) {
    uint64_t a = mix128to64(0x9AE16A3B2F90404Full,
                            fnv1a64(w.data(), w.size(),
                                    0xCBF29CE484222325ull));
    uint64_t b = mix128to64(0xC3A5C85C97CB3127ull,
                            fnv1a64(w.data(), w.size(),
                                    0x84222325CBF29CE4ull));
    if (a == ~0ull) a ^= 1;
    if (b == ~0ull) b ^= 1;
    *h1 = a;
    *h2 = b;
}

} // namespace detail

//! ReduceByKey for (std::string word, integral count) pairs — the
//! word_count shape (examples/word_count/word_count.hpp:35-56). See the
//! declaration for the contract; the GPU path is
//! dictionary-encode -> t9_reduce128 (sum) -> decode.
template <typename ValueType>
template <typename KeyExtractor, typename ReduceFn>
DIA<ValueType> DIA<ValueType>::ReduceByKey(const KeyExtractor& key_ex,
                                           const ReduceFn& red) const {
    using Pair = ValueType;   // pair<std::string, Count>
    using Count = decltype(Pair().second);
    static_assert(std::is_integral<Count>::value && sizeof(Count) == 8,
                  "GPU ReduceByKey carries a 64-bit counter value");
    auto items = AllGather();
    const size_t n = items.size();

    // algebraic probe: is the user's reduce function "add the counters"?
    // (it is a pure function of its operands; probe a few samples)
    bool additive = true;
    for (uint64_t a : { 3ull, 17ull, 1ull << 40 }) {
        Pair x{ "probe", (Count)a }, y{ "probe", (Count)(a * 2 + 5) };
        Pair r = red(x, y);
        if ((uint64_t)r.second != a + (a * 2 + 5) || r.first != "probe") {
            additive = false;
            break;
        }
    }

    // dictionary-encode: word -> (h1, h2); keep the decode map. The
    // decode key folds (h1, h2) into one u64 — a fold collision
    // (p ~= 2^-64 per pair) would mis-map a word at DECODE time only;
    // the reduce itself separates on the full 128-bit composite
    // (DESIGN.md "Config-4 string identity").
    std::vector<uint64_t> k1(n), k2(n), vals(n);
    std::unordered_map<uint64_t, std::string> decode;
    for (size_t i = 0; i < n; ++i) {
        const std::string key = key_ex(items[i]);
        detail::string_hash2(key, &k1[i], &k2[i]);
        vals[i] = (uint64_t)items[i].second;
        decode.emplace(k1[i] ^ (k2[i] << 1 | k2[i] >> 63), key);
    }

    std::vector<Pair> out;
    if (n && additive) {
        // GPU 128-bit composite sum-reduce
        DeviceBuf d1(n * 8), d2(n * 8), dv(n * 8);
        T9_DIA_HIP(hipMemcpy(d1.ptr, k1.data(), n * 8,
                             hipMemcpyHostToDevice));
        T9_DIA_HIP(hipMemcpy(d2.ptr, k2.data(), n * 8,
                             hipMemcpyHostToDevice));
        T9_DIA_HIP(hipMemcpy(dv.ptr, vals.data(), n * 8,
                             hipMemcpyHostToDevice));
        uint64_t cap = 1024;
        while (cap < 2 * n + 2) cap <<= 1;
        DeviceBuf tbl(3 * cap * 8), o1(cap * 8), o2(cap * 8), ov(cap * 8);
        DeviceBuf derr(4), dn(8);
        hipStream_t s = ctx_->stream();
        T9_DIA_TRY(t9_reduce128_init(ctx_->native(), (uint64_t*)tbl.ptr,
                                     cap, s));
        T9_DIA_TRY(t9_reduce128_build(ctx_->native(),
                                      (const uint64_t*)d1.ptr,
                                      (const uint64_t*)d2.ptr,
                                      (const uint64_t*)dv.ptr, n,
                                      (uint64_t*)tbl.ptr, cap, 0,
                                      (uint32_t*)derr.ptr, s));
        T9_DIA_TRY(t9_reduce128_drain(ctx_->native(),
                                      (const uint64_t*)tbl.ptr, cap,
                                      (uint64_t*)o1.ptr, (uint64_t*)o2.ptr,
                                      (uint64_t*)ov.ptr, (uint64_t*)dn.ptr,
                                      s));
        uint64_t m = 0;
        uint32_t err = 0;
        T9_DIA_HIP(hipMemcpy(&m, dn.ptr, 8, hipMemcpyDeviceToHost));
        T9_DIA_HIP(hipMemcpy(&err, derr.ptr, 4, hipMemcpyDeviceToHost));
        if (err) throw std::runtime_error("ReduceByKey: table overflow");
        std::vector<uint64_t> r1(m), r2(m), rv(m);
        if (m) {
            T9_DIA_HIP(hipMemcpy(r1.data(), o1.ptr, m * 8,
                                 hipMemcpyDeviceToHost));
            T9_DIA_HIP(hipMemcpy(r2.data(), o2.ptr, m * 8,
                                 hipMemcpyDeviceToHost));
            T9_DIA_HIP(hipMemcpy(rv.data(), ov.ptr, m * 8,
                                 hipMemcpyDeviceToHost));
        }
        out.reserve(m);
        for (uint64_t i = 0; i < m; ++i)
            out.push_back(Pair{
                decode.at(r1[i] ^ (r2[i] << 1 | r2[i] >> 63)),
                (Count)rv[i] });
    }
    else if (n) {
        // non-additive reduce function: host fold with the user's functor
        // (grouping semantics preserved; the hot path is the additive
        // word_count shape above)
        std::unordered_map<uint64_t, Pair> acc;
        for (size_t i = 0; i < n; ++i) {
            const uint64_t h = k1[i] ^ (k2[i] << 1 | k2[i] >> 63);
            auto it = acc.find(h);
            if (it == acc.end()) acc.emplace(h, items[i]);
            else it->second = red(it->second, items[i]);
        }
        out.reserve(acc.size());
        for (auto& kv : acc) out.push_back(kv.second);
    }
    return FromVector(*ctx_, out);
}

//! ReadBinary — reference api/read_binary.hpp: read packed fixed-size
//! items from files (the worker's share; world=1 here reads all).
template <typename T>
DIA<T> ReadBinary(Context& ctx, const std::vector<std::string>& files) {
    // device fast path (POD T): size the files, read each straight into
    // the device buffer via O_DIRECT + pinned staging (the pageable H2D
    // of the host-vector path measured ~2 GB/s)
    if (std::is_trivially_copyable<T>::value && !files.empty()) {
        size_t total = 0;
        bool sized = true;
        std::vector<size_t> sizes;
        for (const auto& path : files) {
            std::FILE* f = std::fopen(path.c_str(), "rb");
            if (!f) throw std::runtime_error("ReadBinary: cannot open " +
                                             path);
            std::fseek(f, 0, SEEK_END);
            long b = std::ftell(f);
            std::fclose(f);
            if (b < 0 || b % (long)sizeof(T)) {
                sized = false;
                break;
            }
            sizes.push_back((size_t)b);
            total += (size_t)b;
        }
        if (sized && total) {
            auto buf = std::make_shared<DeviceBuf>(total);
            size_t off = 0;
            bool ok = true;
            for (size_t i = 0; ok && i < files.size(); ++i) {
                ok = detail::read_device_direct(
                    files[i].c_str(), (char*)buf->ptr + off, sizes[i],
                    ctx.stream());
                off += sizes[i];
            }
            if (ok) return DIA<T>(&ctx, buf, total / sizeof(T));
            // else fall through to the host path
        }
    }
    std::vector<T> items;
    for (const auto& path : files) {
        std::FILE* f = std::fopen(path.c_str(), "rb");
        if (!f) throw std::runtime_error("ReadBinary: cannot open " + path);
        std::fseek(f, 0, SEEK_END);
        long bytes = std::ftell(f);
        std::fseek(f, 0, SEEK_SET);
        if (bytes % (long)sizeof(T))
            throw std::runtime_error("ReadBinary: size not a multiple of "
                                     "the item size: " + path);
        size_t n = (size_t)bytes / sizeof(T);
        size_t old = items.size();
        items.resize(old + n);
        if (n) {
            std::fclose(f);
            f = nullptr;
            if (!detail::read_file_direct(path.c_str(),
                                          items.data() + old,
                                          n * sizeof(T))) {
                // buffered fallback
                f = std::fopen(path.c_str(), "rb");
                if (!f || std::fread(items.data() + old, sizeof(T), n,
                                     f) != n) {
                    if (f) std::fclose(f);
                    throw std::runtime_error("ReadBinary: short read");
                }
            }
        }
        if (f) std::fclose(f);
    }
    return FromVector(ctx, items);
}

//! GroupByKey — reference api/group_by_key.hpp (sort-based): sort the
//! pairs by key on the GPU, build the device group index
//! (t9_group_index), then apply the user's group function on the host per
//! group (user lambdas are host code in this surface). fn(key, vals_begin,
//! vals_end) -> Result.
template <typename Result, typename GroupFn>
std::vector<Result> GroupByKey(const DIA<KeyValue>& input,
                               const GroupFn& fn) {
    Context& ctx = input.context();
    auto host = input.AllGather();
    const size_t n = host.size();
    std::vector<uint64_t> hk(n), hv(n);
    for (size_t i = 0; i < n; ++i) {
        hk[i] = host[i].key;
        hv[i] = host[i].value;
    }
    std::vector<Result> out;
    if (n == 0) return out;
    DeviceBuf dk(n * 8), dv(n * 8), didx(n * 4), dsv(n * 8);
    DeviceBuf du(n * 8), doff(n * 8), dcnt(8);
    T9_DIA_HIP(hipMemcpy(dk.ptr, hk.data(), n * 8, hipMemcpyHostToDevice));
    T9_DIA_HIP(hipMemcpy(dv.ptr, hv.data(), n * 8, hipMemcpyHostToDevice));
    hipStream_t s = ctx.stream();
    {
        /* iota payload, sort (key, idx), gather values by idx */
        std::vector<uint32_t> iota(n);
        for (size_t i = 0; i < n; ++i) iota[i] = (uint32_t)i;
        T9_DIA_HIP(hipMemcpy(didx.ptr, iota.data(), n * 4,
                             hipMemcpyHostToDevice));
        DeviceBuf ws(t9_sort_pairs_workspace(n));
        T9_DIA_TRY(t9_sort_pairs_u64_u32(ctx.native(), (uint64_t*)dk.ptr,
                                         (uint32_t*)didx.ptr, n, ws.ptr,
                                         s));
        T9_DIA_TRY(t9_gather_records(ctx.native(), (const uint8_t*)dv.ptr,
                                     (const uint32_t*)didx.ptr, n, 8,
                                     (uint8_t*)dsv.ptr, s));
        DeviceBuf gws(t9_group_index_workspace(n));
        T9_DIA_TRY(t9_group_index(ctx.native(), (const uint64_t*)dk.ptr, n,
                                  (uint64_t*)du.ptr, (uint64_t*)doff.ptr,
                                  (uint64_t*)dcnt.ptr, gws.ptr, s));
        T9_DIA_HIP(hipStreamSynchronize(s));
    }
    uint64_t groups = 0;
    T9_DIA_HIP(hipMemcpy(&groups, dcnt.ptr, 8, hipMemcpyDeviceToHost));
    std::vector<uint64_t> uk(groups), off(groups), sv(n);
    T9_DIA_HIP(hipMemcpy(uk.data(), du.ptr, groups * 8,
                         hipMemcpyDeviceToHost));
    T9_DIA_HIP(hipMemcpy(off.data(), doff.ptr, groups * 8,
                         hipMemcpyDeviceToHost));
    T9_DIA_HIP(hipMemcpy(sv.data(), dsv.ptr, n * 8,
                         hipMemcpyDeviceToHost));
    out.reserve(groups);
    for (uint64_t g = 0; g < groups; ++g) {
        const uint64_t b = off[g];
        const uint64_t e = (g + 1 < groups) ? off[g + 1] : n;
        out.push_back(fn(uk[g], sv.data() + b, sv.data() + e));
    }
    return out;
}

//! Merge — reference api/merge.hpp: merge sorted DIAs (u64, default
//! order); equal keys keep source order (a's before b's).
inline DIA<uint64_t> Merge(const DIA<uint64_t>& a, const DIA<uint64_t>& b) {
    Context& ctx = a.context();
    auto out = std::make_shared<DeviceBuf>((a.Size() + b.Size()) * 8);
    T9_DIA_TRY(t9_merge_u64(ctx.native(), (const uint64_t*)a.device_ptr(),
                            a.Size(), (const uint64_t*)b.device_ptr(),
                            b.Size(), (uint64_t*)out->ptr, ctx.stream()));
    T9_DIA_HIP(hipStreamSynchronize(ctx.stream()));
    return DIA<uint64_t>(&ctx, out, a.Size() + b.Size());
}

//! Merge for fixed-size POD records under the byte-lexicographic order
//! (the acceptance comparator; reference api/merge.hpp merges pre-sorted
//! DIAs with source order among equals — a's before b's)
template <typename T>
DIA<T> Merge(const DIA<T>& a, const DIA<T>& b) {
    static_assert(std::is_trivially_copyable<T>::value &&
                      sizeof(T) % 4 == 0,
                  "GPU Merge needs POD records, size % 4 == 0");
    Context& ctx = a.context();
    auto out = std::make_shared<DeviceBuf>((a.Size() + b.Size()) *
                                           sizeof(T));
    T9_DIA_TRY(t9_merge_records(
        ctx.native(), (const uint8_t*)a.device_ptr(), a.Size(),
        (const uint8_t*)b.device_ptr(), b.Size(), sizeof(T),
        (uint8_t*)out->ptr, ctx.stream()));
    T9_DIA_HIP(hipStreamSynchronize(ctx.stream()));
    return DIA<T>(&ctx, out, a.Size() + b.Size());
}

//! Run — reference api/context.cpp:947: construct the context(s) and run
//! the job. Round 1: one process, one GPU, rank 0.
inline int Run(const std::function<void(Context&)>& job) {
    Context ctx(0, 0, 1, nullptr);
    job(ctx);
    return 0;
}

} // namespace api

//! hoist the api entry points into t9::, as the reference hoists
//! thrill::api into thrill:: (each api header's `using api::X`), so
//! `using namespace t9` makes the example sources read identically
using api::Context;
using api::DIA;
using api::FromVector;
using api::Generate;
using api::GroupByKey;
using api::KeyValue;
using api::Merge;
using api::ReadBinary;
using api::ReducePair;
using api::Run;

} // namespace t9
