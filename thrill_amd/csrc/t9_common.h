/* t9_common.h — shared device/host helpers for the thrill_amd HIP kernels.
 * Target: gfx950 (MI355X, CDNA4) only. Wavefront = 64 lanes.
 */
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>

using u8 = uint8_t;
using u16 = uint16_t;
using u32 = uint32_t;
using u64 = uint64_t;

#define T9_WAVE 64

/* errno-style returns across the C ABI */
#define T9_OK 0
#define T9_EINVAL (-22)
#define T9_ENOMEM (-12)
#define T9_EIO (-5)
#define T9_ENOSYS (-38)

#define HIP_TRY(expr)                                                     \
    do {                                                                  \
        hipError_t _e = (expr);                                           \
        if (_e != hipSuccess) {                                           \
            fprintf(stderr, "t9: %s:%d: %s failed: %s\n", __FILE__,       \
                    __LINE__, #expr, hipGetErrorString(_e));              \
            return T9_EIO;                                                \
        }                                                                 \
    } while (0)

#define T9_LAUNCH_CHECK()                                                 \
    do {                                                                  \
        hipError_t _e = hipGetLastError();                                \
        if (_e != hipSuccess) {                                           \
            fprintf(stderr, "t9: %s:%d: kernel launch failed: %s\n",      \
                    __FILE__, __LINE__, hipGetErrorString(_e));           \
            return T9_EIO;                                                \
        }                                                                 \
    } while (0)

struct t9_context {
    int device;
    int rank;
    int world;
    void* comm;     /* ncclComm_t or nullptr */
    int owns_comm;  /* 1 if t9_comm_init created it (t9_destroy frees) */
};

/* splitmix64 random access — must match oracle/t9_oracle.cpp splitmix64_at */
__host__ __device__ inline u64 t9_splitmix64_at(u64 seed, u64 ctr) {
    u64 z = seed + (ctr + 1) * 0x9E3779B97F4A7C15ull;
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    return z ^ (z >> 31);
}

/* Hash128to64 — thrill/common/hash.hpp:64-72 (cityhash, MIT) */
__host__ __device__ inline u64 t9_hash128to64(u64 upper, u64 lower) {
    const u64 k = 0x9DDFEA08EB382D69ull;
    u64 a = (lower ^ upper) * k;
    a ^= (a >> 47);
    u64 b = (upper ^ a) * k;
    b ^= (b >> 47);
    b *= k;
    return b;
}

static inline u64 t9_ceil_div(u64 a, u64 b) { return (a + b - 1) / b; }
static inline u64 t9_align256(u64 x) { return (x + 255) & ~(u64)255; }

/* optional perf-event registry (t9_perf.cpp) */
bool t9perf_on();
void* t9perf_begin(hipStream_t s, const char* cls);
void t9perf_end(void* tok, hipStream_t s);

#define T9_PERF_WRAP(s, cls, launch)                                      \
    do {                                                                  \
        void* _tok = t9perf_on() ? t9perf_begin((s), (cls)) : nullptr;    \
        launch;                                                           \
        if (_tok) t9perf_end(_tok, (s));                                  \
    } while (0)

/* radix sort geometry (shared by t9_sort.hip host code and workspace calc) */
#define T9_RADIX 256
#define T9_KEYS_TILE 4096   /* elems per block, keys-only scatter */
#define T9_PAIRS_TILE 2048  /* elems per block, key+payload scatter */
/* hist rows per scan block — small so B/CHUNK blocks fill the chip (at
 * 512, the 10 GiB workload ran the scan kernels on 26 blocks = chip
 * ~99% idle, ~2 ms/sort of pure latency) */
#define T9_SCAN_CHUNK 32
