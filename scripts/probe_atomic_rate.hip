/* probe_atomic_rate — quantifies the L2-residency factor for global
 * atomics on gfx950: atomicCAS+atomicAdd pairs (the reduce-build cold
 * path's exact op mix) into a table of varying size, uniform random
 * slots. If the rate at a 32 MB (L2-resident) table is much higher than
 * at 256 MB, the round-2 "hash-partitioned build passes" idea
 * (DESIGN.md gap 3b) pays; if not, it dies here.
 *
 * Build: hipcc --offload-arch=gfx950 -O3 scripts/probe_atomic_rate.hip
 *        -o gpurun_out/probe_atomic
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>

typedef uint64_t u64;
typedef uint32_t u32;

__device__ inline u64 mix(u64 x) {
    x ^= x >> 33; x *= 0xff51afd7ed558ccdull;
    x ^= x >> 33; x *= 0xc4ceb9fe1a85ec53ull;
    x ^= x >> 33; return x;
}

/* each lane does ITER CAS+ADD pairs at pseudorandom slots */
__global__ __launch_bounds__(256) void k_storm(u64* t, u64 nslots,
                                               int iter, u64 seed) {
    const u64 gid = (u64)blockIdx.x * 256 + threadIdx.x;
    u64 h = mix(gid ^ seed);
    for (int i = 0; i < iter; ++i) {
        h = mix(h + i);
        const u64 slot = (h % nslots) * 2;
        u64 prev = atomicCAS((unsigned long long*)&t[slot], 0ull, h | 1ull);
        (void)prev;
        atomicAdd((unsigned long long*)&t[slot + 1], 1ull);
    }
}

int main() {
    const u64 MAXB = 512ull << 20;
    u64* t;
    (void)hipMalloc(&t, MAXB);
    const int iter = 64;
    const u64 grid = 2048;
    const u64 ops = grid * 256 * (u64)iter;   /* CAS+ADD pairs */
    for (u64 mb : {8ull, 16ull, 32ull, 64ull, 128ull, 256ull, 512ull}) {
        const u64 nslots = mb * (1ull << 20) / 16;
        (void)hipMemset(t, 0, mb << 20);
        hipLaunchKernelGGL(k_storm, dim3(64), dim3(256), 0, 0, t, nslots,
                           8, 1);   /* warmup */
        (void)hipDeviceSynchronize();
        hipEvent_t a, b;
        (void)hipEventCreate(&a); (void)hipEventCreate(&b);
        (void)hipEventRecord(a);
        hipLaunchKernelGGL(k_storm, dim3((u32)grid), dim3(256), 0, 0, t,
                           nslots, iter, 42);
        (void)hipEventRecord(b);
        (void)hipEventSynchronize(b);
        float ms = 0;
        (void)hipEventElapsedTime(&ms, a, b);
        printf("table=%4llu MB  pairs=%llu  ms=%7.3f  rate=%6.2f G pair/s\n",
               (unsigned long long)mb, (unsigned long long)ops, ms,
               ops / ms / 1e6);
        (void)hipEventDestroy(a); (void)hipEventDestroy(b);
    }
    (void)hipFree(t);
    return 0;
}
