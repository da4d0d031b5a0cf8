/* t9_group.hip — group index over key-sorted data (SURVEY.md §8f item 3:
 * GroupByKey, the sort-based sibling of ReduceByKey —
 * thrill/api/group_by_key.hpp groups all values of a key after a
 * sort/merge; here: the sorted keys are segmented into runs and the
 * (unique key, start offset) index is produced, count in device memory).
 *
 * Three kernels: per-block boundary counts -> single-block scan of block
 * sums -> boundary scatter at global group ids. Group ids are the
 * exclusive prefix over boundary flags, so offsets come out ordered.
 */

#include "t9_common.h"

#define GRP_TILE 8192

__global__ __launch_bounds__(256) void k_grp_count(
    const u64* __restrict__ keys, u64 n, u32* __restrict__ blocksum) {
    __shared__ u32 s;
    if (threadIdx.x == 0) s = 0;
    __syncthreads();
    const u64 base = (u64)blockIdx.x * GRP_TILE;
    const u32 tn = (u32)((n - base < (u64)GRP_TILE) ? (n - base)
                                                    : (u64)GRP_TILE);
    u32 cnt = 0;
    for (u32 i = threadIdx.x; i < tn; i += 256) {
        const u64 g = base + i;
        if (g == 0 || keys[g] != keys[g - 1]) ++cnt;
    }
    atomicAdd(&s, cnt);
    __syncthreads();
    if (threadIdx.x == 0) blocksum[blockIdx.x] = s;
}

/* exclusive scan of B block sums (single block, grid-stride chunks) */
__global__ __launch_bounds__(256) void k_grp_scan(u32* __restrict__ blocksum,
                                                  u64 B,
                                                  u64* __restrict__ total) {
    __shared__ u32 s[256];
    const u32 tid = threadIdx.x;
    u32 carry = 0;
    for (u64 c = 0; c < B; c += 256) {
        const u64 i = c + tid;
        u32 v = (i < B) ? blocksum[i] : 0;
        s[tid] = v;
        __syncthreads();
        for (int off = 1; off < 256; off <<= 1) {
            u32 y = (tid >= (u32)off) ? s[tid - off] : 0;
            __syncthreads();
            s[tid] += y;
            __syncthreads();
        }
        if (i < B) blocksum[i] = s[tid] - v + carry;   /* exclusive */
        carry += s[255];
        __syncthreads();
    }
    if (tid == 0) *total = carry;
}

__global__ __launch_bounds__(256) void k_grp_scatter(
    const u64* __restrict__ keys, u64 n, const u32* __restrict__ blockbase,
    u64* __restrict__ unique_keys, u64* __restrict__ offsets) {
    __shared__ u32 s_base;
    __shared__ u32 s_local;
    if (threadIdx.x == 0) {
        s_base = blockbase[blockIdx.x];
        s_local = 0;
    }
    __syncthreads();
    const u64 base = (u64)blockIdx.x * GRP_TILE;
    const u32 tn = (u32)((n - base < (u64)GRP_TILE) ? (n - base)
                                                    : (u64)GRP_TILE);
    /* two phases to keep group ids ordered: waves claim contiguous id
     * ranges would need tile-ordered ranks; simplest correct: each thread
     * walks a contiguous slice of the tile so boundary order within the
     * slice is preserved, slices claim their counts in slice order. */
    const u32 SLICE = GRP_TILE / 256;
    const u32 t0 = threadIdx.x * SLICE;
    u32 mycnt = 0;
    for (u32 i = t0; i < t0 + SLICE && i < tn; ++i) {
        const u64 g = base + i;
        if (g == 0 || keys[g] != keys[g - 1]) ++mycnt;
    }
    /* exclusive prefix of per-slice counts across the block (LDS scan) */
    __shared__ u32 pre[256];
    pre[threadIdx.x] = mycnt;
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
        u32 y = (threadIdx.x >= (u32)off) ? pre[threadIdx.x - off] : 0;
        __syncthreads();
        pre[threadIdx.x] += y;
        __syncthreads();
    }
    u32 gid = s_base + pre[threadIdx.x] - mycnt;
    for (u32 i = t0; i < t0 + SLICE && i < tn; ++i) {
        const u64 g = base + i;
        if (g == 0 || keys[g] != keys[g - 1]) {
            unique_keys[gid] = keys[g];
            offsets[gid] = g;
            ++gid;
        }
    }
    (void)s_local;
}

namespace {
u64 grp_blocks(u64 n) { return (n + GRP_TILE - 1) / GRP_TILE; }
} // namespace

extern "C" {

u64 t9_group_index_workspace(u64 n) {
    return t9_align256(grp_blocks(n ? n : 1) * 4);
}

/* Build the group index of a key-sorted array: d_unique[g], d_offsets[g]
 * for g in [0, *d_count); offsets are ascending run starts. d_count is a
 * device u64. Capacity of the output arrays must be >= the number of
 * distinct keys (n is always enough). */
int t9_group_index(t9_context* ctx, const u64* d_sorted_keys, u64 n,
                   u64* d_unique, u64* d_offsets, u64* d_count,
                   void* d_workspace, void* stream) {
    (void)ctx;
    if (!d_count) return T9_EINVAL;
    hipStream_t s = (hipStream_t)stream;
    if (n == 0) {
        HIP_TRY(hipMemsetAsync(d_count, 0, 8, s));
        return T9_OK;
    }
    if (!d_sorted_keys || !d_unique || !d_offsets || !d_workspace)
        return T9_EINVAL;
    const u64 B = grp_blocks(n);
    if (B >= (1ull << 31)) return T9_EINVAL;
    u32* blocksum = (u32*)d_workspace;
    hipLaunchKernelGGL(k_grp_count, dim3((u32)B), dim3(256), 0, s,
                       d_sorted_keys, n, blocksum);
    hipLaunchKernelGGL(k_grp_scan, dim3(1), dim3(256), 0, s, blocksum, B,
                       d_count);
    hipLaunchKernelGGL(k_grp_scatter, dim3((u32)B), dim3(256), 0, s,
                       d_sorted_keys, n, blocksum, d_unique, d_offsets);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

} /* extern "C" */
