/* t9_merge.hip — merge of two sorted u64 sequences (SURVEY.md §8f item 4:
 * thrill/api/merge.hpp merges pre-sorted DIAs; its distributed pivot
 * search — merge.hpp:368-520 — splits work across workers, here the
 * merge-path diagonal search splits work across blocks and threads).
 *
 * Stability/tie rule: equal keys from A precede those from B (the
 * reference's MergeNode keeps source order among equals by comparator
 * tie handling). One pass: each 256-thread block owns a TILE of the
 * output, binary-searches its merge-path split on the global diagonal,
 * stages its A/B ranges in LDS, and each thread merge-paths its 16
 * output slots inside LDS. All reads/writes coalesced.
 */

#include "t9_common.h"

#define MRG_TILE 4096

/* smallest i in [lo, hi] such that taking i elements of A and d-i of B is
 * a valid merge-path split for diagonal d: A[i-1] <= B[d-i] fails <=>
 * A[i-1] > B[d-i]; with the A-before-B tie rule the split condition is
 * A[i] <= B[d-i-1] ? advance i. Standard formulation below. */
__device__ inline u64 merge_split(const u64* __restrict__ a, u64 na,
                                  const u64* __restrict__ b, u64 nb,
                                  u64 d) {
    u64 lo = (d > nb) ? d - nb : 0;
    u64 hi = (d < na) ? d : na;
    while (lo < hi) {
        u64 i = (lo + hi) / 2;
        /* element a[i] vs b[d-i-1]: take from A while a[i] <= b[j]
         * (A wins ties) */
        if (a[i] <= b[d - i - 1]) lo = i + 1;
        else hi = i;
    }
    return lo;
}

__global__ __launch_bounds__(256) void k_merge_u64(
    const u64* __restrict__ a, u64 na, const u64* __restrict__ b, u64 nb,
    u64* __restrict__ out) {
    __shared__ u64 s_a[MRG_TILE + 1];
    __shared__ u64 s_b[MRG_TILE + 1];
    __shared__ u64 s_meta[4];
    const u64 n = na + nb;
    const u64 d0 = (u64)blockIdx.x * MRG_TILE;
    if (d0 >= n) return;
    const u64 d1 = (d0 + MRG_TILE < n) ? d0 + MRG_TILE : n;
    const u32 tid = threadIdx.x;
    if (tid == 0) {
        s_meta[0] = merge_split(a, na, b, nb, d0);   /* i0 */
        s_meta[1] = merge_split(a, na, b, nb, d1);   /* i1 */
    }
    __syncthreads();
    const u64 i0 = s_meta[0], i1 = s_meta[1];
    const u64 j0 = d0 - i0, j1 = d1 - i1;
    const u32 la = (u32)(i1 - i0), lb = (u32)(j1 - j0);
    for (u32 t = tid; t < la; t += 256) s_a[t] = a[i0 + t];
    for (u32 t = tid; t < lb; t += 256) s_b[t] = b[j0 + t];
    __syncthreads();

    /* each thread merges its 16 output slots via a local split search */
    const u32 span = MRG_TILE / 256;   /* 16 */
    const u32 od = tid * span;
    if (d0 + od < d1) {
        /* local diagonal split within (s_a[0..la), s_b[0..lb)) */
        u32 lo = (od > lb) ? od - lb : 0;
        u32 hi = (od < la) ? od : la;
        while (lo < hi) {
            u32 i = (lo + hi) / 2;
            if (s_a[i] <= s_b[od - i - 1]) lo = i + 1;
            else hi = i;
        }
        u32 ia = lo, ib = od - lo;
        const u32 end = (u32)((d1 - d0 - od < span) ? (d1 - d0 - od)
                                                    : (u64)span);
        for (u32 k = 0; k < end; ++k) {
            bool take_a =
                ib >= lb || (ia < la && s_a[ia] <= s_b[ib]);
            out[d0 + od + k] = take_a ? s_a[ia++] : s_b[ib++];
        }
    }
}

extern "C" {

/* Merge two sorted u64 sequences into out (size na+nb); equal keys from
 * d_a precede those from d_b (stable source order). */
int t9_merge_u64(t9_context* ctx, const u64* d_a, u64 na, const u64* d_b,
                 u64 nb, u64* d_out, void* stream) {
    (void)ctx;
    hipStream_t s = (hipStream_t)stream;
    const u64 n = na + nb;
    if (n == 0) return T9_OK;
    if (!d_out || (na && !d_a) || (nb && !d_b)) return T9_EINVAL;
    if (na == 0) {
        HIP_TRY(hipMemcpyAsync(d_out, d_b, nb * 8,
                               hipMemcpyDeviceToDevice, s));
        return T9_OK;
    }
    if (nb == 0) {
        HIP_TRY(hipMemcpyAsync(d_out, d_a, na * 8,
                               hipMemcpyDeviceToDevice, s));
        return T9_OK;
    }
    const u64 B = t9_ceil_div(n, MRG_TILE);
    if (B >= (1ull << 31)) return T9_EINVAL;
    hipLaunchKernelGGL(k_merge_u64, dim3((u32)B), dim3(256), 0, s, d_a, na,
                       d_b, nb, d_out);
    T9_LAUNCH_CHECK();
    return T9_OK;
}

} /* extern "C" */
