/* t9_api.cpp — context lifecycle and the RCCL all-to-all-v shuffle.
 *
 * The shuffle replaces the reference's CatStream/MixStream BlockWriters +
 * Multiplexer TCP framing (thrill/data/stream_sink.cpp:97-226,
 * thrill/data/multiplexer.cpp:282-463) with one grouped ncclSend/ncclRecv
 * exchange over xGMI — each GPU pair uses its direct point-to-point link,
 * which is exactly the traffic shape of an all-to-all (SURVEY.md §5
 * "Distributed communication backend").
 */

#include "t9_common.h"

#include <rccl/rccl.h>

#include <cstdlib>
#include <new>

extern "C" {

const char* t9_version(void) { return "thrill_amd 0.1 (gfx950)"; }

int t9_create(t9_context** out, int device, int rank, int world,
              void* comm) {
    if (!out || world < 1 || rank < 0 || rank >= world) return T9_EINVAL;
    HIP_TRY(hipSetDevice(device));
    t9_context* ctx = new (std::nothrow) t9_context;
    if (!ctx) return T9_ENOMEM;
    ctx->device = device;
    ctx->rank = rank;
    ctx->world = world;
    ctx->comm = comm;
    *out = ctx;
    return T9_OK;
}

int t9_destroy(t9_context* ctx) {
    delete ctx;
    return T9_OK;
}

int t9_alltoall(t9_context* ctx, const void* d_send, const u64* send_counts,
                const u64* send_displs, void* d_recv, const u64* recv_counts,
                const u64* recv_displs, u64 elem_size, void* stream) {
    if (!ctx || !d_send || !d_recv || !send_counts || !send_displs ||
        !recv_counts || !recv_displs || elem_size == 0)
        return T9_EINVAL;
    if (ctx->world == 1) {
        /* loopback: single rank exchanges with itself */
        if (send_counts[0] != recv_counts[0]) return T9_EINVAL;
        HIP_TRY(hipMemcpyAsync(
            (char*)d_recv + recv_displs[0] * elem_size,
            (const char*)d_send + send_displs[0] * elem_size,
            send_counts[0] * elem_size, hipMemcpyDeviceToDevice,
            (hipStream_t)stream));
        return T9_OK;
    }
    if (!ctx->comm) return T9_EINVAL;
    ncclComm_t comm = (ncclComm_t)ctx->comm;
    hipStream_t s = (hipStream_t)stream;
    ncclResult_t rc = ncclGroupStart();
    for (int r = 0; r < ctx->world && rc == ncclSuccess; ++r) {
        if (send_counts[r])
            rc = ncclSend((const char*)d_send + send_displs[r] * elem_size,
                          send_counts[r] * elem_size, ncclUint8, r, comm, s);
        if (rc == ncclSuccess && recv_counts[r])
            rc = ncclRecv((char*)d_recv + recv_displs[r] * elem_size,
                          recv_counts[r] * elem_size, ncclUint8, r, comm, s);
    }
    ncclResult_t rce = ncclGroupEnd();
    if (rc != ncclSuccess || rce != ncclSuccess) {
        fprintf(stderr, "t9_alltoall: rccl error: %s\n",
                ncclGetErrorString(rc != ncclSuccess ? rc : rce));
        return T9_EIO;
    }
    return T9_OK;
}

} /* extern "C" */
