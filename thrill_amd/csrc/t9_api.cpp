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
#include <cstring>
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
    ctx->owns_comm = 0;
    *out = ctx;
    return T9_OK;
}

int t9_destroy(t9_context* ctx) {
    if (ctx && ctx->comm && ctx->owns_comm)
        ncclCommDestroy((ncclComm_t)ctx->comm);
    delete ctx;
    return T9_OK;
}

/* ------------------------------------------------------------------ *
 * RCCL communicator bootstrap — so the C ABI owns the xGMI data plane
 * end to end (the reference seam it replaces builds its own TCP mesh in
 * net::tcp::Construct, thrill/api/context.cpp:604-614). Rank 0 calls
 * t9_comm_id, the 128-byte id is distributed out of band (any host
 * channel — the reference distributes its connection endpoints the same
 * way), then every rank calls t9_comm_init.
 * ------------------------------------------------------------------ */

int t9_comm_id_size(void) { return (int)sizeof(ncclUniqueId); }

int t9_comm_id(void* out_id) {
    if (!out_id) return T9_EINVAL;
    ncclResult_t rc = ncclGetUniqueId((ncclUniqueId*)out_id);
    if (rc != ncclSuccess) {
        fprintf(stderr, "t9_comm_id: %s\n", ncclGetErrorString(rc));
        return T9_EIO;
    }
    return T9_OK;
}

int t9_comm_init(t9_context* ctx, const void* id) {
    if (!ctx || !id) return T9_EINVAL;
    if (ctx->comm) return T9_EINVAL;   /* already connected */
    HIP_TRY(hipSetDevice(ctx->device));
    ncclComm_t comm = nullptr;
    ncclResult_t rc = ncclCommInitRank(&comm, ctx->world,
                                       *(const ncclUniqueId*)id, ctx->rank);
    if (rc != ncclSuccess) {
        fprintf(stderr, "t9_comm_init: rank %d/%d: %s\n", ctx->rank,
                ctx->world, ncclGetErrorString(rc));
        return T9_EIO;
    }
    ctx->comm = comm;
    ctx->owns_comm = 1;
    return T9_OK;
}

int t9_alltoall(t9_context* ctx, const void* d_send, const u64* send_counts,
                const u64* send_displs, void* d_recv, const u64* recv_counts,
                const u64* recv_displs, u64 elem_size, void* stream) {
    if (!ctx || !d_send || !d_recv || !send_counts || !send_displs ||
        !recv_counts || !recv_displs || elem_size == 0)
        return T9_EINVAL;
    /* The rank's own share bypasses RCCL: a plain device copy on the same
     * stream is both faster (no protocol round trip for ~1/p of the data)
     * and avoids the self-send path entirely — a 10.7 GB RCCL
     * self-exchange was measured as a hang in round 1 (pipeline.py
     * history, commit 551d307); T9_A2A_SELF=nccl restores the RCCL
     * self-send for investigation (including at world == 1, where it
     * overrides the loopback shortcut if a communicator exists). */
    static const bool self_nccl = [] {
        const char* e = getenv("T9_A2A_SELF");
        return e && strcmp(e, "nccl") == 0;
    }();
    if (ctx->world == 1 && !(self_nccl && ctx->comm)) {
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
    const int me = ctx->rank;
    if (!self_nccl) {
        if (send_counts[me] != recv_counts[me]) return T9_EINVAL;
        if (send_counts[me])
            HIP_TRY(hipMemcpyAsync(
                (char*)d_recv + recv_displs[me] * elem_size,
                (const char*)d_send + send_displs[me] * elem_size,
                send_counts[me] * elem_size, hipMemcpyDeviceToDevice, s));
    }
    /* Chunk every point-to-point transfer at 1 GiB: RCCL's p2p path
     * SILENTLY TRUNCATES single messages at count mod 2^32 bytes
     * (measured on MI355X, scripts/probe_self_nccl.py: a 2^32-byte
     * self-send moves 0 bytes and "succeeds" in 4 ms; 10.7 GB moves the
     * low-32-bit remainder — the same class of failure round 1 observed
     * as a hang through torch's path). Chunks to the same peer inside
     * one group are ordered, so sender and receiver split
     * deterministically and identically. */
    const u64 CHUNK = 1ull << 30;
    ncclResult_t rc = ncclGroupStart();
    for (int r = 0; r < ctx->world && rc == ncclSuccess; ++r) {
        if (r == me && !self_nccl) continue;
        const u64 sb = send_counts[r] * elem_size;
        const char* sp = (const char*)d_send + send_displs[r] * elem_size;
        for (u64 off = 0; off < sb && rc == ncclSuccess; off += CHUNK)
            rc = ncclSend(sp + off,
                          (sb - off < CHUNK) ? (sb - off) : CHUNK,
                          ncclUint8, r, comm, s);
        const u64 rb = recv_counts[r] * elem_size;
        char* rp = (char*)d_recv + recv_displs[r] * elem_size;
        for (u64 off = 0; off < rb && rc == ncclSuccess; off += CHUNK)
            rc = ncclRecv(rp + off,
                          (rb - off < CHUNK) ? (rb - off) : CHUNK,
                          ncclUint8, r, comm, s);
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
