/*
 * cagg_rccl.cpp — the combine step as a device collective: RCCL over xGMI
 * in place of the reference's coordinator merge (the worker partial rows
 * that flow back over libpq, adaptive_executor.c:775-882, merged by
 * coord_combine_agg, aggregate_utils.c:820-1021).
 *
 * Shape: ncclAllGather of each rank's fixed-size partial block, then the
 * SAME strict combine (cagg_combine) applied locally on every rank —
 * all-gather-then-combine rather than ncclAllReduce because the int128
 * fixed-point sums carry across 64-bit lanes (a sum collective would lose
 * carries) and because it is byte-faithful to the coordinator receiving one
 * partial row per shard. Host code is C; callers only distribute the
 * 128-byte unique id (ncclCommInitRank's own bootstrap contract).
 */
#include "internal.h"

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <cstring>

struct cagg_comm {
    ncclComm_t comm = nullptr;
    hipStream_t stream = nullptr;
    void *d_send = nullptr;
    void *d_recv = nullptr;
    size_t cap = 0;            /* bytes of d_send; d_recv is cap * n_ranks */
    int rank = -1;
    int n_ranks = 0;
};

#define RCCL_TRY(x) do { ncclResult_t _r = (x); if (_r != ncclSuccess) { \
    cs_set_err("%s failed: %s", #x, ncclGetErrorString(_r)); return CSTRIPE_ERR; } } while (0)
#define HIPC_TRY(x) do { hipError_t _e = (x); if (_e != hipSuccess) { \
    cs_set_err("%s failed: %s", #x, hipGetErrorString(_e)); return CSTRIPE_ERR; } } while (0)

extern "C" int cagg_comm_unique_id(uint8_t id[CAGG_UNIQUE_ID_BYTES])
{
    static_assert(CAGG_UNIQUE_ID_BYTES == NCCL_UNIQUE_ID_BYTES,
                  "unique id size mismatch");
    ncclUniqueId uid;
    RCCL_TRY(ncclGetUniqueId(&uid));
    memcpy(id, uid.internal, NCCL_UNIQUE_ID_BYTES);
    return CSTRIPE_OK;
}

extern "C" int cagg_comm_init(cagg_comm **out, int n_ranks, int rank,
                              const uint8_t id[CAGG_UNIQUE_ID_BYTES], int device)
{
    if (!out || n_ranks <= 0 || rank < 0 || rank >= n_ranks || !id) {
        cs_set_err("cagg_comm_init: bad args");
        return CSTRIPE_ERR_ARG;
    }
    if (device >= 0)
        HIPC_TRY(hipSetDevice(device));
    auto *c = new cagg_comm();
    c->rank = rank;
    c->n_ranks = n_ranks;
    ncclUniqueId uid;
    memcpy(uid.internal, id, NCCL_UNIQUE_ID_BYTES);
    if (hipStreamCreate(&c->stream) != hipSuccess) {
        cs_set_err("hipStreamCreate failed");
        delete c;
        return CSTRIPE_ERR;
    }
    ncclResult_t r = ncclCommInitRank(&c->comm, n_ranks, uid, rank);
    if (r != ncclSuccess) {
        cs_set_err("ncclCommInitRank failed: %s", ncclGetErrorString(r));
        (void)hipStreamDestroy(c->stream);
        delete c;
        return CSTRIPE_ERR;
    }
    *out = c;
    return CSTRIPE_OK;
}

extern "C" void cagg_comm_destroy(cagg_comm *c)
{
    if (!c) return;
    if (c->d_send) (void)hipFree(c->d_send);
    if (c->d_recv) (void)hipFree(c->d_recv);
    if (c->comm) (void)ncclCommDestroy(c->comm);
    if (c->stream) (void)hipStreamDestroy(c->stream);
    delete c;
}

extern "C" int cagg_comm_rank(const cagg_comm *c) { return c ? c->rank : -1; }
extern "C" int cagg_comm_size(const cagg_comm *c) { return c ? c->n_ranks : 0; }

static int ensure_cap(cagg_comm *c, size_t bytes)
{
    if (c->cap >= bytes) return CSTRIPE_OK;
    if (c->d_send) (void)hipFree(c->d_send);
    if (c->d_recv) (void)hipFree(c->d_recv);
    c->d_send = c->d_recv = nullptr;
    size_t cap = (bytes + 4095) & ~(size_t)4095;
    HIPC_TRY(hipMalloc(&c->d_send, cap));
    HIPC_TRY(hipMalloc(&c->d_recv, cap * (size_t)c->n_ranks));
    c->cap = cap;
    return CSTRIPE_OK;
}

extern "C" int cagg_allgather(cagg_comm *c, const void *src, uint64_t bytes,
                              void *dst)
{
    if (!c || !src || !dst || bytes == 0) {
        cs_set_err("cagg_allgather: bad args");
        return CSTRIPE_ERR_ARG;
    }
    int rc = ensure_cap(c, bytes);
    if (rc != CSTRIPE_OK) return rc;
    HIPC_TRY(hipMemcpyAsync(c->d_send, src, bytes, hipMemcpyHostToDevice, c->stream));
    RCCL_TRY(ncclAllGather(c->d_send, c->d_recv, bytes, ncclUint8, c->comm, c->stream));
    /* ncclAllGather packs rank r's block at offset r * bytes */
    HIPC_TRY(hipMemcpyAsync(dst, c->d_recv, bytes * (uint64_t)c->n_ranks,
                            hipMemcpyDeviceToHost, c->stream));
    HIPC_TRY(hipStreamSynchronize(c->stream));
    return CSTRIPE_OK;
}

extern "C" int cagg_combine_rccl(cagg_comm *c, const cstripe_agg_spec *aggs,
                                 uint32_t n_aggs, const cstripe_partial *local,
                                 cstripe_partial *out)
{
    if (!c || !aggs || n_aggs == 0 || !local || !out) {
        cs_set_err("cagg_combine_rccl: bad args");
        return CSTRIPE_ERR_ARG;
    }
    const uint64_t bytes = (uint64_t)n_aggs * sizeof(cstripe_partial);
    std::vector<cstripe_partial> all((size_t)n_aggs * c->n_ranks);
    int rc = cagg_allgather(c, local, bytes, all.data());
    if (rc != CSTRIPE_OK) return rc;
    return cagg_combine(aggs, n_aggs, all.data(), (uint32_t)c->n_ranks, out);
}
