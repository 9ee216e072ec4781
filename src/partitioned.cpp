/* mpix — partitioned communication (MPIX_Psend/Precv_init, Start, Pready...).
 *
 * Reference counterpart: /root/reference/src/partitioned.cu (231 LoC).
 * The reference passes through to MPI-4.0 MPI_Psend_init/MPI_Pready; this
 * environment has MPI 3.1 (MPICH 3.3.2), so partitioned transfer is
 * implemented natively: each partition is an independent tag-matched
 * message on the shm/xGMI channel, carrying (tag, partition) in its
 * matching key.  This preserves the reference's per-partition flag protocol
 * exactly — a GPU kernel publishes partition p with __device__ MPIX_Pready
 * (system-scope store of PENDING), the proxy sees it and issues that
 * partition's transfer over xGMI immediately, giving fine-grained
 * compute/communication overlap.
 *
 * Flag lifecycle per partition slot (persistent across Start iterations):
 *   send: RESERVED --Pready--> PENDING --proxy--> ISSUED --> COMPLETED
 *         --MPIX_Wait--> RESERVED
 *   recv: RESERVED --Start--> PENDING (proxy posts recv) --> ISSUED
 *         --> COMPLETED --MPIX_Wait--> RESERVED
 */
#include <hip/hip_runtime.h>

#include "internal.h"

namespace mpix {

static int psend_precv_init(bool is_send, void *buf, int partitions,
                            MPI_Count count, MPI_Datatype datatype, int peer,
                            int tag, MPI_Comm comm, MPIX_Request *request)
{
    State *s = g_state;
    if (s == nullptr) return MPI_ERR_OTHER;
    if (request == nullptr || partitions <= 0 || count < 0) return MPI_ERR_ARG;

    int tsz = 0;
    MPIX_CHECK(datatype_size(datatype, &tsz));
    int peer_world = -1;
    uint32_t comm_id = 0;
    bool native_ok = false;
    MPIX_CHECK(resolve_peer(comm, peer, &peer_world, &comm_id, &native_ok));
    if (peer_world == MPI_ANY_SOURCE) {
        MPIX_ERR("partitioned recv requires a concrete source rank");
        return MPI_ERR_RANK;
    }

    /* MPI-4.0 native passthrough (reference parity, partitioned.cu:57-59):
     * runtime opt-in, MPI mode only.  The flag protocol stays ours; the MPI
     * library owns transfer + matching, so arbitrary comms work too. */
    bool mpi_native = false;
#if MPIX_HAVE_MPI_PARTITIONED
    {
        static const int want = [] {
            const char *e = getenv("MPIX_MPI_PARTITIONED");
            return e ? atoi(e) : 0;
        }();
        mpi_native = want && s->mpi_mode;
    }
#endif
    if (!native_ok && !s->mpi_mode) {
        MPIX_ERR("partitioned ops on non-WORLD/SELF comms require MPI mode");
        return MPI_ERR_COMM;
    }

    Request *req = new Request();
    req->kind = ReqKind::PARTITIONED;
    req->is_send = is_send;
    req->n_partitions = partitions;
    req->part_idx.resize(partitions);
    req->buf = buf;
    req->part_bytes = (uint64_t)count * (uint64_t)tsz;
    req->peer = peer;
    req->peer_world = peer_world;
    req->tag = tag;
    req->comm = comm;
    req->comm_id = comm_id;
    req->datatype = datatype;
    req->count_per_part = (int)count;
    req->active = false;

#if MPIX_HAVE_MPI_PARTITIONED
    if (mpi_native) {
        int rc = is_send
            ? MPI_Psend_init(buf, partitions, count, datatype, peer, tag,
                             comm, MPI_INFO_NULL, &req->mpi_preq)
            : MPI_Precv_init(buf, partitions, count, datatype, peer, tag,
                             comm, MPI_INFO_NULL, &req->mpi_preq);
        if (rc != MPI_SUCCESS) {
            MPIX_ERR("MPI_P%s_init failed (%d)", is_send ? "send" : "recv", rc);
            delete req;
            return rc;
        }
        req->mpi_part_native = true;
    }
#else
    (void)mpi_native;
#endif

    for (int p = 0; p < partitions; p++) {
        int idx = slot_allocate();
        if (idx < 0) {
            for (int q = 0; q < p; q++) {
                slot_free(req->part_idx[q]);
            }
            delete req;
            return MPI_ERR_INTERN;
        }
        req->part_idx[p] = idx;
        Op *op = &s->ops[idx];
        op->kind = is_send ? OpKind::PSEND_PART : OpKind::PRECV_PART;
        op->buf = (char *)buf + (uint64_t)p * req->part_bytes;
        op->count = (int)count;
        op->datatype = datatype;
        op->bytes = req->part_bytes;
        op->peer = peer;
        op->peer_world = peer_world;
        op->tag = tag;
        op->comm = comm;
        op->comm_id = comm_id;
        op->partition = p;
        /* native shm/xGMI for WORLD/SELF; MPI transport for arbitrary comms
         * (header-routed partition messages) and the MPI-4.0 passthrough */
        op->native_route = native_ok && !req->mpi_part_native;
        op->req = req;
    }
    /* buffer kind determined once (partitions are slices of one buffer) */
    bool dev = ptr_is_device(buf);
    for (int p = 0; p < partitions; p++)
        s->ops[req->part_idx[p]].buf_is_device = dev;

    *request = (MPIX_Request)req;
    return MPI_SUCCESS;
}

extern "C" int MPIX_Psend_init(const void *buf, int partitions, MPI_Count count,
                               MPI_Datatype datatype, int dest, int tag,
                               MPI_Comm comm, MPI_Info info,
                               MPIX_Request *request)
{
    (void)info;
    return psend_precv_init(true, (void *)buf, partitions, count, datatype,
                            dest, tag, comm, request);
}

extern "C" int MPIX_Precv_init(void *buf, int partitions, MPI_Count count,
                               MPI_Datatype datatype, int source, int tag,
                               MPI_Comm comm, MPI_Info info,
                               MPIX_Request *request)
{
    (void)info;
    return psend_precv_init(false, buf, partitions, count, datatype, source,
                            tag, comm, request);
}

extern "C" int MPIX_Start(MPIX_Request *reqp)
{
    State *s = g_state;
    if (s == nullptr || reqp == nullptr) return MPI_ERR_ARG;
    Request *req = (Request *)*reqp;
    if (req == nullptr || req->kind != ReqKind::PARTITIONED)
        return MPI_ERR_REQUEST;
    if (req->active) {
        MPIX_ERR("MPIX_Start on an active partitioned request");
        return MPI_ERR_REQUEST;
    }
    req->active = true;
    req->start_seq++;
#if MPIX_HAVE_MPI_PARTITIONED
    if (req->mpi_part_native) {
        int rc = MPI_Start(&req->mpi_preq);
        if (rc != MPI_SUCCESS) {
            req->active = false;
            req->start_seq--;
            return rc;
        }
    }
#endif
    for (int p = 0; p < req->n_partitions; p++) {
        int idx = req->part_idx[p];
        Op *op = &s->ops[idx];
        op->ch_done.store(0, std::memory_order_relaxed);
        op->status_saved = false;
        op->pseq = req->start_seq;
        slot_arm(idx); /* proxy dedups repeated arms */
        if (!req->is_send) {
            /* post the partition's receive via the proxy */
            flag_store(idx, MPIX_FLAG_PENDING);
        }
        /* send partitions stay RESERVED until MPIX_Pready */
    }
    return MPI_SUCCESS;
}

extern "C" int MPIX_Startall(int count, MPIX_Request *reqs)
{
    for (int i = 0; i < count; i++) MPIX_CHECK(MPIX_Start(&reqs[i]));
    return MPI_SUCCESS;
}

/* ----------------------------------------------------- host Pready/Parrived */

extern "C" int MPIX_Pready(int partition, void *request)
{
    Request *req = (Request *)request;
    if (req == nullptr || req->kind != ReqKind::PARTITIONED || !req->is_send)
        return MPI_ERR_REQUEST;
    if (partition < 0 || partition >= req->n_partitions) return MPI_ERR_ARG;
    if (!req->active) return MPI_ERR_REQUEST;
    flag_store(req->part_idx[partition], MPIX_FLAG_PENDING);
    return MPI_SUCCESS;
}

extern "C" int MPIX_Parrived(void *request, int partition, int *flag)
{
    Request *req = (Request *)request;
    if (req == nullptr || req->kind != ReqKind::PARTITIONED || req->is_send)
        return MPI_ERR_REQUEST;
    if (partition < 0 || partition >= req->n_partitions || flag == nullptr)
        return MPI_ERR_ARG;
    *flag = (flag_load(req->part_idx[partition]) == MPIX_FLAG_COMPLETED);
    return MPI_SUCCESS;
}

/* -------------------------------------------------------- device prequest */

extern "C" int MPIX_Prequest_create(MPIX_Request request,
                                    MPIX_Prequest *prequest)
{
    State *s = g_state;
    if (s == nullptr || prequest == nullptr) return MPI_ERR_ARG;
    Request *req = (Request *)request;
    if (req == nullptr || req->kind != ReqKind::PARTITIONED)
        return MPI_ERR_REQUEST;
    if (!s->have_gpu) {
        MPIX_ERR("MPIX_Prequest_create requires a GPU");
        return MPI_ERR_OTHER;
    }
    if (req->dev_handle != nullptr) { /* already created */
        *prequest = (MPIX_Prequest)req->dev_handle;
        return MPI_SUCCESS;
    }
    int n = req->n_partitions;
    MPIX_CHECK_HIP(hipMalloc(&req->dev_idx, (size_t)n * sizeof(int32_t)));
    MPIX_CHECK_HIP(hipMemcpy(req->dev_idx, req->part_idx.data(),
                             (size_t)n * sizeof(int32_t),
                             hipMemcpyHostToDevice));
    mpix_prequest_dev_t h{};
    h.n_partitions = n;
    h.idx = req->dev_idx;
    h.flags = s->flags_d;
    MPIX_CHECK_HIP(hipMalloc(&req->dev_handle, sizeof(h)));
    MPIX_CHECK_HIP(hipMemcpy(req->dev_handle, &h, sizeof(h),
                             hipMemcpyHostToDevice));
    *prequest = (MPIX_Prequest)req->dev_handle;
    return MPI_SUCCESS;
}

extern "C" int MPIX_Prequest_free(MPIX_Prequest *prequest)
{
    if (prequest == nullptr) return MPI_ERR_ARG;
    /* The device handle is owned by the Request (freed in MPIX_Request_free);
     * this mirrors the reference where Prequest_free only drops the device
     * mirror (partitioned.cu:192-197). Here we free the device mirror and
     * clear the owning request's pointers. */
    State *s = g_state;
    if (s == nullptr) return MPI_ERR_OTHER;
    mpix_prequest_dev_t *dh = (mpix_prequest_dev_t *)*prequest;
    if (dh == nullptr) return MPI_SUCCESS;
    /* find the owning request via the op table (any partition op) */
    for (size_t i = 0; i < s->nflags; i++) {
        Op *op = &s->ops[i];
        if (op->req != nullptr && op->req->kind == ReqKind::PARTITIONED &&
            op->req->dev_handle == dh) {
            (void)hipFree(op->req->dev_idx);
            (void)hipFree(op->req->dev_handle);
            op->req->dev_idx = nullptr;
            op->req->dev_handle = nullptr;
            break;
        }
    }
    *prequest = MPIX_PREQUEST_NULL;
    return MPI_SUCCESS;
}

} /* namespace mpix */
