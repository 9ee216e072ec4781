/* mpix — stream/graph-enqueued operations + host-side waits.
 *
 * Reference counterpart: /root/reference/src/sendrecv.cu (682 LoC CUDA).
 * MI355X-native rebuild:
 *  - fast path: hipStreamWriteValue32 / hipStreamWaitValue32(Eq) /
 *    hipStreamBatchMemOp against the device-mapped flag pool (functional
 *    probe at init, env kill-switch MPIX_DISABLE_MEMOPS);
 *  - fallback + capture/graph path: tiny gfx950 kernels using system-scope
 *    HIP atomics (host-pinned flags are uncached on-device; the spin loop
 *    backs off with s_sleep so a parked wave doesn't hammer the host link);
 *  - hipGraph support both via stream capture (kernels are capturable) and
 *    via explicit single-node graph construction (MPIX_QUEUE_HIP_GRAPH),
 *    with hipUserObject retain/release driving request cleanup;
 *  - the reference's graph-wait bug (D2: waiting on PENDING instead of
 *    COMPLETED, sendrecv.cu:411) is fixed here: all wait nodes target
 *    COMPLETED.
 */
#include <hip/hip_runtime.h>

#include "internal.h"

namespace mpix {

/* ------------------------------------------------------------ gfx950 kernels
 * All flag kernels are <<<1,1>>>: they touch a single 32-bit word in
 * host-pinned memory.  System scope makes the access visible to the CPU
 * proxy without any cache maintenance; s_sleep parks the wave between
 * probes (each probe is a fabric read over the host link). */

__global__ void k_set_flag(uint32_t *flag, uint32_t val)
{
    __hip_atomic_store(flag, val, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}

__global__ void k_wait_flag(uint32_t *flag, uint32_t val)
{
    while (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM) != val)
        __builtin_amdgcn_s_sleep(16);
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
}

__global__ void k_wait_flag_gte(uint32_t *flag, uint32_t val)
{
    while (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM) < val)
        __builtin_amdgcn_s_sleep(16);
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
}

__global__ void k_wait_and_set(uint32_t *flag, uint32_t val, uint32_t newval)
{
    while (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM) != val)
        __builtin_amdgcn_s_sleep(16);
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
    __hip_atomic_store(flag, newval, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_SYSTEM);
}

/* Waitall fallback: one wavefront polls `count` flags (indices in a pinned
 * array), each lane owning a strided subset — one launch instead of N
 * (the reference left this as dead code, sendrecv.cu:64-72 + TODO:514-520). */
__global__ void k_waitall_and_set(uint32_t *flags, const int32_t *idx,
                                  int count, uint32_t val, uint32_t newval)
{
    for (int i = (int)threadIdx.x; i < count; i += (int)blockDim.x) {
        uint32_t *f = flags + idx[i];
        while (__hip_atomic_load(f, __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_SYSTEM) != val)
            __builtin_amdgcn_s_sleep(16);
        if (newval != 0xFFFFFFFFu)
            __hip_atomic_store(f, newval, __ATOMIC_RELEASE,
                               __HIP_MEMORY_SCOPE_SYSTEM);
    }
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
}

/* ------------------------------------------------------------------ helpers */

/* Record that a spin-wait kernel has been emitted somewhere (stream
 * fallback, capture, or graph node): the transport migrates its copy
 * stream to a greatest-priority one, whose hardware queue no user stream
 * (and hence no spinning wave) can share — see copy_stream() in
 * transport/native.cpp.  The store is ordered before any launch of the
 * kernel it reports. */
static inline void mark_spin_wait()
{
    g_state->spin_wait_kernels.store(true, std::memory_order_release);
}

static bool stream_capturing(hipStream_t stream)
{
    hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
    if (hipStreamIsCapturing(stream, &st) != hipSuccess) {
        (void)hipGetLastError();
        return false;
    }
    return st == hipStreamCaptureStatusActive;
}

/* hipUserObject destructor: releases an enqueued request owned by a graph.
 * HIP schedules this on an internal thread when the owning graph's refcount
 * drops — possibly long after the graph was destroyed, after MPIX_Finalize,
 * or even after a subsequent MPIX_Init built a NEW flag pool where this
 * request's slot index means something else.  So: (a) never spin here (a
 * wedged HIP callback thread stalls all later graph/user-object work in the
 * process), and (b) act on the pool only if the state generation still
 * matches.  In-flight slots are orphaned (the proxy frees them at
 * completion); idle slots are handed over via CLEANUP. */
static void graph_request_destroy(void *ud)
{
    Request *req = (Request *)ud;
    std::lock_guard<std::mutex> lg(lifecycle_mutex());
    State *s = g_state;
    if (s == nullptr || s->gen != req->state_gen) {
        delete req; /* the owning state (and its slots) are gone */
        return;
    }
    int idx = req->flag_idx;
    std::lock_guard<std::mutex> lk(s->completion_mutex);
    if (idx < 0 || idx >= (int)s->nflags || s->ops[idx].req != req) {
        delete req; /* slot already recycled past this request */
        return;
    }
    uint32_t f = flag_load(idx);
    if (f == MPIX_FLAG_PENDING || f == MPIX_FLAG_ISSUED) {
        /* in flight: proxy deletes req + frees the slot at completion */
        s->ops[idx].orphaned.store(true, std::memory_order_relaxed);
    } else {
        /* RESERVED (never launched) or COMPLETED: proxy frees slot + req */
        flag_store(idx, MPIX_FLAG_CLEANUP);
    }
}

/* Attach graph-owned cleanup for `req` to `graph`. */
static int attach_cleanup(hipGraph_t graph, Request *req)
{
    hipUserObject_t uo = nullptr;
    MPIX_CHECK_HIP(hipUserObjectCreate(&uo, req, graph_request_destroy, 1,
                                       hipUserObjectNoDestructorSync));
    MPIX_CHECK_HIP(hipGraphRetainUserObject(graph, uo, 1,
                                            hipGraphUserObjectMove));
    return 0;
}

/* Add a flag trigger/wait node to `g`.  When the graph batch-memOp node is
 * functional (probed at init) the node is a memory operation — crucial:
 * memOp waits park in the queue scheduler instead of occupying a compute
 * queue with a spinning kernel.  A spin-wait kernel node can deadlock the
 * whole protocol when HIP's stream→HSA-queue multiplexing lands it on the
 * same hardware queue as the transport's copy work (graph execution
 * serializes queues with AQL barrier packets) — observed deterministically
 * after a few stream-create cycles (gpurun_out/diag2_*).  The kernel-node
 * fallback remains for memOps-less configs; emitting one migrates the
 * transport's copy stream to its own priority queue (mark_spin_wait) so
 * the blocked-copy cycle cannot form. */
static int add_flag_node(hipGraph_t g, bool is_wait, uint32_t *flag_d,
                         uint32_t val)
{
    State *s = g_state;
    hipGraphNode_t node = nullptr;
    if (s->use_graph_memops) {
        hipStreamBatchMemOpParams p;
        memset(&p, 0, sizeof(p));
        if (is_wait) {
            p.waitValue.operation = hipStreamMemOpWaitValue32;
            p.waitValue.address = flag_d;
            p.waitValue.value = val;
            p.waitValue.flags = hipStreamWaitValueEq;
        } else {
            p.writeValue.operation = hipStreamMemOpWriteValue32;
            p.writeValue.address = flag_d;
            p.writeValue.value = val;
        }
        hipBatchMemOpNodeParams np;
        memset(&np, 0, sizeof(np));
        np.ctx = nullptr; /* current context */
        np.count = 1;
        np.paramArray = &p;
        np.flags = 0;
        hipError_t e = hipGraphAddBatchMemOpNode(&node, g, nullptr, 0, &np);
        if (e != hipSuccess) {
            MPIX_ERR("hipGraphAddBatchMemOpNode failed: %s",
                     hipGetErrorString(e));
            return (int)e;
        }
        return 0;
    }
    if (is_wait) mark_spin_wait();
    hipKernelNodeParams p{};
    /* hipGraphAddKernelNode copies parameter values during the call, so
     * locals are fine here */
    void *kp[2] = {&flag_d, &val};
    p.func = is_wait ? (void *)k_wait_flag : (void *)k_set_flag;
    p.gridDim = dim3(1, 1, 1);
    p.blockDim = dim3(1, 1, 1);
    p.sharedMemBytes = 0;
    p.kernelParams = kp;
    p.extra = nullptr;
    hipError_t e = hipGraphAddKernelNode(&node, g, nullptr, 0, &p);
    if (e != hipSuccess) {
        MPIX_ERR("hipGraphAddKernelNode failed: %s", hipGetErrorString(e));
        return (int)e;
    }
    return 0;
}

/* Build a single-node graph triggering or waiting a flag. */
static int make_flag_graph(hipGraph_t *out, bool is_wait, uint32_t *flag_d,
                           uint32_t val)
{
    hipGraph_t g = nullptr;
    MPIX_CHECK_HIP(hipGraphCreate(&g, 0));
    int rc = add_flag_node(g, is_wait, flag_d, val);
    if (rc != 0) {
        (void)hipGraphDestroy(g);
        return rc;
    }
    *out = g;
    return 0;
}

/* Fire the PENDING trigger for slot idx on a stream/graph queue.
 * On the graph path *graph_out receives the new single-node graph. */
static int fire_trigger(int idx, int qtype, void *queue, Request *req)
{
    State *s = g_state;
    uint32_t *flag_d = s->have_gpu ? s->flags_d + idx : nullptr;

    if (qtype == MPIX_QUEUE_HIP_GRAPH) {
        if (!s->have_gpu) {
            MPIX_ERR("graph queue requires a GPU");
            return MPI_ERR_OTHER;
        }
        hipGraph_t g = nullptr;
        MPIX_CHECK(make_flag_graph(&g, false, flag_d, MPIX_FLAG_PENDING));
        MPIX_CHECK(attach_cleanup(g, req));
        *(hipGraph_t *)queue = g;
        return MPI_SUCCESS;
    }

    /* stream queue */
    if (!s->have_gpu) {
        /* proxy-only path (no GPU): the "stream" reaches the op now */
        return trigger_host(idx);
    }
    hipStream_t stream = queue ? *(hipStream_t *)queue : (hipStream_t)0;
    if (stream_capturing(stream)) {
        if (s->use_capture_memops) {
            MPIX_CHECK_HIP(hipStreamWriteValue32(
                stream, flag_d, (uint32_t)MPIX_FLAG_PENDING, 0));
        } else {
            hipLaunchKernelGGL(k_set_flag, dim3(1), dim3(1), 0, stream,
                               flag_d, (uint32_t)MPIX_FLAG_PENDING);
            MPIX_CHECK_HIP(hipGetLastError());
        }
        /* request lifetime follows the captured graph */
        hipStreamCaptureStatus cst;
        unsigned long long cid = 0;
        hipGraph_t cg = nullptr;
        MPIX_CHECK_HIP(hipStreamGetCaptureInfo_v2(stream, &cst, &cid, &cg,
                                                  nullptr, nullptr));
        if (cg != nullptr) MPIX_CHECK(attach_cleanup(cg, req));
        req->flag_idx = idx; /* graph-owned */
        return MPI_SUCCESS;
    }
    if (s->use_memops) {
        MPIX_CHECK_HIP(hipStreamWriteValue32(stream, flag_d,
                                             (uint32_t)MPIX_FLAG_PENDING, 0));
    } else {
        hipLaunchKernelGGL(k_set_flag, dim3(1), dim3(1), 0, stream, flag_d,
                           (uint32_t)MPIX_FLAG_PENDING);
        MPIX_CHECK_HIP(hipGetLastError());
    }
    return MPI_SUCCESS;
}

/* Common body of Isend/Irecv_enqueue. */
static int enqueue_sendrecv(bool is_send, void *buf, int count,
                            MPI_Datatype datatype, int peer, int tag,
                            MPI_Comm comm, MPIX_Request *request, int qtype,
                            void *queue)
{
    State *s = g_state;
    if (s == nullptr) {
        MPIX_ERR("MPIX_Init not called");
        return MPI_ERR_OTHER;
    }
    if (request == nullptr || count < 0) return MPI_ERR_ARG;
    if (qtype != MPIX_QUEUE_HIP_STREAM && qtype != MPIX_QUEUE_HIP_GRAPH)
        return MPI_ERR_ARG;

    int tsz = 0;
    MPIX_CHECK(datatype_size(datatype, &tsz));
    int peer_world = -1;
    uint32_t comm_id = 0;
    bool native_ok = false;
    MPIX_CHECK(resolve_peer(comm, peer, &peer_world, &comm_id, &native_ok));

    int idx = slot_allocate();
    if (idx < 0) return MPI_ERR_INTERN;

    Op *op = &s->ops[idx];
    op->kind = is_send ? OpKind::ISEND : OpKind::IRECV;
    op->buf = buf;
    op->count = count;
    op->datatype = datatype;
    op->bytes = (uint64_t)count * (uint64_t)tsz;
    op->peer = peer;
    op->peer_world = peer_world;
    op->tag = tag;
    op->comm = comm;
    op->comm_id = comm_id;
    op->buf_is_device = ptr_is_device(buf);
    op->native_route = native_ok;

    Request *req = new Request();
    req->kind = ReqKind::BASIC;
    req->state_gen = s->gen;
    req->flag_idx = idx;
    op->req = req;

    /* fast-wait protocol: stream queues only — graphs (and captures) replay
     * with a fixed wait value, which needs the classic EQ state machine */
    if (s->fast_wait && qtype == MPIX_QUEUE_HIP_STREAM) {
        bool capturing = false;
        if (s->have_gpu && queue != nullptr)
            capturing = stream_capturing(*(hipStream_t *)queue);
        if (!capturing) {
            req->fast = true;
            req->seq = ++s->slot_seq[idx];
            op->fast = true;
        }
    }

    if (s->stats) op->t_enq_ns = now_ns();
    slot_arm(idx);

    int rc = fire_trigger(idx, qtype, queue, req);
    if (rc != MPI_SUCCESS) {
        /* roll back: nothing triggered, proxy will see AVAILABLE and drop */
        delete req;
        slot_free(idx);
        *request = MPIX_REQUEST_NULL;
        return rc;
    }
    *request = (MPIX_Request)req;
    return MPI_SUCCESS;
}

extern "C" int MPIX_Isend_enqueue(const void *buf, int count,
                                  MPI_Datatype datatype, int dest, int tag,
                                  MPI_Comm comm, MPIX_Request *request,
                                  int qtype, void *queue)
{
    return enqueue_sendrecv(true, (void *)buf, count, datatype, dest, tag,
                            comm, request, qtype, queue);
}

extern "C" int MPIX_Irecv_enqueue(void *buf, int count, MPI_Datatype datatype,
                                  int source, int tag, MPI_Comm comm,
                                  MPIX_Request *request, int qtype, void *queue)
{
    return enqueue_sendrecv(false, buf, count, datatype, source, tag, comm,
                            request, qtype, queue);
}

/* ----------------------------------------------------------- wait (stream) */

/* Fast path: if the proxy already completed the op, consume it now and skip
 * all stream work (closes the same race the reference handles with
 * try_complete_wait_op, sendrecv.cu:82-104).  Otherwise posts the user
 * status target for proxy-side delivery.  Returns true if consumed. */
static bool try_complete_now(int idx, MPI_Status *status)
{
    State *s = g_state;
    std::lock_guard<std::mutex> lk(s->completion_mutex);
    Op *op = &s->ops[idx];
    if (flag_load(idx) == MPIX_FLAG_COMPLETED) {
        if (status != nullptr && status != MPI_STATUS_IGNORE)
            *status = op->saved_status;
        flag_store(idx, MPIX_FLAG_CLEANUP);
        return true;
    }
    if (status != nullptr && status != MPI_STATUS_IGNORE)
        op->enq_status_target = status;
    return false;
}

extern "C" int MPIX_Wait_enqueue(MPIX_Request *reqp, MPI_Status *status,
                                 int qtype, void *queue)
{
    State *s = g_state;
    if (s == nullptr || reqp == nullptr) return MPI_ERR_ARG;
    Request *req = (Request *)*reqp;
    if (req == nullptr) return MPI_ERR_REQUEST;
    if (req->kind != ReqKind::BASIC) {
        MPIX_ERR("Wait_enqueue supports enqueued (basic) requests only");
        return MPI_ERR_REQUEST;
    }
    int idx = req->flag_idx;
    uint32_t *flag_d = s->have_gpu ? s->flags_d + idx : nullptr;

    if (req->fast && qtype == MPIX_QUEUE_HIP_STREAM) {
        if (!s->have_gpu) return MPIX_Wait(reqp, status);
        hipStream_t stream = queue ? *(hipStream_t *)queue : (hipStream_t)0;
        if (stream_capturing(stream)) {
            /* the epoch wait value is request-specific; a captured graph
             * would replay it stale (same reason graphs stay classic) */
            MPIX_ERR("MPIX_FAST_WAIT requests cannot be waited inside a "
                     "stream capture");
            return MPI_ERR_REQUEST;
        }
        uint32_t wait_seq;
        {
            std::lock_guard<std::mutex> lk(s->completion_mutex);
            if (seq_load(idx) >= req->seq) {
                /* already completed; slot is long gone, status is in req */
                if (status != nullptr && status != MPI_STATUS_IGNORE)
                    *status = req->fast_status;
                delete req;
                *reqp = MPIX_REQUEST_NULL;
                return MPI_SUCCESS;
            }
            /* copy everything we need BEFORE publishing consume=1: once the
             * proxy sees consume==1 it deletes req at completion, which can
             * happen the moment the mutex is released */
            wait_seq = req->seq;
            req->consume = 1; /* proxy frees req at completion */
            if (status != nullptr && status != MPI_STATUS_IGNORE)
                s->ops[idx].enq_status_target = status;
        }
        uint32_t *seq_d = s->seqs_d + idx;
        if (s->use_memops) {
            MPIX_CHECK_HIP(hipStreamWaitValue32(stream, seq_d, wait_seq,
                                                hipStreamWaitValueGte,
                                                0xFFFFFFFFu));
        } else {
            mark_spin_wait();
            hipLaunchKernelGGL(k_wait_flag_gte, dim3(1), dim3(1), 0, stream,
                               seq_d, wait_seq);
            MPIX_CHECK_HIP(hipGetLastError());
        }
        *reqp = MPIX_REQUEST_NULL;
        return MPI_SUCCESS;
    }

    if (qtype == MPIX_QUEUE_HIP_GRAPH) {
        if (!s->have_gpu) return MPI_ERR_OTHER;
        if (req->fast) {
            MPIX_ERR("MPIX_FAST_WAIT requests cannot be waited on a graph "
                     "queue (graphs use the classic EQ protocol)");
            return MPI_ERR_REQUEST;
        }
        if (status != nullptr && status != MPI_STATUS_IGNORE) {
            std::lock_guard<std::mutex> lk(s->completion_mutex);
            Op *op = &s->ops[idx];
            if (flag_load(idx) == MPIX_FLAG_COMPLETED)
                *status = op->saved_status;
            else
                op->enq_status_target = status;
        }
        hipGraph_t g = nullptr;
        /* graph wait targets COMPLETED (reference bug D2 fixed) */
        MPIX_CHECK(make_flag_graph(&g, true, flag_d, MPIX_FLAG_COMPLETED));
        *(hipGraph_t *)queue = g;
        *reqp = MPIX_REQUEST_NULL; /* ownership: send/recv graph's user object */
        return MPI_SUCCESS;
    }

    if (!s->have_gpu) {
        /* proxy-only path: degenerate to host wait */
        return MPIX_Wait(reqp, status);
    }

    hipStream_t stream = queue ? *(hipStream_t *)queue : (hipStream_t)0;
    bool capturing = stream_capturing(stream);

    if (capturing) {
        if (status != nullptr && status != MPI_STATUS_IGNORE) {
            std::lock_guard<std::mutex> lk(s->completion_mutex);
            s->ops[idx].enq_status_target = status;
        }
        /* poll-only wait (no CLEANUP write): the captured graph relaunches
         * and recycles the flag; cleanup rides the graph user object */
        if (s->use_capture_memops) {
            MPIX_CHECK_HIP(hipStreamWaitValue32(stream, flag_d,
                                                (uint32_t)MPIX_FLAG_COMPLETED,
                                                hipStreamWaitValueEq,
                                                0xFFFFFFFFu));
        } else {
            mark_spin_wait();
            hipLaunchKernelGGL(k_wait_flag, dim3(1), dim3(1), 0, stream,
                               flag_d, (uint32_t)MPIX_FLAG_COMPLETED);
            MPIX_CHECK_HIP(hipGetLastError());
        }
        *reqp = MPIX_REQUEST_NULL;
        return MPI_SUCCESS;
    }

    if (try_complete_now(idx, status)) {
        *reqp = MPIX_REQUEST_NULL;
        return MPI_SUCCESS;
    }
    if (s->use_memops) {
        MPIX_CHECK_HIP(hipStreamWaitValue32(stream, flag_d,
                                            (uint32_t)MPIX_FLAG_COMPLETED,
                                            hipStreamWaitValueEq, 0xFFFFFFFFu));
        MPIX_CHECK_HIP(hipStreamWriteValue32(stream, flag_d,
                                             (uint32_t)MPIX_FLAG_CLEANUP, 0));
    } else {
        mark_spin_wait();
        hipLaunchKernelGGL(k_wait_and_set, dim3(1), dim3(1), 0, stream,
                           flag_d, (uint32_t)MPIX_FLAG_COMPLETED,
                           (uint32_t)MPIX_FLAG_CLEANUP);
        MPIX_CHECK_HIP(hipGetLastError());
    }
    *reqp = MPIX_REQUEST_NULL;
    return MPI_SUCCESS;
}

extern "C" int MPIX_Waitall_enqueue(int count, MPIX_Request *reqs,
                                    MPI_Status *statuses, int qtype,
                                    void *queue)
{
    State *s = g_state;
    if (s == nullptr || (count > 0 && reqs == nullptr)) return MPI_ERR_ARG;
    if (count == 0) return MPI_SUCCESS;

    auto status_at = [&](int i) -> MPI_Status * {
        if (statuses == nullptr || statuses == MPI_STATUSES_IGNORE)
            return nullptr;
        return &statuses[i];
    };

    if (qtype == MPIX_QUEUE_HIP_GRAPH) {
        if (!s->have_gpu) return MPI_ERR_OTHER;
        hipGraph_t g = nullptr;
        MPIX_CHECK_HIP(hipGraphCreate(&g, 0));
        for (int i = 0; i < count; i++) {
            Request *req = (Request *)reqs[i];
            if (req == nullptr || req->kind != ReqKind::BASIC) continue;
            if (req->fast) {
                (void)hipGraphDestroy(g);
                MPIX_ERR("MPIX_FAST_WAIT requests cannot be waited on a "
                         "graph queue");
                return MPI_ERR_REQUEST;
            }
            int idx = req->flag_idx;
            {
                std::lock_guard<std::mutex> lk(s->completion_mutex);
                MPI_Status *st = status_at(i);
                if (st) {
                    if (flag_load(idx) == MPIX_FLAG_COMPLETED)
                        *st = s->ops[idx].saved_status;
                    else
                        s->ops[idx].enq_status_target = st;
                }
            }
            int arc = add_flag_node(g, true, s->flags_d + idx,
                                    MPIX_FLAG_COMPLETED);
            if (arc != 0) {
                (void)hipGraphDestroy(g);
                return arc;
            }
            reqs[i] = MPIX_REQUEST_NULL;
        }
        *(hipGraph_t *)queue = g;
        return MPI_SUCCESS;
    }

    if (!s->have_gpu) return MPIX_Waitall(count, reqs, statuses);

    hipStream_t stream = queue ? *(hipStream_t *)queue : (hipStream_t)0;
    bool capturing = stream_capturing(stream);

    /* batched memOps fast path: one submission for all waits+cleanups */
    if (!capturing && s->use_batch_memops) {
        std::vector<hipStreamBatchMemOpParams> params;
        params.reserve(2 * (size_t)count);
        for (int i = 0; i < count; i++) {
            Request *req = (Request *)reqs[i];
            if (req == nullptr) continue;
            if (req->kind != ReqKind::BASIC) return MPI_ERR_REQUEST;
            if (req->fast) {
                /* fast ops complete via the seq word and free their slot
                 * immediately — a batched EQ-COMPLETED wait on the recycled
                 * flag would hang or fire on an unrelated op.  Route through
                 * the GTE wait, same as the unbatched path. */
                MPIX_Request r1 = (MPIX_Request)req;
                MPIX_CHECK(MPIX_Wait_enqueue(&r1, status_at(i),
                                             MPIX_QUEUE_HIP_STREAM, queue));
                reqs[i] = MPIX_REQUEST_NULL;
                continue;
            }
            int idx = req->flag_idx;
            if (try_complete_now(idx, status_at(i))) {
                reqs[i] = MPIX_REQUEST_NULL;
                continue;
            }
            uint32_t *flag_d = s->flags_d + idx;
            hipStreamBatchMemOpParams w{};
            w.waitValue.operation = hipStreamMemOpWaitValue32;
            w.waitValue.address = flag_d;
            w.waitValue.value = MPIX_FLAG_COMPLETED;
            w.waitValue.flags = hipStreamWaitValueEq;
            params.push_back(w);
            hipStreamBatchMemOpParams wr{};
            wr.writeValue.operation = hipStreamMemOpWriteValue32;
            wr.writeValue.address = flag_d;
            wr.writeValue.value = MPIX_FLAG_CLEANUP;
            params.push_back(wr);
            reqs[i] = MPIX_REQUEST_NULL;
        }
        if (!params.empty())
            MPIX_CHECK_HIP(hipStreamBatchMemOp(stream,
                                               (unsigned)params.size(),
                                               params.data(), 0));
        return MPI_SUCCESS;
    }

    /* unbatched memOps: one (wait, write) pair per request — same engine
     * as MPIX_Wait_enqueue; used when hipStreamBatchMemOp is unavailable
     * (probe_memops found it non-functional on ROCm 7.x mainline) */
    if (!capturing && s->use_memops) {
        for (int i = 0; i < count; i++) {
            Request *req = (Request *)reqs[i];
            if (req == nullptr) continue;
            if (req->kind != ReqKind::BASIC) return MPI_ERR_REQUEST;
            int idx = req->flag_idx;
            if (req->fast) {
                MPIX_Request r1 = (MPIX_Request)req;
                MPIX_CHECK(MPIX_Wait_enqueue(&r1, status_at(i),
                                             MPIX_QUEUE_HIP_STREAM, queue));
                reqs[i] = MPIX_REQUEST_NULL;
                continue;
            }
            if (!try_complete_now(idx, status_at(i))) {
                uint32_t *flag_d = s->flags_d + idx;
                MPIX_CHECK_HIP(hipStreamWaitValue32(
                    stream, flag_d, (uint32_t)MPIX_FLAG_COMPLETED,
                    hipStreamWaitValueEq, 0xFFFFFFFFu));
                MPIX_CHECK_HIP(hipStreamWriteValue32(
                    stream, flag_d, (uint32_t)MPIX_FLAG_CLEANUP, 0));
            }
            reqs[i] = MPIX_REQUEST_NULL;
        }
        return MPI_SUCCESS;
    }

    /* capture path: NO allocations are legal while a stream captures
     * (hipHostMalloc returns hipErrorStreamCaptureUnsupported), so record
     * one poll-only wait kernel per request — flags re-cycle on relaunch,
     * cleanup is owned by the send/recv graph user objects. */
    if (capturing) {
        for (int i = 0; i < count; i++) {
            Request *req = (Request *)reqs[i];
            if (req == nullptr) continue;
            if (req->kind != ReqKind::BASIC) return MPI_ERR_REQUEST;
            if (req->fast) {
                MPIX_ERR("MPIX_FAST_WAIT requests cannot be waited inside "
                         "a stream capture");
                return MPI_ERR_REQUEST;
            }
            int idx = req->flag_idx;
            {
                std::lock_guard<std::mutex> lk(s->completion_mutex);
                MPI_Status *st = status_at(i);
                if (st) s->ops[idx].enq_status_target = st;
            }
            if (s->use_capture_memops) {
                MPIX_CHECK_HIP(hipStreamWaitValue32(
                    stream, s->flags_d + idx, (uint32_t)MPIX_FLAG_COMPLETED,
                    hipStreamWaitValueEq, 0xFFFFFFFFu));
            } else {
                mark_spin_wait();
            hipLaunchKernelGGL(k_wait_flag, dim3(1), dim3(1), 0, stream,
                                   s->flags_d + idx,
                                   (uint32_t)MPIX_FLAG_COMPLETED);
                MPIX_CHECK_HIP(hipGetLastError());
            }
            reqs[i] = MPIX_REQUEST_NULL;
        }
        return MPI_SUCCESS;
    }

    /* kernel fallback (no memOps, not capturing).  The single-wave variant
     * (k_waitall_and_set: one launch polls every flag) is the default for
     * 4+ requests — its r01 "in-situ hang" was the shared-hardware-queue
     * blocking class (see add_flag_node), fixed by the copy-stream design;
     * the forced-mode gpu_ci stage (waitall_wave_kernel) keeps it covered.
     * MPIX_WAITALL_KERNEL=0 forces per-request k_wait_and_set launches,
     * =1 forces the wave kernel even for small counts. */
    static const int wave_env = [] {
        const char *v = getenv("MPIX_WAITALL_KERNEL");
        return v && *v ? atoi(v) : -1;
    }();
    bool force_wave = wave_env >= 0 ? wave_env != 0 : count >= 4;
    if (!force_wave) {
        for (int i = 0; i < count; i++) {
            Request *req = (Request *)reqs[i];
            if (req == nullptr) continue;
            if (req->kind != ReqKind::BASIC) return MPI_ERR_REQUEST;
            int idx = req->flag_idx;
            if (req->fast) {
                MPIX_Request r1 = (MPIX_Request)req;
                MPIX_CHECK(MPIX_Wait_enqueue(&r1, status_at(i),
                                             MPIX_QUEUE_HIP_STREAM, queue));
                reqs[i] = MPIX_REQUEST_NULL;
                continue;
            }
            if (!try_complete_now(idx, status_at(i))) {
                mark_spin_wait();
            hipLaunchKernelGGL(k_wait_and_set, dim3(1), dim3(1), 0,
                                   stream, s->flags_d + idx,
                                   (uint32_t)MPIX_FLAG_COMPLETED,
                                   (uint32_t)MPIX_FLAG_CLEANUP);
                MPIX_CHECK_HIP(hipGetLastError());
            }
            reqs[i] = MPIX_REQUEST_NULL;
        }
        return MPI_SUCCESS;
    }

    /* single-wave path: one launch polls every flag.
     * The index array rides in pinned memory freed by a host callback. */
    int32_t *idx_arr = nullptr;
    MPIX_CHECK_HIP(hipHostMalloc((void **)&idx_arr,
                                 (size_t)count * sizeof(int32_t),
                                 hipHostMallocMapped));
    int n = 0;
    for (int i = 0; i < count; i++) {
        Request *req = (Request *)reqs[i];
        if (req == nullptr) continue;
        if (req->kind != ReqKind::BASIC) {
            (void)hipHostFree(idx_arr);
            return MPI_ERR_REQUEST;
        }
        if (req->fast) { /* same rerouting as the other Waitall branches */
            MPIX_Request r1 = (MPIX_Request)req;
            int rc = MPIX_Wait_enqueue(&r1, status_at(i),
                                       MPIX_QUEUE_HIP_STREAM, queue);
            if (rc != MPI_SUCCESS) {
                (void)hipHostFree(idx_arr);
                return rc;
            }
            reqs[i] = MPIX_REQUEST_NULL;
            continue;
        }
        int idx = req->flag_idx;
        if (try_complete_now(idx, status_at(i))) {
            /* done already */
        } else {
            idx_arr[n++] = idx;
        }
        reqs[i] = MPIX_REQUEST_NULL;
    }
    if (n > 0) {
        int32_t *idx_d = nullptr;
        MPIX_CHECK_HIP(hipHostGetDevicePointer((void **)&idx_d, idx_arr, 0));
        int threads = n < 64 ? 64 : ((n + 63) / 64) * 64;
        if (threads > 1024) threads = 1024;
        mark_spin_wait();
            hipLaunchKernelGGL(k_waitall_and_set, dim3(1), dim3(threads), 0,
                           stream, s->flags_d, idx_d, n,
                           (uint32_t)MPIX_FLAG_COMPLETED,
                           (uint32_t)MPIX_FLAG_CLEANUP);
        MPIX_CHECK_HIP(hipGetLastError());
        MPIX_CHECK_HIP(hipLaunchHostFunc(
            stream, [](void *p) { (void)hipHostFree(p); }, idx_arr));
    } else {
        (void)hipHostFree(idx_arr);
    }
    return MPI_SUCCESS;
}

/* ------------------------------------------------------------- wait (host) */

static int host_wait_basic(Request *req, MPI_Status *status)
{
    State *s = g_state;
    int idx = req->flag_idx;
    if (req->fast) {
        {
            std::lock_guard<std::mutex> lk(s->completion_mutex);
            if (seq_load(idx) < req->seq)
                req->consume = 2; /* claim: proxy must not free req */
        }
        int fspins = 0;
        while (seq_load(idx) < req->seq) {
            if (++fspins > 4096) {
                std::this_thread::yield();
                fspins = 0;
            }
        }
        if (status != nullptr && status != MPI_STATUS_IGNORE)
            *status = req->fast_status;
        delete req;
        return MPI_SUCCESS;
    }
    int spins = 0;
    while (flag_load(idx) != MPIX_FLAG_COMPLETED) {
        if (++spins > 4096) {
            std::this_thread::yield();
            spins = 0;
        }
    }
    {
        std::lock_guard<std::mutex> lk(s->completion_mutex);
        if (status != nullptr && status != MPI_STATUS_IGNORE)
            *status = s->ops[idx].saved_status;
        slot_free(idx);
    }
    delete req;
    return MPI_SUCCESS;
}

static int host_wait_partitioned(Request *req, MPI_Status *status)
{
    State *s = g_state;
    if (!req->active) return MPI_SUCCESS;
    for (int p = 0; p < req->n_partitions; p++) {
        int idx = req->part_idx[p];
        int spins = 0;
        while (flag_load(idx) != MPIX_FLAG_COMPLETED) {
            if (++spins > 4096) {
                std::this_thread::yield();
                spins = 0;
            }
        }
    }
    /* all partitions completed: reset for the next MPIX_Start */
    for (int p = 0; p < req->n_partitions; p++) {
        int idx = req->part_idx[p];
        s->ops[idx].ch_done.store(0, std::memory_order_relaxed);
        s->ops[idx].status_saved = false;
        flag_store(idx, MPIX_FLAG_RESERVED);
    }
    req->active = false;
#if MPIX_HAVE_MPI_PARTITIONED
    if (req->mpi_part_native) {
        /* reference semantics (sendrecv.cu:627): the host wait completes the
         * underlying persistent MPI request after all partitions are done */
        MPIX_CHECK(MPI_Wait(&req->mpi_preq, MPI_STATUS_IGNORE));
    }
#endif
    if (status != nullptr && status != MPI_STATUS_IGNORE) {
        ChStatus cs;
        cs.src = req->is_send ? -1 : req->peer;
        cs.tag = req->tag;
        cs.bytes = req->part_bytes * (uint64_t)req->n_partitions;
        cs.err = MPI_SUCCESS;
        fill_status(status, cs);
    }
    return MPI_SUCCESS;
}

extern "C" int MPIX_Wait(MPIX_Request *reqp, MPI_Status *status)
{
    if (g_state == nullptr || reqp == nullptr) return MPI_ERR_ARG;
    Request *req = (Request *)*reqp;
    if (req == nullptr) return MPI_SUCCESS; /* MPI semantics: null is no-op */
    if (req->kind == ReqKind::BASIC) {
        int rc = host_wait_basic(req, status);
        *reqp = MPIX_REQUEST_NULL;
        return rc;
    }
    /* partitioned requests stay valid (persistent) */
    return host_wait_partitioned(req, status);
}

extern "C" int MPIX_Waitall(int count, MPIX_Request *reqs,
                            MPI_Status *statuses)
{
    for (int i = 0; i < count; i++) {
        MPI_Status *st =
            (statuses == nullptr || statuses == MPI_STATUSES_IGNORE)
                ? MPI_STATUS_IGNORE
                : &statuses[i];
        MPIX_CHECK(MPIX_Wait(&reqs[i], st));
    }
    return MPI_SUCCESS;
}

extern "C" int MPIX_Request_free(MPIX_Request *reqp)
{
    State *s = g_state;
    if (s == nullptr || reqp == nullptr) return MPI_ERR_ARG;
    Request *req = (Request *)*reqp;
    if (req == nullptr) return MPI_SUCCESS;

    if (req->kind == ReqKind::BASIC && req->fast) {
        int idx = req->flag_idx;
        std::lock_guard<std::mutex> lk(s->completion_mutex);
        if (seq_load(idx) >= req->seq)
            delete req;        /* completed: slot already recycled */
        else
            req->consume = 1;  /* proxy frees it at completion */
        *reqp = MPIX_REQUEST_NULL;
        return MPI_SUCCESS;
    }

    if (req->kind == ReqKind::BASIC) {
        int idx = req->flag_idx;
        std::lock_guard<std::mutex> lk(s->completion_mutex);
        uint32_t f = flag_load(idx);
        if (f == MPIX_FLAG_COMPLETED) {
            flag_store(idx, MPIX_FLAG_CLEANUP); /* proxy frees slot+request */
        } else {
            s->ops[idx].orphaned.store(true, std::memory_order_relaxed);
            /* proxy frees at completion */
        }
        *reqp = MPIX_REQUEST_NULL;
        return MPI_SUCCESS;
    }

    /* partitioned: must be inactive (MPI rule); drain defensively anyway */
    for (int p = 0; p < req->n_partitions; p++) {
        int idx = req->part_idx[p];
        uint32_t f;
        while ((f = flag_load(idx)) == MPIX_FLAG_PENDING ||
               f == MPIX_FLAG_ISSUED)
            std::this_thread::yield();
        flag_store(idx, MPIX_FLAG_CLEANUP); /* proxy frees slot */
    }
    if (req->dev_idx) (void)hipFree(req->dev_idx);
    if (req->dev_handle) (void)hipFree(req->dev_handle);
#if MPIX_HAVE_MPI_PARTITIONED
    if (req->mpi_part_native && req->mpi_preq != MPI_REQUEST_NULL)
        (void)MPI_Request_free(&req->mpi_preq);
#endif
    delete req;
    *reqp = MPIX_REQUEST_NULL;
    return MPI_SUCCESS;
}

} /* namespace mpix */
