/* mpix — internal data model: state machine, op/request descriptors, globals.
 *
 * Design notes (vs the reference, /root/reference/include/mpi-acx-internal.h):
 *  - flags are C++ std::atomic<uint32_t> words in host-pinned fine-grained
 *    memory (device writes them with system-scope HIP atomics), not volatile.
 *  - the slot allocator is lock-free CAS with a rotating cursor (fixes the
 *    reference's single-issuer FIXME, triggered.cpp:40-44).
 *  - the proxy watches only ACTIVE slots (an MPSC hand-off ring + a
 *    proxy-owned watch list) instead of scanning the whole pool every pass
 *    (reference scans all 4096 flags, init.cpp:61).
 *  - CLEANUP is handled at the top level of the proxy loop (fixes the
 *    reference's slot leak, D1 in SURVEY.md).
 *  - the data plane is a pluggable Transport (native shm/xGMI or host MPI),
 *    not a hard dependency on a GPU-aware MPI.
 */
#ifndef MPIX_INTERNAL_H
#define MPIX_INTERNAL_H

#include <atomic>
#include <chrono>
#include <cstdint>
#include <memory>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <thread>
#include <vector>

#include <mpi.h>

#include "mpix/mpix.h"
#include "mpix/mpix_abi.h"

/* --------------------------------------------- MPI-4.0 partitioned support
 * Reference parity with /root/reference/Makefile:17-20: when the host MPI
 * implements MPI-4.0 partitioned communication, MPIX_Psend/Precv_init can
 * delegate to MPI_Psend_init/MPI_Pready instead of mpix's own emulation
 * (opt-in at runtime with MPIX_MPI_PARTITIONED=1, MPI mode only).  The gate
 * is automatic on an MPI-4 library; -DMPIX_MPI_PARTITIONED force-compiles
 * the passthrough against declared prototypes on an MPI-3.x toolchain
 * (compile validation only — calling it there aborts at link/bind time). */
#if MPI_VERSION >= 4 || defined(MPIX_MPI_PARTITIONED)
#define MPIX_HAVE_MPI_PARTITIONED 1
#else
#define MPIX_HAVE_MPI_PARTITIONED 0
#endif
#if MPIX_HAVE_MPI_PARTITIONED && MPI_VERSION < 4
extern "C" {
int MPI_Psend_init(const void *buf, int partitions, MPI_Count count,
                   MPI_Datatype datatype, int dest, int tag, MPI_Comm comm,
                   MPI_Info info, MPI_Request *request);
int MPI_Precv_init(void *buf, int partitions, MPI_Count count,
                   MPI_Datatype datatype, int source, int tag, MPI_Comm comm,
                   MPI_Info info, MPI_Request *request);
int MPI_Pready(int partition, MPI_Request request);
int MPI_Parrived(MPI_Request request, int partition, int *flag);
}
#endif

namespace mpix {

/* ------------------------------------------------------------------ errors */

#define MPIX_ERR(...)                                                   \
    do {                                                                \
        fprintf(stderr, "[mpix] %s:%d %s: ", __FILE__, __LINE__, __func__); \
        fprintf(stderr, __VA_ARGS__);                                   \
        fprintf(stderr, "\n");                                          \
    } while (0)

#define MPIX_CHECK(call)                                                \
    do {                                                                \
        int _e = (call);                                                \
        if (_e != 0) { MPIX_ERR("%s failed (%d)", #call, _e); return _e; } \
    } while (0)

#define MPIX_CHECK_HIP(call)                                            \
    do {                                                                \
        hipError_t _e = (call);                                         \
        if (_e != hipSuccess) {                                         \
            MPIX_ERR("%s failed: %s", #call, hipGetErrorString(_e));    \
            return (int)_e;                                             \
        }                                                               \
    } while (0)

#ifdef MPIX_DEBUG
#define MPIX_DBG(...)                                                   \
    do {                                                                \
        fprintf(stderr, "[mpix dbg r%d] ", mpix::g_state ? mpix::g_state->world_rank : -1); \
        fprintf(stderr, __VA_ARGS__);                                   \
        fprintf(stderr, "\n");                                          \
    } while (0)
#else
#define MPIX_DBG(...) do {} while (0)
#endif

/* ------------------------------------------------------------- descriptors */

static inline uint64_t now_ns()
{
    return (uint64_t)std::chrono::duration_cast<std::chrono::nanoseconds>(
        std::chrono::steady_clock::now().time_since_epoch()).count();
}

struct Request;

/* Transport-level completion record (filled by the channel). */
struct ChStatus {
    int src  = -1;   /* world rank of the actual sender */
    int tag  = -1;
    uint64_t bytes = 0;
    int err  = 0;
};

enum class OpKind : int32_t { NONE = 0, ISEND, IRECV, PSEND_PART, PRECV_PART };

/* One in-flight operation, bound 1:1 to a flag slot. */
struct Op {
    OpKind kind = OpKind::NONE;
    void *buf = nullptr;
    uint64_t bytes = 0;
    int count = 0;
    MPI_Datatype datatype = MPI_DATATYPE_NULL;
    int peer = -1;        /* rank as passed by the user (comm-relative) */
    int peer_world = -1;  /* resolved world rank (MPI_ANY_SOURCE passes through) */
    int tag = 0;
    MPI_Comm comm = MPI_COMM_NULL;
    uint32_t comm_id = 0;
    bool buf_is_device = false;
    bool native_route = true;       /* native shm/xGMI channel vs MPI passthrough */
    int partition = -1;             /* >=0 for partitioned partial ops */
    uint32_t pseq = 0;              /* partitioned iteration number */
    Request *req = nullptr;         /* owning request */

    /* transport hand-off (written by channel, read by proxy) */
    uint64_t t_enq_ns = 0;          /* host enqueue time (MPIX_STATS=1) */
    uint64_t t_issue_ns = 0;        /* proxy-side stats (MPIX_STATS=1) */
    std::atomic<int> ch_done{0};
    ChStatus ch_status;
    void *ch_priv = nullptr;        /* channel-private per-op state */

    /* completion/status delivery (guarded by g_state->completion_mutex) */
    bool fast = false;              /* fast-wait protocol op */
    bool waiter_owns_req = false;   /* fast: a host waiter will free req */
    MPI_Status *enq_status_target = nullptr; /* posted by MPIX_Wait*_enqueue */
    MPI_Status saved_status;
    bool status_saved = false;
    /* user called MPIX_Request_free pre-completion.  Atomic: the proxy
     * pre-checks it outside the completion mutex (decisions that free the
     * slot are still made under the mutex). */
    std::atomic<bool> orphaned{false};

    void reset() {
        kind = OpKind::NONE; buf = nullptr; bytes = 0; count = 0;
        datatype = MPI_DATATYPE_NULL; peer = peer_world = -1; tag = 0;
        comm = MPI_COMM_NULL; comm_id = 0; buf_is_device = false;
        native_route = true;
        partition = -1; pseq = 0; req = nullptr;
        t_enq_ns = 0; t_issue_ns = 0;
        ch_done.store(0, std::memory_order_relaxed);
        ch_status = ChStatus{}; ch_priv = nullptr;
        fast = false; waiter_owns_req = false;
        enq_status_target = nullptr; status_saved = false;
        orphaned.store(false, std::memory_order_relaxed);
    }
};

enum class ReqKind : int32_t { BASIC = 0, PARTITIONED };

struct Request {
    ReqKind kind = ReqKind::BASIC;
    /* init-generation of the State this request belongs to.  hipUserObject
     * destructors for graph-owned requests can fire on a HIP-internal thread
     * at any later time — including after MPIX_Finalize or a following
     * MPIX_Init — so they must verify the state generation before touching
     * the flag pool (see graph_request_destroy). */
    uint64_t state_gen = 0;

    /* BASIC */
    int flag_idx = -1;
    /* fast-wait protocol (MPIX_FAST_WAIT=1): completion is a monotonic
     * per-slot sequence number in a second pinned word array, waited with
     * GTE — no CLEANUP write, slot freed by the proxy at completion.
     * The proxy stores the status here (the op is recycled immediately). */
    bool fast = false;
    uint32_t seq = 0;
    MPI_Status fast_status{};
    /* who frees this request (guarded by completion_mutex):
     * 0 = undecided, 1 = proxy at completion, 2 = a host waiter */
    int consume = 0;

    /* PARTITIONED */
    bool is_send = false;
    int n_partitions = 0;
    std::vector<int> part_idx;      /* partition -> flag slot */
    void *buf = nullptr;
    uint64_t part_bytes = 0;        /* bytes per partition */
    int peer = -1, peer_world = -1, tag = 0;
    MPI_Comm comm = MPI_COMM_NULL;
    uint32_t comm_id = 0;
    MPI_Datatype datatype = MPI_DATATYPE_NULL;
    int count_per_part = 0;
    bool active = false;            /* between Start and Wait */
    uint32_t start_seq = 0;         /* number of MPIX_Start calls */
    mpix_prequest_dev_t *dev_handle = nullptr; /* device copy (Prequest_create) */
    int32_t *dev_idx = nullptr;                /* device idx array */
    /* MPI-4.0 native partitioned passthrough (MPIX_HAVE_MPI_PARTITIONED +
     * MPIX_MPI_PARTITIONED=1): the underlying persistent MPI request.  The
     * per-partition flag protocol is unchanged; the proxy routes Pready /
     * Parrived-polls to MPI instead of moving bytes itself. */
    bool mpi_part_native = false;
    MPI_Request mpi_preq = MPI_REQUEST_NULL;
};

/* ----------------------------------------------------------------- channel */

class Transport {
public:
    virtual ~Transport() = default;
    /* Hand an ISEND/IRECV/PSEND_PART/PRECV_PART op to the data plane.
     * Never blocks; completion is signalled via op->ch_done / op->ch_status. */
    virtual int start(Op *op) = 0;
    /* Advance all in-flight traffic. Called from the proxy loop. */
    virtual void progress() = 0;
    virtual const char *name() const = 0;
};

/* --------------------------------------------------------------- MPSC ring */

/* Fixed-size MPSC ring for handing freshly armed slots to the proxy.
 * Producers: any app thread calling an enqueue/init API. Consumer: proxy. */
class SlotRing {
public:
    void init(size_t capacity) {
        cap_ = 1; while (cap_ < capacity) cap_ <<= 1;
        buf_.reset(new std::atomic<int64_t>[cap_]);
        for (size_t i = 0; i < cap_; i++) buf_[i].store(EMPTY, std::memory_order_relaxed);
        head_.store(0); tail_.store(0);
    }
    void push(int idx) {
        uint64_t pos = head_.fetch_add(1, std::memory_order_relaxed);
        /* wait for the slot to be consumed if we lapped (practically never:
         * capacity >= 4 * nflags > max outstanding) */
        auto &cell = buf_[pos & (cap_ - 1)];
        int64_t expected = EMPTY;
        while (!cell.compare_exchange_weak(expected, idx,
                                           std::memory_order_release,
                                           std::memory_order_relaxed)) {
            expected = EMPTY;
            std::this_thread::yield();
        }
    }
    /* proxy-only */
    bool pop(int *idx) {
        uint64_t t = tail_.load(std::memory_order_relaxed);
        auto &cell = buf_[t & (cap_ - 1)];
        int64_t v = cell.load(std::memory_order_acquire);
        if (v == EMPTY) return false;
        cell.store(EMPTY, std::memory_order_relaxed);
        tail_.store(t + 1, std::memory_order_relaxed);
        *idx = (int)v;
        return true;
    }
private:
    static constexpr int64_t EMPTY = -1;
    size_t cap_ = 0;
    std::unique_ptr<std::atomic<int64_t>[]> buf_;
    std::atomic<uint64_t> head_{0};
    std::atomic<uint64_t> tail_{0};
};

/* ------------------------------------------------------------------- state */

struct State {
    /* identity */
    int world_rank = 0;
    int world_size = 1;
    uint64_t gen = 0;           /* init-generation (see lifecycle_mutex) */
    bool mpi_mode = false;      /* MPI_Init was called by the app */
    /* gpu */
    bool have_gpu = false;
    int device_id = -1;
    bool use_memops = false;    /* hipStreamWriteValue32/WaitValue32 path */
    bool use_batch_memops = false;
    bool use_graph_memops = false; /* hipGraphAddBatchMemOpNode functional */
    bool use_capture_memops = false; /* memOps recordable under capture */
    /* sticky: a spin-WAIT kernel has been emitted (graph node, captured, or
     * stream fallback).  While false, no spinning kernel can exist on any
     * HSA queue, so the transport may run its pull copy as a compute
     * kernel; once true, copies ride SDMA (see pull_kernels_safe). */
    std::atomic<bool> spin_wait_kernels{false};
    /* flag pool */
    size_t nflags = 0;
    std::atomic<uint32_t> *flags = nullptr;  /* host view (pinned if GPU) */
    uint32_t *flags_d = nullptr;             /* device view or nullptr */
    bool fast_wait = false;                  /* MPIX_FAST_WAIT=1 */
    std::atomic<uint32_t> *seqs = nullptr;   /* fast: completion seq words */
    uint32_t *seqs_d = nullptr;
    uint32_t *slot_seq = nullptr;            /* fast: per-slot next seq */
    bool flags_pinned = false;
    bool seqs_pinned = false;
    Op *ops = nullptr;                       /* parallel op table */
    std::atomic<uint32_t> alloc_cursor{0};
    /* proxy */
    std::thread proxy;
    std::thread watchdog;            /* MPIX_WATCHDOG debug dumper */
    std::atomic<bool> proxy_stop{false};
    SlotRing armed;                          /* slots newly allocated */
    std::mutex completion_mutex;             /* status-delivery race guard */
    /* data plane */
    Transport *t_native = nullptr;   /* shm + xGMI/IPC intra-node channel */
    Transport *t_mpi = nullptr;      /* host-MPI passthrough (MPI mode only) */
    /* proxy tuning */
    int spin_before_yield = 2000;
    /* stats */
    std::atomic<uint64_t> ops_issued{0};
    std::atomic<uint64_t> ops_completed{0};
    std::atomic<uint64_t> proxy_passes{0};   /* heartbeat (watchdog) */
    /* issue->complete latency histogram, log2 us buckets 0..19 (>=0.5 ms
     * capped), proxy-thread-only (MPIX_STATS=1) */
    bool stats = false;
    uint64_t lat_hist[20] = {0};
    uint64_t lat_sum_ns = 0;
    /* per-leg sums (proxy thread only): enqueue->PENDING-detected (includes
     * the GPU-side trigger when the stream is idle), detect->transport-done,
     * done->completion-published */
    uint64_t leg_trig_ns = 0, leg_xfer_ns = 0, leg_compl_ns = 0;
    uint64_t leg_n = 0;
};

extern State *g_state;

/* Serializes MPIX_Init/Finalize state publication against asynchronously
 * scheduled hipUserObject destructors (graph-owned request cleanup).  A
 * leaked function-local static so it outlives any HIP runtime thread that
 * might still fire a destructor during process exit. */
std::mutex &lifecycle_mutex();

/* ------------------------------------------------------------- flag helpers */

static inline uint32_t flag_load(int idx) {
    return g_state->flags[idx].load(std::memory_order_acquire);
}
static inline void flag_store(int idx, uint32_t v) {
    g_state->flags[idx].store(v, std::memory_order_release);
}
static inline uint32_t seq_load(int idx) {
    return g_state->seqs[idx].load(std::memory_order_acquire);
}
static inline void seq_store(int idx, uint32_t v) {
    g_state->seqs[idx].store(v, std::memory_order_release);
}
static inline bool flag_cas(int idx, uint32_t expect, uint32_t v) {
    return g_state->flags[idx].compare_exchange_strong(
        expect, v, std::memory_order_acq_rel, std::memory_order_relaxed);
}

/* ------------------------------------------------------------------- slots */

/* Allocate a flag slot: AVAILABLE -> RESERVED (lock-free, multi-producer).
 * Returns slot index or -1 if the pool is exhausted. */
int slot_allocate();
/* Release a slot back to AVAILABLE (resets the op descriptor first). */
void slot_free(int idx);
/* Arm a slot: make the proxy watch it (call after the op is fully described). */
void slot_arm(int idx);

/* ------------------------------------------------------------------- utils */

/* Size in bytes of an MPI datatype; works without MPI_Init for MPICH builtin
 * handles (size lives in bits 8..15 of the handle). */
int datatype_size(MPI_Datatype dt, int *size_out);

/* Resolve a comm-relative rank to a world rank; assigns/validates comm_id.
 * comm_id 0 = WORLD, 1 = SELF; other comms are MPI-mode passthrough only. */
int resolve_peer(MPI_Comm comm, int rank, int *world_rank_out,
                 uint32_t *comm_id_out, bool *native_ok_out);

/* Write an MPI_Status (MPICH layout) from a transport completion. */
void fill_status(MPI_Status *st, const ChStatus &cs);

/* Is `ptr` device memory? (false when no GPU) */
bool ptr_is_device(const void *ptr);

/* proxy entry point (init.cpp spawns it) */
void proxy_main();

/* enqueue.cpp internals shared with init */
int trigger_host(int idx);   /* host-side trigger: flag -> PENDING */

/* transport factories */
Transport *make_native_transport(int world_rank, int world_size, bool mpi_mode,
                                 bool have_gpu, int device_id);
Transport *make_mpi_transport();

/* native transport finalize hook (unlinks shm) */
void native_transport_shutdown(Transport *t);

} /* namespace mpix */

#endif /* MPIX_INTERNAL_H */
