/* mpix — native intra-node data plane: shared-memory control + xGMI data.
 *
 * The reference delegates all data movement to a CUDA-aware host MPI
 * (SURVEY.md §2b).  On an MI355X node the fast path between GPUs is direct
 * peer-to-peer over xGMI, so this transport implements tag-matched
 * point-to-point natively:
 *
 *  control plane  per ordered rank pair (s -> r), an SPSC descriptor ring in
 *                 a POSIX shm segment owned by r, plus a chunked staging
 *                 buffer for host payloads.
 *  device data    sender publishes {hipIpcMemHandle, offset}; the RECEIVER
 *                 pulls with hipMemcpyAsync over xGMI (SDMA) on a private
 *                 stream and acks with a DONE descriptor.  Same-process /
 *                 same-rank transfers skip IPC and use the raw pointer.
 *  host data      sender stages through the shm chunk ring (64 KiB chunks,
 *                 credit-based flow control); the receiver copies chunks
 *                 straight into the posted buffer (or an unexpected-message
 *                 heap buffer).  Send completes when fully staged (buffered-
 *                 send semantics), so no ack round-trip.
 *
 * Matching: (comm_id, src rank, tag, partitioned?, partition) with
 * MPI_ANY_SOURCE / MPI_ANY_TAG wildcards; descriptor rings are FIFO per
 * sender, posted receives match in post order — this preserves MPI
 * non-overtaking per (src, tag, comm) pair, which the reference explicitly
 * does not (README.md:173-176).
 *
 * Single-threaded by construction: only the proxy thread touches this object
 * after construction.
 */
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <hip/hip_runtime.h>

#include <deque>
#include <list>
#include <map>
#include <string>
#include <unordered_map>
#include <vector>

#include "../internal.h"
#include "bootstrap.h"

namespace mpix {

/* Batched pull: one launch copies every partition/message that became
 * ready in the same progress pass (a 64 x 4 MiB partitioned step was
 * latency-bound at ~9 us per serialized launch+event).  All blocks stripe
 * over each copy in turn — balanced regardless of size mix.  Args live in
 * pinned mapped memory (an 8-slot ring reused after the batch's event). */
#define MPIX_PULL_BATCH 64
struct PullArg {
    void *dst;
    const void *src;
    size_t n16;   /* 16-byte vectors */
    size_t tail;  /* leftover bytes after n16*16 */
};

__global__ void k_pull_multi(const PullArg *__restrict__ args, int ncopies)
{
    size_t gid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (int c = 0; c < ncopies; c++) {
        const uint4 *s4 = (const uint4 *)args[c].src;
        uint4 *d4 = (uint4 *)args[c].dst;
        size_t nv = args[c].n16;
        for (size_t v = gid; v < nv; v += stride) d4[v] = s4[v];
        if (gid == 0)
            for (size_t b = 0; b < args[c].tail; b++)
                ((char *)args[c].dst)[nv * 16 + b] =
                    ((const char *)args[c].src)[nv * 16 + b];
    }
}

/* Sender-push threshold for small DEVICE payloads: stage D2H into the shm
 * chunk ring like a host message (receiver H2D's it out), skipping the
 * NOTIFY/pull/ACK round trip.  Measured WORSE (74-97 us vs ~38 us half-RTT:
 * the two proxy-synchronous staging copies cost more than the round trip
 * they remove — profiles/r02_pingpong_devpush.json); kept off, tunable for
 * experiments with MPIX_DEV_PUSH_MAX=<bytes>. */
static bool env_flag(const char *name)
{
    const char *e = getenv(name);
    return e && atoi(e);
}

static uint64_t dev_push_max()
{
    static const uint64_t v = [] {
        const char *e = getenv("MPIX_DEV_PUSH_MAX");
        return e ? (uint64_t)atoll(e) : (uint64_t)0;
    }();
    return v;
}

/* ------------------------------------------------------------- shm layout */

static constexpr uint32_t RING_SLOTS_DEFAULT = 1024;
static constexpr uint64_t CHUNK_BYTES = 64 * 1024;
static constexpr uint32_t STAGE_CHUNKS_DEFAULT = 64; /* 4 MiB per pair */

enum DescType : uint32_t {
    DESC_HOST_CHUNK = 1, /* one staged chunk of a host-buffer message */
    DESC_DEV_NOTIFY = 2, /* device-buffer message advertisement */
    DESC_DONE       = 3, /* receiver ack for DESC_DEV_NOTIFY */
};

enum DescFlags : uint32_t {
    DESCF_PARTITIONED = 1u << 0,
    DESCF_SELF_PTR    = 1u << 1, /* ipc field holds a raw pointer (same proc) */
};

struct alignas(64) Desc {
    uint32_t type = 0;
    uint32_t flags = 0;
    uint64_t token = 0;       /* sender-side slot index */
    int32_t src_world = -1;
    int32_t tag = 0;
    uint32_t comm_id = 0;
    int32_t partition = -1;
    uint64_t msg_bytes = 0;
    uint64_t chunk_off = 0;   /* HOST_CHUNK: offset of this chunk in the msg */
    uint64_t chunk_bytes = 0;
    uint64_t stage_chunk = 0; /* HOST_CHUNK: chunk index in the stage ring */
    uint8_t ipc[64] = {};     /* DEV_NOTIFY: hipIpcMemHandle_t (or raw ptr) */
    uint64_t ipc_off = 0;
    int32_t src_dev = -1;
    uint32_t _pad[9] = {};
};
static_assert(sizeof(Desc) == 192, "Desc layout");

struct alignas(64) InboxHdr {
    alignas(64) std::atomic<uint64_t> head;       /* producer (peer proxy) */
    alignas(64) std::atomic<uint64_t> tail;       /* consumer (my proxy) */
    alignas(64) std::atomic<uint64_t> stage_tail; /* chunks released by me */
};

struct ShmGeom {
    uint32_t ring_slots;
    uint32_t stage_chunks;
    size_t inbox_bytes;   /* hdr + ring + stage, 64-aligned */
    size_t segment_bytes; /* nranks inboxes */
};

static ShmGeom shm_geometry(int nranks)
{
    ShmGeom g;
    g.ring_slots = RING_SLOTS_DEFAULT;
    if (const char *p = getenv("MPIX_SHM_RING")) g.ring_slots = atoi(p);
    g.stage_chunks = STAGE_CHUNKS_DEFAULT;
    if (const char *p = getenv("MPIX_SHM_STAGE_CHUNKS")) g.stage_chunks = atoi(p);
    size_t b = sizeof(InboxHdr) + (size_t)g.ring_slots * sizeof(Desc) +
               (size_t)g.stage_chunks * CHUNK_BYTES;
    g.inbox_bytes = (b + 63) & ~(size_t)63;
    g.segment_bytes = g.inbox_bytes * (size_t)nranks;
    return g;
}

/* views into one inbox */
struct InboxView {
    InboxHdr *hdr;
    Desc *ring;
    char *stage;
};

static InboxView inbox_view(void *seg_base, const ShmGeom &g, int src)
{
    char *p = (char *)seg_base + (size_t)src * g.inbox_bytes;
    InboxView v;
    v.hdr = (InboxHdr *)p;
    v.ring = (Desc *)(p + sizeof(InboxHdr));
    v.stage = (char *)(v.ring + g.ring_slots);
    return v;
}

/* -------------------------------------------------------------- transport */

class NativeTransport : public Transport {
public:
    NativeTransport(int rank, int size, bool mpi_mode, bool have_gpu, int dev)
        : rank_(rank), size_(size), have_gpu_(have_gpu), dev_(dev),
          mpi_mode_(mpi_mode) {}

    int init();
    void shutdown(); /* called from MPIX_Finalize before dtor */
    ~NativeTransport() override;

    int start(Op *op) override;
    void progress() override;
    const char *name() const override { return "native-shm-xgmi"; }

private:
    /* ---- send side ---- */
    struct SendState {
        Op *op;
        uint64_t staged = 0;   /* bytes staged so far (host path) */
        bool notified = false; /* DEV_NOTIFY emitted */
    };
    /* per-destination FIFO queues (preserves non-overtaking) */
    std::map<int, std::deque<SendState>> sendq_;
    /* device sends awaiting DONE: token -> op */
    std::unordered_map<uint64_t, Op *> await_done_;
    /* pending DONE descriptors we could not emit (ring full) */
    std::deque<std::pair<int, uint64_t>> pending_done_;

    /* ---- recv side ---- */
    struct InboundMsg {
        Desc d;                 /* first descriptor (header info) */
        int src;                /* sending world rank */
        Op *op = nullptr;       /* matched recv, or null */
        std::vector<char> heap; /* unexpected host payload buffer */
        uint64_t got = 0;       /* host bytes received so far */
        bool dev_copy_started = false;
    };
    std::list<InboundMsg> inbound_;                 /* arrival order */
    std::unordered_map<uint64_t, InboundMsg *> inbound_by_token_;
    std::vector<Op *> posted_recvs_;                /* post order */

    struct CopyInflight {
        Op *op;
        hipEvent_t ev;
        int src;
        uint64_t token;
        ChStatus st;
    };
    std::list<CopyInflight> copies_;

    /* pulls collected this progress pass, flushed as ONE kernel launch */
    struct PendingPull {
        Op *op;
        int src;
        uint64_t token;
        ChStatus st;
        void *dst;
        const void *csrc;
        uint64_t n;
    };
    std::vector<PendingPull> pend_pulls_;
    struct PullBatch {
        hipEvent_t ev;
        int slot;
        std::vector<PendingPull> items;
    };
    std::list<PullBatch> batches_;
    /* pinned arg ring; per-slot busy flags because batches on different
     * streams (MPIX_COPY_PRIO_STREAM=1) can complete out of order */
    static constexpr int ARG_SLOTS = 8;
    PullArg *args_h_ = nullptr;   /* pinned, ARG_SLOTS * MPIX_PULL_BATCH */
    PullArg *args_d_ = nullptr;
    int arg_slot_ = 0;
    bool arg_busy_[ARG_SLOTS] = {};

    /* ---- shm ---- */
    ShmGeom geom_{};
    std::string my_seg_name_;
    std::vector<void *> seg_;       /* seg_[r] = rank r's segment base */
    std::vector<uint64_t> out_head_; /* producer-local head per dst */
    std::vector<uint64_t> out_stage_head_; /* producer-local chunk cursor */
    std::vector<uint64_t> in_tail_;  /* consumer-local tail per src */

    /* ---- hip ---- */
    /* Copy-stream POOL.  Stream 0 is PLAIN and carries small pulls plus
     * synchronous staging (lowest latency: a priority queue costs ~17 us
     * extra half-RTT, ci_full2); streams 1..3 are created at GREATEST
     * PRIORITY and carry large pulls concurrently — large partitioned
     * transfers were latency-bound at ~9 us/partition serialized on one
     * stream.  The high-priority set can never share a hardware queue
     * with user streams (different priority = different queue), which
     * matters beyond spin kernels: an unsatisfied hipStreamWaitValue32 is
     * a queue-parking packet, so a SAME-priority copy stream that aliases
     * a user stream's queue deadlocks exactly like the graph spin-kernel
     * case (observed when a 4-plain-stream pool pushed total streams past
     * the 4-queue pool: every transfer hung).  Stream 0 carries the same
     * small residual aliasing risk as the original single-stream design
     * (mitigated by the lazy priority upgrade below). */
    static constexpr int N_COPY_STREAMS = 4;
    hipStream_t copy_streams_[N_COPY_STREAMS] = {};
    hipStream_t copy_plain0_ = nullptr; /* pre-upgrade stream 0 (kept) */
    bool prio_switched_ = false;
    static uint64_t pool_min_bytes() {
        static const uint64_t v = [] {
            const char *e = getenv("MPIX_COPY_POOL_MIN");
            return e ? (uint64_t)atoll(e) : (uint64_t)(256 << 10);
        }();
        return v;
    }

    /* The copy stream must never share a hardware queue with a stream that
     * holds a SPIN-WAIT kernel: HIP muxes same-priority streams onto a
     * small HSA queue pool, and GRAPH execution orders its queue with AQL
     * barrier packets — a k_wait_flag graph node parked in front of our
     * pull/blit packet blocks it forever, while that wait needs this very
     * copy to finish (deterministic after stream churn rotated the
     * mapping: gpurun_out/diag5_loop5.log).  Streams of different
     * priorities never share a queue, but an always-priority copy stream
     * costs ~17 us extra half-RTT with two processes on one GPU
     * (gpurun_out/ci_full2: 54 us vs 37.9), so the switch is LAZY: plain
     * stream until the library emits its first spin-wait kernel
     * (mark_spin_wait in enqueue.cpp — ordered before any graph launch
     * that could contain one), then migrate to a greatest-priority stream.
     * In-flight copies on the plain stream finish normally (events). */
    void maybe_switch_prio() {
        if (prio_switched_ || g_state == nullptr ||
            !g_state->spin_wait_kernels.load(std::memory_order_acquire))
            return;
        prio_switched_ = true;
        int lo = 0, hi = 0;
        hipStream_t ps = nullptr;
        if (hipDeviceGetStreamPriorityRange(&lo, &hi) == hipSuccess &&
            hi != lo &&
            hipStreamCreateWithPriority(&ps, hipStreamNonBlocking, hi) ==
                hipSuccess) {
            copy_plain0_ = copy_streams_[0];
            copy_streams_[0] = ps;
        } else {
            (void)hipGetLastError();
        }
    }
    /* stream for an independent pull: small payloads ride the plain
     * stream 0 (latency), large ones rotate over the priority set 1..3
     * (bandwidth; per-message ordering is by the event recorded on the
     * same stream as the copy) */
    hipStream_t copy_stream(uint64_t bytes) {
        maybe_switch_prio();
        if (bytes <= pool_min_bytes() || copy_streams_[1] == nullptr)
            return copy_streams_[0];
        return copy_streams_[1];
    }
    /* fixed stream for synchronous staging (memcpy_auto) */
    hipStream_t copy_stream0() {
        maybe_switch_prio();
        return copy_streams_[0];
    }
    std::vector<hipEvent_t> event_pool_;
    std::unordered_map<std::string, void *> ipc_open_;   /* handle -> ptr */
    std::unordered_map<const void *, std::pair<void *, hipIpcMemHandle_t>>
        ipc_get_;                                        /* buf base -> handle */

    Bootstrap *boot_ = nullptr;

    int rank_, size_;
    bool have_gpu_;
    int dev_;
    bool mpi_mode_;
    bool shut_ = false;

    /* helpers */
    InboxView my_inbox(int src) { return inbox_view(seg_[rank_], geom_, src); }
    InboxView out_box(int dst) { return inbox_view(seg_[dst], geom_, rank_); }

    bool ring_has_space(int dst, uint64_t need) {
        InboxView v = out_box(dst);
        uint64_t tail = v.hdr->tail.load(std::memory_order_acquire);
        return out_head_[dst] + need - tail <= geom_.ring_slots;
    }
    void emit_desc(int dst, const Desc &d) {
        InboxView v = out_box(dst);
        uint64_t h = out_head_[dst];
        v.ring[h % geom_.ring_slots] = d;
        v.hdr->head.store(h + 1, std::memory_order_release);
        out_head_[dst] = h + 1;
    }
    uint64_t stage_free_chunks(int dst) {
        InboxView v = out_box(dst);
        uint64_t st = v.hdr->stage_tail.load(std::memory_order_acquire);
        return geom_.stage_chunks - (out_stage_head_[dst] - st);
    }

    hipEvent_t get_event();
    void put_event(hipEvent_t ev);

    int progress_sends();
    int drain_inbox(int src);
    int progress_copies();
    void flush_pulls();
    void handle_desc(int src, const Desc &d, const char *stage_base);
    void try_match_new_inbound(InboundMsg &m);
    void attach(InboundMsg &m, Op *op);
    void start_dev_copy(InboundMsg &m);
    void finish_host_recv(InboundMsg &m);
    void deliver_chunk(InboundMsg &m, const Desc &d, const char *src_chunk);
    void erase_inbound(InboundMsg *m);
    int memcpy_auto(void *dst, const void *src, size_t n);
    void *map_remote(const Desc &d);
    bool match(const Op *op, const Desc &d, int src) const;
    void complete_send_buffered(Op *op);
};

/* ------------------------------------------------------------------- init */

int NativeTransport::init()
{
    boot_ = make_bootstrap(rank_, size_, mpi_mode_);
    if (!boot_) return -1;

    geom_ = shm_geometry(size_);

    /* create my segment */
    char name[96];
    unsigned seed = (unsigned)getpid() * 2654435761u + (unsigned)rank_;
    snprintf(name, sizeof(name), "/mpix-r%d-%d-%x", rank_, (int)getpid(),
             rand_r(&seed));
    my_seg_name_ = name;
    int fd = shm_open(name, O_CREAT | O_EXCL | O_RDWR, 0600);
    if (fd < 0) {
        MPIX_ERR("shm_open(%s) failed", name);
        return -1;
    }
    if (ftruncate(fd, (off_t)geom_.segment_bytes) != 0) {
        MPIX_ERR("ftruncate(%zu) failed", geom_.segment_bytes);
        close(fd);
        return -1;
    }
    void *base = mmap(nullptr, geom_.segment_bytes, PROT_READ | PROT_WRITE,
                      MAP_SHARED, fd, 0);
    close(fd);
    if (base == MAP_FAILED) {
        MPIX_ERR("mmap own segment failed");
        return -1;
    }
    memset(base, 0, geom_.segment_bytes);

    /* exchange names, open peers */
    struct Blob { char name[96]; };
    Blob mine{};
    snprintf(mine.name, sizeof(mine.name), "%s", name);
    std::vector<Blob> all(size_);
    if (boot_->allgather(&mine, all.data(), sizeof(Blob)) != 0) {
        MPIX_ERR("bootstrap allgather failed");
        return -1;
    }
    seg_.assign(size_, nullptr);
    seg_[rank_] = base;
    for (int r = 0; r < size_; r++) {
        if (r == rank_) continue;
        int pfd = shm_open(all[r].name, O_RDWR, 0600);
        if (pfd < 0) {
            MPIX_ERR("shm_open(peer %s) failed", all[r].name);
            return -1;
        }
        void *pb = mmap(nullptr, geom_.segment_bytes, PROT_READ | PROT_WRITE,
                        MAP_SHARED, pfd, 0);
        close(pfd);
        if (pb == MAP_FAILED) {
            MPIX_ERR("mmap peer segment failed");
            return -1;
        }
        seg_[r] = pb;
    }
    /* everyone mapped everything -> segments can be unlinked */
    boot_->barrier();
    shm_unlink(my_seg_name_.c_str());

    out_head_.assign(size_, 0);
    out_stage_head_.assign(size_, 0);
    in_tail_.assign(size_, 0);

    if (have_gpu_) {
        if (hipStreamCreateWithFlags(&copy_streams_[0],
                                     hipStreamNonBlocking) != hipSuccess) {
            MPIX_ERR("copy stream create failed");
            return -1;
        }
        /* NO standing second stream: every additional hardware queue
         * (even per another process) oversubscribes the device's HW
         * scheduler and adds tens of us to every wait — a standing
         * priority stream alone pushed the 2-process 8-B half-RTT from
         * 36.5 to ~55 us.  Batched pulls serialize on stream 0 at HBM
         * rate, so extra streams buy nothing; MPIX_COPY_PRIO_STREAM=1
         * opts a standing priority stream back in for experiments. */
        if (env_flag("MPIX_COPY_PRIO_STREAM")) {
            int lo = 0, hi = 0;
            if (hipDeviceGetStreamPriorityRange(&lo, &hi) == hipSuccess &&
                hi != lo) {
                if (hipStreamCreateWithPriority(&copy_streams_[1],
                                                hipStreamNonBlocking, hi) !=
                    hipSuccess) {
                    (void)hipGetLastError();
                    copy_streams_[1] = nullptr;
                }
            } else {
                (void)hipGetLastError();
            }
        }
        if (hipHostMalloc((void **)&args_h_,
                          (size_t)ARG_SLOTS * MPIX_PULL_BATCH *
                              sizeof(PullArg),
                          hipHostMallocMapped) != hipSuccess ||
            hipHostGetDevicePointer((void **)&args_d_, args_h_, 0) !=
                hipSuccess) {
            MPIX_ERR("pull-arg ring alloc failed");
            return -1;
        }
    }
    return 0;
}

void NativeTransport::shutdown()
{
    if (shut_) return;
    shut_ = true;
    /* make sure no peer is still reading our segment */
    if (boot_) boot_->barrier();
    for (auto &kv : ipc_open_) (void)hipIpcCloseMemHandle(kv.second);
    ipc_open_.clear();
    for (hipEvent_t ev : event_pool_) (void)hipEventDestroy(ev);
    event_pool_.clear();
    for (int i = 0; i < N_COPY_STREAMS; i++) {
        if (copy_streams_[i]) (void)hipStreamDestroy(copy_streams_[i]);
        copy_streams_[i] = nullptr;
    }
    if (copy_plain0_) (void)hipStreamDestroy(copy_plain0_);
    copy_plain0_ = nullptr;
    if (args_h_) (void)hipHostFree(args_h_);
    args_h_ = nullptr;
    for (int r = 0; r < (int)seg_.size(); r++)
        if (seg_[r]) munmap(seg_[r], geom_.segment_bytes);
    seg_.clear();
    delete boot_;
    boot_ = nullptr;
}

NativeTransport::~NativeTransport() { shutdown(); }

/* ------------------------------------------------------------------ events */

hipEvent_t NativeTransport::get_event()
{
    if (!event_pool_.empty()) {
        hipEvent_t ev = event_pool_.back();
        event_pool_.pop_back();
        return ev;
    }
    hipEvent_t ev = nullptr;
    (void)hipEventCreateWithFlags(&ev, hipEventDisableTiming);
    return ev;
}

void NativeTransport::put_event(hipEvent_t ev) { event_pool_.push_back(ev); }

/* ------------------------------------------------------------------- start */

int NativeTransport::start(Op *op)
{
    switch (op->kind) {
    case OpKind::ISEND:
    case OpKind::PSEND_PART:
        sendq_[op->peer_world].push_back(SendState{op});
        return 0;
    case OpKind::IRECV:
    case OpKind::PRECV_PART: {
        /* match against already-arrived messages first (FIFO arrival order) */
        for (auto &m : inbound_) {
            if (m.op == nullptr && match(op, m.d, m.src)) {
                attach(m, op);
                return 0;
            }
        }
        posted_recvs_.push_back(op);
        return 0;
    }
    default:
        return -1;
    }
}

bool NativeTransport::match(const Op *op, const Desc &d, int src) const
{
    if (op->comm_id != d.comm_id) return false;
    bool want_part = (op->kind == OpKind::PRECV_PART);
    bool is_part = (d.flags & DESCF_PARTITIONED) != 0;
    if (want_part != is_part) return false;
    if (want_part && op->partition != d.partition) return false;
    if (op->peer_world != MPI_ANY_SOURCE && op->peer_world != src) return false;
    if (op->tag != MPI_ANY_TAG && op->tag != d.tag) return false;
    return true;
}

/* ---------------------------------------------------------------- progress */

void NativeTransport::progress()
{
    progress_sends();
    for (int s = 0; s < size_; s++) drain_inbox(s);
    flush_pulls();
    progress_copies();

    /* retry queued DONE acks */
    for (size_t i = 0; i < pending_done_.size();) {
        auto [dst, token] = pending_done_[i];
        if (ring_has_space(dst, 1)) {
            Desc d;
            d.type = DESC_DONE;
            d.token = token;
            d.src_world = rank_;
            emit_desc(dst, d);
            pending_done_.erase(pending_done_.begin() + i);
        } else {
            i++;
        }
    }
}

void NativeTransport::complete_send_buffered(Op *op)
{
    op->ch_status.src = (op->comm_id == 1) ? 0 : rank_;
    op->ch_status.tag = op->tag;
    op->ch_status.bytes = op->bytes;
    op->ch_status.err = MPI_SUCCESS;
    op->ch_done.store(1, std::memory_order_release);
}

int NativeTransport::progress_sends()
{
    for (auto &kv : sendq_) {
        int dst = kv.first;
        auto &q = kv.second;
        while (!q.empty()) {
            SendState &ss = q.front();
            Op *op = ss.op;
            if (op->buf_is_device && op->bytes > dev_push_max()) {
                if (!ring_has_space(dst, 1)) break;
                Desc d;
                d.type = DESC_DEV_NOTIFY;
                d.flags = (op->kind == OpKind::PSEND_PART) ? DESCF_PARTITIONED : 0;
                d.token = (uint64_t)(op - g_state->ops);
                d.src_world = rank_;
                d.tag = op->tag;
                d.comm_id = op->comm_id;
                d.partition = op->partition;
                d.msg_bytes = op->bytes;
                d.src_dev = dev_;
                if (dst == rank_) {
                    d.flags |= DESCF_SELF_PTR;
                    uint64_t p = (uint64_t)(uintptr_t)op->buf;
                    memcpy(d.ipc, &p, 8);
                    d.ipc_off = 0;
                } else {
                    /* IPC handle of the allocation base + offset */
                    void *base = nullptr;
                    hipError_t e = hipPointerGetAttribute(
                        &base, HIP_POINTER_ATTRIBUTE_RANGE_START_ADDR,
                        (hipDeviceptr_t)op->buf);
                    if (e != hipSuccess || base == nullptr) base = op->buf;
                    auto it = ipc_get_.find(base);
                    if (it == ipc_get_.end()) {
                        hipIpcMemHandle_t h;
                        if (hipIpcGetMemHandle(&h, base) != hipSuccess) {
                            MPIX_ERR("hipIpcGetMemHandle failed (buf %p)", base);
                            op->ch_status.err = MPI_ERR_OTHER;
                            op->ch_done.store(1, std::memory_order_release);
                            q.pop_front();
                            continue;
                        }
                        it = ipc_get_.emplace(base, std::make_pair(base, h)).first;
                    }
                    memcpy(d.ipc, &it->second.second, sizeof(hipIpcMemHandle_t));
                    d.ipc_off = (uint64_t)((char *)op->buf - (char *)base);
                }
                emit_desc(dst, d);
                await_done_[d.token] = op;
                q.pop_front();
                continue;
            }
            /* host-staged path */
            bool stalled = false;
            while (ss.staged < op->bytes || op->bytes == 0) {
                if (!ring_has_space(dst, 1) || stage_free_chunks(dst) == 0) {
                    stalled = true;
                    break;
                }
                uint64_t chunk_idx = out_stage_head_[dst] % geom_.stage_chunks;
                uint64_t n = op->bytes - ss.staged;
                if (n > CHUNK_BYTES) n = CHUNK_BYTES;
                InboxView v = out_box(dst);
                if (n > 0) {
                    char *stage_dst = v.stage + chunk_idx * CHUNK_BYTES;
                    const char *payload = (const char *)op->buf + ss.staged;
                    if (op->buf_is_device)
                        memcpy_auto(stage_dst, payload, n); /* D2H */
                    else
                        memcpy(stage_dst, payload, n);
                }
                Desc d;
                d.type = DESC_HOST_CHUNK;
                d.flags = (op->kind == OpKind::PSEND_PART) ? DESCF_PARTITIONED : 0;
                d.token = (uint64_t)(op - g_state->ops);
                d.src_world = rank_;
                d.tag = op->tag;
                d.comm_id = op->comm_id;
                d.partition = op->partition;
                d.msg_bytes = op->bytes;
                d.chunk_off = ss.staged;
                d.chunk_bytes = n;
                d.stage_chunk = chunk_idx;
                out_stage_head_[dst]++;
                emit_desc(dst, d);
                ss.staged += n;
                if (op->bytes == 0) break; /* zero-byte message: one desc */
            }
            if (stalled) break; /* preserve FIFO: don't start next send */
            complete_send_buffered(op);
            q.pop_front();
        }
    }
    return 0;
}

int NativeTransport::drain_inbox(int src)
{
    InboxView v = my_inbox(src);
    uint64_t head = v.hdr->head.load(std::memory_order_acquire);
    uint64_t tail = in_tail_[src];
    int budget = 64;
    while (tail < head && budget-- > 0) {
        const Desc &d = v.ring[tail % geom_.ring_slots];
        handle_desc(src, d, v.stage);
        tail++;
        v.hdr->tail.store(tail, std::memory_order_release);
        in_tail_[src] = tail;
    }
    return 0;
}

void NativeTransport::handle_desc(int src, const Desc &d, const char *stage_base)
{
    switch (d.type) {
    case DESC_DONE: {
        auto it = await_done_.find(d.token);
        if (it == await_done_.end()) {
            MPIX_ERR("DONE for unknown token %lu", (unsigned long)d.token);
            return;
        }
        Op *op = it->second;
        await_done_.erase(it);
        complete_send_buffered(op);
        return;
    }
    case DESC_DEV_NOTIFY: {
        inbound_.emplace_back();
        InboundMsg &m = inbound_.back();
        m.d = d;
        m.src = src;
        inbound_by_token_[((uint64_t)src << 32) ^ d.token] = &m;
        try_match_new_inbound(m);
        return;
    }
    case DESC_HOST_CHUNK: {
        uint64_t key = ((uint64_t)src << 32) ^ d.token;
        InboundMsg *m = nullptr;
        if (d.chunk_off == 0) {
            inbound_.emplace_back();
            m = &inbound_.back();
            m->d = d;
            m->src = src;
            inbound_by_token_[key] = m; /* overwrite OK: FIFO => the previous
                                           msg with this token needs no more
                                           chunks (see erase_inbound) */
            try_match_new_inbound(*m);
        } else {
            auto it = inbound_by_token_.find(key);
            if (it != inbound_by_token_.end()) m = it->second;
        }
        if (m != nullptr)
            deliver_chunk(*m, d, stage_base + d.stage_chunk * CHUNK_BYTES);
        else
            MPIX_ERR("chunk for unknown msg token %lu", (unsigned long)d.token);
        /* release the stage credit unconditionally (no credit leaks) */
        InboxView v = my_inbox(src);
        v.hdr->stage_tail.fetch_add(1, std::memory_order_release);
        if (m != nullptr && m->got >= m->d.msg_bytes) finish_host_recv(*m);
        return;
    }
    default:
        MPIX_ERR("bad descriptor type %u from %d", d.type, src);
    }
}

void NativeTransport::try_match_new_inbound(InboundMsg &m)
{
    for (size_t i = 0; i < posted_recvs_.size(); i++) {
        Op *op = posted_recvs_[i];
        if (match(op, m.d, m.src)) {
            posted_recvs_.erase(posted_recvs_.begin() + i);
            attach(m, op);
            return;
        }
    }
    /* unexpected: host messages buffer into heap as chunks arrive */
    if (m.d.type == DESC_HOST_CHUNK) m.heap.resize(m.d.msg_bytes);
}

void NativeTransport::attach(InboundMsg &m, Op *op)
{
    op->ch_priv = &m;
    m.op = op;
    if (m.d.type == DESC_DEV_NOTIFY) {
        start_dev_copy(m);
        return;
    }
    /* host message: move already-buffered bytes into the user buffer */
    if (m.got > 0 && !m.heap.empty()) {
        uint64_t n = m.got;
        if (n > op->bytes) n = op->bytes;
        if (n > 0) memcpy_auto(op->buf, m.heap.data(), n);
    }
    m.heap.clear();
    m.heap.shrink_to_fit();
    if (m.got >= m.d.msg_bytes) finish_host_recv(m);
}

void NativeTransport::deliver_chunk(InboundMsg &m, const Desc &d,
                                    const char *src_chunk)
{
    if (m.op != nullptr) {
        uint64_t room = (m.op->bytes > d.chunk_off)
                            ? m.op->bytes - d.chunk_off : 0;
        uint64_t n = d.chunk_bytes < room ? d.chunk_bytes : room;
        if (n > 0)
            memcpy_auto((char *)m.op->buf + d.chunk_off, src_chunk, n);
    } else {
        if (m.heap.size() < d.chunk_off + d.chunk_bytes)
            m.heap.resize(m.d.msg_bytes);
        if (d.chunk_bytes > 0)
            memcpy(m.heap.data() + d.chunk_off, src_chunk, d.chunk_bytes);
    }
    m.got += d.chunk_bytes;
    if (m.d.msg_bytes == 0) m.got = 0; /* zero-byte msg: single empty chunk */
}

void NativeTransport::finish_host_recv(InboundMsg &m)
{
    if (m.op == nullptr) return; /* stays buffered until a recv posts */
    Op *op = m.op;
    op->ch_status.src = (op->comm_id == 1) ? 0 : m.src;
    op->ch_status.tag = m.d.tag;
    op->ch_status.bytes =
        m.d.msg_bytes <= op->bytes ? m.d.msg_bytes : op->bytes;
    op->ch_status.err =
        m.d.msg_bytes > op->bytes ? MPI_ERR_TRUNCATE : MPI_SUCCESS;
    op->ch_done.store(1, std::memory_order_release);
    erase_inbound(&m);
}

void *NativeTransport::map_remote(const Desc &d)
{
    if (d.flags & DESCF_SELF_PTR) {
        uint64_t p;
        memcpy(&p, d.ipc, 8);
        return (void *)(uintptr_t)p;
    }
    std::string key((const char *)d.ipc, sizeof(hipIpcMemHandle_t));
    auto it = ipc_open_.find(key);
    void *base = nullptr;
    if (it != ipc_open_.end()) {
        base = it->second;
    } else {
        hipIpcMemHandle_t h;
        memcpy(&h, d.ipc, sizeof(h));
        if (hipIpcOpenMemHandle(&base, h, hipIpcMemLazyEnablePeerAccess) !=
            hipSuccess) {
            MPIX_ERR("hipIpcOpenMemHandle failed (src %d)", d.src_world);
            return nullptr;
        }
        ipc_open_.emplace(std::move(key), base);
    }
    return (char *)base + d.ipc_off;
}

void NativeTransport::start_dev_copy(InboundMsg &m)
{
    Op *op = m.op;
    void *src = map_remote(m.d);
    ChStatus st;
    st.src = (op->comm_id == 1) ? 0 : m.src;
    st.tag = m.d.tag;
    st.bytes = m.d.msg_bytes <= op->bytes ? m.d.msg_bytes : op->bytes;
    st.err = m.d.msg_bytes > op->bytes ? MPI_ERR_TRUNCATE : MPI_SUCCESS;
    if (src == nullptr) {
        st.err = MPI_ERR_OTHER;
        op->ch_status = st;
        op->ch_done.store(1, std::memory_order_release);
        /* still ack so the sender does not hang */
        pending_done_.emplace_back(m.src, m.d.token);
        erase_inbound(&m);
        return;
    }
    uint64_t n = st.bytes;
    hipError_t e = hipSuccess;
    /* The shader pull path requires (a) a device destination — dereferencing
     * a pageable host pointer from a kernel faults on non-XNACK systems —
     * and (b) 16-byte alignment on both sides (user buffers can be
     * arbitrarily offset, e.g. partitioned slices).  Everything else rides
     * hipMemcpyAsync (SDMA / runtime blit). */
    /* pull kernel is deadlock-safe in both copy-stream phases: before the
     * priority switch no spin-wait kernel exists anywhere; after it the
     * copy stream owns its hardware queue (see copy_stream()) */
    bool kernel_ok = op->buf_is_device &&
                     ((((uintptr_t)op->buf) | ((uintptr_t)src)) & 15) == 0;
    if (n > 0 && kernel_ok) {
        /* defer: every pull that matched in this progress pass goes out in
         * ONE k_pull_multi launch (flush_pulls) */
        pend_pulls_.push_back(PendingPull{op, m.src, m.d.token, st,
                                          op->buf, src, n});
        m.dev_copy_started = true;
        erase_inbound(&m);
        return;
    }
    hipStream_t cs = copy_stream(n);
    if (n > 0) {
        e = hipMemcpyAsync(op->buf, src, n, hipMemcpyDefault, cs);
    }
    if (e != hipSuccess) {
        MPIX_ERR("hipMemcpyAsync(pull %lu B) failed: %s", (unsigned long)n,
                 hipGetErrorString(e));
        st.err = MPI_ERR_OTHER;
        op->ch_status = st;
        op->ch_done.store(1, std::memory_order_release);
        pending_done_.emplace_back(m.src, m.d.token);
        erase_inbound(&m);
        return;
    }
    if (getenv("MPIX_TRACE") && atoi(getenv("MPIX_TRACE")))
        fprintf(stderr, "[mpix trace] pull start %lu B via memcpyAsync "
                "(dst=%p src=%p)\n", (unsigned long)n, op->buf, src);
    hipEvent_t ev = get_event();
    hipError_t erec = hipEventRecord(ev, cs);
    if (erec != hipSuccess)
        MPIX_ERR("hipEventRecord(pull) failed: %s", hipGetErrorString(erec));
    copies_.push_back(CopyInflight{op, ev, m.src, m.d.token, st});
    m.dev_copy_started = true;
    erase_inbound(&m);
}

void NativeTransport::flush_pulls()
{
    while (!pend_pulls_.empty()) {
        if (arg_busy_[arg_slot_]) return; /* ring full: next pass */
        int n = (int)pend_pulls_.size();
        if (n > MPIX_PULL_BATCH) n = MPIX_PULL_BATCH;
        PullArg *slot_h = args_h_ + (size_t)arg_slot_ * MPIX_PULL_BATCH;
        PullArg *slot_d = args_d_ + (size_t)arg_slot_ * MPIX_PULL_BATCH;
        uint64_t total = 0;
        for (int i = 0; i < n; i++) {
            const PendingPull &pp = pend_pulls_[i];
            slot_h[i].dst = pp.dst;
            slot_h[i].src = pp.csrc;
            slot_h[i].n16 = pp.n / 16;
            slot_h[i].tail = pp.n % 16;
            total += pp.n;
        }
        hipStream_t cs = copy_stream(total);
        unsigned threads = 256;
        static const unsigned max_blocks = [] {
            const char *e = getenv("MPIX_PULL_BLOCKS");
            /* A/B @256 MiB batches: 512 -> 1411 GB/s, 1024 -> 1700,
             * 2048 -> 1594 (4 blocks/CU fills HBM without thrash) */
            return e && atoi(e) > 0 ? (unsigned)atoi(e) : 1024u;
        }();
        unsigned blocks = (unsigned)((total / 16 + threads - 1) / threads);
        if (blocks == 0) blocks = 1;
        if (blocks > max_blocks) blocks = max_blocks;
        hipLaunchKernelGGL(k_pull_multi, dim3(blocks), dim3(threads), 0, cs,
                           slot_d, n);
        hipError_t e = hipGetLastError();
        if (e != hipSuccess) {
            MPIX_ERR("k_pull_multi launch failed: %s", hipGetErrorString(e));
            for (int i = 0; i < n; i++) {
                PendingPull &pp = pend_pulls_[i];
                pp.st.err = MPI_ERR_OTHER;
                pp.op->ch_status = pp.st;
                pp.op->ch_done.store(1, std::memory_order_release);
                pending_done_.emplace_back(pp.src, pp.token);
            }
            pend_pulls_.erase(pend_pulls_.begin(), pend_pulls_.begin() + n);
            continue;
        }
        PullBatch b;
        b.ev = get_event();
        b.slot = arg_slot_;
        (void)hipEventRecord(b.ev, cs);
        b.items.assign(pend_pulls_.begin(), pend_pulls_.begin() + n);
        pend_pulls_.erase(pend_pulls_.begin(), pend_pulls_.begin() + n);
        batches_.push_back(std::move(b));
        arg_busy_[arg_slot_] = true;
        arg_slot_ = (arg_slot_ + 1) % ARG_SLOTS;
        if (getenv("MPIX_TRACE") && atoi(getenv("MPIX_TRACE")))
            fprintf(stderr, "[mpix trace] pull batch n=%d total=%lu B\n", n,
                    (unsigned long)total);
    }
}

int NativeTransport::progress_copies()
{
    for (auto it = batches_.begin(); it != batches_.end();) {
        hipError_t e = hipEventQuery(it->ev);
        if (e == hipErrorNotReady) {
            ++it;
            continue;
        }
        put_event(it->ev);
        for (PendingPull &pp : it->items) {
            if (e != hipSuccess) pp.st.err = MPI_ERR_OTHER;
            if (ring_has_space(pp.src, 1)) {
                Desc d;
                d.type = DESC_DONE;
                d.token = pp.token;
                d.src_world = rank_;
                emit_desc(pp.src, d);
            } else {
                pending_done_.emplace_back(pp.src, pp.token);
            }
            pp.op->ch_status = pp.st;
            pp.op->ch_done.store(1, std::memory_order_release);
        }
        arg_busy_[it->slot] = false;
        it = batches_.erase(it);
    }
    for (auto it = copies_.begin(); it != copies_.end();) {
        hipError_t e = hipEventQuery(it->ev);
        if (e == hipErrorNotReady) {
            ++it;
            continue;
        }
        if (e != hipSuccess) it->st.err = MPI_ERR_OTHER;
        if (getenv("MPIX_TRACE") && atoi(getenv("MPIX_TRACE")))
            fprintf(stderr, "[mpix trace] pull done (e=%d)\n", (int)e);
        put_event(it->ev);
        /* ack the sender, then complete the recv */
        if (ring_has_space(it->src, 1)) {
            Desc d;
            d.type = DESC_DONE;
            d.token = it->token;
            d.src_world = rank_;
            emit_desc(it->src, d);
        } else {
            pending_done_.emplace_back(it->src, it->token);
        }
        it->op->ch_status = it->st;
        it->op->ch_done.store(1, std::memory_order_release);
        it = copies_.erase(it);
    }
    return 0;
}

void NativeTransport::erase_inbound(InboundMsg *m)
{
    /* erase the token mapping only if it still points at THIS message — a
     * later message may have legitimately reused the slot-index token */
    auto it = inbound_by_token_.find(((uint64_t)m->src << 32) ^ m->d.token);
    if (it != inbound_by_token_.end() && it->second == m)
        inbound_by_token_.erase(it);
    for (auto it = inbound_.begin(); it != inbound_.end(); ++it) {
        if (&*it == m) {
            inbound_.erase(it);
            return;
        }
    }
}

int NativeTransport::memcpy_auto(void *dst, const void *src, size_t n)
{
    if (!have_gpu_) {
        memcpy(dst, src, n);
        return 0;
    }
    /* hipMemcpyDefault resolves host/device direction via unified addressing;
     * plain memcpy when both sides are host saves the runtime call.
     * Device copies MUST ride the proxy-private non-blocking stream: a
     * synchronous hipMemcpy runs on the legacy NULL stream, which
     * serializes against every blocking user stream — including one parked
     * on hipStreamWaitValue32 waiting for THIS proxy to make progress
     * (deadlock observed with MPIX_DEV_PUSH_MAX; the reference documents
     * the same hazard class, README.md:140-150). */
    bool dd = ptr_is_device(dst);
    if (!dd && !ptr_is_device(src)) {
        memcpy(dst, src, n);
        return 0;
    }
    hipStream_t cs = copy_stream0();
    if (hipMemcpyAsync(dst, src, n, hipMemcpyDefault, cs) != hipSuccess)
        return -1;
    return hipStreamSynchronize(cs) == hipSuccess ? 0 : -1;
}

/* ----------------------------------------------------------------- factory */

Transport *make_native_transport(int world_rank, int world_size, bool mpi_mode,
                                 bool have_gpu, int device_id)
{
    auto *t = new NativeTransport(world_rank, world_size, mpi_mode, have_gpu,
                                  device_id);
    if (t->init() != 0) {
        delete t;
        return nullptr;
    }
    return t;
}

void native_transport_shutdown(Transport *t)
{
    static_cast<NativeTransport *>(t)->shutdown();
}

} /* namespace mpix */
