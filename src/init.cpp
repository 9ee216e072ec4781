/* mpix — MPIX_Init / MPIX_Finalize: runtime bring-up and teardown.
 *
 * Reference counterpart: /root/reference/src/init.cpp:157-275.  Differences:
 *  - dual bootstrap: works inside an MPI program (MPI mode) OR from plain
 *    torchrun-style env vars (RANK/WORLD_SIZE/MASTER_ADDR) with no MPI_Init;
 *  - runs with zero GPUs (flag pool falls back to plain allocation) — the
 *    reference's MPIACX_DISABLE_MEMOPS path still required CUDA;
 *  - the memOps fast path is probed FUNCTIONALLY (a test hipStreamWriteValue32
 *    round-trip), not just via a device attribute;
 *  - proxy thread lives in proxy.cpp and watches only active slots.
 */
#include <hip/hip_runtime.h>

#include <chrono>

#include "internal.h"

namespace mpix {

static int env_int(const char *name, int dflt)
{
    const char *v = getenv(name);
    if (!v || !*v) return dflt;
    return atoi(v);
}

/* Functional probe of hipStreamWriteValue32 / hipStreamWaitValue32 /
 * hipStreamBatchMemOp against the pinned flag pool. */
static void probe_memops(State *s)
{
    s->use_memops = false;
    s->use_batch_memops = false;
    if (!s->have_gpu) return;
    if (env_int("MPIX_DISABLE_MEMOPS", 0)) {
        MPIX_DBG("memOps disabled by MPIX_DISABLE_MEMOPS");
        return;
    }
    int can = 0;
    (void)hipDeviceGetAttribute(&can, hipDeviceAttributeCanUseStreamWaitValue,
                                s->device_id);
    hipStream_t st;
    if (hipStreamCreateWithFlags(&st, hipStreamNonBlocking) != hipSuccess) {
        (void)hipGetLastError();
        return;
    }
    const int probe_idx = (int)s->nflags - 1;
    uint32_t *fd = s->flags_d + probe_idx;
    const uint32_t magic = 0xC0FFEEu;

    do {
        if (hipStreamWriteValue32(st, fd, magic, 0) != hipSuccess) break;
        if (hipStreamSynchronize(st) != hipSuccess) break;
        if (s->flags[probe_idx].load() != magic) break;
        if (hipStreamWaitValue32(st, fd, magic, hipStreamWaitValueEq, 0xFFFFFFFFu)
            != hipSuccess) break;
        if (hipStreamSynchronize(st) != hipSuccess) break;

        /* HOST-write visibility: the production pattern is proxy (CPU)
         * stores COMPLETED -> enqueued WaitValue32 must observe it.  Probe
         * with a deadline; unwedge via a device-side write on a second
         * stream if the polling engine cannot see host stores. */
        if (hipStreamWaitValue32(st, fd, magic + 1, hipStreamWaitValueEq,
                                 0xFFFFFFFFu) != hipSuccess) break;
        s->flags[probe_idx].store(magic + 1, std::memory_order_release);
        {
            bool done = false;
            for (int i = 0; i < 2000; i++) { /* ~2 s deadline */
                hipError_t q = hipStreamQuery(st);
                if (q == hipSuccess) { done = true; break; }
                if (q != hipErrorNotReady) break;
                std::this_thread::sleep_for(std::chrono::milliseconds(1));
            }
            (void)hipGetLastError();
            if (!done) {
                fprintf(stderr, "[mpix] warn: hipStreamWaitValue32 does not "
                        "observe host stores; disabling memOps wait path\n");
                hipStream_t st2;
                if (hipStreamCreateWithFlags(&st2, hipStreamNonBlocking) ==
                    hipSuccess) {
                    (void)hipStreamWriteValue32(st2, fd, magic + 1, 0);
                    (void)hipStreamSynchronize(st2);
                    (void)hipStreamDestroy(st2);
                }
                (void)hipStreamSynchronize(st);
                break;
            }
        }
        s->use_memops = true;

        hipStreamBatchMemOpParams p[1];
        memset(p, 0, sizeof(p));
        p[0].writeValue.operation = hipStreamMemOpWriteValue32;
        p[0].writeValue.address = fd;
        p[0].writeValue.value = magic + 2;
        if (hipStreamBatchMemOp(st, 1, p, 0) != hipSuccess) break;
        if (hipStreamSynchronize(st) != hipSuccess) break;
        if (s->flags[probe_idx].load() != magic + 2) break;
        s->use_batch_memops = true;
    } while (0);
    (void)hipGetLastError();
    (void)hipStreamSynchronize(st);
    (void)hipStreamDestroy(st);
    s->flags[probe_idx].store(MPIX_FLAG_AVAILABLE);
    if (!s->use_memops) {
        fprintf(stderr,
                "[mpix] warn: hipStream memOps unavailable (attr=%d); "
                "falling back to trigger/wait kernels\n", can);
    }
}

/* Free everything MPIX_Init may have allocated up to a failure point. */
static void destroy_pools(State *s)
{
    delete[] s->ops;
    s->ops = nullptr;
    if (s->flags) {
        if (s->flags_pinned) (void)hipHostFree((void *)s->flags);
        else free((void *)s->flags);
        s->flags = nullptr;
    }
    if (s->seqs) {
        if (s->seqs_pinned) (void)hipHostFree((void *)s->seqs);
        else free((void *)s->seqs);
        s->seqs = nullptr;
    }
    free(s->slot_seq);
    s->slot_seq = nullptr;
}

static std::atomic<uint64_t> g_init_gen{0};

/* Probe hipGraphAddBatchMemOpNode semantics: write node fires, wait node
 * (a) passes once its value is published and (b) does NOT pass before.
 * Needed because graph-node spin-wait KERNELS can deadlock against other
 * work via HIP's stream→HSA-queue multiplexing (see add_flag_node). */
static void probe_graph_memops(State *s)
{
    s->use_graph_memops = false;
    if (!s->have_gpu || !s->use_memops) return;
    /* graph batch-memOp nodes ride the same engine as hipStreamBatchMemOp;
     * if that probe failed, do not even launch this one — a broken wait
     * node that never completes would poison its hardware queue for the
     * rest of the process (there is no way to abort a stuck packet) */
    if (!s->use_batch_memops) return;
    if (env_int("MPIX_DISABLE_GRAPH_MEMOPS", 0)) return;
    const int idx = (int)s->nflags - 1;
    uint32_t *fd = s->flags_d + idx;
    const uint32_t magic = 0x6D700000u;
    hipStream_t st = nullptr;
    if (hipStreamCreateWithFlags(&st, hipStreamNonBlocking) != hipSuccess) {
        (void)hipGetLastError();
        return;
    }
    bool ok = false;
    hipGraph_t g = nullptr, g2 = nullptr;
    hipGraphExec_t ge = nullptr, ge2 = nullptr;
    do {
        s->flags[idx].store(0);
        if (hipGraphCreate(&g, 0) != hipSuccess) break;
        auto op_write = [&](uint32_t v) {
            hipStreamBatchMemOpParams p;
            memset(&p, 0, sizeof(p));
            p.writeValue.operation = hipStreamMemOpWriteValue32;
            p.writeValue.address = fd;
            p.writeValue.value = v;
            return p;
        };
        auto op_wait = [&](uint32_t v) {
            hipStreamBatchMemOpParams p;
            memset(&p, 0, sizeof(p));
            p.waitValue.operation = hipStreamMemOpWaitValue32;
            p.waitValue.address = fd;
            p.waitValue.value = v;
            p.waitValue.flags = hipStreamWaitValueEq;
            return p;
        };
        hipStreamBatchMemOpParams pw = op_write(magic), pq = op_wait(magic),
                                  pw2 = op_write(magic + 1);
        hipBatchMemOpNodeParams np;
        memset(&np, 0, sizeof(np));
        np.count = 1;
        hipGraphNode_t n1, n2, n3;
        np.paramArray = &pw;
        if (hipGraphAddBatchMemOpNode(&n1, g, nullptr, 0, &np) != hipSuccess)
            break;
        np.paramArray = &pq;
        if (hipGraphAddBatchMemOpNode(&n2, g, &n1, 1, &np) != hipSuccess)
            break;
        np.paramArray = &pw2;
        if (hipGraphAddBatchMemOpNode(&n3, g, &n2, 1, &np) != hipSuccess)
            break;
        if (hipGraphInstantiate(&ge, g, nullptr, nullptr, 0) != hipSuccess)
            break;
        if (hipGraphLaunch(ge, st) != hipSuccess) break;
        bool done = false;
        for (int i = 0; i < 2000; i++) {
            hipError_t q = hipStreamQuery(st);
            if (q == hipSuccess) { done = true; break; }
            if (q != hipErrorNotReady) break;
            std::this_thread::sleep_for(std::chrono::milliseconds(1));
        }
        (void)hipGetLastError();
        if (!done || s->flags[idx].load() != magic + 1) break;

        /* unsatisfied wait must BLOCK until the host publishes the value */
        if (hipGraphCreate(&g2, 0) != hipSuccess) break;
        hipStreamBatchMemOpParams pq2 = op_wait(magic + 2);
        np.paramArray = &pq2;
        hipGraphNode_t m1;
        if (hipGraphAddBatchMemOpNode(&m1, g2, nullptr, 0, &np) != hipSuccess)
            break;
        if (hipGraphInstantiate(&ge2, g2, nullptr, nullptr, 0) != hipSuccess)
            break;
        if (hipGraphLaunch(ge2, st) != hipSuccess) break;
        std::this_thread::sleep_for(std::chrono::milliseconds(5));
        if (hipStreamQuery(st) != hipErrorNotReady) {
            /* no-op wait: passed without its value — unusable */
            (void)hipGetLastError();
            break;
        }
        s->flags[idx].store(magic + 2, std::memory_order_release);
        done = false;
        for (int i = 0; i < 2000; i++) {
            hipError_t q = hipStreamQuery(st);
            if (q == hipSuccess) { done = true; break; }
            if (q != hipErrorNotReady) break;
            std::this_thread::sleep_for(std::chrono::milliseconds(1));
        }
        (void)hipGetLastError();
        if (!done) {
            fprintf(stderr, "[mpix] warn: graph memOp wait never observed a "
                    "host store; leaving graph memOps off\n");
            break;
        }
        /* relaunch legs: writes must re-fire and waits must re-block (on
         * ROCm 7.x replayed memOp nodes can be one-shot no-ops) */
        s->flags[idx].store(magic, std::memory_order_release);
        if (hipGraphLaunch(ge, st) != hipSuccess) break;
        done = false;
        for (int i = 0; i < 2000; i++) {
            hipError_t q = hipStreamQuery(st);
            if (q == hipSuccess) { done = true; break; }
            if (q != hipErrorNotReady) break;
            std::this_thread::sleep_for(std::chrono::milliseconds(1));
        }
        (void)hipGetLastError();
        if (!done || s->flags[idx].load() != magic + 1) break;
        s->flags[idx].store(0, std::memory_order_release);
        if (hipGraphLaunch(ge2, st) != hipSuccess) break;
        std::this_thread::sleep_for(std::chrono::milliseconds(5));
        if (hipStreamQuery(st) != hipErrorNotReady) {
            (void)hipGetLastError();
            break; /* relaunched wait was a no-op */
        }
        s->flags[idx].store(magic + 2, std::memory_order_release);
        done = false;
        for (int i = 0; i < 2000; i++) {
            hipError_t q = hipStreamQuery(st);
            if (q == hipSuccess) { done = true; break; }
            if (q != hipErrorNotReady) break;
            std::this_thread::sleep_for(std::chrono::milliseconds(1));
        }
        if (!done) break;
        ok = true;
    } while (0);
    (void)hipGetLastError();
    if (ge) (void)hipGraphExecDestroy(ge);
    if (ge2) (void)hipGraphExecDestroy(ge2);
    if (g) (void)hipGraphDestroy(g);
    if (g2) (void)hipGraphDestroy(g2);
    (void)hipStreamDestroy(st); /* deferred if a packet is stuck */
    s->flags[idx].store(MPIX_FLAG_AVAILABLE);
    s->use_graph_memops = ok;
}

/* Probe whether WriteValue32/WaitValue32 can be RECORDED during stream
 * capture (replaces the captured spin-wait kernels when possible). */
static void probe_capture_memops(State *s)
{
    s->use_capture_memops = false;
    if (!s->have_gpu || !s->use_memops) return;
    if (env_int("MPIX_DISABLE_GRAPH_MEMOPS", 0)) return;
    const int idx = (int)s->nflags - 1;
    uint32_t *fd = s->flags_d + idx;
    const uint32_t magic = 0x63617000u;
    hipStream_t st = nullptr;
    if (hipStreamCreateWithFlags(&st, hipStreamNonBlocking) != hipSuccess) {
        (void)hipGetLastError();
        return;
    }
    bool ok = false;
    hipGraph_t g = nullptr;
    hipGraphExec_t ge = nullptr;
    do {
        s->flags[idx].store(0);
        if (hipStreamBeginCapture(st, hipStreamCaptureModeThreadLocal) !=
            hipSuccess)
            break;
        /* write magic, then wait for magic+1 (UNSATISFIED): detects both
         * record failures and no-op replayed waits — on ROCm 7.x the
         * captured WaitValue32 records fine but does not wait (found the
         * hard way: gpurun_out/pytest_gpu_fix2.log, stale data, not hang) */
        bool rec_ok =
            hipStreamWriteValue32(st, fd, magic, 0) == hipSuccess &&
            hipStreamWaitValue32(st, fd, magic + 1, hipStreamWaitValueEq,
                                 0xFFFFFFFFu) == hipSuccess;
        if (hipStreamEndCapture(st, &g) != hipSuccess || !rec_ok || !g)
            break;
        if (hipGraphInstantiate(&ge, g, nullptr, nullptr, 0) != hipSuccess)
            break;
        if (hipGraphLaunch(ge, st) != hipSuccess) break;
        std::this_thread::sleep_for(std::chrono::milliseconds(5));
        if (hipStreamQuery(st) != hipErrorNotReady) {
            (void)hipGetLastError();
            break; /* no-op wait: replay passed without its value */
        }
        if (s->flags[idx].load() != magic) break; /* write node didn't fire */
        s->flags[idx].store(magic + 1, std::memory_order_release);
        bool done = false;
        for (int i = 0; i < 2000; i++) {
            hipError_t q = hipStreamQuery(st);
            if (q == hipSuccess) { done = true; break; }
            if (q != hipErrorNotReady) break;
            std::this_thread::sleep_for(std::chrono::milliseconds(1));
        }
        if (!done) {
            fprintf(stderr, "[mpix] warn: captured WaitValue32 never "
                    "observed a host store; captures use wait kernels\n");
            break;
        }
        /* RELAUNCH must re-fire the write and re-block the wait — on ROCm
         * 7.x replayed captured memOps are one-shot no-ops (found via
         * gpurun_out/diag7_cap1.log: iter 0 ok, iter 1 stale), which would
         * silently break every re-launched captured graph. */
        if (hipGraphLaunch(ge, st) != hipSuccess) break;
        std::this_thread::sleep_for(std::chrono::milliseconds(5));
        if (hipStreamQuery(st) != hipErrorNotReady) {
            (void)hipGetLastError();
            fprintf(stderr, "[mpix] warn: captured memOps do not replay "
                    "(one-shot); captures use trigger/wait kernels\n");
            break;
        }
        if (s->flags[idx].load() != magic) break; /* write didn't re-fire */
        s->flags[idx].store(magic + 1, std::memory_order_release);
        done = false;
        for (int i = 0; i < 2000; i++) {
            hipError_t q = hipStreamQuery(st);
            if (q == hipSuccess) { done = true; break; }
            if (q != hipErrorNotReady) break;
            std::this_thread::sleep_for(std::chrono::milliseconds(1));
        }
        if (!done) break;
        ok = true;
    } while (0);
    (void)hipGetLastError();
    if (ge) (void)hipGraphExecDestroy(ge);
    if (g) (void)hipGraphDestroy(g);
    (void)hipStreamDestroy(st);
    s->flags[idx].store(MPIX_FLAG_AVAILABLE);
    s->use_capture_memops = ok;
}

extern "C" int MPIX_Init(void)
{
    if (g_state != nullptr) {
        MPIX_ERR("MPIX_Init called twice");
        return MPI_ERR_OTHER;
    }
    State *s = new State();
    s->gen = g_init_gen.fetch_add(1, std::memory_order_relaxed) + 1;

    /* --- identity: MPI mode or env mode ------------------------------- */
    int mpi_inited = 0, mpi_finalized = 0;
    MPI_Initialized(&mpi_inited);
    MPI_Finalized(&mpi_finalized);
    if (mpi_finalized) {
        MPIX_ERR("MPI already finalized");
        delete s;
        return MPI_ERR_OTHER;
    }
    if (mpi_inited) {
        s->mpi_mode = true;
        int provided = 0;
        MPI_Query_thread(&provided);
        if (provided < MPI_THREAD_MULTIPLE) {
            MPIX_ERR("MPI mode requires MPI_THREAD_MULTIPLE "
                     "(the proxy thread calls MPI); got %d", provided);
            delete s;
            return MPI_ERR_OTHER;
        }
        MPI_Comm_rank(MPI_COMM_WORLD, &s->world_rank);
        MPI_Comm_size(MPI_COMM_WORLD, &s->world_size);
    } else {
        s->mpi_mode = false;
        s->world_rank = env_int("RANK", 0);
        s->world_size = env_int("WORLD_SIZE", 1);
    }

    /* --- GPU ----------------------------------------------------------- */
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess) {
        (void)hipGetLastError();
        ndev = 0;
    }
    if (ndev > 0 && !env_int("MPIX_FORCE_NO_GPU", 0)) {
        s->have_gpu = true;
        int local = env_int("LOCAL_RANK", s->world_rank);
        s->device_id = local % ndev;
        if (hipSetDevice(s->device_id) != hipSuccess) {
            MPIX_ERR("hipSetDevice(%d) failed", s->device_id);
            delete s;
            return MPI_ERR_OTHER;
        }
    }

    /* --- flag pool + op table ------------------------------------------ */
    long nf = env_int("MPIX_NFLAGS", 4096);
    if (nf < 64) nf = 64;
    s->nflags = (size_t)nf;
    void *raw = nullptr;
    if (s->have_gpu) {
        if (hipHostMalloc(&raw, s->nflags * sizeof(uint32_t),
                          hipHostMallocMapped) != hipSuccess) {
            MPIX_ERR("hipHostMalloc(flag pool) failed");
            delete s;
            return MPI_ERR_OTHER;
        }
        s->flags_pinned = true;
        s->flags = reinterpret_cast<std::atomic<uint32_t> *>(raw);
        void *dptr = nullptr;
        if (hipHostGetDevicePointer(&dptr, raw, 0) != hipSuccess) {
            MPIX_ERR("hipHostGetDevicePointer failed");
            destroy_pools(s);
            delete s;
            return MPI_ERR_OTHER;
        }
        s->flags_d = (uint32_t *)dptr;
    } else {
        raw = calloc(s->nflags, sizeof(uint32_t));
        s->flags_d = nullptr;
    }
    s->flags = reinterpret_cast<std::atomic<uint32_t> *>(raw);
    for (size_t i = 0; i < s->nflags; i++)
        s->flags[i].store(MPIX_FLAG_AVAILABLE, std::memory_order_relaxed);

    /* fast-wait: second pinned word array for completion sequence numbers.
     * Default ON since r02: 37.9 us vs 43.1 us classic half-RTT on the
     * 2-rank device pingpong (profiles/r02_pingpong_*.json) — one GTE
     * memOp replaces the EQ wait + CLEANUP write + proxy slot round-trip.
     * Graph/capture queues always use the classic EQ protocol (replayable).
     * Kill switch: MPIX_FAST_WAIT=0. */
    s->fast_wait = env_int("MPIX_FAST_WAIT", 1) != 0;
    {
        void *sraw = nullptr;
        if (s->have_gpu) {
            if (hipHostMalloc(&sraw, s->nflags * sizeof(uint32_t),
                              hipHostMallocMapped) != hipSuccess) {
                MPIX_ERR("hipHostMalloc(seq pool) failed");
                destroy_pools(s);
                delete s;
                return MPI_ERR_OTHER;
            }
            s->seqs_pinned = true;
            s->seqs = reinterpret_cast<std::atomic<uint32_t> *>(sraw);
            void *sd = nullptr;
            if (hipHostGetDevicePointer(&sd, sraw, 0) != hipSuccess) {
                MPIX_ERR("hipHostGetDevicePointer(seq) failed");
                destroy_pools(s);
                delete s;
                return MPI_ERR_OTHER;
            }
            s->seqs_d = (uint32_t *)sd;
        } else {
            sraw = calloc(s->nflags, sizeof(uint32_t));
        }
        s->seqs = reinterpret_cast<std::atomic<uint32_t> *>(sraw);
        for (size_t i = 0; i < s->nflags; i++)
            s->seqs[i].store(0, std::memory_order_relaxed);
        s->slot_seq = (uint32_t *)calloc(s->nflags, sizeof(uint32_t));
    }
    s->ops = new Op[s->nflags];
    s->armed.init(4 * s->nflags);
    s->spin_before_yield = env_int("MPIX_PROXY_SPIN", 2000);
    s->stats = env_int("MPIX_STATS", 0) != 0;

    {
        std::lock_guard<std::mutex> lg(lifecycle_mutex());
        g_state = s; /* utilities below use g_state */
    }

    /* --- memOps fast path ---------------------------------------------- */
    probe_memops(s);
    probe_graph_memops(s);
    probe_capture_memops(s);
    if (s->have_gpu && env_int("MPIX_TRACE", 0))
        fprintf(stderr,
                "[mpix r%d] probes: memops=%d batch=%d graph=%d capture=%d\n",
                s->world_rank, s->use_memops, s->use_batch_memops,
                s->use_graph_memops, s->use_capture_memops);

    /* --- data plane ----------------------------------------------------- */
    s->t_native = make_native_transport(s->world_rank, s->world_size,
                                        s->mpi_mode, s->have_gpu, s->device_id);
    if (s->t_native == nullptr) {
        MPIX_ERR("native transport bring-up failed");
        std::lock_guard<std::mutex> lg(lifecycle_mutex());
        g_state = nullptr;
        destroy_pools(s);
        delete s;
        return MPI_ERR_OTHER;
    }
    if (s->mpi_mode) s->t_mpi = make_mpi_transport();

    /* --- proxy ----------------------------------------------------------- */
    s->proxy_stop.store(false);
    s->proxy = std::thread(proxy_main);

    /* MPIX_WATCHDOG=<secs>: periodic flag-pool dump for hang diagnosis.
     * Detached-by-join at finalize via proxy_stop. */
    if (int wd = env_int("MPIX_WATCHDOG", 0)) {
        s->watchdog = std::thread([s, wd] {
            int tick = 0;
            while (!s->proxy_stop.load(std::memory_order_acquire)) {
                for (int ms = 0; ms < wd * 1000 &&
                     !s->proxy_stop.load(std::memory_order_acquire); ms += 100)
                    std::this_thread::sleep_for(std::chrono::milliseconds(100));
                if (s->proxy_stop.load(std::memory_order_acquire)) break;
                fprintf(stderr, "[mpix watchdog r%d t+%ds] active slots:",
                        s->world_rank, ++tick * wd);
                for (size_t i = 0; i < s->nflags; i++) {
                    uint32_t f = s->flags[i].load(std::memory_order_relaxed);
                    if (f != MPIX_FLAG_AVAILABLE)
                        fprintf(stderr, " [%zu]=%u(kind%d,peer%d,tag%d,done%d)",
                                i, f, (int)s->ops[i].kind, s->ops[i].peer_world,
                                s->ops[i].tag,
                                s->ops[i].ch_done.load(
                                    std::memory_order_relaxed));
                }
                fprintf(stderr, " issued=%lu completed=%lu passes=%lu\n",
                        (unsigned long)s->ops_issued.load(
                            std::memory_order_relaxed),
                        (unsigned long)s->ops_completed.load(
                            std::memory_order_relaxed),
                        (unsigned long)s->proxy_passes.load(
                            std::memory_order_relaxed));
            }
        });
    }

    MPIX_DBG("init done: rank %d/%d mpi_mode=%d gpu=%d dev=%d memops=%d batch=%d "
             "nflags=%zu", s->world_rank, s->world_size, s->mpi_mode,
             s->have_gpu, s->device_id, s->use_memops, s->use_batch_memops,
             s->nflags);
    return MPI_SUCCESS;
}

extern "C" int MPIX_Query_config(int *have_gpu, int *use_memops,
                                 int *use_batch_memops, int *mpi_mode,
                                 int *nflags)
{
    State *s = g_state;
    if (s == nullptr) return MPI_ERR_OTHER;
    if (have_gpu) *have_gpu = s->have_gpu;
    if (use_memops) *use_memops = s->use_memops;
    if (use_batch_memops) *use_batch_memops = s->use_batch_memops;
    if (mpi_mode) *mpi_mode = s->mpi_mode;
    if (nflags) *nflags = (int)s->nflags;
    return MPI_SUCCESS;
}

extern "C" int MPIX_Finalize(void)
{
    State *s = g_state;
    if (s == nullptr) return MPI_ERR_OTHER;

    s->proxy_stop.store(true);
    if (s->proxy.joinable()) s->proxy.join();
    if (s->watchdog.joinable()) s->watchdog.join();

    /* From here on, state teardown must exclude late hipUserObject
     * destructors (graph_request_destroy): they check g_state + generation
     * under the same mutex. */
    std::lock_guard<std::mutex> lg(lifecycle_mutex());

    size_t leaked = 0;
    for (size_t i = 0; i < s->nflags; i++) {
        uint32_t f = s->flags[i].load(std::memory_order_relaxed);
        if (f == MPIX_FLAG_CLEANUP) {
            /* consumed after the proxy exited (e.g. a graph destroyed right
             * before finalize): free the request here */
            Op *op = &s->ops[i];
            if ((op->kind == OpKind::ISEND || op->kind == OpKind::IRECV) &&
                op->req != nullptr)
                delete op->req;
            s->flags[i].store(MPIX_FLAG_AVAILABLE, std::memory_order_relaxed);
            continue;
        }
        if (f != MPIX_FLAG_AVAILABLE) leaked++;
    }
    if (leaked)
        fprintf(stderr, "[mpix] warn: %zu flag slot(s) still in use at "
                "MPIX_Finalize\n", leaked);

    if (s->stats) {
        uint64_t n = s->ops_completed.load(std::memory_order_relaxed);
        fprintf(stderr,
                "[mpix stats r%d] ops issued=%lu completed=%lu mean "
                "issue->complete %.1f us\n", s->world_rank,
                (unsigned long)s->ops_issued.load(std::memory_order_relaxed),
                (unsigned long)n,
                n ? (double)s->lat_sum_ns / n / 1e3 : 0.0);
        fprintf(stderr, "[mpix stats r%d] latency histogram (log2 us):",
                s->world_rank);
        for (int b = 0; b < 20; b++)
            if (s->lat_hist[b])
                fprintf(stderr, " [%d]=%lu", b,
                        (unsigned long)s->lat_hist[b]);
        fprintf(stderr, "\n");
        if (s->leg_n) {
            double k = 1e3 * (double)s->leg_n;
            fprintf(stderr,
                    "[mpix stats r%d] per-leg mean us over %lu ops: "
                    "trigger->detect %.1f, transport %.1f, publish %.1f\n",
                    s->world_rank, (unsigned long)s->leg_n,
                    (double)s->leg_trig_ns / k, (double)s->leg_xfer_ns / k,
                    (double)s->leg_compl_ns / k);
        }
    }

    if (s->t_native) {
        native_transport_shutdown(s->t_native);
        delete s->t_native;
    }
    delete s->t_mpi;

    destroy_pools(s);

    g_state = nullptr;
    delete s;
    return MPI_SUCCESS;
}

} /* namespace mpix */
