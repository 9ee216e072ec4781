/* mpix — the CPU proxy/progress thread.
 *
 * Reference counterpart: progress_thread_fn (/root/reference/src/init.cpp:55-154).
 * Redesigned:
 *  - watches only ACTIVE slots (armed-slot ring + watch list) instead of a
 *    full O(nflags) pool scan per pass;
 *  - CLEANUP handled at the TOP level of the loop (fixes reference defect D1,
 *    where stream-waited slots leaked until finalize);
 *  - issue/complete go through the pluggable Transport (native shm/xGMI or
 *    host-MPI), never direct MPI calls;
 *  - exponential idle backoff (spin -> yield) so an idle proxy does not pin
 *    a core at 100% forever (tunable via MPIX_PROXY_SPIN).
 *
 * Flag lifecycle driven here:
 *   PENDING   -> transport->start(op) -> ISSUED
 *   ISSUED    -> op->ch_done          -> COMPLETED (+status save/delivery)
 *   COMPLETED -> (orphaned only) free
 *   CLEANUP   -> free slot (+request for basic ops)
 */
#include <hip/hip_runtime.h>

#include "internal.h"

namespace mpix {

static inline Transport *route(Op *op)
{
    State *s = g_state;
    if (op->native_route || s->t_mpi == nullptr) return s->t_native;
    return s->t_mpi;
}

/* Transition ISSUED -> COMPLETED: save/deliver status under the completion
 * mutex (closes the proxy-vs-Wait race the reference handles with
 * mpiacx_op_completion_mutex + try_complete_wait_op, sendrecv.cu:82-104). */
static bool complete_op(int idx, Op *op)
{
    State *s = g_state;
    std::lock_guard<std::mutex> lk(s->completion_mutex);
    fill_status(&op->saved_status, op->ch_status);
    op->status_saved = true;
    if (op->enq_status_target != nullptr) {
        *op->enq_status_target = op->saved_status;
    }
    s->ops_completed.fetch_add(1, std::memory_order_relaxed);
    if (op->orphaned.load(std::memory_order_relaxed) &&
        (op->kind == OpKind::ISEND || op->kind == OpKind::IRECV)) {
        /* user already called MPIX_Request_free: nobody will wait */
        delete op->req;
        slot_free(idx);
        return true; /* slot gone */
    }
    flag_store(idx, MPIX_FLAG_COMPLETED);
    return false;
}

/* MPIX_TRACE=1: state-transition tracing to stderr (runtime-gated so the
 * production build carries observability without a debug rebuild). */
static bool trace_on()
{
    static const int v = [] {
        const char *e = getenv("MPIX_TRACE");
        return e ? atoi(e) : 0;
    }();
    return v != 0;
}

#define MPIX_TRACE_EV(fmt, ...)                                           \
    do {                                                                  \
        if (trace_on())                                                   \
            fprintf(stderr, "[mpix trace r%d] " fmt "\n",                 \
                    g_state->world_rank, ##__VA_ARGS__);                  \
    } while (0)

void proxy_main()
{
    State *s = g_state;
    if (s->have_gpu) (void)hipSetDevice(s->device_id);

    std::vector<int> watch;
    watch.reserve(s->nflags);
    std::vector<uint8_t> watched(s->nflags, 0);
    size_t dead = 0; /* tombstones in `watch` */
    int idle = 0;

    /* Order-preserving removal: the walk below must see slots in ARM order,
     * because two sends to one peer that both read PENDING in the same pass
     * are issued in walk order — swap-remove compaction reordered the list
     * and broke per-(src,tag) FIFO (caught by tests/test_soak.py).
     * vector::erase kept order but cost O(n) per removal (quadratic under
     * churn with many active slots), so removal is a tombstone and the list
     * is compacted — still in order — once mostly dead. */
    auto drop = [&](size_t i) {
        watched[watch[i]] = 0;
        watch[i] = -1;
        dead++;
    };
    auto compact = [&] {
        if (dead < 64 || dead * 2 < watch.size()) return;
        size_t w = 0;
        for (size_t i = 0; i < watch.size(); i++)
            if (watch[i] >= 0) watch[w++] = watch[i];
        watch.resize(w);
        dead = 0;
    };

    while (true) {
        bool did = false;
        s->proxy_passes.fetch_add(1, std::memory_order_relaxed);

        /* adopt newly armed slots */
        int nidx;
        while (s->armed.pop(&nidx)) {
            if (!watched[nidx]) {
                watched[nidx] = 1;
                watch.push_back(nidx);
            }
            did = true;
        }

        /* drive the data plane */
        s->t_native->progress();
        if (s->t_mpi) s->t_mpi->progress();

        /* walk active slots */
        for (size_t i = 0; i < watch.size(); i++) {
            int idx = watch[i];
            if (idx < 0) continue; /* tombstone */
            Op *op = &s->ops[idx];
            uint32_t f = flag_load(idx);
            switch (f) {
            case MPIX_FLAG_AVAILABLE:
                /* freed by a host-side wait — stop watching */
                drop(i);
                did = true;
                continue;
            case MPIX_FLAG_RESERVED:
                break; /* described but not yet triggered */
            case MPIX_FLAG_PENDING: {
                /* graph relaunch re-fires the same slot: clear stale
                 * completion state from the previous iteration */
                op->ch_done.store(0, std::memory_order_relaxed);
                op->status_saved = false;
                if (s->stats && op->t_enq_ns) {
                    uint64_t t = now_ns();
                    s->leg_trig_ns += t - op->t_enq_ns;
                    op->t_enq_ns = 0;
                }
                int rc = route(op)->start(op);
                if (rc == 0) {
                    MPIX_TRACE_EV("slot %d %s peer=%d tag=%d part=%d bytes=%lu"
                                  " PENDING->ISSUED", idx, (int)op->kind == 1 ?
                                  "isend" : (int)op->kind == 2 ? "irecv" :
                                  "part", op->peer_world, op->tag,
                                  op->partition, (unsigned long)op->bytes);
                    flag_store(idx, MPIX_FLAG_ISSUED);
                    if (s->stats) op->t_issue_ns = now_ns();
                    s->ops_issued.fetch_add(1, std::memory_order_relaxed);
                    did = true;
                } else if (rc < 0) {
                    MPIX_ERR("transport start failed (op kind %d peer %d): %d",
                             (int)op->kind, op->peer_world, rc);
                    op->ch_status.err = MPI_ERR_OTHER;
                    op->ch_done.store(1, std::memory_order_release);
                    flag_store(idx, MPIX_FLAG_ISSUED);
                } /* rc > 0: transient (ring full) — retry next pass */
                break;
            }
            case MPIX_FLAG_ISSUED:
                if (op->fast &&
                    op->ch_done.load(std::memory_order_acquire)) {
                    /* fast-wait completion: status into the request, seq
                     * word release-store (waiters use GTE), slot freed NOW
                     * — no COMPLETED/CLEANUP round trip */
                    if (s->stats && op->t_issue_ns) {
                        uint64_t d = now_ns() - op->t_issue_ns;
                        s->lat_sum_ns += d;
                        s->leg_xfer_ns += d;
                        uint64_t us = d / 1000;
                        int b = 0;
                        while (us > 1 && b < 19) { us >>= 1; b++; }
                        s->lat_hist[b]++;
                        s->leg_n++;
                    }
                    Request *req = op->req;
                    std::lock_guard<std::mutex> lk(s->completion_mutex);
                    fill_status(&req->fast_status, op->ch_status);
                    if (op->enq_status_target != nullptr)
                        *op->enq_status_target = req->fast_status;
                    bool free_req = (req->consume == 1);
                    MPIX_TRACE_EV("slot %d fast complete seq=%u err=%d", idx,
                                  req->seq, op->ch_status.err);
                    seq_store(idx, req->seq);
                    if (free_req) delete req;
                    s->ops_completed.fetch_add(1, std::memory_order_relaxed);
                    slot_free(idx);
                    drop(i);
                    did = true;
                    continue;
                }
                if (op->ch_done.load(std::memory_order_acquire)) {
                    if (s->stats && op->t_issue_ns) {
                        uint64_t t = now_ns();
                        uint64_t d = t - op->t_issue_ns;
                        s->lat_sum_ns += d;
                        s->leg_xfer_ns += d;
                        uint64_t us = d / 1000;
                        int b = 0;
                        while (us > 1 && b < 19) { us >>= 1; b++; }
                        s->lat_hist[b]++;
                        s->leg_n++;
                    }
                    MPIX_TRACE_EV("slot %d ISSUED->COMPLETED err=%d", idx,
                                  op->ch_status.err);
                    uint64_t tc = s->stats ? now_ns() : 0;
                    if (complete_op(idx, op)) { /* orphan: slot freed */
                        if (s->stats) s->leg_compl_ns += now_ns() - tc;
                        drop(i);
                        did = true;
                        continue;
                    }
                    if (s->stats) s->leg_compl_ns += now_ns() - tc;
                    did = true;
                }
                break;
            case MPIX_FLAG_COMPLETED: {
                /* waiting for a waiter; handle late orphaning */
                if (op->orphaned.load(std::memory_order_relaxed) &&
                    (op->kind == OpKind::ISEND || op->kind == OpKind::IRECV)) {
                    std::lock_guard<std::mutex> lk(s->completion_mutex);
                    if (flag_load(idx) == MPIX_FLAG_COMPLETED &&
                        op->orphaned.load(std::memory_order_relaxed)) {
                        delete op->req;
                        slot_free(idx);
                        drop(i);
                        did = true;
                        continue;
                    }
                }
                break;
            }
            case MPIX_FLAG_CLEANUP: {
                /* waiter consumed the completion (stream wait wrote CLEANUP,
                 * or host wait chose proxy-side free) */
                MPIX_TRACE_EV("slot %d CLEANUP->free", idx);
                std::lock_guard<std::mutex> lk(s->completion_mutex);
                if (op->kind == OpKind::ISEND || op->kind == OpKind::IRECV) {
                    delete op->req;
                }
                slot_free(idx);
                drop(i);
                did = true;
                continue;
            }
            default:
                break;
            }
        }
        compact();

        if (s->proxy_stop.load(std::memory_order_acquire) &&
            watch.size() == dead)
            break;

        if (did) {
            idle = 0;
        } else {
            idle++;
            if (idle > s->spin_before_yield) {
                if (s->proxy_stop.load(std::memory_order_acquire)) {
                    /* shutting down but slots still active: keep draining,
                     * don't burn a core */
                    std::this_thread::sleep_for(std::chrono::microseconds(50));
                    /* bail out if only RESERVED/COMPLETED slots remain: the
                     * app is exiting without waiting (leak warning follows) */
                    bool in_flight = false;
                    for (int idx2 : watch) {
                        if (idx2 < 0) continue; /* tombstone */
                        uint32_t f2 = flag_load(idx2);
                        if (f2 == MPIX_FLAG_PENDING || f2 == MPIX_FLAG_ISSUED ||
                            f2 == MPIX_FLAG_CLEANUP) { in_flight = true; break; }
                    }
                    if (!in_flight) break;
                } else if (idle < s->spin_before_yield + 512) {
                    std::this_thread::yield();
                } else if (idle < s->spin_before_yield + 8192) {
                    /* short naps: worst-case +10 us on a cold trigger */
                    std::this_thread::sleep_for(std::chrono::microseconds(10));
                } else {
                    /* deep idle: cap the wake-up cost at 200 us; an active
                     * stream of ops never reaches here (any progress resets
                     * `idle`), so hot-path latency is unaffected */
                    std::this_thread::sleep_for(std::chrono::microseconds(200));
                }
            }
        }
    }
}

} /* namespace mpix */
