/* mpix — host-MPI passthrough transport.
 *
 * Used only in MPI mode, for communicators other than WORLD/SELF (the native
 * shm/xGMI channel has no registered matching context for them).  Mirrors the
 * reference's proxy-issued MPI_Isend/Irecv/Test (init.cpp:69-140), plus:
 *
 *  - device buffers are staged through pinned host bounce buffers (the
 *    sandbox MPICH 3.3.2 is not GPU-aware);
 *  - partitioned ops on arbitrary communicators (the reference requires
 *    MPI-4.0 for these, partitioned.cu:57-59).  Two modes:
 *      (a) MPI-4.0 native passthrough (MPIX_MPI_PARTITIONED=1 on an MPI-4
 *          library): the proxy routes Pready/Parrived-poll to MPI_Pready /
 *          MPI_Parrived on the persistent request;
 *      (b) MPI-3.1 emulation: each partition travels as one header-prefixed
 *          MPI message on a reserved tag (MPI_TAG_UB); the header carries
 *          {user tag, partition, start-seq} so arrival order never matters.
 *          Receives are probe-driven (no pre-posted pool), so partition
 *          sizes may differ freely across concurrent requests.
 */
#include <hip/hip_runtime.h>

#include <list>
#include <vector>

#include "../internal.h"

namespace mpix {

/* partition-message header (emulation mode) */
struct PartHdr {
    uint32_t magic = 0x4d505850u; /* "MPXP" */
    int32_t partition = -1;
    int32_t tag = 0;
    uint32_t pseq = 0;
    uint64_t bytes = 0;
};
static_assert(sizeof(PartHdr) == 24, "PartHdr layout");

static int reserved_part_tag()
{
    static int tag = [] {
        void *v = nullptr;
        int found = 0, t = 32767; /* MPI minimum guarantee as fallback */
        if (MPI_Comm_get_attr(MPI_COMM_WORLD, MPI_TAG_UB, &v, &found) ==
                MPI_SUCCESS && found && v != nullptr)
            t = *(int *)v;
        return t;
    }();
    return tag;
}

class MpiTransport : public Transport {
public:
    int start(Op *op) override {
        switch (op->kind) {
        case OpKind::ISEND:
        case OpKind::IRECV:
            return start_basic(op);
        case OpKind::PSEND_PART:
            return start_psend_part(op);
        case OpKind::PRECV_PART:
            return start_precv_part(op);
        default:
            return -1;
        }
    }

    void progress() override {
        progress_basic();
        progress_parrived();
        progress_part_probe();
    }

    const char *name() const override { return "mpi-passthrough"; }

public:
    ~MpiTransport() override {
        for (hipEvent_t ev : evpool_) (void)hipEventDestroy(ev);
        if (xstream_) (void)hipStreamDestroy(xstream_);
    }

private:
    /* staging phases for device-buffer bounces (async on xstream_ so the
     * proxy never blocks inside a copy — round-1 review finding) */
    enum Phase { POSTED = 0, STAGE_OUT, STAGE_IN, PRE_POST };
    struct Out {
        Op *op = nullptr;
        MPI_Request req = MPI_REQUEST_NULL;
        void *bounce = nullptr;
        bool bounce_pinned = false;
        bool is_recv = false;
        bool is_part_send = false;
        int phase = POSTED;
        hipEvent_t ev = nullptr;       /* staging-copy completion */
        uint64_t recv_bytes = 0;       /* STAGE_IN: payload size */
        MPI_Status st{};               /* STAGE_IN: saved recv status */
    };
    static void free_bounce(Out &o) {
        if (!o.bounce) return;
        if (o.bounce_pinned) (void)hipHostFree(o.bounce);
        else free(o.bounce);
        o.bounce = nullptr;
    }
    std::list<Out> outstanding_;

    /* staging stream: priority stream = own hardware queue (immune to the
     * graph spin-kernel barrier hazard, see transport/native.cpp) */
    hipStream_t xstream_ = nullptr;
    std::vector<hipEvent_t> evpool_;
    hipStream_t xstream() {
        if (xstream_ == nullptr) {
            int lo = 0, hi = 0;
            if (hipDeviceGetStreamPriorityRange(&lo, &hi) != hipSuccess ||
                hi == lo ||
                hipStreamCreateWithPriority(&xstream_, hipStreamNonBlocking,
                                            hi) != hipSuccess) {
                (void)hipGetLastError();
                (void)hipStreamCreateWithFlags(&xstream_,
                                               hipStreamNonBlocking);
            }
        }
        return xstream_;
    }
    hipEvent_t get_ev() {
        if (!evpool_.empty()) {
            hipEvent_t ev = evpool_.back();
            evpool_.pop_back();
            return ev;
        }
        hipEvent_t ev = nullptr;
        (void)hipEventCreateWithFlags(&ev, hipEventDisableTiming);
        return ev;
    }
    void put_ev(hipEvent_t ev) { evpool_.push_back(ev); }
    /* ordering key: MPI non-overtaking is per (comm, dest); a staged send
     * must not be overtaken by a later send that finished staging first */
    static uint64_t okey(const Op *op) {
        return (uint64_t)(uintptr_t)op->comm * 2654435761u ^
               (uint64_t)(uint32_t)op->peer;
    }
    std::vector<Op *> parrived_;      /* MPI-4 passthrough recv partitions */
    std::vector<Op *> part_recvs_;    /* emulation: pending PRECV_PART ops */
    struct PartMsg {                  /* emulation: early partition arrivals */
        MPI_Comm comm;
        int src;
        PartHdr hdr;
        std::vector<char> payload;
    };
    std::list<PartMsg> part_unexpected_;

    static void complete(Op *op, int src, int tag, uint64_t bytes, int err) {
        op->ch_status.src = src;
        op->ch_status.tag = tag;
        op->ch_status.bytes = bytes;
        op->ch_status.err = err;
        op->ch_done.store(1, std::memory_order_release);
    }

    /* ------------------------------------------------------ basic ops */

    int start_basic(Op *op) {
        Out o;
        o.op = op;
        o.is_recv = (op->kind == OpKind::IRECV);
        void *buf = op->buf;
        if (op->buf_is_device) {
            if (hipHostMalloc(&o.bounce, op->bytes ? op->bytes : 1, 0) !=
                hipSuccess) {
                MPIX_ERR("bounce alloc failed (%lu B)", (unsigned long)op->bytes);
                return -1;
            }
            o.bounce_pinned = true;
            if (!o.is_recv) {
                /* async D2H; the MPI_Isend posts from progress() once the
                 * copy lands (FIFO per (comm,dest) to keep non-overtaking) */
                if (op->bytes > 0 &&
                    hipMemcpyAsync(o.bounce, op->buf, op->bytes,
                                   hipMemcpyDeviceToHost, xstream()) !=
                        hipSuccess) {
                    free_bounce(o);
                    return -1;
                }
                o.ev = get_ev();
                (void)hipEventRecord(o.ev, xstream());
                o.phase = STAGE_OUT;
                outstanding_.push_back(o);
                return 0;
            }
            buf = o.bounce;
        }
        if (!o.is_recv && send_pending_before(op)) {
            /* an earlier send to this (comm,dest) is still staging: posting
             * now would overtake it — queue for ordered posting instead */
            o.phase = PRE_POST;
            outstanding_.push_back(o);
            return 0;
        }
        int rc;
        if (o.is_recv)
            rc = MPI_Irecv(buf, op->count, op->datatype, op->peer, op->tag,
                           op->comm, &o.req);
        else
            rc = MPI_Isend(buf, op->count, op->datatype, op->peer, op->tag,
                           op->comm, &o.req);
        if (rc != MPI_SUCCESS) {
            free_bounce(o);
            return -1;
        }
        outstanding_.push_back(o);
        return 0;
    }

    bool send_pending_before(const Op *op) const {
        uint64_t k = okey(op);
        for (const Out &o : outstanding_)
            if (!o.is_recv && (o.phase == STAGE_OUT || o.phase == PRE_POST) &&
                okey(o.op) == k)
                return true;
        return false;
    }

    /* post the MPI send for a staged/queued entry */
    int post_send(Out &o) {
        if (o.is_part_send)
            return MPI_Isend(o.bounce,
                             (int)(sizeof(PartHdr) + o.op->bytes), MPI_BYTE,
                             o.op->peer, reserved_part_tag(), o.op->comm,
                             &o.req);
        void *buf = o.bounce ? o.bounce : o.op->buf;
        return MPI_Isend(buf, o.op->count, o.op->datatype, o.op->peer,
                         o.op->tag, o.op->comm, &o.req);
    }

    void progress_basic() {
        /* keys whose next send must wait (FIFO posting per (comm,dest)) */
        std::vector<uint64_t> blocked;
        auto is_blocked = [&](uint64_t k) {
            for (uint64_t x : blocked)
                if (x == k) return true;
            return false;
        };
        for (auto it = outstanding_.begin(); it != outstanding_.end();) {
            if (it->phase == STAGE_OUT || it->phase == PRE_POST) {
                uint64_t k = okey(it->op);
                if (is_blocked(k)) {
                    ++it;
                    continue;
                }
                if (it->ev != nullptr) {
                    hipError_t e = hipEventQuery(it->ev);
                    if (e == hipErrorNotReady) {
                        blocked.push_back(k);
                        ++it;
                        continue;
                    }
                    put_ev(it->ev);
                    it->ev = nullptr;
                    if (e != hipSuccess) {
                        complete(it->op, -1, it->op->tag, 0, MPI_ERR_OTHER);
                        free_bounce(*it);
                        it = outstanding_.erase(it);
                        continue;
                    }
                }
                if (post_send(*it) != MPI_SUCCESS) {
                    complete(it->op, -1, it->op->tag, 0, MPI_ERR_OTHER);
                    free_bounce(*it);
                    it = outstanding_.erase(it);
                    continue;
                }
                it->phase = POSTED;
                ++it;
                continue;
            }
            if (it->phase == STAGE_IN) {
                hipError_t e = hipEventQuery(it->ev);
                if (e == hipErrorNotReady) {
                    ++it;
                    continue;
                }
                put_ev(it->ev);
                it->ev = nullptr;
                complete(it->op, it->st.MPI_SOURCE, it->st.MPI_TAG,
                         it->recv_bytes,
                         e == hipSuccess ? it->st.MPI_ERROR : MPI_ERR_OTHER);
                free_bounce(*it);
                it = outstanding_.erase(it);
                continue;
            }
            int done = 0;
            MPI_Status st;
            if (MPI_Test(&it->req, &done, &st) != MPI_SUCCESS) {
                it->op->ch_status.err = MPI_ERR_OTHER;
                it->op->ch_done.store(1, std::memory_order_release);
                free_bounce(*it);
                it = outstanding_.erase(it);
                continue;
            }
            if (!done) {
                ++it;
                continue;
            }
            Op *op = it->op;
            if (it->is_part_send) {
                complete(op, -1, op->tag, op->bytes, MPI_SUCCESS);
            } else if (it->is_recv) {
                int cnt = 0, tsz = 0;
                MPI_Get_count(&st, op->datatype, &cnt);
                datatype_size(op->datatype, &tsz);
                uint64_t n = (uint64_t)cnt * (uint64_t)tsz;
                if (it->bounce && n > 0) {
                    /* async H2D; completion in the STAGE_IN phase above */
                    if (hipMemcpyAsync(op->buf, it->bounce, n,
                                       hipMemcpyHostToDevice, xstream()) ==
                        hipSuccess) {
                        it->ev = get_ev();
                        (void)hipEventRecord(it->ev, xstream());
                        it->st = st;
                        it->recv_bytes = n;
                        it->phase = STAGE_IN;
                        ++it;
                        continue;
                    }
                    st.MPI_ERROR = MPI_ERR_OTHER;
                }
                complete(op, st.MPI_SOURCE, st.MPI_TAG, n, st.MPI_ERROR);
            } else {
                complete(op, -1, op->tag, op->bytes, MPI_SUCCESS);
            }
            free_bounce(*it);
            it = outstanding_.erase(it);
        }
    }

    /* ------------------------------------------- partitioned: send side */

    int start_psend_part(Op *op) {
#if MPIX_HAVE_MPI_PARTITIONED
        if (op->req != nullptr && op->req->mpi_part_native) {
            /* reference init.cpp:82-86: Pready then mark completed; the
             * persistent request itself completes in the host MPIX_Wait */
            int rc = MPI_Pready(op->partition, op->req->mpi_preq);
            if (rc != MPI_SUCCESS) return -1;
            complete(op, -1, op->tag, op->bytes, MPI_SUCCESS);
            return 0;
        }
#endif
        /* emulation: header-prefixed message on the reserved tag */
        Out o;
        o.op = op;
        o.is_part_send = true;
        size_t n = sizeof(PartHdr) + op->bytes;
        if (op->buf_is_device) {
            /* pinned staging for the D2H copy */
            if (hipHostMalloc(&o.bounce, n, 0) != hipSuccess) o.bounce = nullptr;
            else o.bounce_pinned = true;
        }
        if (o.bounce == nullptr) o.bounce = malloc(n);
        if (o.bounce == nullptr) {
            MPIX_ERR("partition bounce alloc failed (%zu B)", n);
            return -1;
        }
        PartHdr h;
        h.partition = op->partition;
        h.tag = op->tag;
        h.pseq = op->pseq;
        h.bytes = op->bytes;
        memcpy(o.bounce, &h, sizeof(h));
        if (op->bytes > 0) {
            hipError_t e = op->buf_is_device
                ? hipMemcpy((char *)o.bounce + sizeof(h), op->buf, op->bytes,
                            hipMemcpyDeviceToHost)
                : (memcpy((char *)o.bounce + sizeof(h), op->buf, op->bytes),
                   hipSuccess);
            if (e != hipSuccess) {
                free_bounce(o);
                return -1;
            }
        }
        if (MPI_Isend(o.bounce, (int)n, MPI_BYTE, op->peer,
                      reserved_part_tag(), op->comm, &o.req) != MPI_SUCCESS) {
            free_bounce(o);
            return -1;
        }
        outstanding_.push_back(o);
        return 0;
    }

    /* ------------------------------------------- partitioned: recv side */

    int start_precv_part(Op *op) {
#if MPIX_HAVE_MPI_PARTITIONED
        if (op->req != nullptr && op->req->mpi_part_native) {
            parrived_.push_back(op);
            return 0;
        }
#endif
        /* early arrival already buffered? */
        for (auto it = part_unexpected_.begin(); it != part_unexpected_.end();
             ++it) {
            if (it->comm == op->comm && it->src == op->peer &&
                it->hdr.tag == op->tag && it->hdr.partition == op->partition &&
                it->hdr.pseq == op->pseq) {
                deliver_part(op, it->hdr, it->payload.data(), it->src);
                part_unexpected_.erase(it);
                return 0;
            }
        }
        part_recvs_.push_back(op);
        return 0;
    }

    void progress_parrived() {
#if MPIX_HAVE_MPI_PARTITIONED
        for (size_t i = 0; i < parrived_.size();) {
            Op *op = parrived_[i];
            int f = 0;
            int rc = MPI_Parrived(op->req->mpi_preq, op->partition, &f);
            if (rc != MPI_SUCCESS) {
                complete(op, op->peer, op->tag, 0, MPI_ERR_OTHER);
                f = 1;
            } else if (f) {
                complete(op, op->peer, op->tag, op->bytes, MPI_SUCCESS);
            }
            if (f) {
                parrived_[i] = parrived_.back();
                parrived_.pop_back();
            } else {
                i++;
            }
        }
#endif
    }

    void deliver_part(Op *op, const PartHdr &h, const char *payload, int src) {
        uint64_t n = h.bytes <= op->bytes ? h.bytes : op->bytes;
        int err = h.bytes > op->bytes ? MPI_ERR_TRUNCATE : MPI_SUCCESS;
        if (n > 0) {
            hipError_t e = op->buf_is_device
                ? hipMemcpy(op->buf, payload, n, hipMemcpyHostToDevice)
                : (memcpy(op->buf, payload, n), hipSuccess);
            if (e != hipSuccess) err = MPI_ERR_OTHER;
        }
        complete(op, src, h.tag, n, err);
    }

    void progress_part_probe() {
        /* Probe only communicators with a pending partition recv: probing
         * unconditionally would touch comms the user may already have freed
         * (messages for not-yet-started requests just wait, correctly, in
         * the MPI library's unexpected queue). */
        if (part_recvs_.empty()) return;
        std::vector<MPI_Comm> comms;
        for (Op *op : part_recvs_) {
            bool seen = false;
            for (MPI_Comm c : comms)
                if (c == op->comm) { seen = true; break; }
            if (!seen) comms.push_back(op->comm);
        }
        for (MPI_Comm comm : comms) {
            int flag = 0;
            MPI_Status st;
            while (MPI_Iprobe(MPI_ANY_SOURCE, reserved_part_tag(), comm,
                              &flag, &st) == MPI_SUCCESS && flag) {
                int n = 0;
                MPI_Get_count(&st, MPI_BYTE, &n);
                std::vector<char> buf((size_t)n);
                if (MPI_Recv(buf.data(), n, MPI_BYTE, st.MPI_SOURCE,
                             reserved_part_tag(), comm,
                             MPI_STATUS_IGNORE) != MPI_SUCCESS)
                    break;
                if ((size_t)n < sizeof(PartHdr)) {
                    MPIX_ERR("short partition message (%d B) from %d", n,
                             st.MPI_SOURCE);
                    continue;
                }
                PartHdr h;
                memcpy(&h, buf.data(), sizeof(h));
                if (h.magic != 0x4d505850u) {
                    MPIX_ERR("bad partition magic from %d (user traffic on "
                             "the reserved tag MPI_TAG_UB?)", st.MPI_SOURCE);
                    continue;
                }
                bool matched = false;
                for (size_t i = 0; i < part_recvs_.size(); i++) {
                    Op *op = part_recvs_[i];
                    if (op->comm == comm && op->peer == st.MPI_SOURCE &&
                        op->tag == h.tag && op->partition == h.partition &&
                        op->pseq == h.pseq) {
                        deliver_part(op, h, buf.data() + sizeof(h),
                                     st.MPI_SOURCE);
                        part_recvs_.erase(part_recvs_.begin() + (long)i);
                        matched = true;
                        break;
                    }
                }
                if (!matched) {
                    PartMsg m;
                    m.comm = comm;
                    m.src = st.MPI_SOURCE;
                    m.hdr = h;
                    m.payload.assign(buf.begin() + sizeof(PartHdr), buf.end());
                    part_unexpected_.push_back(std::move(m));
                }
            }
        }
    }
};

Transport *make_mpi_transport() { return new MpiTransport(); }

} /* namespace mpix */
