/* mpix — host-MPI passthrough transport.
 *
 * Used only in MPI mode, for communicators other than WORLD/SELF (the native
 * shm/xGMI channel has no registered matching context for them).  Mirrors the
 * reference's proxy-issued MPI_Isend/Irecv/Test (init.cpp:69-140), with one
 * addition: because the sandbox MPICH 3.3.2 is not GPU-aware, device buffers
 * are staged automatically through pinned host bounce buffers.
 */
#include <hip/hip_runtime.h>

#include <list>

#include "../internal.h"

namespace mpix {

class MpiTransport : public Transport {
public:
    int start(Op *op) override {
        if (op->kind == OpKind::PSEND_PART || op->kind == OpKind::PRECV_PART) {
            MPIX_ERR("partitioned ops require MPI_COMM_WORLD/SELF "
                     "(native channel)");
            return -1;
        }
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
            if (!o.is_recv &&
                hipMemcpy(o.bounce, op->buf, op->bytes, hipMemcpyDeviceToHost)
                    != hipSuccess) {
                (void)hipHostFree(o.bounce);
                return -1;
            }
            buf = o.bounce;
        }
        int rc;
        if (o.is_recv)
            rc = MPI_Irecv(buf, op->count, op->datatype, op->peer, op->tag,
                           op->comm, &o.req);
        else
            rc = MPI_Isend(buf, op->count, op->datatype, op->peer, op->tag,
                           op->comm, &o.req);
        if (rc != MPI_SUCCESS) {
            if (o.bounce) (void)hipHostFree(o.bounce);
            return -1;
        }
        outstanding_.push_back(o);
        return 0;
    }

    void progress() override {
        for (auto it = outstanding_.begin(); it != outstanding_.end();) {
            int done = 0;
            MPI_Status st;
            if (MPI_Test(&it->req, &done, &st) != MPI_SUCCESS) {
                it->op->ch_status.err = MPI_ERR_OTHER;
                it->op->ch_done.store(1, std::memory_order_release);
                if (it->bounce) (void)hipHostFree(it->bounce);
                it = outstanding_.erase(it);
                continue;
            }
            if (!done) {
                ++it;
                continue;
            }
            Op *op = it->op;
            if (it->is_recv) {
                int cnt = 0, tsz = 0;
                MPI_Get_count(&st, op->datatype, &cnt);
                datatype_size(op->datatype, &tsz);
                if (it->bounce) {
                    uint64_t n = (uint64_t)cnt * (uint64_t)tsz;
                    if (n > 0)
                        (void)hipMemcpy(op->buf, it->bounce, n,
                                        hipMemcpyHostToDevice);
                }
                op->ch_status.src = st.MPI_SOURCE;
                op->ch_status.tag = st.MPI_TAG;
                op->ch_status.bytes = (uint64_t)cnt * (uint64_t)tsz;
                op->ch_status.err = st.MPI_ERROR;
            } else {
                op->ch_status.src = -1; /* send status: fields undefined */
                op->ch_status.tag = op->tag;
                op->ch_status.bytes = op->bytes;
                op->ch_status.err = MPI_SUCCESS;
            }
            if (it->bounce) (void)hipHostFree(it->bounce);
            op->ch_done.store(1, std::memory_order_release);
            it = outstanding_.erase(it);
        }
    }

    const char *name() const override { return "mpi-passthrough"; }

private:
    struct Out {
        Op *op = nullptr;
        MPI_Request req = MPI_REQUEST_NULL;
        void *bounce = nullptr;
        bool is_recv = false;
    };
    std::list<Out> outstanding_;
};

Transport *make_mpi_transport() { return new MpiTransport(); }

} /* namespace mpix */
