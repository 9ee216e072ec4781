/* mpix — global state, lock-free slot allocator, datatype/comm utilities.
 *
 * Replaces the reference's single-issuer linear-scan allocator
 * (/root/reference/src/triggered.cpp:35-67) with a CAS allocator safe for
 * concurrent enqueue threads, and adds MPICH-builtin datatype decoding so the
 * library runs without MPI_Init (env/torchrun mode).
 */
#include <hip/hip_runtime.h>

#include "internal.h"
#include "status_codec.h"

namespace mpix {

State *g_state = nullptr;

std::mutex &lifecycle_mutex()
{
    /* intentionally leaked: HIP user-object destructors may fire after
     * static destructors have begun */
    static std::mutex *m = new std::mutex();
    return *m;
}

/* ------------------------------------------------------------------- slots */

int slot_allocate()
{
    State *s = g_state;
    const uint32_t n = (uint32_t)s->nflags;
    uint32_t start = s->alloc_cursor.fetch_add(1, std::memory_order_relaxed) % n;
    for (uint32_t probe = 0; probe < n; probe++) {
        uint32_t idx = (start + probe) % n;
        uint32_t expect = MPIX_FLAG_AVAILABLE;
        if (s->flags[idx].compare_exchange_strong(expect, MPIX_FLAG_RESERVED,
                                                  std::memory_order_acq_rel,
                                                  std::memory_order_relaxed)) {
            return (int)idx;
        }
    }
    MPIX_ERR("flag pool exhausted (%zu slots; raise MPIX_NFLAGS)", s->nflags);
    return -1;
}

void slot_free(int idx)
{
    g_state->ops[idx].reset();
    flag_store(idx, MPIX_FLAG_AVAILABLE);
}

void slot_arm(int idx)
{
    g_state->armed.push(idx);
}

/* ------------------------------------------------------------------- utils */

int datatype_size(MPI_Datatype dt, int *size_out)
{
    if (g_state && g_state->mpi_mode) {
        int sz = 0;
        if (MPI_Type_size(dt, &sz) != MPI_SUCCESS) return MPI_ERR_TYPE;
        *size_out = sz;
        return MPI_SUCCESS;
    }
    /* env mode (no MPI_Init): ABI-specific decode, see status_codec.h */
    int sz = builtin_datatype_size(dt);
    if (sz > 0) { *size_out = sz; return MPI_SUCCESS; }
    MPIX_ERR("non-builtin datatype requires MPI mode (handle 0x%lx)",
             (unsigned long)(uintptr_t)dt);
    return MPI_ERR_TYPE;
}

int resolve_peer(MPI_Comm comm, int rank, int *world_rank_out,
                 uint32_t *comm_id_out, bool *native_ok_out)
{
    State *s = g_state;
    if (comm == MPI_COMM_WORLD) {
        *comm_id_out = 0;
        *world_rank_out = rank; /* world-relative already (ANY_SOURCE passes) */
        *native_ok_out = true;
        return MPI_SUCCESS;
    }
    if (comm == MPI_COMM_SELF) {
        *comm_id_out = 1;
        *world_rank_out = (rank == 0) ? s->world_rank
                        : (rank == MPI_ANY_SOURCE ? s->world_rank : -1);
        if (*world_rank_out < 0) return MPI_ERR_RANK;
        *native_ok_out = true;
        return MPI_SUCCESS;
    }
    if (!s->mpi_mode) {
        MPIX_ERR("non-WORLD/SELF communicator requires MPI mode");
        return MPI_ERR_COMM;
    }
    /* MPI mode, arbitrary comm: route through the MPI transport (matching is
     * then done by the MPI library itself). Translate rank for bookkeeping. */
    *comm_id_out = 0xffffffffu;
    if (rank == MPI_ANY_SOURCE) {
        *world_rank_out = MPI_ANY_SOURCE;
    } else {
        MPI_Group g, gw;
        MPI_Comm_group(comm, &g);
        MPI_Comm_group(MPI_COMM_WORLD, &gw);
        int wr = MPI_UNDEFINED;
        MPI_Group_translate_ranks(g, 1, &rank, gw, &wr);
        MPI_Group_free(&g);
        MPI_Group_free(&gw);
        if (wr == MPI_UNDEFINED) return MPI_ERR_RANK;
        *world_rank_out = wr;
    }
    *native_ok_out = false;
    return MPI_SUCCESS;
}

void fill_status(MPI_Status *st, const ChStatus &cs)
{
    if (st == nullptr) return;
    status_encode(st, cs.src, cs.tag, cs.err, cs.bytes);
}

bool ptr_is_device(const void *ptr)
{
    if (!g_state || !g_state->have_gpu) return false;
    hipPointerAttribute_t attr;
    hipError_t e = hipPointerGetAttributes(&attr, ptr);
    if (e != hipSuccess) {
        (void)hipGetLastError(); /* clear: plain host malloc returns error */
        return false;
    }
    return attr.type == hipMemoryTypeDevice;
}

/* Host-side trigger: the no-GPU / host-bootstrap equivalent of the on-stream
 * PENDING write (proxy-only path; BASELINE config 1). */
int trigger_host(int idx)
{
    flag_store(idx, MPIX_FLAG_PENDING);
    return MPI_SUCCESS;
}

} /* namespace mpix */
