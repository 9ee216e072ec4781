/* mpix — out-of-band bootstrap: fixed-size blob allgather + barrier.
 *
 * Two modes:
 *  - MPI mode: MPI_Allgather / MPI_Barrier on MPI_COMM_WORLD.
 *  - env mode: rank 0 runs a tiny TCP store on MASTER_ADDR:MASTER_PORT+offset
 *    (torchrun-compatible env), other ranks connect once and keep the socket.
 *
 * Used only at init/finalize (shm segment-name exchange, unlink barrier) —
 * never on the data path.
 */
#ifndef MPIX_BOOTSTRAP_H
#define MPIX_BOOTSTRAP_H

#include <cstddef>

namespace mpix {

class Bootstrap {
public:
    virtual ~Bootstrap() = default;
    /* Gather `blob_size` bytes from every rank into all[size*blob_size]
     * (rank order). Collective. Returns 0 on success. */
    virtual int allgather(const void *mine, void *all, size_t blob_size) = 0;
    virtual int barrier() = 0;
};

Bootstrap *make_bootstrap(int rank, int size, bool mpi_mode);

} /* namespace mpix */

#endif
