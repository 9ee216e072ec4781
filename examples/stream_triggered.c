/* Minimal stream-triggered exchange: each rank sends a message to its right
 * neighbor with start and completion both ordered by a HIP stream — the
 * host enqueues everything up front and synchronizes once.
 *
 * Build:  hipcc -O2 --offload-arch=gfx950 -I../include -I/opt/conda/include -x hip \
 *             stream_triggered.c -x none -L.. -lmpix \
 *             /opt/conda/lib/libmpi.so -Wl,-rpath,.. -o stream_triggered
 * Run:    mpiexec -np 2 ./stream_triggered
 */
#include <stdio.h>
#include <hip/hip_runtime.h>
#include <mpi.h>
#include "mpix/mpix.h"

int main(int argc, char **argv)
{
    int provided, rank, size;
    MPI_Init_thread(&argc, &argv, MPI_THREAD_MULTIPLE, &provided);
    MPI_Comm_rank(MPI_COMM_WORLD, &rank);
    MPI_Comm_size(MPI_COMM_WORLD, &size);
    int ndev = 0;
    (void)hipGetDeviceCount(&ndev);
    if (ndev > 0) (void)hipSetDevice(rank % ndev);
    MPIX_Init();

    const int n = 1 << 20;
    int *send, *recv;
    hipStream_t stream = 0;
    if (ndev > 0) {
        (void)hipStreamCreate(&stream);
        (void)hipMalloc((void **)&send, n * sizeof(int));
        (void)hipMalloc((void **)&recv, n * sizeof(int));
        (void)hipMemset(send, rank + 1, n * sizeof(int));
    } else {
        send = (int *)calloc(n, sizeof(int));
        recv = (int *)calloc(n, sizeof(int));
    }

    int right = (rank + 1) % size, left = (rank - 1 + size) % size;
    MPIX_Request sreq, rreq;
    /* everything below is asynchronous: the stream drives the exchange */
    MPIX_Isend_enqueue(send, n, MPI_INT, right, 0, MPI_COMM_WORLD, &sreq,
                       MPIX_QUEUE_HIP_STREAM, &stream);
    MPIX_Irecv_enqueue(recv, n, MPI_INT, left, 0, MPI_COMM_WORLD, &rreq,
                       MPIX_QUEUE_HIP_STREAM, &stream);
    MPIX_Request both[2] = {sreq, rreq};
    MPIX_Waitall_enqueue(2, both, MPI_STATUSES_IGNORE,
                         MPIX_QUEUE_HIP_STREAM, &stream);
    /* ... more kernels could be enqueued here, ordered after the exchange */
    if (ndev > 0) (void)hipStreamSynchronize(stream);
    else { MPIX_Waitall(2, both, MPI_STATUSES_IGNORE); }

    printf("rank %d: exchange complete\n", rank);
    MPIX_Finalize();
    MPI_Finalize();
    return 0;
}
