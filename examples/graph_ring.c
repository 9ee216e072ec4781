/* Explicit hipGraph construction with mpix: the library hands back
 * single-node graphs for the send trigger, recv trigger and waits; the
 * application composes them with hipGraphAddChildGraphNode and relaunches
 * the instantiated graph — each launch re-runs the whole exchange
 * (reference pattern: test/src/ring-all-graph-construction.c; the mpix
 * graph wait targets COMPLETED, fixing the reference's D2 defect).
 *
 * Build (after `make` at the repo root):
 *   hipcc -O2 --offload-arch=gfx950 -I../include -I/opt/conda/include \
 *       -x hip graph_ring.c -x none -L.. -lmpix /opt/conda/lib/libmpi.so \
 *       -Wl,-rpath,.. -Wl,-rpath,/usr/lib/x86_64-linux-gnu \
 *       -Wl,-rpath,/opt/conda/lib -o graph_ring
 *   mpiexec -np 2 ./graph_ring
 */
#include <stdio.h>
#include <stdlib.h>

#include <hip/hip_runtime.h>
#include <mpi.h>

#include "mpix/mpix.h"

#define N 1024
#define LAUNCHES 4

#define OK(c)                                                        \
    do {                                                             \
        if (!(c)) {                                                  \
            fprintf(stderr, "FAILED: %s (line %d)\n", #c, __LINE__); \
            MPI_Abort(MPI_COMM_WORLD, 1);                            \
        }                                                            \
    } while (0)

int main(int argc, char **argv)
{
    int provided, rank, size, ndev = 0;
    MPI_Init_thread(&argc, &argv, MPI_THREAD_MULTIPLE, &provided);
    MPI_Comm_rank(MPI_COMM_WORLD, &rank);
    MPI_Comm_size(MPI_COMM_WORLD, &size);
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev == 0) {
        if (rank == 0) printf("graph_ring: SKIP (no GPU)\n");
        MPI_Finalize();
        return 0;
    }
    (void)hipSetDevice(rank % ndev);
    OK(MPIX_Init() == 0);

    int right = (rank + 1) % size, left = (rank - 1 + size) % size;
    int *sbuf, *rbuf;
    OK(hipMalloc((void **)&sbuf, N * sizeof(int)) == hipSuccess);
    OK(hipMalloc((void **)&rbuf, N * sizeof(int)) == hipSuccess);

    /* the library returns one single-node graph per enqueued operation */
    MPIX_Request reqs[2];
    hipGraph_t g_send, g_recv, g_wait;
    OK(MPIX_Isend_enqueue(sbuf, N, MPI_INT, right, 7, MPI_COMM_WORLD,
                          &reqs[0], MPIX_QUEUE_HIP_GRAPH, &g_send) == 0);
    OK(MPIX_Irecv_enqueue(rbuf, N, MPI_INT, left, 7, MPI_COMM_WORLD,
                          &reqs[1], MPIX_QUEUE_HIP_GRAPH, &g_recv) == 0);
    OK(MPIX_Waitall_enqueue(2, reqs, MPI_STATUSES_IGNORE,
                            MPIX_QUEUE_HIP_GRAPH, &g_wait) == 0);

    /* compose send -> recv -> wait and instantiate once */
    hipGraph_t parent;
    hipGraphNode_t n_send, n_recv, n_wait;
    OK(hipGraphCreate(&parent, 0) == hipSuccess);
    OK(hipGraphAddChildGraphNode(&n_send, parent, NULL, 0, g_send) ==
       hipSuccess);
    OK(hipGraphAddChildGraphNode(&n_recv, parent, &n_send, 1, g_recv) ==
       hipSuccess);
    OK(hipGraphAddChildGraphNode(&n_wait, parent, &n_recv, 1, g_wait) ==
       hipSuccess);
    hipGraphExec_t exec;
    OK(hipGraphInstantiate(&exec, parent, NULL, NULL, 0) == hipSuccess);

    hipStream_t stream;
    OK(hipStreamCreate(&stream) == hipSuccess);
    int *host = (int *)malloc(N * sizeof(int));
    int errors = 0;
    for (int it = 0; it < LAUNCHES; it++) {
        for (int i = 0; i < N; i++) host[i] = rank * 1000 + it;
        OK(hipMemcpy(sbuf, host, N * sizeof(int), hipMemcpyHostToDevice) ==
           hipSuccess);
        MPI_Barrier(MPI_COMM_WORLD);
        OK(hipGraphLaunch(exec, stream) == hipSuccess);
        OK(hipStreamSynchronize(stream) == hipSuccess);
        OK(hipMemcpy(host, rbuf, N * sizeof(int), hipMemcpyDeviceToHost) ==
           hipSuccess);
        for (int i = 0; i < N; i++)
            if (host[i] != left * 1000 + it) errors++;
    }
    printf("rank %d: %d relaunches, %d errors\n", rank, LAUNCHES, errors);

    OK(hipGraphExecDestroy(exec) == hipSuccess);
    OK(hipGraphDestroy(parent) == hipSuccess);
    /* drop the child-graph references so the request user-objects fire */
    (void)hipGraphDestroy(g_send);
    (void)hipGraphDestroy(g_recv);
    (void)hipGraphDestroy(g_wait);
    free(host);
    (void)hipFree(sbuf);
    (void)hipFree(rbuf);
    (void)hipStreamDestroy(stream);
    MPIX_Finalize();
    int total = 0;
    MPI_Allreduce(&errors, &total, 1, MPI_INT, MPI_SUM, MPI_COMM_WORLD);
    MPI_Finalize();
    return total ? 1 : 0;
}
