/* Graph-captured ring: hipStreamBeginCapture over enqueue + waitall, then
 * instantiate and relaunch the graph several times — each launch re-runs the
 * whole trigger/issue/complete flag cycle.  Coverage parity with
 * /root/reference/test/src/ring-all-graph.c, written fresh for HIP.
 * GPU-only (graph capture needs a device); SKIPs cleanly without one.
 */
#include "common.h"

#define N 512
#define ITERS 4

int main(int argc, char **argv)
{
    T_CHECK(t_setup(&argc, &argv) == 0);
    if (!g_have_gpu) {
        if (g_rank == 0) printf("ring_all_graph: SKIP (no GPU)\n");
        MPIX_Finalize();
        MPI_Finalize();
        return 0;
    }
    int right = (g_rank + 1) % g_size;
    int left = (g_rank - 1 + g_size) % g_size;

    hipStream_t stream;
    T_HIP(hipStreamCreate(&stream));
    int *sbuf = t_alloc(N), *rbuf = t_alloc(N);
    T_CHECK(sbuf && rbuf);

    T_HIP(hipStreamBeginCapture(stream, hipStreamCaptureModeGlobal));
    MPIX_Request reqs[2];
    T_CHECK(MPIX_Isend_enqueue(sbuf, N, MPI_INT, right, 4, MPI_COMM_WORLD,
                               &reqs[0], MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Irecv_enqueue(rbuf, N, MPI_INT, left, 4, MPI_COMM_WORLD,
                               &reqs[1], MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Waitall_enqueue(2, reqs, MPI_STATUSES_IGNORE,
                                 MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    hipGraph_t graph;
    T_HIP(hipStreamEndCapture(stream, &graph));
    hipGraphExec_t gexec;
    T_HIP(hipGraphInstantiate(&gexec, graph, NULL, NULL, 0));

    for (int it = 0; it < ITERS; it++) {
        t_fill(sbuf, 1000 * g_rank + it, N);
        t_fill(rbuf, -1, N);
        t_sync();
        MPI_Barrier(MPI_COMM_WORLD);
        T_HIP(hipGraphLaunch(gexec, stream));
        T_HIP(hipStreamSynchronize(stream));
        t_verify(rbuf, 1000 * left + it, N);
    }

    T_HIP(hipGraphExecDestroy(gexec));
    T_HIP(hipGraphDestroy(graph));
    t_free(sbuf);
    t_free(rbuf);
    (void)hipStreamDestroy(stream);
    return t_teardown("ring_all_graph");
}
