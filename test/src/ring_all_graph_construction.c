/* Explicit graph construction: MPIX_QUEUE_HIP_GRAPH calls return single-node
 * graphs which the application composes send -> recv -> wait with
 * hipGraphAddChildGraphNode, instantiates once, and relaunches.
 * Coverage parity with
 * /root/reference/test/src/ring-all-graph-construction.c (fresh for HIP;
 * the reference's D2 defect — graph wait on the trigger value instead of
 * COMPLETED — is fixed in this library and exercised here via the
 * single-request MPIX_Wait_enqueue graph path as well).
 * GPU-only; SKIPs cleanly without a device.
 */
#include "common.h"

#define N 256
#define ITERS 4

int main(int argc, char **argv)
{
    T_CHECK(t_setup(&argc, &argv) == 0);
    if (!g_have_gpu) {
        if (g_rank == 0) printf("ring_all_graph_construction: SKIP (no GPU)\n");
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

    /* library returns one single-node graph per call */
    hipGraph_t g_send, g_recv, g_wait_s, g_wait_r;
    MPIX_Request reqs[2];
    T_CHECK(MPIX_Isend_enqueue(sbuf, N, MPI_INT, right, 5, MPI_COMM_WORLD,
                               &reqs[0], MPIX_QUEUE_HIP_GRAPH, &g_send) == 0);
    T_CHECK(MPIX_Irecv_enqueue(rbuf, N, MPI_INT, left, 5, MPI_COMM_WORLD,
                               &reqs[1], MPIX_QUEUE_HIP_GRAPH, &g_recv) == 0);
    /* one per-request graph wait (exercises the D2-fixed path) + one waitall */
    T_CHECK(MPIX_Wait_enqueue(&reqs[0], MPI_STATUS_IGNORE,
                              MPIX_QUEUE_HIP_GRAPH, &g_wait_s) == 0);
    T_CHECK(MPIX_Waitall_enqueue(1, &reqs[1], MPI_STATUSES_IGNORE,
                                 MPIX_QUEUE_HIP_GRAPH, &g_wait_r) == 0);

    /* compose: send -> recv -> wait(send) -> wait(recv) */
    hipGraph_t parent;
    T_HIP(hipGraphCreate(&parent, 0));
    hipGraphNode_t n_send, n_recv, n_ws, n_wr;
    T_HIP(hipGraphAddChildGraphNode(&n_send, parent, NULL, 0, g_send));
    T_HIP(hipGraphAddChildGraphNode(&n_recv, parent, &n_send, 1, g_recv));
    T_HIP(hipGraphAddChildGraphNode(&n_ws, parent, &n_recv, 1, g_wait_s));
    T_HIP(hipGraphAddChildGraphNode(&n_wr, parent, &n_ws, 1, g_wait_r));

    hipGraphExec_t gexec;
    T_HIP(hipGraphInstantiate(&gexec, parent, NULL, NULL, 0));

    for (int it = 0; it < ITERS; it++) {
        t_fill(sbuf, 50 * g_rank + it, N);
        t_fill(rbuf, -1, N);
        t_sync();
        MPI_Barrier(MPI_COMM_WORLD);
        T_HIP(hipGraphLaunch(gexec, stream));
        T_HIP(hipStreamSynchronize(stream));
        t_verify(rbuf, 50 * left + it, N);
    }

    T_HIP(hipGraphExecDestroy(gexec));
    T_HIP(hipGraphDestroy(parent));
    t_free(sbuf);
    t_free(rbuf);
    (void)hipStreamDestroy(stream);
    return t_teardown("ring_all_graph_construction");
}
