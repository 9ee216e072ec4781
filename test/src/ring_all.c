/* Waitall ring: several enqueued send/recv pairs completed with one
 * MPIX_Waitall_enqueue (batched memOps / waitall kernel on GPU, host wait
 * fallback without one).  Coverage parity with
 * /root/reference/test/src/ring-all.c, written fresh for HIP.
 */
#include "common.h"

#define N 512
#define PAIRS 4

int main(int argc, char **argv)
{
    T_CHECK(t_setup(&argc, &argv) == 0);
    int right = (g_rank + 1) % g_size;
    int left = (g_rank - 1 + g_size) % g_size;

    hipStream_t stream = 0;
    if (g_have_gpu) T_HIP(hipStreamCreate(&stream));

    int *sbuf[PAIRS], *rbuf[PAIRS];
    MPIX_Request reqs[2 * PAIRS];
    for (int i = 0; i < PAIRS; i++) {
        sbuf[i] = t_alloc(N);
        rbuf[i] = t_alloc(N);
        T_CHECK(sbuf[i] && rbuf[i]);
        t_fill(sbuf[i], 100 * g_rank + i, N);
        t_fill(rbuf[i], -1, N);
    }
    t_sync();

    for (int i = 0; i < PAIRS; i++) {
        T_CHECK(MPIX_Isend_enqueue(sbuf[i], N, MPI_INT, right, i,
                                   MPI_COMM_WORLD, &reqs[2 * i],
                                   MPIX_QUEUE_HIP_STREAM, &stream) == 0);
        T_CHECK(MPIX_Irecv_enqueue(rbuf[i], N, MPI_INT, left, i,
                                   MPI_COMM_WORLD, &reqs[2 * i + 1],
                                   MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    }
    T_CHECK(MPIX_Waitall_enqueue(2 * PAIRS, reqs, MPI_STATUSES_IGNORE,
                                 MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    if (g_have_gpu) T_HIP(hipStreamSynchronize(stream));

    for (int i = 0; i < PAIRS; i++) {
        t_verify(rbuf[i], 100 * left + i, N);
        t_free(sbuf[i]);
        t_free(rbuf[i]);
    }
    if (g_have_gpu) (void)hipStreamDestroy(stream);
    return t_teardown("ring_all");
}
