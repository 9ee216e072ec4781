/* Basic ring: stream-enqueued Isend/Irecv, first completed with
 * MPIX_Wait_enqueue on the stream, then with host-side MPIX_Wait; the
 * received payload and the MPI_Status fields (SOURCE/TAG/ERROR) are both
 * verified.  Coverage parity with /root/reference/test/src/ring.c (written
 * fresh for HIP; degrades to host buffers + host waits without a GPU).
 */
#include "common.h"

#define N 1000

int main(int argc, char **argv)
{
    T_CHECK(t_setup(&argc, &argv) == 0);
    int right = (g_rank + 1) % g_size;
    int left = (g_rank - 1 + g_size) % g_size;

    hipStream_t stream = 0;
    if (g_have_gpu) T_HIP(hipStreamCreate(&stream));

    int *sbuf = t_alloc(N), *rbuf = t_alloc(N);
    T_CHECK(sbuf && rbuf);

    /* -------- phase 1: stream wait (falls back to host path GPU-less) */
    t_fill(sbuf, 10 * g_rank + 1, N);
    t_fill(rbuf, -1, N);
    t_sync();

    MPIX_Request sreq, rreq;
    T_CHECK(MPIX_Isend_enqueue(sbuf, N, MPI_INT, right, 7, MPI_COMM_WORLD,
                               &sreq, MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Irecv_enqueue(rbuf, N, MPI_INT, left, 7, MPI_COMM_WORLD,
                               &rreq, MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Wait_enqueue(&sreq, MPI_STATUS_IGNORE,
                              MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Wait_enqueue(&rreq, MPI_STATUS_IGNORE,
                              MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    if (g_have_gpu) T_HIP(hipStreamSynchronize(stream));
    t_verify(rbuf, 10 * left + 1, N);
    T_CHECK(sreq == MPIX_REQUEST_NULL && rreq == MPIX_REQUEST_NULL);

    /* -------- phase 2: host wait + status verification */
    t_fill(sbuf, 20 * g_rank + 3, N);
    t_fill(rbuf, -1, N);
    t_sync();

    MPI_Status st;
    memset(&st, 0xff, sizeof(st));
    T_CHECK(MPIX_Isend_enqueue(sbuf, N, MPI_INT, right, 9, MPI_COMM_WORLD,
                               &sreq, MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Irecv_enqueue(rbuf, N, MPI_INT, left, 9, MPI_COMM_WORLD,
                               &rreq, MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Wait(&rreq, &st) == 0);
    T_CHECK(MPIX_Wait(&sreq, MPI_STATUS_IGNORE) == 0);
    t_sync();
    t_verify(rbuf, 20 * left + 3, N);
    T_CHECK(st.MPI_SOURCE == left);
    T_CHECK(st.MPI_TAG == 9);
    T_CHECK(st.MPI_ERROR == MPI_SUCCESS);
    int cnt = -1;
    MPI_Get_count(&st, MPI_INT, &cnt);
    T_CHECK(cnt == N);

    t_free(sbuf);
    t_free(rbuf);
    if (g_have_gpu) (void)hipStreamDestroy(stream);
    return t_teardown("ring");
}
