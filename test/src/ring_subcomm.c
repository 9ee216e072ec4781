/* Ring over a non-WORLD communicator: MPI_Comm_split into even/odd halves,
 * then stream-enqueued send/recv within each sub-communicator.  This routes
 * through the MPI-passthrough transport (src/transport/mpi.cpp) — the
 * native shm/xGMI channel only carries WORLD/SELF — including the device
 * bounce-buffer staging when GPUs are present.  Status SOURCE must be
 * comm-relative (MPI semantics), which the passthrough inherits from the
 * host MPI's own matching.
 */
#include "common.h"

#define N 600

int main(int argc, char **argv)
{
    T_CHECK(t_setup(&argc, &argv) == 0);

    MPI_Comm sub;
    MPI_Comm_split(MPI_COMM_WORLD, g_rank % 2, g_rank, &sub);
    int srank, ssize;
    MPI_Comm_rank(sub, &srank);
    MPI_Comm_size(sub, &ssize);
    int right = (srank + 1) % ssize;
    int left = (srank - 1 + ssize) % ssize;

    hipStream_t stream = 0;
    if (g_have_gpu) T_HIP(hipStreamCreate(&stream));
    int *sbuf = t_alloc(N), *rbuf = t_alloc(N);
    T_CHECK(sbuf && rbuf);
    t_fill(sbuf, 31 * g_rank + 7, N);
    t_fill(rbuf, -1, N);
    t_sync();

    MPIX_Request reqs[2];
    MPI_Status st;
    T_CHECK(MPIX_Isend_enqueue(sbuf, N, MPI_INT, right, 13, sub, &reqs[0],
                               MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Irecv_enqueue(rbuf, N, MPI_INT, left, 13, sub, &reqs[1],
                               MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Wait(&reqs[1], &st) == 0);
    T_CHECK(MPIX_Wait(&reqs[0], MPI_STATUS_IGNORE) == 0);
    t_sync();

    /* left neighbor in MY subcomm has world rank left*2 + (g_rank%2) */
    int left_world = left * 2 + (g_rank % 2);
    if (left_world >= g_size) left_world = left; /* odd world sizes */
    t_verify(rbuf, 31 * left_world + 7, N);
    T_CHECK(st.MPI_SOURCE == left);
    T_CHECK(st.MPI_TAG == 13);

    t_free(sbuf);
    t_free(rbuf);
    if (g_have_gpu) (void)hipStreamDestroy(stream);
    MPI_Comm_free(&sub);
    return t_teardown("ring_subcomm");
}
