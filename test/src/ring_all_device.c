/* Device-buffer ring completed with HOST MPIX_Waitall — the reference's
 * workaround for its cudaMemcpy-vs-blocked-stream deadlock caveat
 * (/root/reference/test/src/ring-all-device.c:93-103, README.md:140-150).
 * The mpix native transport needs no such workaround (the receiver-side
 * SDMA pull runs on a proxy-private stream), but the pattern is kept for
 * coverage parity.  GPU-only: prints SKIP and exits 0 without a device.
 */
#include "common.h"

#define N 4096

int main(int argc, char **argv)
{
    T_CHECK(t_setup(&argc, &argv) == 0);
    if (!g_have_gpu) {
        if (g_rank == 0) printf("ring_all_device: SKIP (no GPU)\n");
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
    t_fill(sbuf, 7 * g_rank + 5, N);
    t_fill(rbuf, -1, N);
    t_sync();

    MPIX_Request reqs[2];
    MPI_Status sts[2];
    T_CHECK(MPIX_Isend_enqueue(sbuf, N, MPI_INT, right, 3, MPI_COMM_WORLD,
                               &reqs[0], MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Irecv_enqueue(rbuf, N, MPI_INT, left, 3, MPI_COMM_WORLD,
                               &reqs[1], MPIX_QUEUE_HIP_STREAM, &stream) == 0);
    T_CHECK(MPIX_Waitall(2, reqs, sts) == 0);
    t_sync();
    t_verify(rbuf, 7 * left + 5, N);
    T_CHECK(sts[1].MPI_SOURCE == left && sts[1].MPI_TAG == 3);

    t_free(sbuf);
    t_free(rbuf);
    (void)hipStreamDestroy(stream);
    return t_teardown("ring_all_device");
}
