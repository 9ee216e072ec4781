/* Partitioned ring over a non-WORLD communicator.  The reference can only
 * do this on an MPI-4.0 library (MPI_Psend_init passes the user comm
 * through, /root/reference/src/partitioned.cu:57-59); mpix additionally
 * emulates it on MPI-3.1 via the MPI-passthrough transport: each partition
 * travels as a header-routed message on the reserved tag (MPI_TAG_UB), so
 * Pready order, reuse iterations and differing partition sizes all work
 * (src/transport/mpi.cpp).  Host Pready/Parrived only (host buffers): the
 * device-kernel variants are covered by ring_partitioned on WORLD.
 *
 * Exercises: out-of-order Pready publication, persistent-request reuse
 * across 4 iterations, and a second concurrent partitioned request on the
 * same (comm, peer, tag) disambiguated by start-seq.
 */
#include "common.h"

#define PARTS 8
#define PER 512
#define ITERS 4

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

    int n = PARTS * PER;
    int *sbuf = (int *)malloc((size_t)n * sizeof(int));
    int *rbuf = (int *)malloc((size_t)n * sizeof(int));
    T_CHECK(sbuf && rbuf);

    MPIX_Request ps, pr;
    T_CHECK(MPIX_Psend_init(sbuf, PARTS, PER, MPI_INT, right, 5, sub,
                            MPI_INFO_NULL, &ps) == 0);
    T_CHECK(MPIX_Precv_init(rbuf, PARTS, PER, MPI_INT, left, 5, sub,
                            MPI_INFO_NULL, &pr) == 0);

    int left_world = left * 2 + (g_rank % 2);
    if (left_world >= g_size) left_world = left; /* odd world sizes */

    for (int it = 0; it < ITERS; it++) {
        for (int i = 0; i < n; i++) {
            sbuf[i] = 1000 * g_rank + 10 * it + i / PER;
            rbuf[i] = -1;
        }
        T_CHECK(MPIX_Start(&pr) == 0);
        T_CHECK(MPIX_Start(&ps) == 0);
        /* publish partitions out of order (stride walk) */
        for (int k = 0; k < PARTS; k++) {
            int p = (k * 3 + it) % PARTS;
            T_CHECK(MPIX_Pready(p, ps) == 0);
        }
        /* poll a couple of partitions with Parrived before the full wait */
        int seen = 0;
        while (seen < 2) {
            int f = 0;
            T_CHECK(MPIX_Parrived(pr, seen, &f) == 0);
            if (f) seen++;
        }
        T_CHECK(MPIX_Wait(&pr, MPI_STATUS_IGNORE) == 0);
        T_CHECK(MPIX_Wait(&ps, MPI_STATUS_IGNORE) == 0);
        for (int i = 0; i < n; i++) {
            int want = 1000 * left_world + 10 * it + i / PER;
            if (rbuf[i] != want) {
                if (g_errors < 5)
                    fprintf(stderr, "[r%d] it %d: rbuf[%d]=%d want %d\n",
                            g_rank, it, i, rbuf[i], want);
                g_errors++;
            }
        }
    }

    /* concurrent second request, same (comm, peer, tag), smaller parts */
    {
        int m = PARTS * 16;
        int *s2 = (int *)malloc((size_t)m * sizeof(int));
        int *r2 = (int *)malloc((size_t)m * sizeof(int));
        T_CHECK(s2 && r2);
        for (int i = 0; i < m; i++) { s2[i] = 7 * g_rank + i; r2[i] = -1; }
        MPIX_Request ps2, pr2;
        T_CHECK(MPIX_Psend_init(s2, PARTS, 16, MPI_INT, right, 5, sub,
                                MPI_INFO_NULL, &ps2) == 0);
        T_CHECK(MPIX_Precv_init(r2, PARTS, 16, MPI_INT, left, 5, sub,
                                MPI_INFO_NULL, &pr2) == 0);
        /* interleave with another iteration of the big request */
        for (int i = 0; i < n; i++) {
            sbuf[i] = 1000 * g_rank + 10 * ITERS + i / PER;
            rbuf[i] = -1;
        }
        T_CHECK(MPIX_Start(&pr) == 0);
        T_CHECK(MPIX_Start(&pr2) == 0);
        T_CHECK(MPIX_Start(&ps) == 0);
        T_CHECK(MPIX_Start(&ps2) == 0);
        for (int p = 0; p < PARTS; p++) {
            T_CHECK(MPIX_Pready(p, ps2) == 0);
            T_CHECK(MPIX_Pready(PARTS - 1 - p, ps) == 0);
        }
        T_CHECK(MPIX_Wait(&pr2, MPI_STATUS_IGNORE) == 0);
        T_CHECK(MPIX_Wait(&pr, MPI_STATUS_IGNORE) == 0);
        T_CHECK(MPIX_Wait(&ps, MPI_STATUS_IGNORE) == 0);
        T_CHECK(MPIX_Wait(&ps2, MPI_STATUS_IGNORE) == 0);
        for (int i = 0; i < m; i++)
            if (r2[i] != 7 * left_world + i) { g_errors++; break; }
        for (int i = 0; i < n; i++)
            if (rbuf[i] != 1000 * left_world + 10 * ITERS + i / PER) {
                g_errors++;
                break;
            }
        T_CHECK(MPIX_Request_free(&ps2) == 0);
        T_CHECK(MPIX_Request_free(&pr2) == 0);
        free(s2);
        free(r2);
    }

    T_CHECK(MPIX_Request_free(&ps) == 0);
    T_CHECK(MPIX_Request_free(&pr) == 0);
    free(sbuf);
    free(rbuf);
    MPI_Comm_free(&sub);
    return t_teardown("ring_partitioned_subcomm");
}
