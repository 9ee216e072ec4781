/* Host-only concurrency stress for ThreadSanitizer runs:
 * many app threads race enqueue/wait/free against the proxy thread over a
 * small flag pool.  Build: make -C test tsan   Run:
 *   MPIX_FORCE_NO_GPU=1 MPIX_NFLAGS=128 ./bin/stress_tsan
 * (single process, env-mode bootstrap; exercises the CAS slot allocator,
 * the MPSC armed ring, the completion mutex, and transport staging.)
 */
#include <pthread.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include "mpix/mpix.h"

#define THREADS 8
#define ITERS 200

static int g_errors = 0;

static void *worker(void *arg)
{
    long tid = (long)arg;
    int buf[64], out[64];
    for (int it = 0; it < ITERS; it++) {
        int tag = (int)(tid * 10000 + it);
        for (int i = 0; i < 64; i++) buf[i] = tag + i;
        MPIX_Request sr, rr;
        if (MPIX_Isend_enqueue(buf, 64, MPI_INT, 0, tag, MPI_COMM_WORLD, &sr,
                               MPIX_QUEUE_HIP_STREAM, NULL) != 0) {
            __sync_fetch_and_add(&g_errors, 1);
            continue;
        }
        if (MPIX_Irecv_enqueue(out, 64, MPI_INT, 0, tag, MPI_COMM_WORLD, &rr,
                               MPIX_QUEUE_HIP_STREAM, NULL) != 0) {
            __sync_fetch_and_add(&g_errors, 1);
            MPIX_Request_free(&sr);
            continue;
        }
        MPI_Status st;
        MPIX_Wait(&rr, &st);
        if (it % 3 == 0) {
            MPIX_Request_free(&sr); /* orphan path: proxy-side free */
        } else {
            MPIX_Wait(&sr, MPI_STATUS_IGNORE);
        }
        for (int i = 0; i < 64; i++)
            if (out[i] != tag + i) {
                __sync_fetch_and_add(&g_errors, 1);
                break;
            }
        if (st.MPI_TAG != tag) __sync_fetch_and_add(&g_errors, 1);
    }
    return NULL;
}

int main(void)
{
    setenv("RANK", "0", 0);
    setenv("WORLD_SIZE", "1", 0);
    if (MPIX_Init() != 0) {
        fprintf(stderr, "MPIX_Init failed\n");
        return 2;
    }
    pthread_t th[THREADS];
    for (long t = 0; t < THREADS; t++)
        pthread_create(&th[t], NULL, worker, (void *)t);
    for (int t = 0; t < THREADS; t++)
        pthread_join(th[t], NULL);
    MPIX_Finalize();
    printf("stress_tsan: %s (%d threads x %d iters, errors=%d)\n",
           g_errors ? "FAIL" : "PASS", THREADS, ITERS, g_errors);
    return g_errors ? 1 : 0;
}
