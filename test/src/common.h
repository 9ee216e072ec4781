/* Shared harness for the mpix C integration tests.
 *
 * Mirrors the reference suite's conventions (/root/reference/test/src/:
 * self-verifying ring programs, errors max-reduced across ranks into the
 * exit code) with MI355X specifics: hipSetDevice(rank % ndev) when GPUs are
 * present, and every program degrades to host buffers + host waits when
 * hipGetDeviceCount() == 0 so the suite runs in GPU-less CI.
 */
#ifndef MPIX_TEST_COMMON_H
#define MPIX_TEST_COMMON_H

#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include <hip/hip_runtime.h>
#include <mpi.h>

#include "mpix/mpix.h"

#define T_CHECK(cond)                                                      \
    do {                                                                   \
        if (!(cond)) {                                                     \
            fprintf(stderr, "[r%d] %s:%d check failed: %s\n", g_rank,      \
                    __FILE__, __LINE__, #cond);                            \
            g_errors++;                                                    \
        }                                                                  \
    } while (0)

#define T_HIP(call)                                                       \
    do {                                                                   \
        hipError_t _e = (call);                                            \
        if (_e != hipSuccess) {                                            \
            fprintf(stderr, "[r%d] %s:%d %s: %s\n", g_rank, __FILE__,      \
                    __LINE__, #call, hipGetErrorString(_e));               \
            g_errors++;                                                    \
        }                                                                  \
    } while (0)

static int g_rank = -1, g_size = 0, g_errors = 0, g_have_gpu = 0;

static int t_setup(int *argc, char ***argv)
{
    int provided = 0;
    MPI_Init_thread(argc, argv, MPI_THREAD_MULTIPLE, &provided);
    if (provided < MPI_THREAD_MULTIPLE) {
        fprintf(stderr, "MPI_THREAD_MULTIPLE unavailable (%d)\n", provided);
        MPI_Abort(MPI_COMM_WORLD, 1);
    }
    MPI_Comm_rank(MPI_COMM_WORLD, &g_rank);
    MPI_Comm_size(MPI_COMM_WORLD, &g_size);
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) == hipSuccess && ndev > 0) {
        g_have_gpu = 1;
        (void)hipSetDevice(g_rank % ndev);
    }
    return MPIX_Init();
}

/* Max-reduce errors across ranks; returns process exit code. */
static int t_teardown(const char *name)
{
    int total = 0;
    MPI_Allreduce(&g_errors, &total, 1, MPI_INT, MPI_MAX, MPI_COMM_WORLD);
    MPIX_Finalize();
    if (g_rank == 0)
        printf("%s: %s (%d ranks%s)\n", name, total ? "FAIL" : "PASS",
               g_size, g_have_gpu ? ", GPU" : ", host-only");
    MPI_Finalize();
    return total ? 1 : 0;
}

/* Allocate an int32 buffer on device (if available) else host. */
static int *t_alloc(size_t n)
{
    int *p = NULL;
    if (g_have_gpu) {
        if (hipMalloc((void **)&p, n * sizeof(int)) != hipSuccess) return NULL;
    } else {
        p = (int *)malloc(n * sizeof(int));
    }
    return p;
}

static void t_free(int *p)
{
    if (g_have_gpu) (void)hipFree(p); else free(p);
}

static void t_fill(int *dst, int val, size_t n)
{
    if (g_have_gpu) {
        int *tmp = (int *)malloc(n * sizeof(int));
        for (size_t i = 0; i < n; i++) tmp[i] = val;
        (void)hipMemcpy(dst, tmp, n * sizeof(int), hipMemcpyHostToDevice);
        free(tmp);
    } else {
        for (size_t i = 0; i < n; i++) dst[i] = val;
    }
}

/* Verify every element == val; count mismatches into g_errors. */
static void t_verify(const int *src, int val, size_t n)
{
    const int *view = src;
    int *tmp = NULL;
    if (g_have_gpu) {
        tmp = (int *)malloc(n * sizeof(int));
        (void)hipMemcpy(tmp, src, n * sizeof(int), hipMemcpyDeviceToHost);
        view = tmp;
    }
    size_t bad = 0;
    for (size_t i = 0; i < n; i++)
        if (view[i] != val) bad++;
    if (bad) {
        fprintf(stderr, "[r%d] %zu/%zu elements wrong (want %d, got %d...)\n",
                g_rank, bad, n, val, view[0]);
        g_errors++;
    }
    free(tmp);
}

static void t_sync(void)
{
    if (g_have_gpu) (void)hipDeviceSynchronize();
}

#endif /* MPIX_TEST_COMMON_H */
