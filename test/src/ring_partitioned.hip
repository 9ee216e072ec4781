/* Partitioned ring with kernel-side triggering: persistent Psend/Precv
 * requests reused over ITERS iterations; a gfx950 kernel fills each
 * partition and publishes it with __device__ MPIX_Pready (system-scope
 * release store); the receiver kernel polls __device__ MPIX_Parrived with
 * s_sleep backoff, then verifies the payload on-device.  Coverage parity
 * with /root/reference/test/src/ring-partitioned.cu (fresh CDNA4 HIP: one
 * 64-wide wavefront per partition instead of <<<1,10>>> single threads).
 * Host-only fallback (no GPU): host MPIX_Pready / MPIX_Parrived loop.
 */
#include "common.h"
#include "mpix/mpix_device.h"

#define PARTS 16
#define PER 4096 /* int32 per partition */
#define ITERS 10

/* One wavefront per partition: fill the partition's payload, then lane 0
 * publishes it.  The release store in MPIX_Pready orders the HBM writes. */
__global__ void fill_and_pready(int *buf, int base, void *dpreq)
{
    int part = blockIdx.x;
    int *p = buf + (size_t)part * PER;
    for (int i = threadIdx.x; i < PER; i += blockDim.x)
        p[i] = base + part;
    __syncthreads();
    if (threadIdx.x == 0)
        MPIX_Pready(part, dpreq);
}

/* One wavefront per partition: lane 0 spins on Parrived, then all lanes
 * verify the partition payload, accumulating mismatches into *errs. */
__global__ void wait_and_verify(const int *buf, int base, void *dpreq,
                                int *errs)
{
    int part = blockIdx.x;
    if (threadIdx.x == 0)
        MPIX_Parrived_spin(dpreq, part);
    __syncthreads();
    const int *p = buf + (size_t)part * PER;
    int bad = 0;
    for (int i = threadIdx.x; i < PER; i += blockDim.x)
        if (p[i] != base + part) bad++;
    if (bad) atomicAdd(errs, bad);
}

int main(int argc, char **argv)
{
    T_CHECK(t_setup(&argc, &argv) == 0);
    int right = (g_rank + 1) % g_size;
    int left = (g_rank - 1 + g_size) % g_size;

    int *sbuf = t_alloc(PARTS * PER), *rbuf = t_alloc(PARTS * PER);
    T_CHECK(sbuf && rbuf);

    MPIX_Request psend, precv;
    T_CHECK(MPIX_Psend_init(sbuf, PARTS, PER, MPI_INT, right, 11,
                            MPI_COMM_WORLD, MPI_INFO_NULL, &psend) == 0);
    T_CHECK(MPIX_Precv_init(rbuf, PARTS, PER, MPI_INT, left, 11,
                            MPI_COMM_WORLD, MPI_INFO_NULL, &precv) == 0);

    if (g_have_gpu) {
        MPIX_Prequest dps, dpr;
        T_CHECK(MPIX_Prequest_create(psend, &dps) == 0);
        T_CHECK(MPIX_Prequest_create(precv, &dpr) == 0);
        int *errs;
        T_HIP(hipMalloc(&errs, sizeof(int)));
        T_HIP(hipMemset(errs, 0, sizeof(int)));
        hipStream_t stream;
        T_HIP(hipStreamCreate(&stream));

        for (int it = 0; it < ITERS; it++) {
            MPIX_Request active[2] = {precv, psend};
            T_CHECK(MPIX_Startall(2, active) == 0);
            int sbase = 100000 * g_rank + 1000 * it;
            int rbase = 100000 * left + 1000 * it;
            hipLaunchKernelGGL(fill_and_pready, dim3(PARTS), dim3(64), 0,
                               stream, sbuf, sbase, dps);
            hipLaunchKernelGGL(wait_and_verify, dim3(PARTS), dim3(64), 0,
                               stream, rbuf, rbase, dpr, errs);
            T_HIP(hipStreamSynchronize(stream));
            T_CHECK(MPIX_Wait(&precv, MPI_STATUS_IGNORE) == 0);
            T_CHECK(MPIX_Wait(&psend, MPI_STATUS_IGNORE) == 0);
        }
        int h_errs = -1;
        T_HIP(hipMemcpy(&h_errs, errs, sizeof(int), hipMemcpyDeviceToHost));
        T_CHECK(h_errs == 0);
        T_CHECK(MPIX_Prequest_free(&dps) == 0);
        T_CHECK(MPIX_Prequest_free(&dpr) == 0);
        (void)hipFree(errs);
        (void)hipStreamDestroy(stream);
    } else {
        /* host path: publish each partition, poll arrivals, verify */
        for (int it = 0; it < ITERS; it++) {
            MPIX_Request active[2] = {precv, psend};
            T_CHECK(MPIX_Startall(2, active) == 0);
            int sbase = 100000 * g_rank + 1000 * it;
            int rbase = 100000 * left + 1000 * it;
            for (int p = 0; p < PARTS; p++) {
                for (int i = 0; i < PER; i++)
                    sbuf[(size_t)p * PER + i] = sbase + p;
                T_CHECK(MPIX_Pready(p, psend) == 0);
            }
            for (int p = 0; p < PARTS; p++) {
                int arrived = 0;
                while (!arrived)
                    T_CHECK(MPIX_Parrived(precv, p, &arrived) == 0);
                for (int i = 0; i < PER; i++)
                    if (rbuf[(size_t)p * PER + i] != rbase + p) {
                        g_errors++;
                        break;
                    }
            }
            T_CHECK(MPIX_Wait(&precv, MPI_STATUS_IGNORE) == 0);
            T_CHECK(MPIX_Wait(&psend, MPI_STATUS_IGNORE) == 0);
        }
    }

    T_CHECK(MPIX_Request_free(&psend) == 0);
    T_CHECK(MPIX_Request_free(&precv) == 0);
    t_free(sbuf);
    t_free(rbuf);
    return t_teardown("ring_partitioned");
}
