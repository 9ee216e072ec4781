/* 3D-halo exchange with kernel-triggered partitioned sends (BASELINE.json
 * config 4): each rank owns an nx*ny*nz f32 box in a 1-D ring decomposition
 * along x.  Per step:
 *
 *   1. MPIX_Startall on 4 persistent partitioned requests (send/recv x left
 *      and right neighbors; each face is split into NPARTS partitions).
 *   2. pack_faces kernel packs both boundary faces into send buffers; one
 *      workgroup per partition, lane 0 publishes the partition with
 *      __device__ MPIX_Pready the moment its tile is packed -> the proxy
 *      streams partitions over xGMI while...
 *   3. ...the interior stencil kernel (7-point Jacobi) runs: this is the
 *      compute the transfer hides behind.
 *   4. unpack_faces kernel polls __device__ MPIX_Parrived_spin per
 *      partition and writes the halo planes; boundary stencil follows.
 *   5. host MPIX_Wait resets the persistent requests for the next step.
 *
 * Reports overlapped step time vs a serialized variant (exchange fully
 * completes before any compute) -> overlap efficiency.  Self-verifying:
 * halo planes are checked against the neighbor's closed-form fill every
 * VERIFY_EVERY steps.
 *
 * Run: mpiexec -np N bench/bin/halo3d [nx ny nz iters]
 * (reference has no benchmarks — SURVEY.md §6; this is a from-scratch
 * MI355X workload exercising the partitioned API end to end)
 */
#include <chrono>
#include <cstdio>
#include <cstdlib>

#include <hip/hip_runtime.h>
#include <mpi.h>

#include "mpix/mpix.h"
#include "mpix/mpix_device.h"

#define NPARTS 64
#define VERIFY_EVERY 8

#define CHECK(cond)                                                       \
    do {                                                                  \
        if (!(cond)) {                                                    \
            fprintf(stderr, "[r%d] %s:%d FAILED: %s\n", rank, __FILE__,   \
                    __LINE__, #cond);                                     \
            MPI_Abort(MPI_COMM_WORLD, 1);                                 \
        }                                                                 \
    } while (0)

#define HIP(call) CHECK((call) == hipSuccess)

/* grid[x][y][z] flattened as ((x+1)*ny*nz + y*nz + z) with 2 halo planes in
 * x (x = -1 and x = nx).  Faces are the x=0 and x=nx-1 planes (ny*nz). */

/* Pack one partition per workgroup; publish it as soon as it is staged. */
__global__ void pack_faces(const float *grid, float *sendL, float *sendR,
                           int nx, int ny, int nz, int iter,
                           void *dpsL, void *dpsR)
{
    const size_t face = (size_t)ny * nz;
    const size_t per = face / NPARTS;
    int part = blockIdx.x;
    const float *planeL = grid + face;                 /* x = 0   */
    const float *planeR = grid + (size_t)nx * face;    /* x = nx-1 */
    for (size_t i = threadIdx.x; i < per; i += blockDim.x) {
        sendL[part * per + i] = planeL[part * per + i];
        sendR[part * per + i] = planeR[part * per + i];
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        MPIX_Pready(part, dpsL);
        MPIX_Pready(part, dpsR);
    }
}

/* 7-point Jacobi over interior cells (x in [1, nx-2]) — the overlap work. */
__global__ void stencil_interior(const float *in, float *out,
                                 int nx, int ny, int nz)
{
    const size_t face = (size_t)ny * nz;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t n = (size_t)(nx - 2) * face;
    if (i >= n) return;
    int x = 1 + (int)(i / face);
    size_t yz = i % face;
    int y = (int)(yz / nz), z = (int)(yz % nz);
    size_t c = (size_t)(x + 1) * face + yz;
    float v = in[c];
    float acc = v * 0.4f;
    acc += 0.1f * in[c - face] + 0.1f * in[c + face];            /* x+-1 */
    acc += 0.1f * ((y > 0) ? in[c - nz] : v);
    acc += 0.1f * ((y < ny - 1) ? in[c + nz] : v);
    acc += 0.1f * ((z > 0) ? in[c - 1] : v);
    acc += 0.1f * ((z < nz - 1) ? in[c + 1] : v);
    out[c] = acc;
}

/* Wait for each incoming partition, then write it into the halo plane. */
__global__ void unpack_faces(float *grid, const float *recvL,
                             const float *recvR, int nx, int ny, int nz,
                             void *dprL, void *dprR)
{
    const size_t face = (size_t)ny * nz;
    const size_t per = face / NPARTS;
    int part = blockIdx.x;
    float *haloL = grid;                                /* x = -1 */
    float *haloR = grid + (size_t)(nx + 1) * face;      /* x = nx */
    if (threadIdx.x == 0) MPIX_Parrived_spin(dprL, part);
    __syncthreads();
    for (size_t i = threadIdx.x; i < per; i += blockDim.x)
        haloL[part * per + i] = recvL[part * per + i];
    if (threadIdx.x == 0) MPIX_Parrived_spin(dprR, part);
    __syncthreads();
    for (size_t i = threadIdx.x; i < per; i += blockDim.x)
        haloR[part * per + i] = recvR[part * per + i];
}

__global__ void fill_grid(float *grid, int nx, int ny, int nz, int rank,
                          int iter)
{
    const size_t face = (size_t)ny * nz;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t n = (size_t)nx * face;
    if (i >= n) return;
    /* interior planes x=0..nx-1 live at offset face.. */
    grid[face + i] = (float)rank * 1000.f + (float)iter + (float)(i % 97);
}

__global__ void check_halo(const float *grid, int nx, int ny, int nz,
                           int left, int right, int iter, int *errs)
{
    const size_t face = (size_t)ny * nz;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= face) return;
    /* halo-L holds left neighbor's x=nx-1 plane: global idx (nx-1)*face+i */
    float expL = (float)left * 1000.f + (float)iter +
                 (float)(((size_t)(nx - 1) * face + i) % 97);
    float expR = (float)right * 1000.f + (float)iter + (float)(i % 97);
    if (grid[i] != expL) atomicAdd(errs, 1);
    if (grid[(size_t)(nx + 1) * face + i] != expR) atomicAdd(errs, 1);
}

int main(int argc, char **argv)
{
    int provided, rank, size;
    MPI_Init_thread(&argc, &argv, MPI_THREAD_MULTIPLE, &provided);
    MPI_Comm_rank(MPI_COMM_WORLD, &rank);
    MPI_Comm_size(MPI_COMM_WORLD, &size);

    int nx = argc > 1 ? atoi(argv[1]) : 256;
    int ny = argc > 2 ? atoi(argv[2]) : 256;
    int nz = argc > 3 ? atoi(argv[3]) : 256;
    int iters = argc > 4 ? atoi(argv[4]) : 20;
    int warmup = 5;

    int ndev = 0;
    HIP(hipGetDeviceCount(&ndev));
    CHECK(ndev > 0);
    HIP(hipSetDevice(rank % ndev));
    CHECK(MPIX_Init() == 0);

    int left = (rank - 1 + size) % size, right = (rank + 1) % size;
    const size_t face = (size_t)ny * nz;
    CHECK(face % NPARTS == 0);
    const size_t vol = (size_t)(nx + 2) * face;

    float *grid, *grid2, *sendL, *sendR, *recvL, *recvR;
    HIP(hipMalloc(&grid, vol * sizeof(float)));
    HIP(hipMalloc(&grid2, vol * sizeof(float)));
    HIP(hipMalloc(&sendL, face * sizeof(float)));
    HIP(hipMalloc(&sendR, face * sizeof(float)));
    HIP(hipMalloc(&recvL, face * sizeof(float)));
    HIP(hipMalloc(&recvR, face * sizeof(float)));
    int *errs;
    HIP(hipMalloc(&errs, sizeof(int)));
    HIP(hipMemset(errs, 0, sizeof(int)));

    const int count = (int)(face / NPARTS);
    MPIX_Request psL, psR, prL, prR;
    /* tags: 0 = rightward traffic (my R face -> right's halo-L),
     *       1 = leftward  traffic (my L face -> left's halo-R) */
    CHECK(MPIX_Psend_init(sendR, NPARTS, count, MPI_FLOAT, right, 0,
                          MPI_COMM_WORLD, MPI_INFO_NULL, &psR) == 0);
    CHECK(MPIX_Psend_init(sendL, NPARTS, count, MPI_FLOAT, left, 1,
                          MPI_COMM_WORLD, MPI_INFO_NULL, &psL) == 0);
    CHECK(MPIX_Precv_init(recvL, NPARTS, count, MPI_FLOAT, left, 0,
                          MPI_COMM_WORLD, MPI_INFO_NULL, &prL) == 0);
    CHECK(MPIX_Precv_init(recvR, NPARTS, count, MPI_FLOAT, right, 1,
                          MPI_COMM_WORLD, MPI_INFO_NULL, &prR) == 0);
    MPIX_Prequest dpsL, dpsR, dprL, dprR;
    CHECK(MPIX_Prequest_create(psL, &dpsL) == 0);
    CHECK(MPIX_Prequest_create(psR, &dpsR) == 0);
    CHECK(MPIX_Prequest_create(prL, &dprL) == 0);
    CHECK(MPIX_Prequest_create(prR, &dprR) == 0);

    hipStream_t st;
    HIP(hipStreamCreate(&st));
    const size_t interior = (size_t)(nx - 2) * face;
    const int TPB = 256;
    const int packT = 256;

    auto step = [&](int it, bool overlap, bool verify) {
        MPIX_Request act[4] = {prL, prR, psL, psR};
        CHECK(MPIX_Startall(4, act) == 0);
        hipLaunchKernelGGL(fill_grid, dim3((unsigned)((vol + TPB - 1) / TPB)),
                           dim3(TPB), 0, st, grid, nx, ny, nz, rank, it);
        hipLaunchKernelGGL(pack_faces, dim3(NPARTS), dim3(packT), 0, st,
                           grid, sendL, sendR, nx, ny, nz, it, dpsL, dpsR);
        if (!overlap) {
            /* serialize: finish the whole exchange before any compute */
            HIP(hipStreamSynchronize(st));
            CHECK(MPIX_Wait(&psL, MPI_STATUS_IGNORE) == 0);
            CHECK(MPIX_Wait(&psR, MPI_STATUS_IGNORE) == 0);
        }
        hipLaunchKernelGGL(stencil_interior,
                           dim3((unsigned)((interior + TPB - 1) / TPB)),
                           dim3(TPB), 0, st, grid, grid2, nx, ny, nz);
        hipLaunchKernelGGL(unpack_faces, dim3(NPARTS), dim3(packT), 0, st,
                           grid, recvL, recvR, nx, ny, nz, dprL, dprR);
        if (verify)
            hipLaunchKernelGGL(check_halo,
                               dim3((unsigned)((face + TPB - 1) / TPB)),
                               dim3(TPB), 0, st, grid, nx, ny, nz, left,
                               right, it, errs);
        HIP(hipStreamSynchronize(st));
        CHECK(MPIX_Wait(&prL, MPI_STATUS_IGNORE) == 0);
        CHECK(MPIX_Wait(&prR, MPI_STATUS_IGNORE) == 0);
        CHECK(MPIX_Wait(&psL, MPI_STATUS_IGNORE) == 0);
        CHECK(MPIX_Wait(&psR, MPI_STATUS_IGNORE) == 0);
    };

    auto run = [&](bool overlap) {
        for (int it = 0; it < warmup; it++) step(it, overlap, it == 0);
        HIP(hipDeviceSynchronize());
        MPI_Barrier(MPI_COMM_WORLD);
        auto t0 = std::chrono::steady_clock::now();
        for (int it = 0; it < iters; it++)
            step(warmup + it, overlap, (it % VERIFY_EVERY) == 0);
        MPI_Barrier(MPI_COMM_WORLD);
        double dt = std::chrono::duration<double>(
                        std::chrono::steady_clock::now() - t0).count();
        double mx;
        MPI_Allreduce(&dt, &mx, 1, MPI_DOUBLE, MPI_MAX, MPI_COMM_WORLD);
        return mx / iters;
    };

    double t_overlap = run(true);
    double t_serial = run(false);

    int h_errs = -1;
    HIP(hipMemcpy(&h_errs, errs, sizeof(int), hipMemcpyDeviceToHost));
    int tot_errs = 0;
    MPI_Allreduce(&h_errs, &tot_errs, 1, MPI_INT, MPI_MAX, MPI_COMM_WORLD);

    if (rank == 0) {
        double bytes = 2.0 * face * sizeof(float); /* per rank per step */
        printf("{\"bench\": \"halo3d\", \"ranks\": %d, \"grid\": [%d,%d,%d], "
               "\"nparts\": %d, \"halo_bytes_per_rank\": %.0f, "
               "\"ms_overlap\": %.4f, \"ms_serial\": %.4f, "
               "\"overlap_speedup\": %.4f, \"verify_errors\": %d}\n",
               size, nx, ny, nz, NPARTS, bytes, t_overlap * 1e3,
               t_serial * 1e3, t_serial / t_overlap, tot_errs);
    }

    CHECK(MPIX_Prequest_free(&dpsL) == 0);
    CHECK(MPIX_Prequest_free(&dpsR) == 0);
    CHECK(MPIX_Prequest_free(&dprL) == 0);
    CHECK(MPIX_Prequest_free(&dprR) == 0);
    CHECK(MPIX_Request_free(&psL) == 0);
    CHECK(MPIX_Request_free(&psR) == 0);
    CHECK(MPIX_Request_free(&prL) == 0);
    CHECK(MPIX_Request_free(&prR) == 0);
    (void)hipFree(grid); (void)hipFree(grid2);
    (void)hipFree(sendL); (void)hipFree(sendR);
    (void)hipFree(recvL); (void)hipFree(recvR);
    (void)hipFree(errs);
    (void)hipStreamDestroy(st);
    MPIX_Finalize();
    MPI_Finalize();
    return tot_errs ? 1 : 0;
}
