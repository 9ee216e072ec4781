/* Stream-enqueued ping-pong latency / bandwidth sweep (BASELINE.json
 * config 2) at the C API level — no Python overhead, so this is the
 * library's own half-RTT.  Rank pairs (0<->1); self-loopback at np=1.
 *
 * For each message size: W warmup + K timed round trips of
 *   MPIX_Isend_enqueue -> MPIX_Wait_enqueue -> MPIX_Irecv_enqueue ->
 *   MPIX_Wait_enqueue  (device buffers, hipStream memOps fast path),
 * one hipStreamSynchronize per iteration so each RTT is fully retired
 * before the next starts (osu_latency discipline, not pipelined).
 *
 * Run: mpiexec -np 2 bench/bin/pingpong [max_log2_bytes iters]
 */
#include <chrono>
#include <cstdio>
#include <cstdlib>

#include <hip/hip_runtime.h>
#include <mpi.h>

#include "mpix/mpix.h"

#define CHECK(cond)                                                       \
    do {                                                                  \
        if (!(cond)) {                                                    \
            fprintf(stderr, "[r%d] %s:%d FAILED: %s\n", rank, __FILE__,   \
                    __LINE__, #cond);                                     \
            MPI_Abort(MPI_COMM_WORLD, 1);                                 \
        }                                                                 \
    } while (0)

#define HIP(call) CHECK((call) == hipSuccess)

int main(int argc, char **argv)
{
    int provided, rank, size;
    MPI_Init_thread(&argc, &argv, MPI_THREAD_MULTIPLE, &provided);
    MPI_Comm_rank(MPI_COMM_WORLD, &rank);
    MPI_Comm_size(MPI_COMM_WORLD, &size);

    int max_log2 = argc > 1 ? atoi(argv[1]) : 24; /* up to 16 MiB */
    int iters = argc > 2 ? atoi(argv[2]) : 100;
    int warmup = iters / 5 + 5;

    int ndev = 0;
    HIP(hipGetDeviceCount(&ndev));
    CHECK(ndev > 0);
    HIP(hipSetDevice(rank % ndev));
    CHECK(MPIX_Init() == 0);

    int peer = (size >= 2) ? (rank ^ 1) : rank;
    bool self = (peer == rank);
    hipStream_t st;
    HIP(hipStreamCreate(&st));
    char *buf;
    HIP(hipMalloc(&buf, (size_t)1 << max_log2));

    if (rank == 0)
        printf("{\"bench\": \"pingpong\", \"ranks\": %d, \"mode\": \"%s\", "
               "\"sweep\": [\n", size, self ? "self" : "pair");

    for (int lg = 3; lg <= max_log2; lg += 3) {
        size_t n = (size_t)1 << lg;
        int it_n = n > (1 << 20) ? iters / 4 + 4 : iters;

        auto one = [&](int tag) {
            MPIX_Request s_req, r_req;
            if (self) {
                CHECK(MPIX_Isend_enqueue(buf, (int)n, MPI_BYTE, rank, tag,
                                         MPI_COMM_WORLD, &s_req,
                                         MPIX_QUEUE_HIP_STREAM, &st) == 0);
                CHECK(MPIX_Irecv_enqueue(buf, (int)n, MPI_BYTE, rank, tag,
                                         MPI_COMM_WORLD, &r_req,
                                         MPIX_QUEUE_HIP_STREAM, &st) == 0);
                CHECK(MPIX_Wait_enqueue(&s_req, MPI_STATUS_IGNORE,
                                        MPIX_QUEUE_HIP_STREAM, &st) == 0);
                CHECK(MPIX_Wait_enqueue(&r_req, MPI_STATUS_IGNORE,
                                        MPIX_QUEUE_HIP_STREAM, &st) == 0);
            } else if (rank % 2 == 0) {
                CHECK(MPIX_Isend_enqueue(buf, (int)n, MPI_BYTE, peer, tag,
                                         MPI_COMM_WORLD, &s_req,
                                         MPIX_QUEUE_HIP_STREAM, &st) == 0);
                CHECK(MPIX_Wait_enqueue(&s_req, MPI_STATUS_IGNORE,
                                        MPIX_QUEUE_HIP_STREAM, &st) == 0);
                CHECK(MPIX_Irecv_enqueue(buf, (int)n, MPI_BYTE, peer, tag,
                                         MPI_COMM_WORLD, &r_req,
                                         MPIX_QUEUE_HIP_STREAM, &st) == 0);
                CHECK(MPIX_Wait_enqueue(&r_req, MPI_STATUS_IGNORE,
                                        MPIX_QUEUE_HIP_STREAM, &st) == 0);
            } else {
                CHECK(MPIX_Irecv_enqueue(buf, (int)n, MPI_BYTE, peer, tag,
                                         MPI_COMM_WORLD, &r_req,
                                         MPIX_QUEUE_HIP_STREAM, &st) == 0);
                CHECK(MPIX_Wait_enqueue(&r_req, MPI_STATUS_IGNORE,
                                        MPIX_QUEUE_HIP_STREAM, &st) == 0);
                CHECK(MPIX_Isend_enqueue(buf, (int)n, MPI_BYTE, peer, tag,
                                         MPI_COMM_WORLD, &s_req,
                                         MPIX_QUEUE_HIP_STREAM, &st) == 0);
                CHECK(MPIX_Wait_enqueue(&s_req, MPI_STATUS_IGNORE,
                                        MPIX_QUEUE_HIP_STREAM, &st) == 0);
            }
            HIP(hipStreamSynchronize(st));
        };

        for (int i = 0; i < warmup; i++) one(i);
        MPI_Barrier(MPI_COMM_WORLD);
        auto t0 = std::chrono::steady_clock::now();
        for (int i = 0; i < it_n; i++) one(warmup + i);
        double dt = std::chrono::duration<double>(
                        std::chrono::steady_clock::now() - t0).count();
        double mx;
        MPI_Allreduce(&dt, &mx, 1, MPI_DOUBLE, MPI_MAX, MPI_COMM_WORLD);
        /* pair iteration = 2 messages (1 RTT); self iteration = 1 message */
        double legs = self ? 1.0 : 2.0;
        double half_rtt_us = mx / it_n / legs * 1e6;
        double gbps = n * legs * it_n / mx / 1e9;
        if (rank == 0)
            printf("  {\"bytes\": %zu, \"half_rtt_us\": %.3f, "
                   "\"gbps\": %.3f, \"iters\": %d}%s\n",
                   n, half_rtt_us, gbps, it_n, lg + 3 <= max_log2 ? "," : "");
    }
    if (rank == 0) printf("]}\n");

    (void)hipFree(buf);
    (void)hipStreamDestroy(st);
    MPIX_Finalize();
    MPI_Finalize();
    return 0;
}
