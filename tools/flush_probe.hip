/* FLUSH_REMOTE_WRITES decision probe (SURVEY.md hard-part 4).
 *
 * The reference optionally passes CU_STREAM_WAIT_VALUE_FLUSH to its stream
 * wait (/root/reference/src/sendrecv.cu:377-380) so that DATA written by a
 * remote agent before the flag is guaranteed visible to kernels running
 * after the wait.  ROCm documents hipStreamWaitValueFlush as unsupported,
 * so mpix must establish whether the MI355X needs any flush analog for its
 * protocol: proxy (CPU) writes payload into host-pinned memory, then
 * release-stores the flag; the GPU passes hipStreamWaitValue32 (or the
 * k_wait acquire-fenced spin kernel) and reads the payload.
 *
 * This probe runs that exact pattern back-to-back TRIALS times for both
 * wait mechanisms and for device-resident payload destinations (host write
 * -> pinned staging -> device reads after flag), counting stale reads.
 * Result feeds profiles/ and the ARCHITECTURE notes: zero stale reads =>
 * host-pinned fine-grained memory is coherent with system-scope acquire on
 * gfx950 and no flush op is required.
 */
#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <thread>

#include <hip/hip_runtime.h>

#define HIP(c)                                                            \
    do {                                                                  \
        hipError_t e_ = (c);                                              \
        if (e_ != hipSuccess) {                                           \
            fprintf(stderr, "%s:%d %s: %s\n", __FILE__, __LINE__, #c,     \
                    hipGetErrorString(e_));                               \
            exit(1);                                                      \
        }                                                                 \
    } while (0)

#define N 1024
#define TRIALS 2000

__global__ void k_wait_flag_acq(uint32_t *flag, uint32_t val)
{
    while (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM) < val)
        __builtin_amdgcn_s_sleep(8);
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
}

/* read payload AFTER the wait (separate kernel: stream order is the only
 * ordering, exactly like user kernels after MPIX_Wait_enqueue) */
__global__ void k_check(const uint32_t *payload, uint32_t want, int *errs)
{
    int i = (int)(blockIdx.x * blockDim.x + threadIdx.x);
    if (i < N && payload[i] != want) atomicAdd(errs, 1);
}

int main(int argc, char **argv)
{
    int use_memops = argc > 1 ? atoi(argv[1]) : 1;

    uint32_t *payload_h, *payload_d, *flag_h, *flag_d;
    HIP(hipHostMalloc((void **)&payload_h, N * sizeof(uint32_t),
                      hipHostMallocMapped));
    HIP(hipHostGetDevicePointer((void **)&payload_d, payload_h, 0));
    HIP(hipHostMalloc((void **)&flag_h, sizeof(uint32_t), hipHostMallocMapped));
    HIP(hipHostGetDevicePointer((void **)&flag_d, flag_h, 0));
    int *errs;
    HIP(hipMalloc(&errs, sizeof(int)));
    HIP(hipMemset(errs, 0, sizeof(int)));
    *flag_h = 0;

    hipStream_t st;
    HIP(hipStreamCreateWithFlags(&st, hipStreamNonBlocking));

    /* writer thread plays the proxy: payload store -> release flag store */
    std::atomic<uint32_t> go{0};
    std::thread writer([&] {
        for (uint32_t t = 1; t <= TRIALS; t++) {
            while (go.load(std::memory_order_acquire) < t)
                ;
            for (int i = 0; i < N; i++) payload_h[i] = t;
            std::atomic_thread_fence(std::memory_order_release);
            __atomic_store_n(flag_h, t, __ATOMIC_RELEASE);
        }
    });

    auto t0 = std::chrono::steady_clock::now();
    for (uint32_t t = 1; t <= TRIALS; t++) {
        if (use_memops) {
            HIP(hipStreamWaitValue32(st, flag_d, t, hipStreamWaitValueGte,
                                     0xFFFFFFFFu));
        } else {
            hipLaunchKernelGGL(k_wait_flag_acq, dim3(1), dim3(1), 0, st,
                               flag_d, t);
            HIP(hipGetLastError());
        }
        hipLaunchKernelGGL(k_check, dim3((N + 255) / 256), dim3(256), 0, st,
                           payload_d, t, errs);
        HIP(hipGetLastError());
        go.store(t, std::memory_order_release); /* writer may fire any time
                                                   relative to the wait */
        HIP(hipStreamSynchronize(st));
    }
    double us = std::chrono::duration<double, std::micro>(
                    std::chrono::steady_clock::now() - t0).count() / TRIALS;
    writer.join();

    int h_errs = -1;
    HIP(hipMemcpy(&h_errs, errs, sizeof(int), hipMemcpyDeviceToHost));
    printf("{\"probe\": \"flush_remote_writes\", \"mech\": \"%s\", "
           "\"trials\": %d, \"stale_reads\": %d, \"wait_roundtrip_us\": %.2f, "
           "\"verdict\": \"%s\"}\n",
           use_memops ? "hipStreamWaitValue32" : "spin-kernel", TRIALS, h_errs,
           us, h_errs == 0 ? "no flush needed" : "FLUSH REQUIRED");
    return h_errs == 0 ? 0 : 2;
}
