/* Diagnostic: host<->device flag visibility on gfx950, every spin bounded
 * by a cycle budget so a failed direction can NEVER wedge the GPU.
 *
 *   A. device k_set -> host load observes            (GPU->host)
 *   B. host store -> device spin kernel observes     (host->GPU, the
 *      direction the enqueue/partitioned kernel paths depend on)
 *   C. waitall-style: one wave, lanes spin on distinct flags written by
 *      the host at staggered times
 *   D. B with plain volatile load (no system-scope atomic) for comparison
 *
 * Prints PASS/FAIL per direction.  Exit code = number of failures of the
 * directions the library requires (A, B, C).
 */
#include <chrono>
#include <cstdio>
#include <thread>

#include <hip/hip_runtime.h>

#define HIP(call)                                                         \
    do {                                                                  \
        hipError_t e_ = (call);                                           \
        if (e_ != hipSuccess) {                                           \
            fprintf(stderr, "%s:%d %s: %s\n", __FILE__, __LINE__, #call,  \
                    hipGetErrorString(e_));                               \
            return 2;                                                     \
        }                                                                 \
    } while (0)

/* ~3e9 cycles at 2 GHz ≈ 1.5 s bound per spin */
#define SPIN_BUDGET 3000000000ull

__device__ static inline uint64_t rt_clock()
{
    return __builtin_amdgcn_s_memtime();
}

__global__ void k_set(uint32_t *flag, uint32_t val)
{
    __hip_atomic_store(flag, val, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}

__global__ void k_spin_atomic(uint32_t *flag, uint32_t val, int *ok)
{
    uint64_t t0 = rt_clock();
    while (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM) != val) {
        __builtin_amdgcn_s_sleep(16);
        if (rt_clock() - t0 > SPIN_BUDGET) { *ok = 0; return; }
    }
    *ok = 1;
}

__global__ void k_spin_volatile(volatile uint32_t *flag, uint32_t val, int *ok)
{
    uint64_t t0 = rt_clock();
    while (*flag != val) {
        __builtin_amdgcn_s_sleep(16);
        if (rt_clock() - t0 > SPIN_BUDGET) { *ok = 0; return; }
    }
    *ok = 1;
}

__global__ void k_spin_many(uint32_t *flags, int count, uint32_t val, int *ok)
{
    int i = threadIdx.x;
    if (i < count) {
        uint64_t t0 = rt_clock();
        while (__hip_atomic_load(&flags[i], __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_SYSTEM) != val) {
            __builtin_amdgcn_s_sleep(16);
            if (rt_clock() - t0 > SPIN_BUDGET) { atomicAnd(ok, 0); return; }
        }
    }
}

int main()
{
    uint32_t *flags_h = nullptr;
    HIP(hipHostMalloc((void **)&flags_h, 64 * sizeof(uint32_t),
                      hipHostMallocMapped));
    uint32_t *flags_d = nullptr;
    HIP(hipHostGetDevicePointer((void **)&flags_d, flags_h, 0));
    int *ok_h = nullptr;
    HIP(hipHostMalloc((void **)&ok_h, sizeof(int), hipHostMallocMapped));
    int *ok_d = nullptr;
    HIP(hipHostGetDevicePointer((void **)&ok_d, ok_h, 0));
    hipStream_t st;
    HIP(hipStreamCreate(&st));
    int fails = 0;

    /* A: device -> host */
    flags_h[0] = 0;
    hipLaunchKernelGGL(k_set, dim3(1), dim3(1), 0, st, flags_d, 42u);
    HIP(hipStreamSynchronize(st));
    bool a = (__atomic_load_n(&flags_h[0], __ATOMIC_ACQUIRE) == 42u);
    printf("A device-store -> host-load      : %s\n", a ? "PASS" : "FAIL");
    if (!a) fails++;

    /* B: host -> device atomic spin */
    flags_h[1] = 0; *ok_h = -1;
    hipLaunchKernelGGL(k_spin_atomic, dim3(1), dim3(1), 0, st, flags_d + 1,
                       7u, ok_d);
    std::this_thread::sleep_for(std::chrono::milliseconds(100));
    __atomic_store_n(&flags_h[1], 7u, __ATOMIC_RELEASE);
    HIP(hipStreamSynchronize(st));
    printf("B host-store -> device atomic    : %s\n", *ok_h == 1 ? "PASS" : "FAIL");
    if (*ok_h != 1) fails++;

    /* C: one wave, many flags, staggered host stores */
    for (int i = 0; i < 16; i++) flags_h[8 + i] = 0;
    *ok_h = 1;
    hipLaunchKernelGGL(k_spin_many, dim3(1), dim3(64), 0, st, flags_d + 8,
                       16, 9u, ok_d);
    for (int i = 0; i < 16; i++) {
        std::this_thread::sleep_for(std::chrono::milliseconds(5));
        __atomic_store_n(&flags_h[8 + i], 9u, __ATOMIC_RELEASE);
    }
    HIP(hipStreamSynchronize(st));
    printf("C host-stores -> waitall wave    : %s\n", *ok_h == 1 ? "PASS" : "FAIL");
    if (*ok_h != 1) fails++;

    /* D: volatile variant (informational) */
    flags_h[2] = 0; *ok_h = -1;
    hipLaunchKernelGGL(k_spin_volatile, dim3(1), dim3(1), 0, st, flags_d + 2,
                       5u, ok_d);
    std::this_thread::sleep_for(std::chrono::milliseconds(100));
    __atomic_store_n(&flags_h[2], 5u, __ATOMIC_RELEASE);
    HIP(hipStreamSynchronize(st));
    printf("D host-store -> device volatile  : %s (informational)\n",
           *ok_h == 1 ? "PASS" : "FAIL");

    (void)hipStreamDestroy(st);
    (void)hipHostFree(flags_h);
    (void)hipHostFree(ok_h);
    printf("spinvis fails=%d\n", fails);
    return fails;
}
