/* mpix — __device__ partitioned-communication functions for gfx950 (CDNA4).
 *
 * Reference parity: __device__ MPIX_Pready / MPIX_Parrived
 * (/root/reference/src/partitioned.cu:200-231), rebuilt as CDNA4 HIP:
 *
 *  - The flag pool is host-pinned fine-grained memory.  Device stores/loads
 *    use __hip_atomic_* with __HIP_MEMORY_SCOPE_SYSTEM so they are uncached
 *    on the device side and visible to the CPU proxy without any flush.
 *  - MPIX_Pready is a system-scope RELEASE store: all of the kernel's prior
 *    writes to the partition's payload (in HBM) are ordered before the flag
 *    becomes visible to the proxy thread that will issue the transfer.
 *  - MPIX_Parrived is a relaxed poll + ACQUIRE fence on success, so payload
 *    reads issued after a true return see the peer's data.
 *  - MPIX_Parrived_spin adds an s_sleep backoff so a polling wave does not
 *    hammer the PCIe/xGMI host link (64-wide wavefronts: one lane polls,
 *    the wave sleeps between probes).
 *
 * Include this header from HIP translation units (device compile).
 */
#ifndef MPIX_DEVICE_H
#define MPIX_DEVICE_H

#include <hip/hip_runtime.h>
#include "mpix_abi.h"

#if defined(__HIPCC__)

extern "C" {

__device__ __forceinline__ int MPIX_Pready(int partition, void *prequest)
{
    mpix_prequest_dev_t *preq = (mpix_prequest_dev_t *)prequest;
    if (preq == nullptr || partition < 0 || partition >= preq->n_partitions)
        return 1;
    uint32_t *flag = &preq->flags[preq->idx[partition]];
    __hip_atomic_store(flag, MPIX_FLAG_PENDING, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_SYSTEM);
    return 0;
}

__device__ __forceinline__ int MPIX_Parrived(void *prequest, int partition,
                                             int *flag_out)
{
    mpix_prequest_dev_t *preq = (mpix_prequest_dev_t *)prequest;
    if (preq == nullptr || partition < 0 || partition >= preq->n_partitions)
        return 1;
    uint32_t *flag = &preq->flags[preq->idx[partition]];
    uint32_t v = __hip_atomic_load(flag, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_SYSTEM);
    int arrived = (v == MPIX_FLAG_COMPLETED);
    if (arrived) {
        /* order subsequent payload reads after the flag observation */
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
    }
    *flag_out = arrived;
    return 0;
}

/* Blocking convenience poll with s_sleep backoff (not in the reference API;
 * the reference's tests open-code this spin — test/src/ring-partitioned.cu:42-47).
 * Safe to call from one lane or a whole wave. */
__device__ __forceinline__ void MPIX_Parrived_spin(void *prequest, int partition)
{
    mpix_prequest_dev_t *preq = (mpix_prequest_dev_t *)prequest;
    uint32_t *flag = &preq->flags[preq->idx[partition]];
    while (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM) != MPIX_FLAG_COMPLETED) {
        __builtin_amdgcn_s_sleep(32); /* ~32*64 clk idle between host-link probes */
    }
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
}

} /* extern "C" */

#endif /* __HIPCC__ */
#endif /* MPIX_DEVICE_H */
