/* mpix — shared host/device ABI: flag state machine + device prequest layout.
 *
 * The 6-state per-slot lifecycle mirrors the reference's protocol
 * (/root/reference/include/mpi-acx-internal.h:143-210) but is implemented
 * with C11/HIP atomics rather than volatile int:
 *
 *   AVAILABLE  slot free
 *   RESERVED   slot allocated, operation described, trigger not yet fired
 *   PENDING    trigger fired (GPU stream/kernel or host) — proxy must issue
 *   ISSUED     proxy handed the operation to the transport; completion pending
 *   COMPLETED  transport finished; waiters may proceed
 *   CLEANUP    waiter consumed the completion; proxy frees the slot
 */
#ifndef MPIX_ABI_H
#define MPIX_ABI_H

#include <stdint.h>

#define MPIX_FLAG_AVAILABLE 0u
#define MPIX_FLAG_RESERVED  1u
#define MPIX_FLAG_PENDING   2u
#define MPIX_FLAG_ISSUED    3u
#define MPIX_FLAG_COMPLETED 4u
#define MPIX_FLAG_CLEANUP   5u

/* Device-resident partitioned-request handle, built by MPIX_Prequest_create.
 * Lives in device memory; `idx` maps partition -> flag-pool slot; `flags` is
 * the device-mapped alias of the host-pinned flag pool. */
typedef struct mpix_prequest_dev {
    int32_t n_partitions;
    int32_t _pad;
    const int32_t *idx;
    uint32_t *flags;
} mpix_prequest_dev_t;

#endif /* MPIX_ABI_H */
