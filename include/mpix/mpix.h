/* mpix — accelerator-triggered MPI extensions for AMD Instinct MI355X (gfx950)
 *
 * Public C API. Provides the same 17-entry-point MPIX_* surface as NVIDIA's
 * MPI-ACX prototype (reference: /root/reference/include/mpi-acx.h:48-104),
 * re-designed for ROCm/HIP on CDNA4:
 *
 *   - Stream/graph-triggered point-to-point: MPIX_Isend_enqueue /
 *     MPIX_Irecv_enqueue / MPIX_Wait[all]_enqueue ordered by a HIP stream or
 *     HIP graph instead of the host.
 *   - Kernel-triggered partitioned communication: MPIX_Psend_init /
 *     MPIX_Precv_init plus __device__ MPIX_Pready / MPIX_Parrived
 *     (see mpix_device.h) so a GPU kernel can publish / poll individual
 *     partitions of a large message.
 *
 * Core mechanism (same protocol family as the reference, new implementation):
 * a CPU proxy thread polls 32-bit flag words in host-pinned, device-mapped
 * memory.  The GPU (via hipStream memOps or tiny gfx950 kernels) flips a flag
 * to PENDING; the proxy issues the communication on the GPU's behalf and
 * flips the flag to COMPLETED, which the GPU (or host) is waiting on.
 *
 * Unlike the reference, the data plane is NOT required to be a GPU-aware MPI:
 * intra-node transfers between GPUs use a native xGMI path (HIP IPC +
 * peer-to-peer SDMA copies) and host transfers use a shared-memory channel,
 * with host-MPI passthrough available when the library is initialized inside
 * an MPI program.  See README.md / ARCHITECTURE.md.
 */
#ifndef MPIX_H
#define MPIX_H

#include <mpi.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef void *MPIX_Request;
typedef void *MPIX_Prequest;

#define MPIX_REQUEST_NULL  NULL
#define MPIX_PREQUEST_NULL NULL

/* Library init / teardown.  May be called either
 *  (a) after MPI_Init_thread(MPI_THREAD_MULTIPLE)  — "MPI mode": rank/size
 *      come from MPI_COMM_WORLD and host-buffer traffic may ride MPI, or
 *  (b) without MPI, with RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT in the
 *      environment (torchrun-style)                — "env mode": the native
 *      shared-memory + xGMI data plane carries everything.
 * Both modes work with zero GPUs (host-buffer, proxy-only path). */
int MPIX_Init(void);
int MPIX_Finalize(void);

/* Runtime introspection (not in the reference API): how MPIX_Init resolved
 * this process — GPU presence, hipStream memOps probe results, bootstrap
 * mode.  Any pointer may be NULL. */
int MPIX_Query_config(int *have_gpu, int *use_memops, int *use_batch_memops,
                      int *mpi_mode, int *nflags);

/* ENQUEUED OPERATIONS ******************************************************/

enum {
    MPIX_QUEUE_HIP_STREAM, /* queue is a hipStream_t*  */
    MPIX_QUEUE_HIP_GRAPH   /* queue is a hipGraph_t*: the call RETURNS a
                              single-node graph the caller composes */
};

int MPIX_Isend_enqueue(const void *buf, int count, MPI_Datatype datatype,
                       int dest, int tag, MPI_Comm comm, MPIX_Request *request,
                       int qtype, void *queue);

int MPIX_Irecv_enqueue(void *buf, int count, MPI_Datatype datatype,
                       int source, int tag, MPI_Comm comm, MPIX_Request *request,
                       int qtype, void *queue);

int MPIX_Wait_enqueue(MPIX_Request *req, MPI_Status *status, int qtype,
                      void *queue);
int MPIX_Waitall_enqueue(int count, MPIX_Request *reqs, MPI_Status *statuses,
                         int qtype, void *queue);

/* PARTITIONED OPERATIONS ***************************************************/

int MPIX_Psend_init(const void *buf, int partitions, MPI_Count count,
                    MPI_Datatype datatype, int dest, int tag, MPI_Comm comm,
                    MPI_Info info, MPIX_Request *request);

int MPIX_Precv_init(void *buf, int partitions, MPI_Count count,
                    MPI_Datatype datatype, int source, int tag, MPI_Comm comm,
                    MPI_Info info, MPIX_Request *request);

/* Build a device-resident handle for __device__ MPIX_Pready/Parrived. */
int MPIX_Prequest_create(MPIX_Request request, MPIX_Prequest *prequest);
int MPIX_Prequest_free(MPIX_Prequest *prequest);

/* HELPERS FOR PARTITIONED OPERATIONS ***************************************/

int MPIX_Start(MPIX_Request *request);
int MPIX_Startall(int count, MPIX_Request *request);

int MPIX_Wait(MPIX_Request *req, MPI_Status *status);
int MPIX_Waitall(int count, MPIX_Request *reqs, MPI_Status *statuses);

int MPIX_Request_free(MPIX_Request *request);

/* Host-side partitioned publish / poll (device versions in mpix_device.h).
 * The request argument is void* so one signature serves host (MPIX_Request*)
 * and device (MPIX_Prequest*) — reference parity: mpi-acx.h:96-103. */
int MPIX_Pready(int partition, void *request);
int MPIX_Parrived(void *request, int partition, int *flag);

#ifdef __cplusplus
}
#endif

#if defined(__HIPCC__) && defined(__HIP_DEVICE_COMPILE__)
/* device overloads live in mpix_device.h; include it from HIP TUs */
#endif

#endif /* MPIX_H */
