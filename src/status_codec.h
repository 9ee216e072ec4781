/* mpix — MPI_Status field codec: the one place that knows the host MPI's
 * ABI for status objects and builtin datatype handles.
 *
 * Why this exists: mpix fills MPI_Status objects ITSELF (the native
 * shm/xGMI transport completes receives without any MPI call), and in env
 * (torchrun) mode it must size builtin datatypes before/without MPI_Init.
 * Both require ABI knowledge the MPI standard does not expose:
 *
 *  - MPICH (and ABI-compatible derivatives: Intel MPI, MVAPICH, Cray MPT)
 *    lay MPI_Status out as {int count_lo; int count_hi_and_cancelled;
 *    int MPI_SOURCE; int MPI_TAG; int MPI_ERROR;} with the received BYTE
 *    count in count_lo, and encode builtin-datatype sizes in handle bits
 *    8..15 (e.g. MPI_INT = 0x4c000405 -> 4 bytes).
 *  - Open MPI's MPI_Status is {int MPI_SOURCE; int MPI_TAG; int MPI_ERROR;
 *    int _cancelled; size_t _ucount;} with the byte count in _ucount, and
 *    datatype handles are pointers (no size in the handle).
 *
 * The build autodetects MPICH via its header macro; an Open MPI build gets
 * the OMPI branch below (set-count via _ucount; datatype sizing falls back
 * to MPI_Type_size, which on OMPI works after MPI_Init only — env mode on
 * OMPI additionally uses a small table of the predefined handles).  If
 * neither ABI is recognized the build fails loudly rather than writing
 * through a wrong layout.
 */
#ifndef MPIX_STATUS_CODEC_H
#define MPIX_STATUS_CODEC_H

#include <cstring>

#include <mpi.h>

namespace mpix {

/* ------------------------------------------------------------ status write */

/* Write source/tag/error and the received byte count into an MPI_Status so
 * that MPI_Get_count / MPI_Get_elements work on it. */
static inline void status_encode(MPI_Status *st, int src, int tag, int err,
                                 unsigned long long bytes)
{
#if defined(MPICH) || defined(MPICH_VERSION) || defined(MPICH2)
    /* MPICH ABI: byte count in count_lo (+ count_hi_and_cancelled 0) */
    memset(st, 0, sizeof(*st));
    st->count_lo = (int)bytes;
    st->count_hi_and_cancelled = 0;
    st->MPI_SOURCE = src;
    st->MPI_TAG = tag;
    st->MPI_ERROR = err;
#elif defined(OPEN_MPI)
    /* Open MPI ABI: byte count in _ucount */
    memset(st, 0, sizeof(*st));
    st->MPI_SOURCE = src;
    st->MPI_TAG = tag;
    st->MPI_ERROR = err;
    st->_ucount = (size_t)bytes;
#else
#error "mpix: unrecognized MPI ABI — add a status_encode branch for it"
#endif
}

/* Read back the byte count from a status we (or the MPI library) filled. */
static inline unsigned long long status_bytes(const MPI_Status &st)
{
#if defined(MPICH) || defined(MPICH_VERSION) || defined(MPICH2)
    return (unsigned long long)(unsigned int)st.count_lo;
#elif defined(OPEN_MPI)
    return (unsigned long long)st._ucount;
#else
#error "mpix: unrecognized MPI ABI — add a status_bytes branch for it"
#endif
}

/* --------------------------------------------------------- datatype sizing */

/* Size of a builtin datatype without requiring MPI_Init (env mode). */
static inline int builtin_datatype_size(MPI_Datatype dt)
{
#if defined(MPICH) || defined(MPICH_VERSION) || defined(MPICH2)
    /* builtin handles are 0x4c00ssii: size in bits 8..15 */
    unsigned long h = (unsigned long)(uintptr_t)dt;
    if ((h & 0xff000000ul) == 0x4c000000ul) {
        int sz = (int)((h >> 8) & 0xfful);
        if (sz > 0) return sz;
    }
    return -1;
#else
    /* pointer-handle ABIs: compare against the predefined globals */
    if (dt == MPI_BYTE || dt == MPI_CHAR || dt == MPI_SIGNED_CHAR ||
        dt == MPI_UNSIGNED_CHAR || dt == MPI_INT8_T || dt == MPI_UINT8_T)
        return 1;
    if (dt == MPI_SHORT || dt == MPI_UNSIGNED_SHORT || dt == MPI_INT16_T ||
        dt == MPI_UINT16_T)
        return 2;
    if (dt == MPI_INT || dt == MPI_UNSIGNED || dt == MPI_FLOAT ||
        dt == MPI_INT32_T || dt == MPI_UINT32_T)
        return 4;
    if (dt == MPI_LONG || dt == MPI_UNSIGNED_LONG || dt == MPI_DOUBLE ||
        dt == MPI_INT64_T || dt == MPI_UINT64_T || dt == MPI_LONG_LONG ||
        dt == MPI_UNSIGNED_LONG_LONG)
        return 8;
    return -1;
#endif
}

} /* namespace mpix */

#endif /* MPIX_STATUS_CODEC_H */
