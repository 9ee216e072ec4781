/* mpix — pybind11 bindings over the MPIX_* C API.
 *
 * Deliberately torch-free: buffers and streams come in as raw addresses
 * (torch tensors supply .data_ptr() / stream.cuda_stream on the Python side,
 * see mpix/__init__.py).  Compiled by hipcc so it can also host the small
 * device test kernels that exercise __device__ MPIX_Pready / MPIX_Parrived
 * (the Python analog of the reference's test/src/ring-partitioned.cu:38-47).
 */
#include <hip/hip_runtime.h>

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <stdexcept>
#include <vector>

#include "mpix/mpix.h"
#include "mpix/mpix_device.h"
#include "../src/status_codec.h"

namespace py = pybind11;

#define PY_CHECK(call)                                                        \
    do {                                                                      \
        int _rc = (call);                                                     \
        if (_rc != 0) throw std::runtime_error(#call " failed, rc=" +         \
                                               std::to_string(_rc));          \
    } while (0)

#define PY_CHECK_HIP(call)                                                    \
    do {                                                                      \
        hipError_t _e = (call);                                               \
        if (_e != hipSuccess)                                                 \
            throw std::runtime_error(std::string(#call " failed: ") +         \
                                     hipGetErrorString(_e));                  \
    } while (0)

/* ---------------------------------------------------------------- handles */

struct MxRequest {
    MPIX_Request req = MPIX_REQUEST_NULL;
};

struct MxStatus {
    MPI_Status st{};
    bool valid = false;
};

struct MxPrequest {
    MPIX_Prequest preq = MPIX_PREQUEST_NULL;
};

static py::dict status_to_dict(const MPI_Status &st)
{
    py::dict d;
    d["source"] = st.MPI_SOURCE;
    d["tag"] = st.MPI_TAG;
    d["error"] = st.MPI_ERROR;
    d["count_bytes"] = mpix::status_bytes(st); /* ABI codec, status_codec.h */
    return d;
}

/* ----------------------------------------------------- device test kernels */

__global__ void k_test_pready_all(mpix_prequest_dev_t *preq)
{
    int p = (int)(blockIdx.x * blockDim.x + threadIdx.x);
    if (p < preq->n_partitions) (void)MPIX_Pready(p, preq);
}

__global__ void k_test_wait_arrived_all(mpix_prequest_dev_t *preq)
{
    int p = (int)(blockIdx.x * blockDim.x + threadIdx.x);
    if (p < preq->n_partitions) MPIX_Parrived_spin(preq, p);
}

/* Fill a device buffer with a rank/iteration pattern, publishing each
 * partition with MPIX_Pready as soon as its tile is written — the
 * compute/communication-overlap idiom (one workgroup per partition). */
__global__ void k_test_fill_and_pready(int32_t *buf, int n_per_part,
                                       int32_t base, mpix_prequest_dev_t *preq)
{
    int part = (int)blockIdx.x;
    int64_t off = (int64_t)part * n_per_part;
    for (int i = (int)threadIdx.x; i < n_per_part; i += (int)blockDim.x)
        buf[off + i] = base + part;
    __syncthreads();
    if (threadIdx.x == 0) (void)MPIX_Pready(part, preq);
}

/* Spin until each partition arrives, then verify the payload; one workgroup
 * per partition; errors accumulated into *errs. */
__global__ void k_test_wait_and_check(const int32_t *buf, int n_per_part,
                                      int32_t base, mpix_prequest_dev_t *preq,
                                      int *errs)
{
    int part = (int)blockIdx.x;
    if (threadIdx.x == 0) MPIX_Parrived_spin(preq, part);
    __syncthreads();
    int64_t off = (int64_t)part * n_per_part;
    int bad = 0;
    for (int i = (int)threadIdx.x; i < n_per_part; i += (int)blockDim.x)
        if (buf[off + i] != base + part) bad++;
    if (bad) atomicAdd(errs, bad);
}

/* ----------------------------------------------------------------- module */

PYBIND11_MODULE(_C, m)
{
    m.doc() = "mpix: MI355X-native accelerator-triggered MPI extensions";

    m.attr("QUEUE_STREAM") = (int)MPIX_QUEUE_HIP_STREAM;
    m.attr("QUEUE_GRAPH") = (int)MPIX_QUEUE_HIP_GRAPH;
    m.attr("ANY_SOURCE") = (int)MPI_ANY_SOURCE;
    m.attr("ANY_TAG") = (int)MPI_ANY_TAG;

    py::class_<MxRequest>(m, "Request")
        .def("is_null", [](const MxRequest &r) {
            return r.req == MPIX_REQUEST_NULL;
        });
    py::class_<MxStatus>(m, "Status")
        .def(py::init<>())
        .def("as_dict", [](const MxStatus &s) { return status_to_dict(s.st); });
    py::class_<MxPrequest>(m, "Prequest")
        .def("address", [](const MxPrequest &p) {
            return (uintptr_t)p.preq;
        });

    m.def("init", [] { PY_CHECK(MPIX_Init()); });
    m.def("finalize", [] { PY_CHECK(MPIX_Finalize()); });

    m.def("isend_enqueue",
          [](uintptr_t buf, int64_t nbytes, int dest, int tag, int qtype,
             uintptr_t stream) {
              auto *r = new MxRequest();
              /* stream/graph handle passed by address-of, reference-style */
              void *q = (void *)&stream;
              PY_CHECK(MPIX_Isend_enqueue((const void *)buf, (int)nbytes,
                                          MPI_BYTE, dest, tag, MPI_COMM_WORLD,
                                          &r->req, qtype, q));
              return r;
          },
          py::return_value_policy::take_ownership);

    m.def("irecv_enqueue",
          [](uintptr_t buf, int64_t nbytes, int source, int tag, int qtype,
             uintptr_t stream) {
              auto *r = new MxRequest();
              void *q = (void *)&stream;
              PY_CHECK(MPIX_Irecv_enqueue((void *)buf, (int)nbytes, MPI_BYTE,
                                          source, tag, MPI_COMM_WORLD,
                                          &r->req, qtype, q));
              return r;
          },
          py::return_value_policy::take_ownership);

    /* graph-construction variants: return the created hipGraph_t address */
    m.def("isend_graph", [](uintptr_t buf, int64_t nbytes, int dest, int tag) {
        auto *r = new MxRequest();
        hipGraph_t g = nullptr;
        PY_CHECK(MPIX_Isend_enqueue((const void *)buf, (int)nbytes, MPI_BYTE,
                                    dest, tag, MPI_COMM_WORLD, &r->req,
                                    MPIX_QUEUE_HIP_GRAPH, &g));
        return py::make_tuple(
            py::cast(r, py::return_value_policy::take_ownership),
            (uintptr_t)g);
    });
    m.def("irecv_graph", [](uintptr_t buf, int64_t nbytes, int source, int tag) {
        auto *r = new MxRequest();
        hipGraph_t g = nullptr;
        PY_CHECK(MPIX_Irecv_enqueue((void *)buf, (int)nbytes, MPI_BYTE, source,
                                    tag, MPI_COMM_WORLD, &r->req,
                                    MPIX_QUEUE_HIP_GRAPH, &g));
        return py::make_tuple(
            py::cast(r, py::return_value_policy::take_ownership),
            (uintptr_t)g);
    });
    m.def("wait_graph", [](MxRequest *r) {
        /* per-request wait graph (the reference's composition style in
         * test/src/ring-all-graph-construction.c uses one per request) */
        hipGraph_t g = nullptr;
        PY_CHECK(MPIX_Wait_enqueue(&r->req, MPI_STATUS_IGNORE,
                                   MPIX_QUEUE_HIP_GRAPH, &g));
        return (uintptr_t)g;
    });
    m.def("waitall_graph", [](std::vector<MxRequest *> reqs) {
        std::vector<MPIX_Request> rr;
        for (auto *r : reqs) rr.push_back(r->req);
        hipGraph_t g = nullptr;
        PY_CHECK(MPIX_Waitall_enqueue((int)rr.size(), rr.data(),
                                      MPI_STATUSES_IGNORE,
                                      MPIX_QUEUE_HIP_GRAPH, &g));
        for (size_t i = 0; i < rr.size(); i++) reqs[i]->req = rr[i];
        return (uintptr_t)g;
    });

    m.def("wait_enqueue", [](MxRequest *r, MxStatus *st, uintptr_t stream) {
        PY_CHECK(MPIX_Wait_enqueue(&r->req, st ? &st->st : MPI_STATUS_IGNORE,
                                   MPIX_QUEUE_HIP_STREAM, &stream));
        if (st) st->valid = true;
    }, py::arg("req"), py::arg("status") = nullptr, py::arg("stream") = 0);

    m.def("waitall_enqueue", [](std::vector<MxRequest *> reqs, uintptr_t stream) {
        std::vector<MPIX_Request> rr;
        for (auto *r : reqs) rr.push_back(r->req);
        PY_CHECK(MPIX_Waitall_enqueue((int)rr.size(), rr.data(),
                                      MPI_STATUSES_IGNORE,
                                      MPIX_QUEUE_HIP_STREAM, &stream));
        for (size_t i = 0; i < rr.size(); i++) reqs[i]->req = rr[i];
    });

    m.def("wait", [](MxRequest *r) {
        MPI_Status st;
        memset(&st, 0, sizeof(st));
        {
            py::gil_scoped_release nogil;
            PY_CHECK(MPIX_Wait(&r->req, &st));
        }
        return status_to_dict(st);
    });

    m.def("request_free", [](MxRequest *r) {
        PY_CHECK(MPIX_Request_free(&r->req));
    });

    /* ------------------------------ partitioned ------------------------- */

    m.def("psend_init", [](uintptr_t buf, int partitions, int64_t bytes_per_part,
                           int dest, int tag) {
        auto *r = new MxRequest();
        PY_CHECK(MPIX_Psend_init((const void *)buf, partitions,
                                 (MPI_Count)bytes_per_part, MPI_BYTE, dest,
                                 tag, MPI_COMM_WORLD, MPI_INFO_NULL, &r->req));
        return r;
    }, py::return_value_policy::take_ownership);

    m.def("precv_init", [](uintptr_t buf, int partitions, int64_t bytes_per_part,
                           int source, int tag) {
        auto *r = new MxRequest();
        PY_CHECK(MPIX_Precv_init((void *)buf, partitions,
                                 (MPI_Count)bytes_per_part, MPI_BYTE, source,
                                 tag, MPI_COMM_WORLD, MPI_INFO_NULL, &r->req));
        return r;
    }, py::return_value_policy::take_ownership);

    m.def("start", [](MxRequest *r) { PY_CHECK(MPIX_Start(&r->req)); });
    m.def("pready", [](int partition, MxRequest *r) {
        PY_CHECK(MPIX_Pready(partition, (void *)r->req));
    });
    m.def("parrived", [](MxRequest *r, int partition) {
        int f = 0;
        PY_CHECK(MPIX_Parrived((void *)r->req, partition, &f));
        return (bool)f;
    });
    m.def("prequest_create", [](MxRequest *r) {
        auto *p = new MxPrequest();
        PY_CHECK(MPIX_Prequest_create(r->req, &p->preq));
        return p;
    }, py::return_value_policy::take_ownership);
    m.def("prequest_free", [](MxPrequest *p) {
        PY_CHECK(MPIX_Prequest_free(&p->preq));
    });

    /* ------------------------------ introspection ----------------------- */

    m.def("world", [] {
        int rank = 0, size = 1;
        const char *r = getenv("RANK"), *s = getenv("WORLD_SIZE");
        int mi = 0;
        MPI_Initialized(&mi);
        if (mi) {
            MPI_Comm_rank(MPI_COMM_WORLD, &rank);
            MPI_Comm_size(MPI_COMM_WORLD, &size);
        } else {
            if (r) rank = atoi(r);
            if (s) size = atoi(s);
        }
        return py::make_tuple(rank, size);
    });
    m.def("config", [] {
        int gpu = 0, mo = 0, bmo = 0, mm = 0, nf = 0;
        PY_CHECK(MPIX_Query_config(&gpu, &mo, &bmo, &mm, &nf));
        py::dict d;
        d["have_gpu"] = (bool)gpu;
        d["use_memops"] = (bool)mo;
        d["use_batch_memops"] = (bool)bmo;
        d["mpi_mode"] = (bool)mm;
        d["nflags"] = nf;
        return d;
    });
    m.def("have_gpu", [] {
        int n = 0;
        if (hipGetDeviceCount(&n) != hipSuccess) { (void)hipGetLastError(); n = 0; }
        return n > 0;
    });

    /* ------------------------------ graph helpers ----------------------- */

    m.def("graph_chain_instantiate", [](std::vector<uintptr_t> graphs) {
        /* compose child graphs in dependency order (the reference's
         * ring-all-graph-construction.c:74-96 pattern) and instantiate */
        hipGraph_t parent = nullptr;
        PY_CHECK_HIP(hipGraphCreate(&parent, 0));
        hipGraphNode_t prev = nullptr;
        for (uintptr_t ga : graphs) {
            hipGraphNode_t node = nullptr;
            PY_CHECK_HIP(hipGraphAddChildGraphNode(
                &node, parent, prev ? &prev : nullptr, prev ? 1 : 0,
                (hipGraph_t)ga));
            prev = node;
        }
        hipGraphExec_t exec = nullptr;
        PY_CHECK_HIP(hipGraphInstantiate(&exec, parent, nullptr, nullptr, 0));
        return py::make_tuple((uintptr_t)parent, (uintptr_t)exec);
    });
    m.def("graph_instantiate", [](uintptr_t g) {
        hipGraphExec_t exec = nullptr;
        PY_CHECK_HIP(hipGraphInstantiate(&exec, (hipGraph_t)g, nullptr,
                                         nullptr, 0));
        return (uintptr_t)exec;
    });
    m.def("graph_launch", [](uintptr_t exec, uintptr_t stream) {
        PY_CHECK_HIP(hipGraphLaunch((hipGraphExec_t)exec,
                                    (hipStream_t)stream));
    });
    m.def("graph_exec_destroy", [](uintptr_t exec) {
        PY_CHECK_HIP(hipGraphExecDestroy((hipGraphExec_t)exec));
    });
    m.def("graph_destroy", [](uintptr_t g) {
        PY_CHECK_HIP(hipGraphDestroy((hipGraph_t)g));
    });
    m.def("stream_begin_capture", [](uintptr_t stream) {
        PY_CHECK_HIP(hipStreamBeginCapture((hipStream_t)stream,
                                           hipStreamCaptureModeGlobal));
    });
    m.def("stream_end_capture", [](uintptr_t stream) {
        hipGraph_t g = nullptr;
        PY_CHECK_HIP(hipStreamEndCapture((hipStream_t)stream, &g));
        hipGraphExec_t exec = nullptr;
        PY_CHECK_HIP(hipGraphInstantiate(&exec, g, nullptr, nullptr, 0));
        return py::make_tuple((uintptr_t)g, (uintptr_t)exec);
    });

    /* ------------------------------ test kernels ------------------------ */

    m.def("launch_pready_all", [](MxPrequest *p, int partitions,
                                  uintptr_t stream) {
        int threads = 64;
        int blocks = (partitions + threads - 1) / threads;
        hipLaunchKernelGGL(k_test_pready_all, dim3(blocks), dim3(threads), 0,
                           (hipStream_t)stream,
                           (mpix_prequest_dev_t *)p->preq);
        PY_CHECK_HIP(hipGetLastError());
    });
    m.def("launch_wait_arrived_all", [](MxPrequest *p, int partitions,
                                        uintptr_t stream) {
        int threads = 64;
        int blocks = (partitions + threads - 1) / threads;
        hipLaunchKernelGGL(k_test_wait_arrived_all, dim3(blocks), dim3(threads),
                           0, (hipStream_t)stream,
                           (mpix_prequest_dev_t *)p->preq);
        PY_CHECK_HIP(hipGetLastError());
    });
    m.def("launch_fill_and_pready", [](uintptr_t buf, int n_per_part,
                                       int32_t base, MxPrequest *p,
                                       int partitions, uintptr_t stream) {
        hipLaunchKernelGGL(k_test_fill_and_pready, dim3(partitions), dim3(256),
                           0, (hipStream_t)stream, (int32_t *)buf, n_per_part,
                           base, (mpix_prequest_dev_t *)p->preq);
        PY_CHECK_HIP(hipGetLastError());
    });
    m.def("launch_wait_and_check", [](uintptr_t buf, int n_per_part,
                                      int32_t base, MxPrequest *p,
                                      int partitions, uintptr_t errs,
                                      uintptr_t stream) {
        hipLaunchKernelGGL(k_test_wait_and_check, dim3(partitions), dim3(256),
                           0, (hipStream_t)stream, (const int32_t *)buf,
                           n_per_part, base, (mpix_prequest_dev_t *)p->preq,
                           (int *)errs);
        PY_CHECK_HIP(hipGetLastError());
    });
}
