"""mpix — MI355X-native accelerator-triggered MPI extensions.

Python surface over the MPIX_* C API (see include/mpix/mpix.h).  Buffers may
be torch tensors (CPU or HIP), numpy arrays, or raw (address, nbytes) pairs;
streams are torch.cuda.Stream objects or raw hipStream_t addresses.

The native extension (mpix/_C.so) is REQUIRED — there is no Python/eager
fallback.  Build with `make python` (hipcc, gfx950).
"""
from __future__ import annotations

import os

# Load torch's bundled HIP runtime FIRST if torch is present: torch ships its
# own libamdhip64 (same SONAME libamdhip64.so.7 as /opt/rocm's). If _C.so
# loads the /opt/rocm runtime first, torch later loads a SECOND copy (its
# DT_NEEDED is the unversioned "libamdhip64.so", which never matches an
# already-loaded soname) and two HSA runtimes in one process leave
# torch.cuda.is_available() == False on a live GPU. Importing torch first
# means _C.so's libamdhip64.so.7 dependency resolves to torch's
# already-loaded copy and both stacks share one runtime.
try:  # pragma: no cover - torch is optional for pure C-API consumers
    import torch as _torch  # noqa: F401
except ImportError:
    pass

try:
    from . import _C  # type: ignore[attr-defined]
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "mpix native extension not built. Run `make python` at the repo root "
        f"(hipcc --offload-arch=gfx950). Original error: {e}"
    ) from e

QUEUE_STREAM = _C.QUEUE_STREAM
QUEUE_GRAPH = _C.QUEUE_GRAPH
ANY_SOURCE = _C.ANY_SOURCE
ANY_TAG = _C.ANY_TAG

Request = _C.Request
Status = _C.Status
Prequest = _C.Prequest


def _addr_len(buf, nbytes=None):
    """Resolve (address, nbytes) from a tensor / ndarray / (addr, len)."""
    if hasattr(buf, "data_ptr"):  # torch tensor
        addr = buf.data_ptr()
        n = buf.numel() * buf.element_size()
    elif hasattr(buf, "ctypes"):  # numpy array
        addr = buf.ctypes.data
        n = buf.nbytes
    elif isinstance(buf, tuple):
        addr, n = buf
    else:
        raise TypeError(f"unsupported buffer type {type(buf)}")
    if nbytes is not None:
        n = nbytes
    return int(addr), int(n)


def _stream_addr(stream):
    if stream is None:
        return 0
    if hasattr(stream, "cuda_stream"):  # torch.cuda.Stream
        return int(stream.cuda_stream)
    return int(stream)


def init():
    """Initialize mpix.  Call after MPI_Init (MPI mode) or with
    RANK/WORLD_SIZE/MASTER_ADDR in the environment (torchrun mode)."""
    _C.init()


def finalize():
    _C.finalize()


def world():
    """(rank, world_size)."""
    return _C.world()


def have_gpu():
    return _C.have_gpu()


def config():
    """Resolved runtime config (after init): gpu, memOps probe, mode."""
    return _C.config()


def isend_enqueue(buf, dest, tag=0, stream=None, nbytes=None):
    addr, n = _addr_len(buf, nbytes)
    return _C.isend_enqueue(addr, n, dest, tag, QUEUE_STREAM,
                            _stream_addr(stream))


def irecv_enqueue(buf, source, tag=0, stream=None, nbytes=None):
    addr, n = _addr_len(buf, nbytes)
    return _C.irecv_enqueue(addr, n, source, tag, QUEUE_STREAM,
                            _stream_addr(stream))


def isend_graph(buf, dest, tag=0, nbytes=None):
    """Graph-construction enqueue: returns (request, graph_handle)."""
    addr, n = _addr_len(buf, nbytes)
    return _C.isend_graph(addr, n, dest, tag)


def irecv_graph(buf, source, tag=0, nbytes=None):
    addr, n = _addr_len(buf, nbytes)
    return _C.irecv_graph(addr, n, source, tag)


def waitall_graph(reqs):
    return _C.waitall_graph(list(reqs))


def wait_graph(req):
    """Per-request wait graph (explicit-construction composition)."""
    return _C.wait_graph(req)


def wait_enqueue(req, stream=None, status=None):
    _C.wait_enqueue(req, status, _stream_addr(stream))


def waitall_enqueue(reqs, stream=None):
    _C.waitall_enqueue(list(reqs), _stream_addr(stream))


def wait(req):
    """Host-side wait; returns a status dict."""
    return _C.wait(req)


def waitall(reqs):
    return [wait(r) for r in reqs]


def request_free(req):
    _C.request_free(req)


# ----------------------------- partitioned --------------------------------

def psend_init(buf, partitions, dest, tag=0, nbytes=None):
    addr, n = _addr_len(buf, nbytes)
    assert n % partitions == 0, "buffer must divide evenly into partitions"
    return _C.psend_init(addr, partitions, n // partitions, dest, tag)


def precv_init(buf, partitions, source, tag=0, nbytes=None):
    addr, n = _addr_len(buf, nbytes)
    assert n % partitions == 0, "buffer must divide evenly into partitions"
    return _C.precv_init(addr, partitions, n // partitions, source, tag)


def start(req):
    _C.start(req)


def startall(reqs):
    for r in reqs:
        _C.start(r)


def pready(partition, req):
    _C.pready(partition, req)


def parrived(req, partition):
    return _C.parrived(req, partition)


def prequest_create(req):
    return _C.prequest_create(req)


def prequest_free(preq):
    _C.prequest_free(preq)


# ----------------------------- graph helpers ------------------------------

graph_chain_instantiate = _C.graph_chain_instantiate
graph_instantiate = _C.graph_instantiate
graph_launch = _C.graph_launch
graph_exec_destroy = _C.graph_exec_destroy
graph_destroy = _C.graph_destroy
stream_begin_capture = _C.stream_begin_capture
stream_end_capture = _C.stream_end_capture

# ----------------------------- test kernels -------------------------------

launch_pready_all = _C.launch_pready_all
launch_wait_arrived_all = _C.launch_wait_arrived_all
launch_fill_and_pready = _C.launch_fill_and_pready
launch_wait_and_check = _C.launch_wait_and_check
