"""Multi-process CPU ring tests (proxy-only path, host buffers, env
bootstrap on 127.0.0.1).  Mirrors the reference's integration suite
(test/src/ring.c, ring-all.c) without a GPU; the same ring bodies run
on-device in the gpu-marked tests."""
import numpy as np

from conftest import run_ranks


def _ring_host_wait(rank, size):
    import mpix
    mpix.init()
    try:
        n = 1000
        right = (rank + 1) % size
        left = (rank - 1 + size) % size
        send = np.full(n, rank * 10 + 1, dtype=np.int32)
        recv = np.zeros(n, dtype=np.int32)
        rs = mpix.isend_enqueue(send, dest=right, tag=1)
        rr = mpix.irecv_enqueue(recv, source=left, tag=1)
        st = mpix.wait(rr)
        mpix.wait(rs)
        assert (recv == left * 10 + 1).all(), "payload"
        assert st["source"] == left and st["tag"] == 1 and st["error"] == 0
        assert st["count_bytes"] == n * 4
    finally:
        mpix.finalize()


def test_ring_2rank():
    run_ranks(2, _ring_host_wait)


def test_ring_4rank():
    run_ranks(4, _ring_host_wait)


def _ring_multi_iter(rank, size):
    import mpix
    mpix.init()
    try:
        right = (rank + 1) % size
        left = (rank - 1 + size) % size
        for it in range(20):
            send = np.full(256, rank + it * 100, dtype=np.int32)
            recv = np.zeros(256, dtype=np.int32)
            rr = mpix.irecv_enqueue(recv, source=left, tag=it)
            rs = mpix.isend_enqueue(send, dest=right, tag=it)
            mpix.wait(rr)
            mpix.wait(rs)
            assert (recv == left + it * 100).all()
    finally:
        mpix.finalize()


def test_ring_many_iterations():
    run_ranks(2, _ring_multi_iter)


def _ring_large(rank, size):
    import mpix
    mpix.init()
    try:
        right = (rank + 1) % size
        left = (rank - 1 + size) % size
        n = 5 * 1024 * 1024  # 20 MiB: crosses the 4 MiB staging ring
        rng = np.random.default_rng(seed=rank)
        send = rng.integers(0, 2**31 - 1, n, dtype=np.int32)
        recv = np.zeros(n, dtype=np.int32)
        rr = mpix.irecv_enqueue(recv, source=left, tag=0)
        rs = mpix.isend_enqueue(send, dest=right, tag=0)
        mpix.wait(rr)
        mpix.wait(rs)
        expect = np.random.default_rng(seed=left).integers(
            0, 2**31 - 1, n, dtype=np.int32)
        assert (recv == expect).all()
    finally:
        mpix.finalize()


def test_ring_large_chunked():
    run_ranks(2, _ring_large)


def _pingpong(rank, size):
    import mpix
    mpix.init()
    try:
        buf = np.zeros(64, dtype=np.int64)
        for it in range(50):
            if rank == 0:
                buf[:] = it
                rs = mpix.isend_enqueue(buf, dest=1, tag=it)
                mpix.wait(rs)
                rr = mpix.irecv_enqueue(buf, source=1, tag=it)
                mpix.wait(rr)
                assert (buf == it + 1).all()
            else:
                rr = mpix.irecv_enqueue(buf, source=0, tag=it)
                mpix.wait(rr)
                assert (buf == it).all()
                buf[:] = it + 1
                rs = mpix.isend_enqueue(buf, dest=0, tag=it)
                mpix.wait(rs)
    finally:
        mpix.finalize()


def test_pingpong_host():
    run_ranks(2, _pingpong)


def _partitioned_ring(rank, size):
    """Host-triggered partitioned ring, persistent requests reused over
    iterations (reference: test/src/ring-partitioned.cu)."""
    import mpix
    mpix.init()
    try:
        parts, per = 10, 128
        right = (rank + 1) % size
        left = (rank - 1 + size) % size
        send = np.zeros(parts * per, dtype=np.int32)
        recv = np.zeros_like(send)
        ps = mpix.psend_init(send, parts, dest=right, tag=42)
        pr = mpix.precv_init(recv, parts, source=left, tag=42)
        for it in range(10):
            send[:] = rank * 1000 + it
            mpix.start(pr)
            mpix.start(ps)
            for p in range(parts):
                mpix.pready(p, ps)
            # poll parrived on a partition before full wait
            while not mpix.parrived(pr, parts - 1):
                pass
            mpix.wait(pr)
            mpix.wait(ps)
            assert (recv == left * 1000 + it).all()
        mpix.request_free(ps)
        mpix.request_free(pr)
    finally:
        mpix.finalize()


def test_partitioned_ring_2rank():
    run_ranks(2, _partitioned_ring)


def _anysource_gather(rank, size):
    import mpix
    mpix.init()
    try:
        if rank == 0:
            got = set()
            for _ in range(size - 1):
                buf = np.zeros(1, dtype=np.int32)
                rr = mpix.irecv_enqueue(buf, source=mpix.ANY_SOURCE, tag=5)
                st = mpix.wait(rr)
                assert st["source"] == buf[0]
                got.add(int(buf[0]))
            assert got == set(range(1, size))
        else:
            buf = np.array([rank], dtype=np.int32)
            rs = mpix.isend_enqueue(buf, dest=0, tag=5)
            mpix.wait(rs)
    finally:
        mpix.finalize()


def test_any_source_2rank():
    run_ranks(3, _anysource_gather)
