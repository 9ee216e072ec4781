"""MPIX_FAST_WAIT=1 tier: the epoch/GTE wait protocol (single-memOp waits,
slots recycled at completion).  Runs the protocol-critical subset in fast
mode so both state machines stay covered by default CI."""
import os

import numpy as np
import pytest

from conftest import run_ranks


@pytest.fixture
def mpix_fast(monkeypatch):
    monkeypatch.setenv("RANK", "0")
    monkeypatch.setenv("WORLD_SIZE", "1")
    monkeypatch.setenv("MPIX_FAST_WAIT", "1")
    import mpix
    mpix.init()
    assert mpix.config()["nflags"] >= 64
    yield mpix
    mpix.finalize()


def test_fast_roundtrip_status(mpix_fast):
    mpix = mpix_fast
    src = np.arange(50, dtype=np.int32)
    dst = np.zeros(50, dtype=np.int32)
    rs = mpix.isend_enqueue(src, dest=0, tag=5)
    rr = mpix.irecv_enqueue(dst, source=0, tag=5)
    st = mpix.wait(rr)
    mpix.wait(rs)
    assert (dst == src).all()
    assert st["source"] == 0 and st["tag"] == 5 and st["count_bytes"] == 200


def test_fast_wait_after_completion(mpix_fast):
    """Late wait: the slot is recycled long before the wait — status must
    come from the request, not the (reused) slot."""
    import time
    mpix = mpix_fast
    src = np.full(8, 9, dtype=np.int32)
    dst = np.zeros(8, dtype=np.int32)
    rs = mpix.isend_enqueue(src, dest=0, tag=6)
    rr = mpix.irecv_enqueue(dst, source=0, tag=6)
    time.sleep(0.05)  # let the proxy complete + recycle both slots
    # churn the pool so the original slots get reused with new epochs
    for i in range(100):
        a = np.full(4, i, dtype=np.int32)
        b = np.zeros(4, dtype=np.int32)
        r1 = mpix.isend_enqueue(a, dest=0, tag=100 + i)
        r2 = mpix.irecv_enqueue(b, source=0, tag=100 + i)
        mpix.wait(r2)
        mpix.wait(r1)
    st = mpix.wait(rr)
    mpix.wait(rs)
    assert (dst == 9).all()
    assert st["tag"] == 6 and st["count_bytes"] == 32


def test_fast_request_free_orphan(mpix_fast):
    mpix = mpix_fast
    src = np.arange(16, dtype=np.int32)
    dst = np.zeros(16, dtype=np.int32)
    rs = mpix.isend_enqueue(src, dest=0, tag=7)
    mpix.request_free(rs)  # never waited: proxy owns the cleanup
    rr = mpix.irecv_enqueue(dst, source=0, tag=7)
    mpix.wait(rr)
    assert (dst == src).all()


def test_fast_slot_recycling_many(mpix_fast):
    """Far more operations than pool slots, all in flight waits consumed
    late — exercises epoch monotonicity across heavy slot reuse."""
    mpix = mpix_fast
    n_ops = 3000
    batch = 64
    for b in range(0, n_ops, batch):
        reqs = []
        bufs = []
        for i in range(batch):
            tag = b + i
            a = np.full(4, tag, dtype=np.int32)
            d = np.zeros(4, dtype=np.int32)
            bufs.append((a, d, tag))
            reqs.append(mpix.isend_enqueue(a, dest=0, tag=tag))
            reqs.append(mpix.irecv_enqueue(d, source=0, tag=tag))
        for r in reqs:
            mpix.wait(r)
        for a, d, tag in bufs:
            assert (d == tag).all()


def _fast_ring(rank, size):
    os.environ["MPIX_FAST_WAIT"] = "1"
    import mpix
    mpix.init()
    try:
        right = (rank + 1) % size
        left = (rank - 1 + size) % size
        for it in range(50):
            src = np.full(257, rank * 100 + it, dtype=np.int32)
            dst = np.zeros(257, dtype=np.int32)
            rs = mpix.isend_enqueue(src, dest=right, tag=it % 5)
            rr = mpix.irecv_enqueue(dst, source=left, tag=it % 5)
            st = mpix.wait(rr)
            mpix.wait(rs)
            assert (dst == left * 100 + it).all()
            assert st["source"] == left
    finally:
        mpix.finalize()


def test_fast_ring_2rank():
    run_ranks(2, _fast_ring)


def test_fast_ring_4rank():
    run_ranks(4, _fast_ring)
