"""Single-rank (loopback) CPU tests of the full enqueue -> proxy -> transport
-> wait pipeline. Reference analog: the self-contained checks inside
test/src/ring.c run at world_size 1 (every rank sends to itself)."""
import numpy as np
import pytest


def test_send_recv_roundtrip(mpix_env):
    mpix = mpix_env
    send = np.arange(1024, dtype=np.int32)
    recv = np.zeros_like(send)
    rs = mpix.isend_enqueue(send, dest=0, tag=7)
    rr = mpix.irecv_enqueue(recv, source=0, tag=7)
    st = mpix.wait(rr)
    mpix.wait(rs)
    assert (recv == send).all()
    assert st["source"] == 0
    assert st["tag"] == 7
    assert st["error"] == 0
    assert st["count_bytes"] == send.nbytes


def test_recv_before_send(mpix_env):
    mpix = mpix_env
    send = np.full(10, 42, dtype=np.int64)
    recv = np.zeros_like(send)
    rr = mpix.irecv_enqueue(recv, source=0, tag=1)
    rs = mpix.isend_enqueue(send, dest=0, tag=1)
    mpix.wait(rr)
    mpix.wait(rs)
    assert (recv == send).all()


def test_unexpected_message_buffering(mpix_env):
    """Send completes (buffered) before any recv is posted."""
    mpix = mpix_env
    send = np.random.randint(0, 1000, 5000).astype(np.int32)
    recv = np.zeros_like(send)
    rs = mpix.isend_enqueue(send, dest=0, tag=3)
    mpix.wait(rs)  # buffered-send completion without a matching recv
    rr = mpix.irecv_enqueue(recv, source=0, tag=3)
    mpix.wait(rr)
    assert (recv == send).all()


def test_tag_matching_order(mpix_env):
    mpix = mpix_env
    a = np.array([1], dtype=np.int32)
    b = np.array([2], dtype=np.int32)
    ra = np.zeros(1, dtype=np.int32)
    rb = np.zeros(1, dtype=np.int32)
    s1 = mpix.isend_enqueue(a, dest=0, tag=100)
    s2 = mpix.isend_enqueue(b, dest=0, tag=200)
    r2 = mpix.irecv_enqueue(rb, source=0, tag=200)
    r1 = mpix.irecv_enqueue(ra, source=0, tag=100)
    for r in (r1, r2, s1, s2):
        mpix.wait(r)
    assert ra[0] == 1 and rb[0] == 2


def test_any_source_any_tag(mpix_env):
    mpix = mpix_env
    send = np.array([9, 8, 7], dtype=np.int32)
    recv = np.zeros_like(send)
    rs = mpix.isend_enqueue(send, dest=0, tag=55)
    rr = mpix.irecv_enqueue(recv, source=mpix.ANY_SOURCE, tag=mpix.ANY_TAG)
    st = mpix.wait(rr)
    mpix.wait(rs)
    assert (recv == send).all()
    assert st["source"] == 0 and st["tag"] == 55


def test_large_message_chunked(mpix_env):
    """Larger than the 4 MiB staging ring: exercises chunked flow control."""
    mpix = mpix_env
    n = 3 * 1024 * 1024  # 12 MiB of int32
    send = np.random.randint(-2**31, 2**31 - 1, n, dtype=np.int64).astype(
        np.int32)
    recv = np.zeros_like(send)
    rr = mpix.irecv_enqueue(recv, source=0, tag=9)
    rs = mpix.isend_enqueue(send, dest=0, tag=9)
    mpix.wait(rr)
    mpix.wait(rs)
    assert (recv == send).all()


def test_zero_byte_message(mpix_env):
    mpix = mpix_env
    send = np.zeros(0, dtype=np.int8)
    recv = np.zeros(0, dtype=np.int8)
    rs = mpix.isend_enqueue(send, dest=0, tag=2)
    rr = mpix.irecv_enqueue(recv, source=0, tag=2)
    st = mpix.wait(rr)
    mpix.wait(rs)
    assert st["count_bytes"] == 0


def test_truncation_error(mpix_env):
    mpix = mpix_env
    send = np.arange(100, dtype=np.int32)
    recv = np.zeros(10, dtype=np.int32)
    rs = mpix.isend_enqueue(send, dest=0, tag=4)
    rr = mpix.irecv_enqueue(recv, source=0, tag=4)
    st = mpix.wait(rr)
    mpix.wait(rs)
    assert st["error"] != 0  # MPI_ERR_TRUNCATE
    assert (recv == send[:10]).all()


def test_request_free_orphan(mpix_env):
    mpix = mpix_env
    send = np.arange(16, dtype=np.int32)
    recv = np.zeros_like(send)
    rs = mpix.isend_enqueue(send, dest=0, tag=5)
    rr = mpix.irecv_enqueue(recv, source=0, tag=5)
    mpix.request_free(rs)  # fire-and-forget
    mpix.wait(rr)
    assert (recv == send).all()


def test_many_outstanding(mpix_env):
    mpix = mpix_env
    n_msgs = 200
    sends = [np.full(64, i, dtype=np.int32) for i in range(n_msgs)]
    recvs = [np.zeros(64, dtype=np.int32) for _ in range(n_msgs)]
    rrs = [mpix.irecv_enqueue(recvs[i], source=0, tag=i) for i in range(n_msgs)]
    rss = [mpix.isend_enqueue(sends[i], dest=0, tag=i) for i in range(n_msgs)]
    for r in rrs + rss:
        mpix.wait(r)
    for i in range(n_msgs):
        assert (recvs[i] == i).all()


def test_partitioned_host_loopback(mpix_env):
    """Host-side Pready/Parrived over 8 partitions, 3 Start iterations
    (persistent-request reuse; reference: ring-partitioned.cu:101-127)."""
    mpix = mpix_env
    parts, per = 8, 256
    send = np.zeros(parts * per, dtype=np.int32)
    recv = np.zeros_like(send)
    ps = mpix.psend_init(send, parts, dest=0, tag=11)
    pr = mpix.precv_init(recv, parts, source=0, tag=11)
    for it in range(3):
        send[:] = np.arange(parts * per, dtype=np.int32) + it * 1000
        mpix.start(pr)
        mpix.start(ps)
        for p in range(parts):
            mpix.pready(p, ps)
        mpix.wait(pr)
        assert (recv == send).all()
        # parrived reports completed before the reset in wait(ps)? wait(pr)
        # already consumed recv completions; check send side finishes too
        mpix.wait(ps)
    mpix.request_free(ps)
    mpix.request_free(pr)


def test_partitioned_out_of_order_pready(mpix_env):
    mpix = mpix_env
    parts, per = 16, 64
    send = np.arange(parts * per, dtype=np.int32)
    recv = np.zeros_like(send)
    ps = mpix.psend_init(send, parts, dest=0, tag=12)
    pr = mpix.precv_init(recv, parts, source=0, tag=12)
    mpix.start(pr)
    mpix.start(ps)
    import random
    order = list(range(parts))
    random.shuffle(order)
    for p in order:
        mpix.pready(p, ps)
    mpix.wait(pr)
    mpix.wait(ps)
    assert (recv == send).all()
    mpix.request_free(ps)
    mpix.request_free(pr)


def test_wildcard_recv_ignores_partitioned(mpix_env):
    """A wildcard basic receive must never match partition messages —
    partitioned traffic matches in its own channel (src/transport/
    native.cpp match(): want_part != is_part excludes cross-matching)."""
    import numpy as np
    mpix = mpix_env
    parts, per = 4, 64
    psend = np.arange(parts * per, dtype=np.int32)
    precv = np.zeros_like(psend)
    wild = np.full(per, -1, dtype=np.int32)
    # wildcard basic recv posted FIRST, same tag as the partitioned pair
    rw = mpix.irecv_enqueue(wild, source=mpix.ANY_SOURCE, tag=mpix.ANY_TAG)
    ps = mpix.psend_init(psend, parts, dest=0, tag=5)
    pr = mpix.precv_init(precv, parts, source=0, tag=5)
    mpix.start(pr)
    mpix.start(ps)
    for p in range(parts):
        mpix.pready(p, ps)
    mpix.wait(pr)
    mpix.wait(ps)
    assert (precv == psend).all()
    assert (wild == -1).all(), "wildcard recv stole a partition message"
    # the wildcard recv still works for a real basic message
    basic = np.full(per, 77, dtype=np.int32)
    rs = mpix.isend_enqueue(basic, dest=0, tag=99)
    st = mpix.wait(rw)
    mpix.wait(rs)
    assert (wild == 77).all()
    assert st["tag"] == 99
    mpix.request_free(ps)
    mpix.request_free(pr)
