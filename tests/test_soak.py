"""Randomized (seeded, deterministic) protocol soak tests: random op
sequences checked against a simple FIFO-matching reference model.  Covers
interleavings the directed tests cannot: out-of-order posts, tag collisions,
unexpected-message buffering under load, mixed sizes crossing the 64 KiB
chunk boundary, many-iteration slot reuse."""
import numpy as np
import pytest

from conftest import run_ranks


def _soak_loopback(mpix, seed, iters):
    rng = np.random.default_rng(seed)
    pending_sends = []   # (req, tag, value, n)
    pending_recvs = []   # (req, buf, expected_value, n)
    sent_by_tag = {}     # tag -> list of (value, n) in send order
    recv_by_tag = {}     # tag -> count consumed
    for it in range(iters):
        action = rng.integers(0, 3)
        tag = int(rng.integers(0, 8))  # small tag space -> collisions
        n = int(rng.choice([1, 7, 64, 1000, 20000]))
        if action == 0:  # send
            val = int(rng.integers(0, 1 << 30))
            buf = np.full(n, val, dtype=np.int32)
            r = mpix.isend_enqueue(buf, dest=0, tag=tag)
            pending_sends.append((r, buf))
            sent_by_tag.setdefault(tag, []).append((val, n))
        elif action == 1:  # recv for the oldest unconsumed send of this tag
            q = sent_by_tag.get(tag, [])
            consumed = recv_by_tag.get(tag, 0)
            if consumed >= len(q):
                continue
            val, sn = q[consumed]
            recv_by_tag[tag] = consumed + 1
            buf = np.zeros(sn, dtype=np.int32)
            r = mpix.irecv_enqueue(buf, source=0, tag=tag)
            pending_recvs.append((r, buf, val, sn))
        else:  # drain one pending recv
            if pending_recvs:
                r, buf, val, sn = pending_recvs.pop(0)
                st = mpix.wait(r)
                assert st["count_bytes"] == sn * 4
                assert (buf == val).all()
    # drain everything (recvs for unconsumed sends first)
    for tag, q in sent_by_tag.items():
        for val, sn in q[recv_by_tag.get(tag, 0):]:
            buf = np.zeros(sn, dtype=np.int32)
            r = mpix.irecv_enqueue(buf, source=0, tag=tag)
            mpix.wait(r)
            assert (buf == val).all(), f"tag {tag}"
    for r, _buf in pending_sends:
        mpix.wait(r)
    for r, buf, val, sn in pending_recvs:
        mpix.wait(r)
        assert (buf == val).all()


@pytest.mark.parametrize("seed", [1, 2, 3])
def test_soak_loopback(mpix_env, seed):
    _soak_loopback(mpix_env, seed, iters=300)


def _soak_pair(rank, size):
    """Both ranks push randomized same-seed streams at each other; the
    receive order within (src, tag) must equal send order."""
    import mpix
    mpix.init()
    try:
        peer = 1 - rank
        rng_mine = np.random.default_rng(100 + rank)
        rng_peer = np.random.default_rng(100 + peer)

        def plan(rng):
            ops = []
            for i in range(120):
                tag = int(rng.integers(0, 4))
                n = int(rng.choice([1, 33, 900, 17000]))
                val = int(rng.integers(0, 1 << 30))
                ops.append((tag, n, val))
            return ops

        mine, theirs = plan(rng_mine), plan(rng_peer)
        # send buffers must stay alive until the send completes (MPI rule)
        sbufs = [np.full(n, v, dtype=np.int32) for t, n, v in mine]
        sreqs = [(mpix.isend_enqueue(sbufs[i], dest=peer, tag=mine[i][0]),
                  sbufs[i]) for i in range(len(mine))]
        # post recvs in the peer's send order per tag (FIFO guarantee)
        rreqs = []
        for t, n, v in theirs:
            buf = np.zeros(n, dtype=np.int32)
            rreqs.append((mpix.irecv_enqueue(buf, source=peer, tag=t),
                          buf, v))
        for r, buf, v in rreqs:
            mpix.wait(r)
            assert (buf == v).all()
        for r, _ in sreqs:
            mpix.wait(r)
    finally:
        mpix.finalize()


def test_soak_2rank():
    run_ranks(2, _soak_pair, timeout=240)


def _soak_partitioned(rank, size):
    """Randomized partitioned traffic: random partition counts and sizes,
    Pready published in random order, persistent reuse over iterations,
    interleaved with plain sends on overlapping tags."""
    import mpix
    mpix.init()
    try:
        # geometry must agree across ranks -> shared seed; per-rank rng
        # only randomizes publish/poll ORDER
        geom = np.random.default_rng(500)
        rng = np.random.default_rng(900 + rank)
        peer = (rank + 1) % size
        left = (rank - 1 + size) % size
        for round_ in range(6):
            parts = int(geom.choice([1, 3, 8, 32]))
            per = int(geom.choice([4, 100, 5000]))
            sbuf = np.zeros(parts * per, dtype=np.int32)
            rbuf = np.zeros(parts * per, dtype=np.int32)
            ps = mpix.psend_init(sbuf, parts, dest=peer, tag=3)
            pr = mpix.precv_init(rbuf, parts, source=left, tag=3)
            for it in range(4):
                base = rank * 1000000 + round_ * 10000 + it * 100
                lbase = left * 1000000 + round_ * 10000 + it * 100
                mpix.start(pr)
                mpix.start(ps)
                # a plain same-tag message must not confuse partitioned match
                extra = np.full(16, base + 77, dtype=np.int32)
                xr = mpix.isend_enqueue(extra, dest=peer, tag=3)
                order = rng.permutation(parts)
                for p in order:
                    sbuf[p * per:(p + 1) * per] = base + p
                    mpix.pready(int(p), ps)
                # poll arrivals in a different random order
                for p in rng.permutation(parts):
                    while not mpix.parrived(pr, int(p)):
                        pass
                    seg = rbuf[p * per:(p + 1) * per]
                    assert (seg == lbase + p).all(), f"part {p} round {round_}"
                xbuf = np.zeros(16, dtype=np.int32)
                rr = mpix.irecv_enqueue(xbuf, source=left, tag=3)
                mpix.wait(rr)
                assert (xbuf == lbase + 77).all()
                mpix.wait(xr)
                mpix.wait(pr)
                mpix.wait(ps)
            mpix.request_free(ps)
            mpix.request_free(pr)
    finally:
        mpix.finalize()


def test_soak_partitioned_2rank():
    run_ranks(2, _soak_partitioned, timeout=300)
