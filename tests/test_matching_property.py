"""Property-based matching-semantics tests (hypothesis, derandomized so CI
is deterministic): arbitrary interleavings of sends/recvs/waits against a
pure-python MPI matching model — posted receives match arrivals per
(source, tag) in FIFO order, wildcards match in post order."""
import os

import numpy as np
from hypothesis import given, settings, strategies as st

import pytest


@pytest.fixture(scope="module")
def mpix_mod():
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    import mpix
    mpix.init()
    yield mpix
    mpix.finalize()


# an operation stream: send(tag) | recv(tag) | recv(ANY)
ops_strategy = st.lists(
    st.one_of(
        st.tuples(st.just("send"), st.integers(0, 3)),
        st.tuples(st.just("recv"), st.integers(0, 3)),
        st.tuples(st.just("recv_any"), st.just(-1)),
    ),
    min_size=1, max_size=40,
)


class Model:
    """Pure-python matching model: sends queue per tag in program order;
    a recv(tag) consumes the oldest unconsumed send of that tag; recv(ANY)
    consumes the oldest unconsumed send overall (arrival order == program
    order in loopback)."""

    def __init__(self):
        self.sends = []  # (serial, tag, consumed?)

    def send(self, tag, serial):
        self.sends.append([serial, tag, False])

    def recv(self, tag):
        for rec in self.sends:
            if not rec[2] and (tag == -1 or rec[1] == tag):
                rec[2] = True
                return rec[0]
        return None


@settings(max_examples=120, derandomize=True, deadline=None)
@given(ops=ops_strategy)
def test_matching_follows_model(mpix_mod, ops):
    mpix = mpix_mod
    model = Model()
    serial = 0
    keepalive = []  # send buffers must outlive completion
    pending = []    # (req, buf, expected_serial)
    for kind, tag in ops:
        if kind == "send":
            buf = np.full(8, serial, dtype=np.int32)
            keepalive.append((mpix.isend_enqueue(buf, dest=0, tag=tag), buf))
            model.send(tag, serial)
            serial += 1
        else:
            want = model.recv(-1 if kind == "recv_any" else tag)
            if want is None:
                continue  # would block forever; model says no match exists
            buf = np.zeros(8, dtype=np.int32)
            src = mpix.ANY_SOURCE if kind == "recv_any" else 0
            rtag = mpix.ANY_TAG if kind == "recv_any" else tag
            r = mpix.irecv_enqueue(buf, source=src, tag=rtag)
            pending.append((r, buf, want))
    # drain every posted recv and check it got the modelled message
    for r, buf, want in pending:
        mpix.wait(r)
        assert (buf == want).all(), f"expected send #{want}, got {buf[0]}"
    # drain unmatched sends so finalize is clean
    for rec in model.sends:
        if not rec[2]:
            buf = np.zeros(8, dtype=np.int32)
            r = mpix.irecv_enqueue(buf, source=0, tag=rec[1])
            mpix.wait(r)
            assert (buf == rec[0]).all()
            rec[2] = True
    for r, _ in keepalive:
        mpix.wait(r)
