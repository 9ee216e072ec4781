"""mpix test harness.

Markers:
  gpu — requires an MI355X (run with `pytest -m gpu` on a GPU box).
Everything else runs on CPU (the proxy-only path; BASELINE config 1).

Multi-process tests spawn real processes with torchrun-style env vars
(RANK / WORLD_SIZE / MASTER_ADDR=127.0.0.1) — the same bootstrap the
driver's torch.distributed.run launch uses.
"""
import multiprocessing as mp
import os
import socket
import sys
import traceback

try:  # torch's HIP runtime must load before mpix._C (see mpix/__init__.py)
    import torch  # noqa: F401
except ImportError:
    pass

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X)")


def pytest_collection_modifyitems(config, items):
    """Skip gpu tests when no GPU is present; on a GPU box, give every gpu
    test a hard per-test timeout (method=thread interrupts wedged HIP calls
    by killing the process with a stack dump) so one hang can never zero
    the whole suite's record."""
    try:
        import mpix
        has_gpu = mpix.have_gpu()
    except Exception:
        has_gpu = False
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" not in item.keywords:
            continue
        if not has_gpu:
            item.add_marker(skip)
        elif item.get_closest_marker("timeout") is None:
            item.add_marker(pytest.mark.timeout(300, method="thread"))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _rank_entry(fn, rank, world_size, port, q, args):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        fn(rank, world_size, *args)
        q.put((rank, None))
    except BaseException:
        q.put((rank, traceback.format_exc()))
        sys.exit(1)


def run_ranks(world_size, fn, *args, timeout=120):
    """Run fn(rank, world_size, *args) in `world_size` fresh processes."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_rank_entry, args=(fn, r, world_size, port, q, args))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in procs:
        try:
            rank, err = q.get(timeout=timeout)
            results[rank] = err
        except Exception:
            break
    for p in procs:
        p.join(timeout=10)
        if p.is_alive():
            p.terminate()
            p.join()
    errs = [f"rank {r}:\n{e}" for r, e in sorted(results.items()) if e]
    missing = [r for r in range(world_size) if r not in results]
    assert not errs, "\n".join(errs)
    assert not missing, f"ranks {missing} did not report (hang/crash)"


@pytest.fixture
def mpix_env(monkeypatch):
    """Single-rank in-process mpix init/finalize."""
    monkeypatch.setenv("RANK", "0")
    monkeypatch.setenv("WORLD_SIZE", "1")
    import mpix
    mpix.init()
    yield mpix
    mpix.finalize()
