#!/usr/bin/env python3
"""mpix flagship benchmark — BASELINE.json metrics on MI355X.

Measures, on N ranks (one per GPU, torchrun layout):
  1. stream-enqueued ping-pong half-RTT (us) — MPIX_Isend/Irecv_enqueue +
     MPIX_Wait_enqueue on hipStream memOps, device buffers, rank pairs
     (0<->1, 2<->3, ...); loopback-self at N=1.
  2. Psend bandwidth (GB/s) — HEADLINE `value`: partitioned ring exchange,
     device buffers, __device__ MPIX_Pready trigger kernel, 64 partitions.
     `value` is the whole-job aggregate over all N ranks (weak scaling:
     per-rank message size is fixed as N grows).

Run:  python bench.py [--gpus N] [--steps K] [--warmup W]
The driver launches N>1 via `python -m torch.distributed.run --nnodes=1
--nproc-per-node N ... bench.py --gpus N`; we read RANK/LOCAL_RANK/
WORLD_SIZE/MASTER_* from the environment (mpix's TCP bootstrap uses
MASTER_PORT+31, next to torchrun's c10d store).

Timing discipline (driver contract): W untimed warmup steps, then EXACTLY K
timed steps bracketed by barrier + torch.cuda.synchronize() on both sides;
elapsed is max-reduced over ranks; rank 0 prints ONE JSON line.

Reference (NVIDIA/mpi-acx) publishes no numbers — vs_baseline
is null; these are the self-measured targets BASELINE.json defines.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--msg-mib", type=int, default=256,
                   help="per-rank Psend message size (MiB)")
    p.add_argument("--partitions", type=int, default=64)
    p.add_argument("--pp-iters", type=int, default=200,
                   help="ping-pong iterations per timed block")
    p.add_argument("--pp-bytes", type=int, default=8,
                   help="ping-pong message size (bytes)")
    return p.parse_args()


def main():
    args = parse_args()
    import torch
    import mpix

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    os.environ.setdefault("RANK", str(rank))
    os.environ.setdefault("WORLD_SIZE", str(world))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")

    use_gpu = torch.cuda.is_available()
    if mpix.have_gpu() and not use_gpu:
        # never run a silently degraded CPU bench on a GPU box
        raise RuntimeError("mpix sees a GPU but torch.cuda does not — "
                           "HIP runtime clash (import order?)")
    if use_gpu:
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        dev = "cuda"
    else:
        dev = "cpu"  # CPU sanity mode: host buffers through the same paths

    dist = None
    if world > 1:
        import torch.distributed as tdist
        tdist.init_process_group("gloo", rank=rank, world_size=world)
        dist = tdist

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    def max_over_ranks(x):
        if dist is None:
            return x
        t = torch.tensor([x], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return float(t.item())

    def trace(msg):
        print(f"[bench r{rank}] {msg}", file=sys.stderr, flush=True)

    mpix.init()
    trace(f"init done: {mpix.config()}")
    stream = torch.cuda.current_stream() if use_gpu else None

    # ------------------------------------------------ 1. ping-pong half-RTT
    n_pp = max(args.pp_bytes // 4, 1)
    buf = torch.zeros(n_pp, dtype=torch.int32, device=dev)
    if world == 1:
        peer, pingpong_role = 0, "self"
    else:
        pair = rank ^ 1
        peer = pair if pair < world else rank  # odd world: last rank loops back
        pingpong_role = "ping" if rank % 2 == 0 else "pong"
        if peer == rank:
            pingpong_role = "self"

    def pp_iter():
        if pingpong_role == "self":
            rs = mpix.isend_enqueue(buf, dest=rank, tag=1, stream=stream)
            rr = mpix.irecv_enqueue(buf, source=rank, tag=1, stream=stream)
            mpix.waitall_enqueue([rs, rr], stream=stream)
        elif pingpong_role == "ping":
            rs = mpix.isend_enqueue(buf, dest=peer, tag=1, stream=stream)
            mpix.wait_enqueue(rs, stream=stream)
            rr = mpix.irecv_enqueue(buf, source=peer, tag=1, stream=stream)
            mpix.wait_enqueue(rr, stream=stream)
        else:
            rr = mpix.irecv_enqueue(buf, source=peer, tag=1, stream=stream)
            mpix.wait_enqueue(rr, stream=stream)
            rs = mpix.isend_enqueue(buf, dest=peer, tag=1, stream=stream)
            mpix.wait_enqueue(rs, stream=stream)

    trace("pingpong warmup")
    for _ in range(max(args.warmup * 4, 20)):
        pp_iter()
    barrier_sync()
    trace("pingpong timed")
    t0 = time.perf_counter()
    for _ in range(args.pp_iters):
        pp_iter()
    barrier_sync()
    pp_elapsed = max_over_ranks(time.perf_counter() - t0)
    # one pair iteration = 2 messages = 1 RTT; loopback iteration = 1 message
    legs = 2.0 if pingpong_role != "self" else 1.0
    half_rtt_us = pp_elapsed / args.pp_iters / legs * 1e6

    # --------------------------------------- 2. Psend ring bandwidth (value)
    parts = args.partitions
    msg_bytes = args.msg_mib * (1 << 20)
    n_words = msg_bytes // 4
    right = (rank + 1) % world
    left = (rank - 1 + world) % world
    send = torch.zeros(n_words, dtype=torch.int32, device=dev)
    recv = torch.zeros_like(send)
    ps = mpix.psend_init(send, parts, dest=right, tag=9)
    pr = mpix.precv_init(recv, parts, source=left, tag=9)
    dps = mpix.prequest_create(ps) if use_gpu else None

    def psend_step():
        mpix.start(pr)
        mpix.start(ps)
        if use_gpu:
            # kernel-triggered: one __device__ MPIX_Pready per partition
            mpix.launch_pready_all(dps, parts, stream.cuda_stream)
        else:
            for p in range(parts):
                mpix.pready(p, ps)
        mpix.wait(pr)
        mpix.wait(ps)

    trace(f"pingpong done: half_rtt={half_rtt_us:.2f}us; psend warmup")
    for _ in range(args.warmup):
        psend_step()
    barrier_sync()
    trace("psend timed")
    t0 = time.perf_counter()
    for _ in range(args.steps):
        psend_step()
    barrier_sync()
    elapsed = max_over_ranks(time.perf_counter() - t0)
    ms_per_step = elapsed / args.steps * 1e3
    # aggregate: every rank moves msg_bytes per step
    agg_gbps = world * msg_bytes * args.steps / elapsed / 1e9

    if dps is not None:
        mpix.prequest_free(dps)
    mpix.request_free(ps)
    mpix.request_free(pr)
    mpix.finalize()
    if dist is not None:
        dist.destroy_process_group()

    if rank == 0:
        print(json.dumps({
            "metric": "Psend GB/s over xGMI (+ stream-enqueued ping-pong half-RTT us)",
            "value": round(agg_gbps, 3),
            "unit": "GB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "int32",
            "data": "synthetic",
            "config": {
                "model": "partitioned-ring",
                "pattern": f"ring x{world} ranks, device buffers" if use_gpu
                           else f"ring x{world} ranks, host buffers (no GPU)",
                "msg_bytes_per_rank": msg_bytes,
                "partitions": parts,
                "trigger": "__device__ MPIX_Pready kernel" if use_gpu
                           else "host MPIX_Pready",
                "pingpong_half_rtt_us": round(half_rtt_us, 3),
                "pingpong_bytes": args.pp_bytes,
                "pingpong_iters": args.pp_iters,
                "pingpong_mode": pingpong_role if world == 1 else "pairs",
                "parallelism": f"spmd{world}",
            },
        }))


if __name__ == "__main__":
    main()
