"""Kernel-triggered partitioned send from Python: a GPU kernel publishes
each partition with __device__ MPIX_Pready the moment its tile is computed,
so the transfer overlaps the rest of the kernel.  Run (single GPU, or one
process per GPU under torchrun):

    RANK=0 WORLD_SIZE=1 python examples/kernel_partitioned.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")

import torch
import mpix

mpix.init()
print("config:", mpix.config())
rank, world = mpix.world()
right = (rank + 1) % world
left = (rank - 1 + world) % world

parts, per = 64, 1 << 16
dev = "cuda" if torch.cuda.is_available() else "cpu"
send = torch.zeros(parts * per, dtype=torch.int32, device=dev)
recv = torch.zeros_like(send)

ps = mpix.psend_init(send, parts, dest=right, tag=1)
pr = mpix.precv_init(recv, parts, source=left, tag=1)

for it in range(3):
    mpix.start(pr)
    mpix.start(ps)
    base = 1000 * (rank + 1) + it
    if dev == "cuda":
        dps = mpix.prequest_create(ps) if it == 0 else dps
        stream = torch.cuda.current_stream()
        # fill each partition and publish it from inside the kernel
        mpix.launch_fill_and_pready(send.data_ptr(), per, base, dps, parts,
                                    stream.cuda_stream)
        torch.cuda.synchronize()
    else:
        for p in range(parts):
            send[p * per:(p + 1) * per] = base + p
            mpix.pready(p, ps)
    mpix.wait(pr)
    mpix.wait(ps)
    expect = 1000 * (left + 1) + it
    ok = all(bool((recv[p * per:(p + 1) * per] == expect + p).all())
             for p in range(parts))
    print(f"iter {it}: {'OK' if ok else 'MISMATCH'}")

if dev == "cuda":
    mpix.prequest_free(dps)
mpix.request_free(ps)
mpix.request_free(pr)
mpix.finalize()
