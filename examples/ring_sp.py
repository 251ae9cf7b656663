#!/usr/bin/env python3
"""Sequence-parallel traffic patterns (ring send/recv + all-to-all) over
the uccl_amd engine — the building blocks of ring attention / Ulysses.
The reference treats SP/CP the same way: as the traffic its library
accelerates, not as a model-layer feature.

Launch: torch.distributed.run, one rank per GPU."""
import os

import torch
import torch.distributed as dist

import uccl_amd.collective as ucol


def main():
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    if world > 1:
        dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) %
                          torch.cuda.device_count())
    comm = ucol.init()

    # ring KV rotation (ring attention inner loop shape)
    heads, seq, dim = 8, 2048, 128
    kv = torch.randn(2, heads, seq // max(world, 1), dim,
                     device="cuda", dtype=torch.bfloat16)
    nxt = torch.empty_like(kv)
    for _ in range(max(world - 1, 0)):
        dst = (rank + 1) % world
        src = (rank - 1 + world) % world
        if rank % 2 == 0:
            comm.send(kv, dst)
            comm.recv(nxt, src)
        else:
            comm.recv(nxt, src)
            comm.send(kv, dst)
        kv, nxt = nxt, kv
    torch.cuda.synchronize()

    # Ulysses-style head<->sequence all-to-all
    x = torch.randn(world * heads * dim, device="cuda",
                    dtype=torch.bfloat16)
    y = torch.empty_like(x)
    comm.all_to_all(y, x)
    torch.cuda.synchronize()
    if rank == 0:
        print("ring_sp OK")
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
