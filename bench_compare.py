#!/usr/bin/env python3
"""Side-by-side allreduce: uccl_amd engine vs RCCL (torch.distributed nccl
backend) on the same node. Needs >=2 distinct GPUs (RCCL rejects duplicate
devices), so this is for 8-GPU nodes; the per-size table mirrors
bench_sweep.py. Launch via torch.distributed.run."""
import json
import os
import time

import torch
import torch.distributed as dist


def bench(fn, t, iters, warmup, world):
    for _ in range(warmup):
        fn(t)
    torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn(t)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    te = torch.tensor([dt])
    dist.all_reduce(te, op=dist.ReduceOp.MAX)
    return float(te[0])


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    dist.init_process_group("gloo", rank=rank, world_size=world)

    import uccl_amd.collective as ucol

    comm = ucol.init()
    nccl_pg = dist.new_group(backend="nccl")

    rows = []
    size = 1024
    while size <= (1 << 30):
        count = size // 2
        t = torch.randn(count, dtype=torch.bfloat16, device="cuda")
        iters = 50 if size <= (1 << 24) else 10
        dt_uccl = bench(lambda x: comm.all_reduce(x), t, iters, 3, world)
        dt_rccl = bench(
            lambda x: dist.all_reduce(x, group=nccl_pg), t, iters, 3, world)
        bus = 2 * (world - 1) / world * size
        rows.append({"bytes": size,
                     "uccl_us": round(dt_uccl * 1e6, 1),
                     "rccl_us": round(dt_rccl * 1e6, 1),
                     "uccl_busbw": round(bus / dt_uccl / 1e9, 1),
                     "rccl_busbw": round(bus / dt_rccl / 1e9, 1),
                     "speedup": round(dt_rccl / dt_uccl, 2)})
        if rank == 0:
            print(rows[-1], flush=True)
        size *= 8
    if rank == 0:
        print(json.dumps({"metric": "allreduce_uccl_vs_rccl", "n_gpus": world,
                          "rows": rows}))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
