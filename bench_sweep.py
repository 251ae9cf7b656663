#!/usr/bin/env python3
"""AllReduce size sweep, rccl-tests style (BASELINE configs 1-2:
all_reduce_perf 8B-1GB bf16 on 1..8 MI355X over xGMI).

Prints one table row per size: size, time/op (us), algbw, busbw. Launch
multi-rank via torch.distributed.run (gloo rendezvous); single-rank runs
the staged world=1 path (UCCL_WORLD1_STAGED).
"""

from __future__ import annotations

import argparse
import json
import os
import time


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--min-bytes", type=int, default=8)
    ap.add_argument("--max-bytes", type=int, default=1 << 30)
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "f32", "f8"])
    args = ap.parse_args()

    os.environ.setdefault("UCCL_WORLD1_STAGED", "1")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29573")

    import torch

    assert torch.cuda.is_available()
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    if world > 1:
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) %
                          torch.cuda.device_count())

    import uccl_amd.collective as ucol

    comm = ucol.init()
    dt = {"bf16": torch.bfloat16, "f32": torch.float32,
          "f8": getattr(torch, "float8_e4m3fn", torch.bfloat16)}[args.dtype]
    es = torch.tensor([], dtype=dt).element_size()

    rows = []
    size = args.min_bytes
    while size <= args.max_bytes:
        count = max(size // es, 1)
        if dt == getattr(torch, "float8_e4m3fn", None):
            t = torch.randn(count, dtype=torch.float32, device="cuda").to(dt)
        else:
            t = torch.randn(count, dtype=dt, device="cuda")
        iters = args.iters if size > (1 << 20) else args.iters * 5
        for _ in range(args.warmup):
            comm.all_reduce(t)
        torch.cuda.synchronize()
        if world > 1:
            import torch.distributed as dist

            dist.barrier()
        t0 = time.perf_counter()
        for _ in range(iters):
            comm.all_reduce(t)
        torch.cuda.synchronize()
        dt_s = (time.perf_counter() - t0) / iters
        if world > 1:
            import torch.distributed as dist

            te = torch.tensor([dt_s])
            dist.all_reduce(te, op=dist.ReduceOp.MAX)
            dt_s = float(te[0])
        nbytes = count * es
        algbw = nbytes / dt_s / 1e9
        busbw = algbw * (2 * (world - 1) / world) if world > 1 else algbw
        rows.append((nbytes, dt_s * 1e6, algbw, busbw))
        if rank == 0:
            print(f"{nbytes:>12}  {dt_s*1e6:>10.2f} us  algbw {algbw:>8.2f} "
                  f"GB/s  busbw {busbw:>8.2f} GB/s", flush=True)
        size *= 4

    if rank == 0:
        print(json.dumps({
            "metric": "allreduce_sweep",
            "n_gpus": world,
            "dtype": args.dtype,
            "rows": [{"bytes": b, "us": round(u, 2),
                      "algbw_GBps": round(a, 2), "busbw_GBps": round(bb, 2)}
                     for b, u, a, bb in rows],
        }))
    if world > 1:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
