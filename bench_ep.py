#!/usr/bin/env python3
"""EP dispatch/combine latency benchmark (BASELINE.json config:
"DeepEP low-latency dispatch+combine, 8 experts x 4096 tokens x 7168
hidden bf16, 8 MI355X intranode"). Default topk=8 — the reference
DeepEP workload shape (experimental/misc/ep_results.md:18, 256 experts
top-8 moves 4x the data of top-2; the r1 default of top-2 understated
the work per token).

Runs at any world size (1..8); for N>1 launch one rank per GPU via
torch.distributed.run with gloo rendezvous. Prints one JSON line with p50
latencies per op.
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import time


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--tokens", type=int, default=4096)
    p.add_argument("--hidden", type=int, default=7168)
    p.add_argument("--experts", type=int, default=8)
    p.add_argument("--topk", type=int, default=8)
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    args = p.parse_args()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29572")
    # proxy lane streams must not share HW queues with spin-wait kernels
    # (see tests/workers/ep_worker.py header note)
    os.environ.setdefault("GPU_MAX_HW_QUEUES", "8")

    import torch

    assert torch.cuda.is_available(), "needs a GPU"
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    if world > 1:
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) %
                          torch.cuda.device_count())

    import uccl_amd.ep as uep

    E = max(args.experts, world)  # at least one expert per rank
    buf = uep.Buffer(num_experts=E, topk=args.topk, hidden=args.hidden,
                     max_tokens=args.tokens, dtype=torch.bfloat16)

    g = torch.Generator().manual_seed(7 + rank)
    x = torch.randn(args.tokens, args.hidden, generator=g).bfloat16().cuda()
    topk_idx = torch.stack([
        torch.randperm(E, generator=g)[:args.topk]
        for _ in range(args.tokens)
    ]).cuda()
    topk_w = torch.rand(args.tokens, args.topk, generator=g).cuda()

    d_lat, c_lat = [], []
    for i in range(args.warmup + args.iters):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        recv_x, counts = buf.dispatch(x, topk_idx)
        torch.cuda.synchronize()
        t1 = time.perf_counter()
        out = buf.combine(recv_x, topk_idx, topk_w)
        torch.cuda.synchronize()
        t2 = time.perf_counter()
        if i >= args.warmup:
            d_lat.append((t1 - t0) * 1e6)
            c_lat.append((t2 - t1) * 1e6)
    del out

    if rank == 0:
        # bytes moved per dispatch: every (token, k) row once
        row_bytes = args.hidden * 2
        disp_bytes = args.tokens * args.topk * row_bytes
        p50d = statistics.median(d_lat)
        p50c = statistics.median(c_lat)
        print(json.dumps({
            "metric": "ep_ll_dispatch_combine_p50_us",
            "dispatch_p50_us": round(p50d, 1),
            "combine_p50_us": round(p50c, 1),
            "total_p50_us": round(p50d + p50c, 1),
            "dispatch_GBps": round(disp_bytes / (p50d * 1e-6) / 1e9, 2),
            "combine_GBps": round(disp_bytes / (p50c * 1e-6) / 1e9, 2),
            "n_gpus": world,
            "config": {
                "tokens": args.tokens, "hidden": args.hidden,
                "experts": E, "topk": args.topk, "dtype": "bf16",
                "data": "synthetic",
            },
        }))
    if world > 1:
        import torch.distributed as dist

        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
