#!/usr/bin/env python3
"""Flagship benchmark for uccl_amd: AllReduce busbw over xGMI (BASELINE.json
config "rccl-tests all_reduce_perf ... bf16 on 1/2/4/8 MI355X").

One step = one in-place AllReduce of a fixed per-GPU bf16 buffer through the
uccl_amd xGMI collective engine (weak scaling: per-GPU buffer size is fixed
as N grows).

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W                (single GPU)
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints exactly one JSON line with the aggregate metric.

Metric convention: busbw = 2*(N-1)/N * bytes / t (the rccl-tests/nccl-tests
definition). For N=1 that expression is identically zero, so the N=1 value
reported is the local staged-engine algbw (bytes / t with the full kernel
path forced via UCCL_WORLD1_STAGED); config carries
"n1_value_is_local_algbw": true so the scaling judge can account for the
convention switch.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--mbytes", type=int, default=256,
                   help="per-GPU buffer size in MiB (bf16)")
    p.add_argument("--symmetric", dest="symmetric", action="store_true",
                   default=True,
                   help="use the zero-copy symmetric-tensor path (default)")
    p.add_argument("--staged", dest="symmetric", action="store_false",
                   help="force the staged (non-registered buffer) path")
    return p.parse_args()


def main():
    args = parse_args()
    os.environ.setdefault("UCCL_WORLD1_STAGED", "1")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")

    import torch

    if not torch.cuda.is_available():
        print(json.dumps({"error": "no GPU available; bench requires MI355X"}))
        sys.exit(1)

    world = int(os.environ.get("WORLD_SIZE", args.gpus))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))

    import torch.distributed as dist

    use_dist = world > 1
    if use_dist:
        dist.init_process_group("gloo", rank=rank, world_size=world)

    torch.cuda.set_device(local_rank % torch.cuda.device_count())

    if args.symmetric:  # must be set before the Communicator allocates
        os.environ.setdefault("UCCL_SYM_USER_MB", str(args.mbytes + 64))

    import uccl_amd.collective as ucol

    comm = ucol.init()

    nbytes = args.mbytes * (1 << 20)
    count = nbytes // 2  # bf16
    if args.symmetric and world > 1:
        t = comm.symmetric_tensor([count], torch.bfloat16)
        t.copy_(torch.randn(count, dtype=torch.bfloat16))
    else:
        t = torch.randn(count, dtype=torch.bfloat16, device="cuda")

    def barrier():
        torch.cuda.synchronize()
        if use_dist:
            dist.barrier()

    # correctness guards outside the timed region, each exercising exactly
    # the algorithm the timed loop would use (sizes are above the LL
    # threshold so the gate matches the measured path):
    #   staged two-shot (4 MiB > oneshot threshold) must pass, else no
    #   number is published; the symmetric gate decides sym vs staged.
    expect = float(world)

    def check_ones(tensor):
        tensor.fill_(1.0)
        comm.all_reduce(tensor)
        torch.cuda.synchronize()
        return bool(torch.allclose(tensor,
                                   torch.full_like(tensor, expect)))

    chk = torch.ones(2 << 20, dtype=torch.bfloat16, device="cuda")
    try:
        staged_ok = check_ones(chk)
    except Exception as e:  # engine raised: report, don't traceback
        print(json.dumps({"error": f"allreduce gate raised: {e}"}))
        sys.exit(2)
    if not staged_ok:
        print(json.dumps({"error": "allreduce correctness check failed",
                          "got": float(chk[0])}))
        sys.exit(2)
    if args.symmetric and world > 1:
        # The sym-vs-staged decision MUST be collective: a single rank
        # falling back alone would run a different flag protocol than its
        # peers and wedge the scaling run. Consensus in two steps so the
        # collective gate itself is only entered by ALL ranks together:
        #   1) every rank allocates the probe tensor; gloo-MIN agreement
        #   2) all ranks run the collective gate; gloo-MIN on the result
        def all_agree(ok: bool) -> bool:
            flag = torch.tensor([1 if ok else 0])
            dist.all_reduce(flag, op=dist.ReduceOp.MIN)
            return bool(flag[0])

        schk = None
        try:
            schk = comm.symmetric_tensor([1 << 20], torch.bfloat16)
        except Exception as e:
            print(f"[bench rank {rank}] symmetric alloc RAISED: {e}",
                  file=sys.stderr, flush=True)
        sym_ok = all_agree(schk is not None)
        if sym_ok:
            try:
                sym_ok = check_ones(schk)
            except Exception as e:
                sym_ok = False
                print(f"[bench rank {rank}] symmetric gate RAISED: {e}",
                      file=sys.stderr, flush=True)
            sym_ok = all_agree(sym_ok)
        if not sym_ok:
            # per-rank diagnostic so a multi-GPU bring-up shows WHICH rank
            # failed the zero-copy gate and the run is auditable
            print(f"[bench rank {rank}] symmetric path gate failed "
                  "somewhere in the job; all ranks falling back to the "
                  "staged engine together", file=sys.stderr, flush=True)
            args.symmetric = False
            t = torch.randn(count, dtype=torch.bfloat16, device="cuda")

    for _ in range(args.warmup):
        comm.all_reduce(t)
    barrier()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        comm.all_reduce(t)
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    barrier()

    # max over ranks
    if use_dist:
        te = torch.tensor([elapsed])
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te[0])

    ms_per_step = elapsed / args.steps * 1e3
    if world > 1:
        busbw = (2.0 * (world - 1) / world) * nbytes / (elapsed / args.steps)
    else:
        busbw = nbytes / (elapsed / args.steps)
    value = busbw / 1e9

    if rank == 0:
        print(json.dumps({
            "metric": "allreduce_busbw_GBps",
            "value": round(value, 2),
            "unit": "GB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "allreduce-sweep-flagship",
                "buffer_mib_per_gpu": args.mbytes,
                "global_batch": None,
                "seq_len": None,
                "parallelism": f"xgmi-fullmesh-{world}gpu",
                "engine": ("uccl_amd symmetric zero-copy twoshot"
                           if args.symmetric else
                           "uccl_amd twoshot RS+AG push kernels"),
                "symmetric": bool(args.symmetric and world > 1
                                  and comm.is_symmetric(t)),
                "n1_value_is_local_algbw": world == 1,
            },
        }))
    if use_dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
