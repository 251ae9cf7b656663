#!/usr/bin/env python3
"""DDP training demo over the native "uccl" backend (parity with the
reference's examples/ddp_train.py, which demos DDP over the UCCL plugin).

Launch:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 examples/ddp_train.py
"""
import os

import torch
import torch.distributed as dist
from torch.nn.parallel import DistributedDataParallel as DDP

import uccl_amd.collective as ucol


def main():
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) %
                          torch.cuda.device_count())
    ucol.register_torch_backend()
    dist.init_process_group("uccl", rank=rank, world_size=world)

    model = torch.nn.Sequential(
        torch.nn.Linear(1024, 4096), torch.nn.ReLU(),
        torch.nn.Linear(4096, 1024)).cuda().bfloat16()
    ddp = DDP(model)
    opt = torch.optim.SGD(ddp.parameters(), lr=1e-3)

    g = torch.Generator().manual_seed(1 + rank)
    for step in range(20):
        x = torch.randn(64, 1024, generator=g).bfloat16().cuda()
        loss = ddp(x).float().square().mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
        if rank == 0 and step % 5 == 0:
            print(f"step {step} loss {loss.item():.4f}", flush=True)
    dist.destroy_process_group()
    if rank == 0:
        print("ddp_train OK")


if __name__ == "__main__":
    main()
