#!/usr/bin/env python3
"""MoE expert-parallel demo: dispatch -> per-expert MLP -> combine through
uccl_amd.ep (parity with the reference's ep/bench test flow). Launch with
torch.distributed.run, one rank per GPU."""
import os

import torch
import torch.distributed as dist

import uccl_amd.ep as uep


def main():
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    if world > 1:
        dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) %
                          torch.cuda.device_count())

    T, H, E, K = 1024, 1024, 8 * world, 2
    buf = uep.Buffer(num_experts=E, topk=K, hidden=H, max_tokens=T,
                     dtype=torch.bfloat16)
    local_E = E // world
    experts = [torch.nn.Linear(H, H).cuda().bfloat16()
               for _ in range(local_E)]

    g = torch.Generator().manual_seed(7 + rank)
    x = torch.randn(T, H, generator=g).bfloat16().cuda()
    topk_idx = torch.stack([torch.randperm(E, generator=g)[:K]
                            for _ in range(T)]).cuda()
    topk_w = torch.softmax(torch.rand(T, K, generator=g), -1).cuda()

    recv_x, counts = buf.dispatch(x, topk_idx)
    expert_out = torch.empty_like(recv_x)
    for le in range(local_E):
        with torch.no_grad():
            expert_out[le] = experts[le](recv_x[le])
    y = buf.combine(expert_out, topk_idx, topk_w)
    torch.cuda.synchronize()
    if rank == 0:
        print(f"moe_ep OK: y {tuple(y.shape)} mean {y.float().mean():.4f}")
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
