import os, sys, torch
import torch.distributed as dist
rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
dist.init_process_group("gloo", rank=rank, world_size=world)
torch.cuda.set_device(0)
import uccl_amd.ep as uep
T, H, K, E = 64, 256, 4, 8 * world
buf = uep.Buffer(num_experts=E, topk=K, hidden=H, max_tokens=128,
                 dtype=torch.bfloat16)
g = torch.Generator().manual_seed(1 + rank)
for it in range(4):
    x = torch.randn(T, H, generator=g, dtype=torch.float32).to(torch.bfloat16)
    topk = torch.stack([torch.randperm(E, generator=g)[:K] for _ in range(T)])
    w = torch.rand(T, K, generator=g, dtype=torch.float32)
    rx, c = buf.dispatch(x.cuda(), topk.cuda())
    torch.cuda.synchronize()
    print(f"[rank {rank}] it{it} dispatch done", flush=True)
    out = buf.combine(rx.clone(), topk.cuda(), w.cuda())
    torch.cuda.synchronize()
    print(f"[rank {rank}] it{it} combine done", flush=True)
print(f"[rank {rank}] REPRO OK", flush=True)
dist.barrier(); dist.destroy_process_group()
