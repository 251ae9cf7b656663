import os, sys, torch
import torch.distributed as dist
rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
dist.init_process_group("gloo", rank=rank, world_size=world)
torch.cuda.set_device(0)
import uccl_amd.ep as uep
T, H, K, E, maxT = 256, 512, 4, 8 * world, 512
buf = uep.Buffer(num_experts=E, topk=K, hidden=H, max_tokens=maxT,
                 dtype=torch.bfloat16)
g = torch.Generator().manual_seed(3 + rank)
def mk():
    x = torch.randn(T, H, generator=g, dtype=torch.float32).to(torch.bfloat16)
    topk = torch.stack([torch.randperm(E, generator=g)[:K] for _ in range(T)])
    w = torch.rand(T, K, generator=g, dtype=torch.float32)
    return x.cuda(), topk.cuda(), w.cuda()
mode = os.environ.get("REPRO_MODE", "matmul")
def step(tag):
    print(f"[rank {rank}] {tag}", flush=True)
for it in range(3):
    x, topk, w = mk()
    step(f"it{it} send...")
    buf.dispatch_send(x, topk)
    step(f"it{it} sent")
    if mode == "matmul":
        d = torch.randn(1024, 1024, device="cuda") @ \
            torch.randn(1024, 1024, device="cuda")
        torch.cuda.synchronize()
        step(f"it{it} matmul done")
    c = buf.dispatch_recv()
    torch.cuda.synchronize()
    step(f"it{it} recv done counts_sum={int(c.sum())}")
    rx = buf.recv_x_view()
    out = buf.combine(rx.clone(), topk, w)
    torch.cuda.synchronize()
    step(f"it{it} combine done")
# eager double dispatch (no combine between)
x, topk, w = mk()
step("double: d1")
r1, c1 = buf.dispatch(x, topk)
torch.cuda.synchronize()
step("double: d2")
r2, c2 = buf.dispatch(x, topk)
torch.cuda.synchronize()
step("double: done")
print(f"[rank {rank}] PHASE REPRO OK", flush=True)
dist.barrier(); dist.destroy_process_group()
