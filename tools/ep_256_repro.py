import os, sys, torch
import torch.distributed as dist
rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
dist.init_process_group("gloo", rank=rank, world_size=world)
torch.cuda.set_device(0)
import uccl_amd.ep as uep
def step(t): print(f"[rank {rank}] {t}", flush=True)

# first buffer like the worker's main one (proxy threads already running)
buf = uep.Buffer(num_experts=8 * world, topk=4, hidden=512, max_tokens=512,
                 dtype=torch.bfloat16)
g = torch.Generator().manual_seed(5 + rank)
x = torch.randn(64, 512, generator=g).to(torch.bfloat16).cuda()
tk = torch.stack([torch.randperm(8 * world, generator=g)[:4]
                  for _ in range(64)]).cuda()
w = torch.rand(64, 4, generator=g).cuda()
r0, c0 = buf.dispatch(x, tk)
o0 = buf.combine(r0.clone(), tk, w)
torch.cuda.synchronize()
step("buf1 warm OK")

# second buffer: 256 experts like the worker section
E2 = 256
buf2 = uep.Buffer(num_experts=E2, topk=8, hidden=256, max_tokens=128,
                  dtype=torch.bfloat16)
step("buf2 created")
x5 = torch.randn(64, 256, generator=g).to(torch.bfloat16).cuda()
t5 = torch.stack([torch.randperm(E2, generator=g)[:8]
                  for _ in range(64)]).cuda()
w5 = torch.rand(64, 8, generator=g).cuda()
step("dispatch...")
rx5, c5 = buf2.dispatch(x5, t5)
torch.cuda.synchronize()
step(f"dispatch done sum={int(c5.sum())}")
step("combine...")
out5 = buf2.combine(rx5.clone(), t5, w5)
torch.cuda.synchronize()
step("combine done")
buf2.close()
step("buf2 closed")

# sync cmds on buf1
b = buf._b
step("quiet...")
b.quiet(); torch.cuda.synchronize()
step("quiet done")
step("barrier...")
b.barrier(); torch.cuda.synchronize()
step("barrier done")
for dst in range(world):
    if dst != rank:
        b.atomic_add(dst, rank + 1)
step("atomic pushed")
b.barrier(); torch.cuda.synchronize()
got = b.read_sync_word(2)
want = sum(r + 1 for r in range(world) if r != rank)
step(f"atomic got={got} want={want}")
assert got == want
print(f"[rank {rank}] 256 REPRO OK", flush=True)
dist.barrier(); dist.destroy_process_group()
