"""Worker: torch.distributed with the native "uccl" c10d backend."""
import os, signal
signal.alarm(int(os.environ.get("UCCL_TEST_ALARM", "200")))

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(int(os.environ.get("UCCL_TEST_DEVICE", "0")))
    import uccl_amd.collective as ucol
    ucol.register_torch_backend()
    dist.init_process_group("uccl", rank=rank, world_size=world)

    t = torch.full((10000,), float(rank + 1), device="cuda")
    dist.all_reduce(t)
    torch.cuda.synchronize()
    want = sum(r + 1 for r in range(world))
    assert torch.allclose(t, torch.full_like(t, want)), t[:4]

    ta = torch.full((4096,), float(rank + 1), device="cuda")
    dist.all_reduce(ta, op=dist.ReduceOp.AVG)
    torch.cuda.synchronize()
    assert torch.allclose(ta, torch.full_like(ta, want / world)), ta[:4]

    b = torch.full((64,), float(rank), device="cuda")
    dist.broadcast(b, src=0)
    torch.cuda.synchronize()
    assert torch.allclose(b, torch.zeros_like(b))

    inp = torch.full((256,), float(rank), device="cuda")
    out = torch.empty(256 * world, device="cuda")
    dist.all_gather_into_tensor(out, inp)
    torch.cuda.synchronize()
    for r in range(world):
        assert torch.allclose(out[r * 256:(r + 1) * 256],
                              torch.full((256,), float(r), device="cuda"))

    big = torch.full((512 * world,), float(rank + 1), device="cuda")
    small = torch.empty(512, device="cuda")
    dist.reduce_scatter_tensor(small, big)
    torch.cuda.synchronize()
    assert torch.allclose(small, torch.full_like(small, float(want)))

    src = torch.arange(world * 8, dtype=torch.float32, device="cuda") + rank * 1000
    dst = torch.empty_like(src)
    dist.all_to_all_single(dst, src)
    torch.cuda.synchronize()
    for r in range(world):
        seg = dst[r * 8:(r + 1) * 8].cpu()
        want_seg = torch.arange(rank * 8, rank * 8 + 8, dtype=torch.float32) + r * 1000
        assert torch.equal(seg, want_seg), (r, seg, want_seg)

    # reduce / gather / scatter
    r = torch.full((128,), float(rank + 1), device="cuda")
    dist.reduce(r, dst=0)
    torch.cuda.synchronize()
    if rank == 0:
        assert torch.allclose(r, torch.full_like(r, float(want)))

    gin = torch.full((64,), float(rank * 10), device="cuda")
    gouts = [torch.empty(64, device="cuda") for _ in range(world)] \
        if rank == 0 else []
    dist.gather(gin, gouts if rank == 0 else None, dst=0)
    torch.cuda.synchronize()
    if rank == 0:
        for rr in range(world):
            assert torch.allclose(gouts[rr],
                                  torch.full((64,), float(rr * 10),
                                             device="cuda"))

    sout = torch.empty(32, device="cuda")
    sins = [torch.full((32,), float(100 + rr), device="cuda")
            for rr in range(world)] if rank == 0 else None
    dist.scatter(sout, sins, src=0)
    torch.cuda.synchronize()
    assert torch.allclose(sout, torch.full_like(sout, float(100 + rank)))

    # subgroup: each rank in its own single-member group (creator is
    # invoked per group with a prefixed store). new_group must be called
    # by ALL ranks with the SAME list, once per group:
    solo = None
    for r in range(world):
        g = dist.new_group([r], backend="uccl")
        if r == rank:
            solo = g
    t1 = torch.full((16,), float(rank), device="cuda")
    dist.all_reduce(t1, group=solo)
    torch.cuda.synchronize()
    assert torch.allclose(t1, torch.full_like(t1, float(rank)))

    dist.barrier()

    # DDP end-to-end: tiny model, grads must match single-process reference
    model = torch.nn.Sequential(
        torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 8)
    ).cuda()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    ddp = torch.nn.parallel.DistributedDataParallel(model)
    g = torch.Generator().manual_seed(500 + rank)
    x = torch.randn(16, 32, generator=g).cuda()
    y = ddp(x).square().mean()
    y.backward()
    torch.cuda.synchronize()
    # reference grads: average of per-rank grads computed locally
    ref_model = torch.nn.Sequential(
        torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 8)
    ).cuda()
    with torch.no_grad():
        for rp, p in zip(ref_model.parameters(), model.parameters()):
            rp.copy_(p)
    grads_acc = [torch.zeros_like(p) for p in ref_model.parameters()]
    for r in range(world):
        gr = torch.Generator().manual_seed(500 + r)
        xr = torch.randn(16, 32, generator=gr).cuda()
        ref_model.zero_grad()
        ref_model(xr).square().mean().backward()
        for acc, p in zip(grads_acc, ref_model.parameters()):
            acc += p.grad / world
    for p, ref in zip(model.parameters(), grads_acc):
        assert torch.allclose(p.grad, ref, atol=1e-5), \
            (p.grad - ref).abs().max()

    # FSDP end-to-end (exercises _allgather_base/_reduce_scatter_base)
    try:
        from torch.distributed.fsdp import FullyShardedDataParallel as FSDP

        fm = torch.nn.Sequential(
            torch.nn.Linear(64, 128), torch.nn.ReLU(),
            torch.nn.Linear(128, 16)).cuda()
        for p_ in fm.parameters():
            dist.broadcast(p_.data, src=0)
        fsdp = FSDP(fm, device_id=torch.cuda.current_device())
        gx = torch.Generator().manual_seed(900 + rank)
        loss = fsdp(torch.randn(8, 64, generator=gx).cuda()).square().mean()
        loss.backward()
        torch.cuda.synchronize()
        if rank == 0:
            print("FSDP OK", flush=True)
    except ImportError:
        pass

    dist.barrier()
    if rank == 0:
        print("PG BACKEND ALL OK", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
