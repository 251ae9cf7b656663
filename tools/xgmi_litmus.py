#!/usr/bin/env python3
"""Peer-HBM visibility litmus for real multi-GPU xGMI.

Every multi-rank GPU result so far ran N processes on ONE MI355X over
HIP IPC; the cross-device claim (dispatch-boundary release publishes
stores to PEER HBM; acquire on the other GPU sees them) is exactly the
class of bug the reference hit on AMD (ep/README.md:139). This script
proves it in minutes on any >=2-GPU lease:

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 tools/xgmi_litmus.py

Checks, each across REAL xGMI (one process per GPU):
  1. host-sequenced send/recv (release-store flag + payload, acquire
     read) with pattern verification both directions
  2. LL-packet allreduce (relaxed 64-bit flagged packets)
  3. one-shot fullmesh allreduce (dispatch-boundary visibility)
  4. two-shot RS+AG push (peer-write visibility + flag rounds)
  5. symmetric zero-copy path
  6. EP dispatch/combine over xGMI slot arrays

Exit 0 + "XGMI LITMUS PASSED" on success; each failure names the rank,
the path and the first mismatching element.
"""

import os
import sys

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local = int(os.environ.get("LOCAL_RANK", rank))
    assert world >= 2, "litmus needs >=2 ranks (one per GPU)"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    ngpu = torch.cuda.device_count()
    assert ngpu >= world, (
        f"litmus wants one REAL GPU per rank (have {ngpu}, world {world}); "
        "on a 1-GPU box this script does not prove anything new")
    torch.cuda.set_device(local)

    import uccl_amd.collective as ucol
    comm = ucol.init()

    def report(name, ok, detail=""):
        print(f"[rank {rank}] {name}: {'OK' if ok else 'FAIL ' + detail}",
              flush=True)
        if not ok:
            dist.destroy_process_group()
            sys.exit(1)

    # 1. send/recv pattern both directions (flag + data ordering)
    n = 1 << 20
    pat = torch.arange(n, dtype=torch.float32, device="cuda") * (rank + 1)
    got = torch.zeros(n, dtype=torch.float32, device="cuda")
    peer = rank ^ 1
    if peer < world:
        if rank < peer:
            comm.send(pat, peer)
            comm.recv(got, peer)
        else:
            comm.recv(got, peer)
            comm.send(pat, peer)
        torch.cuda.synchronize()
        want = torch.arange(n, dtype=torch.float32, device="cuda") * (peer + 1)
        ok = torch.equal(got, want)
        bad = (got != want).nonzero()
        report("send/recv", ok,
               f"first bad elem {bad[0].item() if len(bad) else '?'}")

    # 2..4. allreduce across its three algorithm paths
    for label, count in (("ll", 1000), ("oneshot", 200000),
                         ("twoshot", 3 << 20)):
        x = torch.full((count,), float(rank + 1), dtype=torch.bfloat16,
                       device="cuda")
        comm.all_reduce(x)
        torch.cuda.synchronize()
        want = float(world * (world + 1) // 2)
        ok = bool(torch.all(x == want))
        report(f"allreduce[{label}]", ok, f"got {float(x[0])} want {want}")

    # 5. symmetric zero-copy
    st = comm.symmetric_tensor([1 << 20], torch.bfloat16)
    st.fill_(float(rank + 2))
    comm.all_reduce(st)
    torch.cuda.synchronize()
    want = float(sum(r + 2 for r in range(world)))
    report("allreduce[symmetric]", bool(torch.all(st == want)),
           f"got {float(st[0])}")

    # 6. EP dispatch/combine
    import uccl_amd.ep as uep
    E, K, H, T = 4 * world, 2, 512, 64
    buf = uep.Buffer(num_experts=E, topk=K, hidden=H, max_tokens=128,
                     dtype=torch.bfloat16)
    g = torch.Generator().manual_seed(7 + rank)
    x = torch.randn(T, H, generator=g).to(torch.bfloat16).cuda()
    topk = torch.stack([torch.randperm(E, generator=g)[:K]
                        for _ in range(T)]).cuda()
    w = torch.rand(T, K, generator=g).cuda()
    rx, counts = buf.dispatch(x, topk)
    out = buf.combine(rx.clone(), topk, w)
    torch.cuda.synchronize()
    ref = (w.sum(dim=1, keepdim=True).cpu() * x.float().cpu()).to(
        torch.bfloat16)
    diff = (out.cpu().float() - ref.float()).abs().max().item()
    report("ep dispatch/combine", diff < 0.1, f"maxdiff {diff}")

    dist.barrier()
    if rank == 0:
        print("XGMI LITMUS PASSED", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
