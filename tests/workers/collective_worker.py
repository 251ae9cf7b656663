"""Worker for the 2-process collective correctness battery.

Launched by tests/test_collective_2proc_gpu.py with RANK/WORLD_SIZE env.
Both ranks may share one physical GPU (gpurun boxes have a single MI355X):
HIP IPC handles open fine across processes on the same device, so the full
multi-rank protocol (flags, LL slots, parity scratch) is exercised even
when xGMI itself is not.

Every collective result is checked against a torch fp32/exact reference
computed from deterministic per-rank inputs.
"""

from __future__ import annotations

import os
import signal

signal.alarm(int(os.environ.get("UCCL_TEST_ALARM", "240")))

import torch
import torch.distributed as dist


def make_input(rank: int, count: int, dtype, seed_tag: int) -> torch.Tensor:
    g = torch.Generator().manual_seed(1234 + 97 * rank + seed_tag)
    if dtype in (torch.int32, torch.int64):
        return torch.randint(-1000000, 1000000, (count,), generator=g,
                             dtype=dtype).cuda()
    if dtype == torch.float64:
        return torch.randn(count, generator=g, dtype=torch.float64).cuda()
    return torch.randn(count, generator=g, dtype=torch.float32).to(dtype).cuda()


def expected_sum(world: int, count: int, dtype, seed_tag: int) -> torch.Tensor:
    # fp32 (or int64) reference accumulation on CPU, then cast once — this is
    # exactly what the kernels' AccumV16 fp32 accumulators implement.
    if dtype in (torch.int32, torch.int64):
        acc = torch.zeros(count, dtype=torch.int64)
        for r in range(world):
            g = torch.Generator().manual_seed(1234 + 97 * r + seed_tag)
            acc += torch.randint(-1000000, 1000000, (count,), generator=g,
                                 dtype=dtype).to(torch.int64)
        return acc.to(dtype)
    if dtype == torch.float64:
        acc = torch.zeros(count, dtype=torch.float64)
        for r in range(world):
            g = torch.Generator().manual_seed(1234 + 97 * r + seed_tag)
            acc += torch.randn(count, generator=g, dtype=torch.float64)
        return acc
    acc = torch.zeros(count, dtype=torch.float32)
    for r in range(world):
        g = torch.Generator().manual_seed(1234 + 97 * r + seed_tag)
        acc += torch.randn(count, generator=g, dtype=torch.float32).to(
            dtype).to(torch.float32)
    return acc.to(dtype)


def check(name: str, got: torch.Tensor, want: torch.Tensor, tol: float):
    got = got.cpu()
    want = want.cpu()
    if tol == 0:
        ok = torch.equal(got, want)  # raw-dtype exact compare
    else:
        ok = torch.allclose(got.float(), want.float(), rtol=tol, atol=tol)
    if not ok:
        diff = (got - want).abs().max().item()
        raise AssertionError(f"{name}: max diff {diff}")
    print(f"[rank {dist.get_rank()}] {name} OK", flush=True)


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(int(os.environ.get("UCCL_TEST_DEVICE", "0")))

    import uccl_amd.collective as ucol

    # small heap so the large message below genuinely exercises chunking
    light_heap = 64 if os.environ.get("UCCL_TEST_LIGHT", "0") == "1" else 192
    comm = ucol.init(device=torch.cuda.current_device(),
                     heap_bytes=light_heap * (1 << 20))

    light = os.environ.get("UCCL_TEST_LIGHT", "0") == "1"
    # fixed-order accumulation makes fp32/fp64/int results bit-exact
    tol = {torch.float32: 0.0, torch.int32: 0.0, torch.int64: 0.0,
           torch.float64: 0.0, torch.bfloat16: 1e-2, torch.float16: 1e-3}

    seed = 0
    # --- allreduce across the three algorithm paths -------------------------
    # LL (<=32KB), one-shot (<=2MB), two-shot (>2MB), chunked two-shot
    cases = [(1000, "ll"), (100000, "oneshot"), (3 << 20, "twoshot"),
             (1000003, "oneshot-odd")]
    dtypes = (torch.float32, torch.bfloat16, torch.float16, torch.int32,
              torch.int64, torch.float64)
    if light:
        cases = [(1000, "ll"), (100000, "oneshot"), (3 << 20, "twoshot")]
        dtypes = (torch.float32, torch.bfloat16)
    for count, label in cases:
        for dtype in dtypes:
            seed += 1
            x = make_input(rank, count, dtype, seed)
            comm.all_reduce(x)
            torch.cuda.synchronize()
            check(f"allreduce[{label},{dtype}]", x,
                  expected_sum(world, count, dtype, seed), tol[dtype])

    # chunked two-shot: shrink scratch via a dedicated small-heap comm is
    # heavy; instead pick a size above the default 2MB oneshot threshold and
    # large enough to need >1 chunk only if scratch < size. Default scratch
    # is tens of MB, so force chunking with a big-ish message.
    seed += 1
    count = (8 if light else 48) << 20  # > per-parity scratch -> chunked
    x = make_input(rank, count, torch.float32, seed)
    comm.all_reduce(x)
    torch.cuda.synchronize()
    check("allreduce[chunked]", x,
          expected_sum(world, count, torch.float32, seed), 0.0)

    # --- allreduce reduction ops (min / max / prod) across all algo paths ---
    def expected_op(op, count, dtype, seed_tag):
        ins = torch.stack([
            make_input(r, count, dtype, seed_tag).cpu().float()
            for r in range(world)])
        if op == "min":
            ref = ins.min(dim=0).values
        elif op == "max":
            ref = ins.max(dim=0).values
        else:
            ref = ins.prod(dim=0)  # fp32 accumulation, like the kernels
        return ref.to(dtype)

    op_cases = [(5000, "ll"), (200000, "oneshot"), ((3 << 20) + 5, "twoshot")]
    op_dtypes = (torch.float32, torch.bfloat16, torch.int32)
    if light:
        op_cases = op_cases[:2]
        op_dtypes = (torch.float32, torch.bfloat16)
    for count, label in op_cases:
        for op in ("min", "max", "prod"):
            for dtype in op_dtypes:
                if dtype == torch.int32 and op == "prod":
                    continue  # int overflow wraps; not a meaningful check
                seed += 1
                x = make_input(rank, count, dtype, seed)
                comm.all_reduce(x, op=op)
                torch.cuda.synchronize()
                check(f"allreduce[{op},{label},{dtype}]", x,
                      expected_op(op, count, dtype, seed), tol[dtype])

    # reduce_scatter with a non-sum op
    seed += 1
    n = 8192
    mine = make_input(rank, world * n, torch.float32, seed)
    rs_out = torch.empty(n, dtype=torch.float32, device="cuda")
    comm.reduce_scatter(rs_out, mine, op="max")
    torch.cuda.synchronize()
    full = expected_op("max", world * n, torch.float32, seed)
    check("reduce_scatter[max]", rs_out, full[rank * n:(rank + 1) * n], 0.0)

    # --- allgather ----------------------------------------------------------
    seed += 1
    n = 12345
    mine = make_input(rank, n, torch.float32, seed)
    out = torch.empty(world * n, dtype=torch.float32, device="cuda")
    comm.all_gather(out, mine)
    torch.cuda.synchronize()
    want = torch.cat([
        make_input(r, n, torch.float32, seed).cpu() for r in range(world)])
    check("allgather", out, want, 0.0)

    # --- chunked allgather (count > per-parity scratch => strided pulls) ---
    if not light:
        seed += 1
        n = 3 << 20  # 12MB fp32 > 192MB-heap parity scratch? cap=(192-48)/4=36MB; force via more
        n = 12 << 20  # 48MB fp32 > 36MB parity scratch -> chunked path
        mine = make_input(rank, n, torch.float32, seed)
        out = torch.empty(world * n, dtype=torch.float32, device="cuda")
        comm.all_gather(out, mine)
        torch.cuda.synchronize()
        for r in range(world):
            want = make_input(r, n, torch.float32, seed).cpu()
            got = out[r * n:(r + 1) * n].cpu()
            assert torch.equal(got, want), f"chunked allgather rank {r}"
        print(f"[rank {rank}] allgather[chunked] OK", flush=True)
        del out, mine

    # --- chunked alltoall ---------------------------------------------------
    if not light:
        seed += 1
        per = (10 << 20) // world  # total 40MB fp32 > cap/world chunks
        inp = make_input(rank, per * world, torch.float32, seed)
        out = torch.empty_like(inp)
        comm.all_to_all(out, inp)
        torch.cuda.synchronize()
        want = torch.cat([
            make_input(r, per * world, torch.float32,
                       seed)[rank * per:(rank + 1) * per].cpu()
            for r in range(world)
        ])
        check("alltoall[chunked]", out, want, 0.0)
        del out, inp

    # --- reduce_scatter -----------------------------------------------------
    seed += 1
    per = 4096
    inp = make_input(rank, per * world, torch.float32, seed)
    out = torch.empty(per, dtype=torch.float32, device="cuda")
    comm.reduce_scatter(out, inp)
    torch.cuda.synchronize()
    full = expected_sum(world, per * world, torch.float32, seed)
    check("reduce_scatter", out, full[rank * per:(rank + 1) * per], 0.0)

    # --- broadcast ----------------------------------------------------------
    seed += 1
    x = make_input(rank, 9999, torch.bfloat16, seed)
    comm.broadcast(x, root=0)
    torch.cuda.synchronize()
    check("broadcast", x, make_input(0, 9999, torch.bfloat16, seed).cpu(),
          0.0)

    # --- all_to_all ---------------------------------------------------------
    seed += 1
    per = 2048
    inp = make_input(rank, per * world, torch.float32, seed)
    out = torch.empty_like(inp)
    comm.all_to_all(out, inp)
    torch.cuda.synchronize()
    want = torch.cat([
        make_input(r, per * world, torch.float32,
                   seed)[rank * per:(rank + 1) * per].cpu()
        for r in range(world)
    ])
    check("all_to_all", out, want, 0.0)

    # --- chunked broadcast / reduce_scatter --------------------------------
    if not light:
        seed += 1
        n = 24 << 20  # 96MB fp32 > 36MB parity scratch -> multi-chunk bcast
        x = make_input(rank, n, torch.float32, seed)
        comm.broadcast(x, root=1 if world > 1 else 0)
        torch.cuda.synchronize()
        check("broadcast[chunked]", x,
              make_input(1 if world > 1 else 0, n, torch.float32,
                         seed).cpu(), 0.0)
        del x
        seed += 1
        per = 10 << 20  # per-rank 40MB fp32; full input world*40MB chunked
        inp = make_input(rank, per * world, torch.float32, seed)
        out = torch.empty(per, dtype=torch.float32, device="cuda")
        comm.reduce_scatter(out, inp)
        torch.cuda.synchronize()
        full = expected_sum(world, per * world, torch.float32, seed)
        check("reduce_scatter[chunked]", out,
              full[rank * per:(rank + 1) * per], 0.0)
        del inp, out

    # --- send/recv (pairwise ring) ------------------------------------------
    seed += 1
    n = 300000  # spans >1 p2p slot chunk at 2MB slots? 1.2MB -> single chunk
    if world >= 2:
        peer_to = (rank + 1) % world
        peer_from = (rank - 1 + world) % world
        payload = make_input(rank, n, torch.float32, seed)
        got = torch.empty(n, dtype=torch.float32, device="cuda")
        if rank % 2 == 0:
            comm.send(payload, peer_to)
            comm.recv(got, peer_from)
        else:
            comm.recv(got, peer_from)
            comm.send(payload, peer_to)
        torch.cuda.synchronize()
        check("send/recv", got, make_input(peer_from, n, torch.float32,
                                           seed).cpu(), 0.0)

        # multi-chunk send (> 2MB slot)
        seed += 1
        n2 = 1500000  # 6MB fp32 -> 3 chunks
        payload = make_input(rank, n2, torch.float32, seed)
        got = torch.empty(n2, dtype=torch.float32, device="cuda")
        if rank % 2 == 0:
            comm.send(payload, peer_to)
            comm.recv(got, peer_from)
        else:
            comm.recv(got, peer_from)
            comm.send(payload, peer_to)
        torch.cuda.synchronize()
        check("send/recv-chunked", got,
              make_input(peer_from, n2, torch.float32, seed).cpu(), 0.0)

    # --- barrier + interleave stress ---------------------------------------
    comm.barrier()
    for i in range(5 if light else 20):  # rapid-fire parity/seq stress
        seed += 1
        x = make_input(rank, 257, torch.float32, seed)
        comm.all_reduce(x)
        torch.cuda.synchronize()
        check(f"stress[{i}]", x, expected_sum(world, 257, torch.float32,
                                              seed), 0.0)

    # --- symmetric (zero-copy) allreduce -----------------------------------
    for count in ([100000, 3 << 20] if not light else [100000]):
        for dtype in (torch.float32, torch.bfloat16):
            seed += 1
            st = comm.symmetric_tensor([count], dtype)
            st.copy_(make_input(rank, count, dtype, seed))
            assert comm.is_symmetric(st)
            comm.all_reduce(st)
            torch.cuda.synchronize()
            check(f"allreduce[sym,{dtype},{count}]", st,
                  expected_sum(world, count, dtype, seed), tol[dtype])
    # symmetric all_gather (out symmetric) + reduce_scatter (in symmetric)
    seed += 1
    n = 65536
    ag_out = comm.symmetric_tensor([world * n], torch.float32)
    mine = make_input(rank, n, torch.float32, seed)
    comm.all_gather(ag_out, mine)
    torch.cuda.synchronize()
    want = torch.cat([make_input(r, n, torch.float32, seed).cpu()
                      for r in range(world)])
    check("allgather[sym]", ag_out, want, 0.0)

    seed += 1
    per = 32768
    rs_in = comm.symmetric_tensor([world * per], torch.float32)
    rs_in.copy_(make_input(rank, world * per, torch.float32, seed))
    rs_out = torch.empty(per, dtype=torch.float32, device="cuda")
    comm.reduce_scatter(rs_out, rs_in)
    torch.cuda.synchronize()
    full = expected_sum(world, world * per, torch.float32, seed)
    check("reduce_scatter[sym]", rs_out,
          full[rank * per:(rank + 1) * per], 0.0)

    # symmetric all_to_all (out symmetric)
    seed += 1
    per = 16384
    a2a_out = comm.symmetric_tensor([world * per], torch.float32)
    a2a_in = make_input(rank, per * world, torch.float32, seed)
    comm.all_to_all(a2a_out, a2a_in)
    torch.cuda.synchronize()
    want = torch.cat([
        make_input(r, per * world, torch.float32,
                   seed)[rank * per:(rank + 1) * per].cpu()
        for r in range(world)
    ])
    check("alltoall[sym]", a2a_out, want, 0.0)

    # repeated in-place reuse of the same symmetric tensor
    seed += 1
    st = comm.symmetric_tensor([4096], torch.float32)
    st.copy_(make_input(rank, 4096, torch.float32, seed))
    comm.all_reduce(st)
    comm.all_reduce(st)  # sum of sums
    torch.cuda.synchronize()
    once = expected_sum(world, 4096, torch.float32, seed).float()
    check("allreduce[sym,repeat]", st, (once * world).to(torch.float32), 1e-4)

    # --- fp8 (OCP e4m3) allreduce across LL / oneshot / twoshot paths ------
    if hasattr(torch, "float8_e4m3fn"):
        for count in ((1000, 300000) if light else (1000, 300000, 5 << 20)):
            seed += 1
            gens = [torch.Generator().manual_seed(4321 + 13 * r + seed)
                    for r in range(world)]
            vals = [torch.randn(count, generator=g) * 0.5 for g in gens]
            x8 = vals[rank].to(torch.float8_e4m3fn).cuda()
            comm.all_reduce(x8)
            torch.cuda.synchronize()
            ref = sum(v.to(torch.float8_e4m3fn).float() for v in vals)
            ref8 = ref.to(torch.float8_e4m3fn).float()
            got = x8.float().cpu()
            # fp8 quantization of the fp32-accumulated sum: one-ulp slack
            assert torch.allclose(got, ref8, rtol=0.15, atol=0.1), \
                (got - ref8).abs().max()
            print(f"[rank {rank}] fp8 allreduce[{count}] OK", flush=True)

    dist.barrier()
    if rank == 0:
        print("ALL COLLECTIVE TESTS PASSED", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
