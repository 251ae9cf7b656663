"""Worker for the EP dispatch/combine battery. Works at world=1 (launched
directly) or world=N (spawned with RANK/WORLD_SIZE + gloo rendezvous).

Checks uccl_amd.ep against a pure-torch reference of DeepEP low-latency
semantics: per-(expert, source) slot arrays in token order, then weighted
top-k combine reduced in fp32.
"""

from __future__ import annotations

import os
import signal

signal.alarm(int(os.environ.get("UCCL_TEST_ALARM", "240")))

import torch


def rank_inputs(r: int, T: int, H: int, K: int, E: int, dtype, seed: int):
    g = torch.Generator().manual_seed(seed + 1000 * r)
    x = torch.randn(T, H, generator=g, dtype=torch.float32).to(dtype)
    topk = torch.empty(T, K, dtype=torch.int64)
    for t in range(T):
        topk[t] = torch.randperm(E, generator=g)[:K]  # distinct experts
    w = torch.rand(T, K, generator=g, dtype=torch.float32)
    return x, topk, w


def fp8_check():
    """fp8 dispatch: quantized payloads + per-128 scales must dequantize
    back to the input within e4m3 relative error."""
    import uccl_amd.ep as uep

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    T, H, K = 128, 512, 2
    E = 4 * world
    buf = uep.Buffer(num_experts=E, topk=K, hidden=H, max_tokens=256,
                     dtype=torch.bfloat16, use_fp8=True)
    x, topk, w = rank_inputs(rank, T, H, K, E, torch.bfloat16, 77)
    recv_x, counts, recv_scale = buf.dispatch(x.cuda(), topk.cuda())
    torch.cuda.synchronize()
    local_E = E // world
    all_inputs = [rank_inputs(r, T, H, K, E, torch.bfloat16, 77)
                  for r in range(world)]
    counts_cpu = counts.cpu()
    for le in range(local_E):
        e = rank * local_E + le
        for src in range(world):
            sx, stopk, _ = all_inputs[src]
            sel = [t for t in range(T) if (stopk[t] == e).any()]
            n = counts_cpu[le, src].item()
            assert n == len(sel), (n, len(sel))
            q = recv_x[le, src * 256: src * 256 + n].float().cpu()
            sc = recv_scale[le, src * 256: src * 256 + n].cpu()
            deq = q.view(n, H // 128, 128) * sc.unsqueeze(-1)
            want = sx[sel].float().view(n, H // 128, 128)
            err = (deq - want).abs()
            ref = want.abs().amax(dim=-1, keepdim=True).clamp(min=1e-6)
            assert (err / ref).max() <= 0.08, float((err / ref).max())
    print(f"[rank {rank}] EP FP8 OK", flush=True)


def main():
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1:
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(int(os.environ.get("UCCL_TEST_DEVICE", "0")))

    import uccl_amd.ep as uep

    T, H, K = 256, 512, 4
    E = 8 * world  # local_experts = 8
    maxT = 512
    dtype = torch.bfloat16
    seed = int(os.environ.get("UCCL_TEST_SEED", "42"))

    buf = uep.Buffer(num_experts=E, topk=K, hidden=H, max_tokens=maxT,
                     dtype=dtype)
    local_E = E // world

    x, topk, w = rank_inputs(rank, T, H, K, E, dtype, seed)
    xg, topkg = x.cuda(), topk.cuda()

    recv_x, counts = buf.dispatch(xg, topkg)
    torch.cuda.synchronize()

    # ---- reference: recompute all ranks' routing on CPU --------------------
    all_inputs = [rank_inputs(r, T, H, K, E, dtype, seed) for r in range(world)]
    counts_cpu = counts.cpu()
    for le in range(local_E):
        e = rank * local_E + le
        for src in range(world):
            sx, stopk, _ = all_inputs[src]
            sel = [t for t in range(T) if (stopk[t] == e).any()]
            assert counts_cpu[le, src].item() == len(sel), (
                f"count mismatch e={e} src={src}: "
                f"{counts_cpu[le, src].item()} vs {len(sel)}")
            got = recv_x[le, src * maxT: src * maxT + len(sel)].cpu()
            want = sx[sel]
            assert torch.equal(got.view(torch.int16), want.view(torch.int16)), \
                f"payload mismatch e={e} src={src}"
    print(f"[rank {rank}] dispatch OK", flush=True)

    # ---- run a fake 'expert': out = 2*x + expert_id -------------------------
    expert_out = recv_x.clone()
    for le in range(local_E):
        expert_out[le] = (recv_x[le].float() * 2 +
                          (rank * local_E + le)).to(dtype)

    combined = buf.combine(expert_out, topkg, w.cuda())
    torch.cuda.synchronize()

    # reference combine for MY tokens
    ref = torch.zeros(T, H, dtype=torch.float32)
    for t in range(T):
        for k in range(K):
            e = int(topk[t, k])
            fx = (x[t].to(torch.float32) * 2 + e)
            # expert computation happened in `dtype` precision:
            fx = fx.to(dtype).to(torch.float32)
            ref[t] += w[t, k] * fx
    ref = ref.to(dtype)
    got = combined.cpu()
    diff = (got.float() - ref.float()).abs().max().item()
    scale = ref.float().abs().max().item()
    assert diff <= 0.05 * max(scale, 1.0), f"combine diff {diff} scale {scale}"
    print(f"[rank {rank}] combine OK (maxdiff {diff:.4f})", flush=True)

    # ---- repeat to exercise seq tagging across calls ------------------------
    for it in range(3):
        x2, topk2, w2 = rank_inputs(rank, T, H, K, E, dtype, seed + 7 + it)
        rx, cnts = buf.dispatch(x2.cuda(), topk2.cuda())
        out2 = buf.combine(rx, topk2.cuda(), w2.cuda())
        torch.cuda.synchronize()
        assert out2.shape == (T, H)
    if os.environ.get("UCCL_EP_FORCE_PROXY", "0") != "1":
        fp8_check()
    print(f"[rank {rank}] EP ALL OK", flush=True)
    if world > 1:
        import torch.distributed as dist

        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
