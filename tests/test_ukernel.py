"""Planner / lowering / spray-executor tests (CPU; host mock backend).

Mirrors the reference's ukernel test strategy: single-process executor
unit tests over a mock backend (reference experimental/ukernel
src/ccl/README.md — mock-backend test_modules/test_spray_executor)."""

import pytest
import torch

from uccl_amd import ukernel as uk


def _inputs(world, elems, seed=0):
    g = torch.Generator().manual_seed(seed)
    # small integers -> fp32 sums are exact, so equality checks are strict
    return [torch.randint(-8, 8, (elems,), generator=g).float()
            for _ in range(world)]


def test_allreduce_rsag_correct():
    world, elems = 4, 4096
    topo = uk.Topology(world)
    ins = _inputs(world, elems)
    expect = torch.stack(ins).sum(0)
    g = uk.lower(uk.plan_allreduce_rsag(topo, elems * 4, 4,
                                        chunk_bytes=1024))
    outs, stats = uk.execute_host(g, ins)
    for o in outs:
        assert torch.equal(o, expect)
    assert stats["tasks_run"] == g.num_tasks


def test_allreduce_oneshot_correct():
    world, elems = 8, 512
    topo = uk.Topology(world)
    ins = _inputs(world, elems, seed=1)
    expect = torch.stack(ins).sum(0)
    g = uk.lower(uk.plan_allreduce_oneshot(topo, elems * 4))
    outs, _ = uk.execute_host(g, ins)
    for o in outs:
        assert torch.equal(o, expect)


def test_sendrecv_spray_uses_relays():
    # 8-rank fullmesh: a pairwise transfer must NOT ride one link only —
    # the planner sprays chunks over the direct link + 6 relay paths
    world, elems = 8, 7 * 1024
    topo = uk.Topology(world)
    ins = _inputs(world, elems, seed=2)
    g = uk.lower(uk.plan_sendrecv(topo, 0, 5, elems * 4, chunk_bytes=4096))
    outs, stats = uk.execute_host(g, ins)
    assert torch.equal(outs[5], ins[0])
    lm = uk.link_matrix(stats, world)
    used_out_links = sum(1 for d in range(world) if lm[0][d] > 0)
    assert used_out_links == 7  # direct + 6 relays
    # capacity-proportional: equal weights -> every source link carries
    # the same share (chunks divide evenly here)
    nz = [lm[0][d] for d in range(world) if lm[0][d] > 0]
    assert max(nz) == min(nz)


def test_sendrecv_degraded_link_shifts_load():
    world, elems = 4, 8 * 1024
    topo = uk.Topology(world)
    topo.set_link_weight(0, 3, 0.25)  # direct link degraded 4x
    ins = _inputs(world, elems, seed=3)
    g = uk.lower(uk.plan_sendrecv(topo, 0, 3, elems * 4, chunk_bytes=2048))
    outs, stats = uk.execute_host(g, ins)
    assert torch.equal(outs[3], ins[0])
    lm = uk.link_matrix(stats, world)
    # each healthy relay path should carry ~4x the direct link's bytes
    assert lm[0][1] > 2 * lm[0][3]
    assert lm[0][2] > 2 * lm[0][3]


def test_broadcast_correct():
    world, elems = 4, 2048
    topo = uk.Topology(world)
    ins = _inputs(world, elems, seed=4)
    g = uk.lower(uk.plan_broadcast(topo, 1, elems * 4, chunk_bytes=2048))
    outs, _ = uk.execute_host(g, ins)
    for o in outs:
        assert torch.equal(o, ins[1])


def test_lower_inserts_signal_wait_pairs():
    topo = uk.Topology(2)
    raw = uk.plan_allreduce_oneshot(topo, 256)
    low = uk.lower(raw)
    # every cross-rank dependency became a signal+wait pair
    assert low.num_tasks > raw.num_tasks
    assert "signal" in low.dump() and "wait" in low.dump()


def test_plan_deterministic():
    topo = uk.Topology(8)
    a = uk.plan_allreduce_rsag(topo, 1 << 16, 4, chunk_bytes=4096)
    b = uk.plan_allreduce_rsag(topo, 1 << 16, 4, chunk_bytes=4096)
    assert a.dump() == b.dump()


def test_ragged_sizes():
    # non-divisible payloads: last shard/chunk shorter
    world = 3
    topo = uk.Topology(world)
    for elems in (1, 5, 1023, 1025):
        ins = _inputs(world, elems, seed=elems)
        expect = torch.stack(ins).sum(0)
        g = uk.lower(uk.plan_allreduce_rsag(topo, elems * 4, 4,
                                            chunk_bytes=512))
        outs, _ = uk.execute_host(g, ins)
        for o in outs:
            assert torch.equal(o, expect)


def test_executor_requeues_waits():
    # oneshot at world 8 has genuine cross-rank racing: waits often poll
    # before their signal lands, exercising the deferred re-queue path
    world, elems = 8, 64 * 1024
    topo = uk.Topology(world)
    ins = _inputs(world, elems, seed=9)
    g = uk.lower(uk.plan_allreduce_oneshot(topo, elems * 4))
    outs, stats = uk.execute_host(g, ins)
    expect = torch.stack(ins).sum(0)
    for o in outs:
        assert torch.equal(o, expect)
    assert stats["tasks_run"] == g.num_tasks


def test_cost_model_monotonic_in_bytes():
    topo = uk.Topology(8)
    t_small = uk.estimate_us(uk.plan_allreduce_rsag(topo, 1 << 16, 4, 4096),
                             topo)
    t_big = uk.estimate_us(uk.plan_allreduce_rsag(topo, 1 << 24, 4, 1 << 20),
                           topo)
    assert 0 < t_small < t_big


def test_cost_model_degraded_link_costs_more():
    nbytes = 32 << 20
    topo = uk.Topology(8)
    base = uk.estimate_us(uk.plan_allreduce_oneshot(topo, nbytes), topo)
    slow = uk.Topology(8)
    slow.set_link_weight(0, 1, 0.1)
    worse = uk.estimate_us(uk.plan_allreduce_oneshot(slow, nbytes), slow)
    assert worse > base


def test_auto_planner_picks_by_size():
    topo = uk.Topology(8)
    # tiny: one-shot (fewer rounds beats per-link efficiency)
    tiny = uk.plan_allreduce_auto(topo, 4096)
    # huge: RS+AG (per-link bytes ~S/4 vs S for one-shot)
    huge = uk.plan_allreduce_auto(topo, 256 << 20, 4, 8 << 20)
    # distinguish the chosen family by its local-op signature: one-shot has
    # world-1 reduces of the FULL payload per rank; rsag reduces shards
    t_one = uk.estimate_us(uk.plan_allreduce_oneshot(topo, 256 << 20), topo)
    t_rsag = uk.estimate_us(
        uk.plan_allreduce_rsag(topo, 256 << 20, 4, 8 << 20), topo)
    assert t_rsag < t_one  # big payloads: RS+AG must win the estimate
    assert huge.num_tasks > tiny.num_tasks


def test_auto_planner_executes():
    world = 4
    topo = uk.Topology(world)
    ins = _inputs(world, 2048, seed=17)
    expect = torch.stack(ins).sum(0)
    g = uk.plan_allreduce_auto(topo, 2048 * 4, 4, 2048)
    outs, _ = uk.execute_host(g, ins)
    for o in outs:
        assert torch.equal(o, expect)


def test_planner_fuzz():
    # randomized geometry sweep across every planner family; fp32 sums on
    # small-int data are exact, so verification is strict equality
    import random

    rng = random.Random(20260912)
    for trial in range(12):
        world = rng.choice([2, 3, 4, 5, 8])
        elems = rng.randint(1, 5000)
        chunk = rng.choice([64, 512, 4096]) * 4
        topo = uk.Topology(world)
        if rng.random() < 0.5 and world > 2:
            a, b = rng.sample(range(world), 2)
            topo.set_link_weight(a, b, rng.choice([0.1, 0.5, 2.0]))
        ins = _inputs(world, elems, seed=trial)
        expect = torch.stack(ins).sum(0)
        fam = rng.choice(["rsag", "oneshot", "auto", "bcast", "send"])
        if fam == "rsag":
            g = uk.lower(uk.plan_allreduce_rsag(topo, elems * 4, 4, chunk))
        elif fam == "oneshot":
            g = uk.lower(uk.plan_allreduce_oneshot(topo, elems * 4))
        elif fam == "auto":
            g = uk.plan_allreduce_auto(topo, elems * 4, 4, chunk)
        elif fam == "bcast":
            g = uk.lower(uk.plan_broadcast(topo, world - 1, elems * 4,
                                           chunk))
        else:
            g = uk.lower(uk.plan_sendrecv(topo, 0, world - 1, elems * 4,
                                          chunk))
        outs, stats = uk.execute_host(g, ins)
        assert stats["tasks_run"] == g.num_tasks, (fam, world, elems)
        if fam in ("rsag", "oneshot", "auto"):
            for o in outs:
                assert torch.equal(o, expect), (fam, world, elems, chunk)
        elif fam == "bcast":
            for o in outs:
                assert torch.equal(o, ins[world - 1]), (world, elems)
        else:
            assert torch.equal(outs[world - 1], ins[0]), (world, elems)


def test_allgather_plan():
    world, elems = 4, 1000  # per-rank contribution
    topo = uk.Topology(world)
    ins = _inputs(world, elems, seed=21)
    g = uk.lower(uk.plan_allgather(topo, elems * 4, chunk_bytes=1024))
    outs, _ = uk.execute_host(g, ins, out_bytes=world * elems * 4)
    expect = torch.cat(ins)
    for o in outs:
        assert torch.equal(o, expect)


def test_reducescatter_plan():
    world, shard = 4, 600  # elems per shard; input = world*shard per rank
    topo = uk.Topology(world)
    ins = _inputs(world, world * shard, seed=22)
    g = uk.lower(uk.plan_reducescatter(topo, shard * 4, 4, chunk_bytes=512))
    outs, _ = uk.execute_host(g, ins, out_bytes=shard * 4)
    total = torch.stack(ins).sum(0)
    for r, o in enumerate(outs):
        assert torch.equal(o, total[r * shard:(r + 1) * shard]), r


def test_alltoall_plan():
    world, seg = 4, 500  # elems per segment
    topo = uk.Topology(world)
    ins = _inputs(world, world * seg, seed=31)
    g = uk.lower(uk.plan_alltoall(topo, seg * 4, chunk_bytes=1024))
    outs, stats = uk.execute_host(g, ins, out_bytes=world * seg * 4)
    for dst in range(world):
        for src in range(world):
            got = outs[dst][src * seg:(src + 1) * seg]
            want = ins[src][dst * seg:(dst + 1) * seg]
            assert torch.equal(got, want), (src, dst)
    # every directed link carried exactly seg bytes
    lm = uk.link_matrix(stats, world)
    for s_ in range(world):
        for d_ in range(world):
            assert lm[s_][d_] == (0 if s_ == d_ else seg * 4)
