"""ElasticBuffer membership logic (CPU; injected fake native buffer).

Reference analog: lite-ep ElasticBuffer rank join/leave
(experimental/lite/lite-ep/csrc/elastic/buffer.hpp)."""

import pytest
import torch

from uccl_amd.ep.elastic import (ElasticBuffer, MembershipChanged,
                                 expert_rank_table)


class FakeNative:
    alive = []

    def __init__(self, group, **cfg):
        self.group = group
        self.cfg = cfg
        self.closed = False
        FakeNative.alive.append(self)

    def dispatch(self, x, topk_idx):
        assert not self.closed
        return ("dispatched", x.shape[0])

    def combine(self, x, topk_idx, topk_weights):
        assert not self.closed
        return ("combined", x.shape[0])

    def close(self):
        self.closed = True


class FakeGroup:
    def __init__(self, world):
        self.world = world


def _make(world=4, **kw):
    return ElasticBuffer(FakeGroup(world), num_experts=32, topk=2,
                         hidden=64, max_tokens=16,
                         factory=lambda g, **cfg: FakeNative(g, **cfg),
                         world_fn=lambda g: g.world if g else 1, **kw)


def test_lazy_build_and_dispatch():
    FakeNative.alive.clear()
    eb = _make()
    assert FakeNative.alive == []  # lazy
    x = torch.zeros(8, 64)
    assert eb.dispatch(x, None) == ("dispatched", 8)
    assert len(FakeNative.alive) == 1
    assert eb.generation == 0


def test_resize_rebuilds_and_remaps():
    FakeNative.alive.clear()
    eb = _make(world=8)
    eb.dispatch(torch.zeros(4, 64), None)
    old = FakeNative.alive[-1]
    assert eb.expert_rank == expert_rank_table(32, 8)

    gen = eb.resize(FakeGroup(4))  # 4 ranks left
    assert gen == 1
    assert old.closed  # old buffer torn down
    assert eb.world == 4
    assert eb.expert_rank == expert_rank_table(32, 4)
    eb.combine(torch.zeros(4, 64), None, None)
    assert len(FakeNative.alive) == 2  # rebuilt lazily on next op
    assert not FakeNative.alive[-1].closed


def test_grow_after_shrink():
    eb = _make(world=2)
    eb.dispatch(torch.zeros(1, 64), None)
    eb.resize(FakeGroup(8))
    assert eb.world == 8 and eb.generation == 1
    eb.dispatch(torch.zeros(1, 64), None)


def test_leaver_raises():
    eb = _make()
    eb.dispatch(torch.zeros(2, 64), None)
    eb.leave()
    with pytest.raises(MembershipChanged):
        eb.dispatch(torch.zeros(2, 64), None)


def test_indivisible_world_rejected():
    eb = _make(world=4)
    with pytest.raises(ValueError):
        eb.resize(FakeGroup(5))  # 32 experts don't divide by 5
    # failed resize must not have torn the working state into limbo
    assert eb.world == 4


def test_expert_rank_table():
    assert expert_rank_table(8, 4) == [0, 0, 1, 1, 2, 2, 3, 3]
    assert expert_rank_table(4, 4) == [0, 1, 2, 3]
