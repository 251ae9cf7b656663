"""Elastic EP buffer: rank join/leave between iterations.

Reference analog: lite-ep's ElasticBuffer
(experimental/lite/lite-ep/csrc/elastic/buffer.hpp — rank-join/leave EP
buffer with a deterministic mode). MI355X design: on a single xGMI node
the symmetric heap must be re-exchanged whenever membership changes, and
buffer construction is cheap (one IPC handshake), so elasticity is
implemented as generation-tracked rebuild — every membership change bumps
the generation, tears down the old native buffer, and lazily constructs a
new one for the surviving group. Routing state (expert -> rank) is
recomputed so callers can rebalance before the next dispatch.

    eb = ElasticBuffer(group, num_experts=64, topk=8, hidden=7168,
                       max_tokens=4096, dtype=torch.bfloat16)
    eb.dispatch(x, topk_idx)            # normal steady-state
    eb.resize(new_group)                # collective: all surviving ranks
    eb.dispatch(x, remapped_topk_idx)   # next iteration, new geometry
"""

from __future__ import annotations

from typing import Callable, Optional

import torch


class MembershipChanged(RuntimeError):
    """Raised when an op is attempted against a stale generation."""


def expert_rank_table(num_experts: int, world: int) -> list:
    """expert id -> owning rank under contiguous block assignment (the
    layout the native dispatch kernels use: num_experts/world local
    experts per rank)."""
    per = num_experts // world
    assert per * world == num_experts, "num_experts must divide by world"
    return [e // per for e in range(num_experts)]


def remap_topk_idx(topk_idx: torch.Tensor, old_world: int,
                   new_world: int, num_experts: int) -> torch.Tensor:
    """Identity under block assignment (expert ids are global); provided
    as the hook where a load-balancing remap would slot in."""
    del old_world, new_world, num_experts
    return topk_idx


class ElasticBuffer:
    def __init__(self, group=None, *, num_experts: int, topk: int,
                 hidden: int, max_tokens: int, dtype=torch.bfloat16,
                 use_fp8: bool = False, deterministic: bool = False,
                 factory: Optional[Callable] = None,
                 world_fn: Optional[Callable] = None):
        """`factory(group, **cfg) -> native buffer` defaults to the native
        EP Buffer; injectable so membership logic is testable without a
        GPU (the same seam the reference's elastic tests use)."""
        self._cfg = dict(num_experts=num_experts, topk=topk, hidden=hidden,
                         max_tokens=max_tokens, dtype=dtype, use_fp8=use_fp8)
        self.deterministic = deterministic
        self._factory = factory or self._default_factory
        self._world_fn = world_fn or self._group_world
        self._group = group
        self._generation = 0
        self._buf = None
        self._world = self._world_fn(group)
        self.expert_rank = expert_rank_table(num_experts, self._world)

    # -- membership ---------------------------------------------------------
    @property
    def generation(self) -> int:
        return self._generation

    @property
    def world(self) -> int:
        return self._world

    def resize(self, new_group) -> int:
        """Collective: every rank of the NEW group calls resize with the
        same group. Ranks not in the new group call `leave()` instead.
        Returns the new generation."""
        new_world = self._world_fn(new_group)
        if self._cfg["num_experts"] % new_world:
            raise ValueError(
                f"num_experts={self._cfg['num_experts']} does not divide "
                f"across world={new_world}")
        self._teardown()
        self._group = new_group
        self._world = new_world
        self._generation += 1
        self.expert_rank = expert_rank_table(self._cfg["num_experts"],
                                             new_world)
        return self._generation

    def leave(self) -> None:
        """Departing rank: release resources; further ops raise."""
        self._teardown()
        self._group = None
        self._world = 0
        self._generation += 1

    # -- data path ----------------------------------------------------------
    def dispatch(self, x, topk_idx):
        return self._native().dispatch(x, topk_idx)

    def combine(self, x, topk_idx, topk_weights):
        return self._native().combine(x, topk_idx, topk_weights)

    def stats(self):
        b = self._buf
        return {"generation": self._generation, "world": self._world,
                "active": b is not None}

    # -- internals ----------------------------------------------------------
    def _native(self):
        if self._group is None:
            raise MembershipChanged(
                "this rank left the EP group (generation "
                f"{self._generation}); no further dispatch/combine")
        if self._buf is None:
            self._buf = self._factory(self._group, **self._cfg)
        return self._buf

    def _teardown(self):
        buf, self._buf = self._buf, None
        if buf is not None and hasattr(buf, "close"):
            buf.close()

    @staticmethod
    def _group_world(group) -> int:
        import torch.distributed as dist

        if group is not None and dist.is_available() and dist.is_initialized():
            return dist.get_world_size(group=group)
        if dist.is_available() and dist.is_initialized():
            return dist.get_world_size()
        return 1

    def _default_factory(self, group, **cfg):
        from uccl_amd.ep import Buffer

        return Buffer(group=group, **cfg)
