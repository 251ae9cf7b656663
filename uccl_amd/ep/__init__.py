"""DeepEP-compatible expert-parallel communication for MI355X.

API parity target: the reference's `deep_ep.Buffer`-replicating Python layer
(ep/bench/buffer.py) — intranode xGMI path. One node, up to 8 ranks:

    buf = uccl_amd.ep.Buffer(group, num_experts=64, topk=8,
                             hidden=7168, max_tokens=4096,
                             dtype=torch.bfloat16)
    recv_x, recv_count, handle = buf.dispatch(x, topk_idx)
    ...run local experts over recv_x[e, :recv_count_total(e)]...
    combined = buf.combine(expert_out, topk_idx, topk_weights)

Dispatch writes each token's hidden vector straight into every destination
rank's per-(expert, source) slot array over xGMI (no proxy, no staging);
recv_x is a zero-copy view of that slot memory. Combine returns expert
outputs to the (token, k) cells of their source ranks and reduces with
top-k weights in fp32.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


def get_dispatch_layout(topk_idx: torch.Tensor, num_experts: int,
                        num_ranks: int):
    """Token routing layout, mirroring the reference's get_dispatch_layout
    (ep/src/layout.cu:10): returns (num_tokens_per_rank [R],
    num_tokens_per_expert [E], is_token_in_rank [T, R] bool)."""
    T, K = topk_idx.shape
    valid = topk_idx >= 0
    flat = topk_idx.clamp(min=0)
    num_tokens_per_expert = torch.zeros(num_experts, dtype=torch.int32,
                                        device=topk_idx.device)
    num_tokens_per_expert.scatter_add_(
        0, flat.reshape(-1),
        valid.reshape(-1).to(torch.int32))
    experts_per_rank = num_experts // num_ranks
    token_rank = flat // experts_per_rank  # [T, K]
    # masked (idx<0) entries are routed to a throwaway bucket so they can
    # never set a real rank bit (scatter order is undefined on duplicates)
    tr = token_rank.masked_fill(~valid, num_ranks)
    is_token_in_rank = torch.zeros(T, num_ranks + 1, dtype=torch.bool,
                                   device=topk_idx.device)
    is_token_in_rank.scatter_(1, tr, torch.ones_like(tr, dtype=torch.bool))
    is_token_in_rank = is_token_in_rank[:, :num_ranks].contiguous()
    num_tokens_per_rank = is_token_in_rank.sum(0).to(torch.int32)
    return num_tokens_per_rank, num_tokens_per_expert, is_token_in_rank


class Buffer:
    def __init__(self, group=None, num_experts: int = 8, topk: int = 2,
                 hidden: int = 7168, max_tokens: int = 4096,
                 dtype: torch.dtype = torch.bfloat16,
                 device: Optional[int] = None, use_fp8: bool = False):
        from uccl_amd import _load_native

        C = _load_native(required=True)
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            self.rank = dist.get_rank(group)
            self.world = dist.get_world_size(group)
        else:
            self.rank, self.world = 0, 1
        if device is None:
            device = torch.cuda.current_device()
        self.device = device
        self.dtype = dtype
        self.num_experts = num_experts
        self.topk = topk
        self.hidden = hidden
        self.max_tokens = max_tokens
        self.use_fp8 = use_fp8
        elem = torch.tensor([], dtype=dtype).element_size()
        self._b = C.EpBuffer(self.rank, self.world, device, num_experts,
                             topk, hidden, max_tokens, elem, use_fp8)
        if self.world > 1:
            handles = [None] * self.world
            dist.all_gather_object(handles, self._b.handle_bytes(),
                                   group=group)
            self._b.connect(handles)

    @property
    def local_experts(self) -> int:
        return self.num_experts // self.world

    def dispatch(self, x: torch.Tensor, topk_idx: torch.Tensor):
        """Returns (packed_recv_x [local_E, world*max_tokens, hidden] view,
        recv_count [local_E, world] int32) — plus recv_scales
        [local_E, world*max_tokens, hidden/128] f32 when use_fp8 (payloads
        are e4m3 with per-128 scales, DeepEP LL fp8 semantics). Slots for
        source rank r live at [e, r*max_tokens : + recv_count[e, r]]."""
        return self._b.dispatch(x, topk_idx)

    def dispatch_send(self, x: torch.Tensor, topk_idx: torch.Tensor,
                      reuse_plan: bool = False) -> None:
        """SEND phase only (DeepEP phase split, internode_ll.cu:62):
        plans (unless reuse_plan replays the cached compaction lists),
        copies tokens into destination slot arrays over xGMI and
        publishes counts. Pair with dispatch_recv(); compute launched in
        between overlaps the communication."""
        self._b.dispatch_send(x, topk_idx, reuse_plan)

    def dispatch_recv(self, counts: Optional[torch.Tensor] = None
                      ) -> torch.Tensor:
        """RECV phase: waits for every (expert, src) count of this seq;
        fills (or allocates) recv_count [local_E, world] int32."""
        if counts is None:
            counts = torch.empty(self.local_experts, self.world,
                                 dtype=torch.int32, device="cuda")
        return self._b.dispatch_recv(counts)

    def recv_x_view(self) -> torch.Tensor:
        """Zero-copy view of the dispatch slot arrays
        [local_E, world*max_tokens, hidden] (valid after dispatch_recv)."""
        return self._b.recv_x_view()

    def recv_scale_view(self) -> torch.Tensor:
        return self._b.recv_scale_view()

    def combine(self, expert_out: torch.Tensor, topk_idx: torch.Tensor,
                topk_weights: torch.Tensor) -> torch.Tensor:
        return self._b.combine(expert_out, topk_idx,
                               topk_weights.float().contiguous())

    def combine_send(self, expert_out: torch.Tensor) -> None:
        """SEND phase of combine: return expert outputs to their source
        (token, k) cells and signal completion to every rank."""
        self._b.combine_send(expert_out)

    def combine_recv(self, topk_idx: torch.Tensor,
                     topk_weights: torch.Tensor,
                     out: Optional[torch.Tensor] = None) -> torch.Tensor:
        """RECV phase of combine: wait for all ranks' returns, then the
        fp32 top-k weighted reduction (into `out` when given)."""
        if out is None:
            out = torch.empty(topk_idx.shape[0], self.hidden,
                              dtype=self.dtype, device="cuda")
        return self._b.combine_recv(out, topk_idx,
                                    topk_weights.float().contiguous())

    # -- normal (rank-granular) mode: DeepEP HT semantics -------------------
    def nrm_dispatch(self, x: torch.Tensor, topk_idx: torch.Tensor,
                     topk_weights: Optional[torch.Tensor] = None):
        """Rank-granular dispatch: each token ships ONCE per destination
        rank (deduped over its top-k experts) together with its topk row
        and weights. Returns (recv_x [world, max_tokens, hidden] view,
        counts [world] int32, recv_topk [world, max_tokens, topk] i64
        view with GLOBAL expert ids, recv_w view). Rows for source r are
        [r, :counts[r]]."""
        w = topk_weights.float().contiguous() if topk_weights is not None             else None
        self._b.nrm_dispatch_send(x, topk_idx, w)
        counts = torch.empty(self.world, dtype=torch.int32, device="cuda")
        self._b.nrm_dispatch_recv(counts)
        return (self._b.nrm_x_view(), counts, self._b.nrm_topk_view(),
                self._b.nrm_w_view())

    def nrm_combine(self, x: torch.Tensor, topk_idx: torch.Tensor,
                    out: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Return one processed row per received token ([world,
        max_tokens, hidden] layout) to its source and reduce over
        contributing ranks (weights are expected to be applied by the
        receiver's expert computation, DeepEP normal-combine style)."""
        self._b.nrm_combine_send(x)
        if out is None:
            out = torch.empty(topk_idx.shape[0], self.hidden,
                              dtype=self.dtype, device="cuda")
        return self._b.nrm_combine_recv(out, topk_idx)

    def close(self):
        """Release the native buffer (symmetric heap + IPC handles)
        promptly instead of waiting for GC — used by ElasticBuffer on
        membership changes."""
        self._b = None

    # DeepEP-compatible aliases (low-latency intranode semantics)
    low_latency_dispatch = dispatch
    low_latency_combine = combine


def __getattr__(name):
    if name in ("ElasticBuffer", "MembershipChanged", "expert_rank_table"):
        from uccl_amd.ep import elastic

        return getattr(elastic, name)
    raise AttributeError(name)
