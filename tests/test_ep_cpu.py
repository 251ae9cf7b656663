"""CPU-tier EP tests: routing-layout math (pure torch, no GPU)."""

import torch

from uccl_amd.ep import get_dispatch_layout


def test_dispatch_layout_basic():
    topk = torch.tensor([[0, 3], [1, 2], [3, 3], [-1, 0]], dtype=torch.int64)
    npr, npe, in_rank = get_dispatch_layout(topk, num_experts=4, num_ranks=2)
    # experts 0,1 -> rank0 ; 2,3 -> rank1
    assert npe.tolist() == [2, 1, 1, 3]
    assert in_rank.tolist() == [[True, True], [True, True], [False, True],
                                [True, False]]
    assert npr.tolist() == [3, 3]


def test_dispatch_layout_random_consistency():
    g = torch.Generator().manual_seed(7)
    T, K, E, R = 512, 8, 64, 8
    topk = torch.randint(0, E, (T, K), generator=g)
    npr, npe, in_rank = get_dispatch_layout(topk, E, R)
    assert npe.sum().item() == T * K
    assert (in_rank.sum(1) >= 1).all()
    assert (npr == in_rank.sum(0).to(torch.int32)).all()
