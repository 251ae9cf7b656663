"""DeepEP-shaped compat layer: static layout helper (CPU)."""

import torch

from uccl_amd.ep.deep_ep_compat import Buffer


def test_compat_layout_shape():
    topk = torch.tensor([[0, 1], [2, 3], [0, 3]], dtype=torch.int64)
    npr, _, npe, in_rank, ev = Buffer.get_dispatch_layout(topk, 4)
    assert npe.tolist() == [2, 1, 1, 2]
    assert in_rank.shape == (3, 1)  # single-rank fallback
