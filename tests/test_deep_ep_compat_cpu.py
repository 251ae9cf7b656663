"""DeepEP-shaped compat layer: static layout helper (CPU)."""

import torch

from uccl_amd.ep.deep_ep_compat import Buffer


def test_compat_layout_shape():
    topk = torch.tensor([[0, 1], [2, 3], [0, 3]], dtype=torch.int64)
    npr, _, npe, in_rank, ev = Buffer.get_dispatch_layout(topk, 4)
    assert npe.tolist() == [2, 1, 1, 2]
    assert in_rank.shape == (3, 1)  # single-rank fallback


def test_compat_maintenance_surface():
    from uccl_amd.ep.deep_ep_compat import Buffer, Config

    b = Buffer(group=None)
    assert b.clean_low_latency_buffer(4096, 7168, 64) is None
    hint = Buffer.get_low_latency_rdma_size_hint(128, 7168, 8, 64)
    assert hint > 128 * 8 * 7168 * 2
    cfg = Buffer.get_dispatch_config(8)
    assert isinstance(cfg, Config) and cfg.num_sms > 0
    assert Buffer.get_combine_config(8).num_sms > 0
    assert b.group_size == 1
    assert b.internode_dispatch is not None
    assert b.internode_combine is not None
