"""Topology utils degrade gracefully without GPUs."""


def test_topology_summary_no_gpu():
    from uccl_amd.utils import topology_summary

    out = topology_summary()
    assert "GPU(s)" in out
