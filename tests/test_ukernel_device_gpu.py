"""ukernel DEVICE backend (persistent worker kernels) vs the host mock.

The spray executor runs the same lowered chunk graphs against the
DeviceBackend (per-rank C2D task FIFOs drained by a persistent HIP
kernel — the reference's persistent_kernel_ops.cu design) and the
HostBackend; outputs must match bitwise (fp32 copy/reduce in fixed
order on both).
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires GPU", allow_module_level=True)

from uccl_amd import _load_native

C = _load_native(required=True)
uk = C


def _inputs(world, elems, seed=0):
    g = torch.Generator().manual_seed(seed)
    return [torch.randn(elems, generator=g) for _ in range(world)]


@pytest.mark.parametrize("world", [2, 4])
def test_sendrecv_spray_device(world):
    elems = 1 << 16
    topo = uk.UkTopology(world)
    g = uk.uk_plan_sendrecv(topo, 0, 1, elems * 4, 16384)
    g = uk.uk_lower(g)
    ins = _inputs(world, elems, seed=11)
    host_outs, _ = uk.uk_execute_host(g, ins, elems * 4)
    dev_outs, st = uk.uk_execute_device(g, ins, elems * 4)
    for r in range(world):
        assert torch.equal(host_outs[r], dev_outs[r]), r
    assert st["tasks_run"] > 0


@pytest.mark.parametrize("world", [2, 4])
def test_allreduce_auto_device(world):
    elems = 1 << 15
    g = uk.uk_plan_allreduce_auto(uk.UkTopology(world), elems * 4, 4, 16384)
    ins = _inputs(world, elems, seed=13)
    host_outs, _ = uk.uk_execute_host(g, ins, elems * 4)
    dev_outs, _ = uk.uk_execute_device(g, ins, elems * 4)
    for r in range(world):
        assert torch.equal(host_outs[r], dev_outs[r]), r


def test_allgather_device():
    world, elems = 4, 4096
    g = uk.uk_plan_allgather(uk.UkTopology(world), elems * 4, 8192)
    g = uk.uk_lower(g)
    ins = _inputs(world, elems, seed=15)
    host_outs, _ = uk.uk_execute_host(g, ins, world * elems * 4)
    dev_outs, _ = uk.uk_execute_device(g, ins, world * elems * 4)
    for r in range(world):
        assert torch.equal(host_outs[r], dev_outs[r]), r
