"""GPU-tier tests: single-process world=1 kernel paths on a real MI355X.

These validate the staged kernel machinery (copy-in, signal/wait flags,
fullmesh reduce, LL packets) without needing multiple ranks. Numerics are
checked against plain PyTorch fp32 references.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires GPU", allow_module_level=True)

os.environ["UCCL_WORLD1_STAGED"] = "1"


@pytest.fixture(scope="module")
def comm():
    import uccl_amd.collective as ucol

    torch.cuda.set_device(0)
    return ucol.Communicator(rank=0, world=1, device=0)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16,
                                   torch.float16, torch.int32])
@pytest.mark.parametrize("count", [16, 1000, 65536, 1 << 20, 1000003])
def test_allreduce_world1_identity(comm, dtype, count):
    if dtype == torch.int32:
        x = torch.randint(-1000, 1000, (count,), dtype=dtype, device="cuda")
    else:
        x = torch.randn(count, dtype=dtype, device="cuda")
    ref = x.clone()
    comm.all_reduce(x)
    torch.cuda.synchronize()
    assert torch.equal(x, ref), f"world=1 allreduce must be identity {dtype}"


def test_allreduce_world1_large_chunked(comm):
    # larger than one parity scratch half -> exercises the chunk loop
    cap = comm._c.scratch_capacity
    count = (cap // 4) * 3  # 1.5x capacity in fp32 elems
    x = torch.randn(count, dtype=torch.float32, device="cuda")
    ref = x.clone()
    comm.all_reduce(x)
    torch.cuda.synchronize()
    assert torch.equal(x, ref)


def test_allgather_world1(comm):
    src = torch.randn(4096, device="cuda")
    dst = torch.empty_like(src)
    comm.all_gather(dst, src)
    torch.cuda.synchronize()
    assert torch.equal(dst, src)


def test_reduce_scatter_world1(comm):
    src = torch.randn(4096, device="cuda")
    dst = torch.empty_like(src)
    comm.reduce_scatter(dst, src)
    torch.cuda.synchronize()
    assert torch.equal(dst, src)


def test_barrier_world1(comm):
    comm.barrier()
    torch.cuda.synchronize()
