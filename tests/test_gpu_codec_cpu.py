"""CPU mirror tests of the GPU rANS coder (gpu_codec.hip host path).

The interleaved 64-lane encode/decode and the 12-bit frequency
normalization are __host__ __device__; this tier proves the coder math
(round-trips across distributions, including heavy-tailed histograms
that can underflow naive normalization) without a GPU.
"""

import pytest
import torch

from uccl_amd import _load_native

C = _load_native(required=False)
if C is None or not hasattr(C, "gpu_codec_host_selftest"):
    pytest.skip("native module missing", allow_module_level=True)


def _check(t):
    assert C.gpu_codec_host_selftest(t.contiguous())


def test_uniform_random():
    g = torch.Generator().manual_seed(1)
    _check(torch.randint(0, 256, (65536,), generator=g, dtype=torch.uint8))


def test_skewed():
    g = torch.Generator().manual_seed(2)
    x = (torch.randn(65536, generator=g).abs() * 12).clamp(max=255)
    _check(x.to(torch.uint8))


def test_constant():
    _check(torch.full((4096,), 7, dtype=torch.uint8))


def test_two_symbols():
    g = torch.Generator().manual_seed(3)
    _check((torch.rand(32768, generator=g) < 0.01).to(torch.uint8))


def test_heavy_tail():
    # 255 rare symbols + one dominant: stresses normalization settling
    g = torch.Generator().manual_seed(4)
    x = torch.zeros(65536, dtype=torch.uint8)
    x[:255] = torch.arange(1, 256, dtype=torch.uint8)
    _check(x)


@pytest.mark.parametrize("n", [1, 63, 64, 65, 1000, 4095])
def test_ragged_sizes(n):
    g = torch.Generator().manual_seed(5 + n)
    _check(torch.randint(0, 256, (n,), generator=g, dtype=torch.uint8))


def test_bf16_exponent_plane():
    # the plane the codec actually wins on: bf16 high bytes of randn
    g = torch.Generator().manual_seed(6)
    x = torch.randn(32768, generator=g).to(torch.bfloat16)
    hi = x.view(torch.uint8).reshape(-1, 2)[:, 1].contiguous()
    _check(hi)
