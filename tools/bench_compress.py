"""Codec microbench: ratio + throughput per dtype/strategy.

    PYTHONPATH=. python tools/bench_compress.py [MB]
"""

import sys
import time

import torch

from uccl_amd import p2p


def run(name, t, strategy=p2p.STRATEGY_SPLIT_DEFLATE):
    nb = t.numel() * t.element_size()
    p2p.compress(t[: 1 << 16], strategy)  # warm the pool/caches
    t0 = time.perf_counter()
    f = p2p.compress(t, strategy)
    t1 = time.perf_counter()
    back = p2p.decompress(f)
    t2 = time.perf_counter()
    assert bool((back.view(torch.uint8) ==
                 t.contiguous().view(-1).view(torch.uint8)).all())
    print(f"{name:28s} ratio {nb / f.numel():5.2f}x  "
          f"comp {nb / 1e6 / (t1 - t0):6.0f} MB/s  "
          f"decomp {nb / 1e6 / (t2 - t1):6.0f} MB/s")


def main():
    mb = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    n = mb << 20
    run("bf16 randn", torch.randn(n // 2).bfloat16())
    run("f16 randn", torch.randn(n // 2).half())
    run("f32 randn", torch.randn(n // 4))
    run("bf16 randn (split only)", torch.randn(n // 2).bfloat16(),
        p2p.STRATEGY_SPLIT_ONLY)
    run("uniform bytes (fallback)",
        torch.randint(0, 256, (n // 2,), dtype=torch.uint8)
        .view(torch.bfloat16))


if __name__ == "__main__":
    main()
