"""Compressed p2p transfer demo (CPU — runs anywhere).

Ships a bf16 tensor between two Endpoints with the lossless plane-split
codec on the wire (reference analog: DietGPU compression on p2p
transfers). Prints the wire size vs the raw size.

    python examples/p2p_compressed.py
"""

import threading
import time

import torch

from uccl_amd import p2p


def main():
    a = p2p.Endpoint(gpu=0, num_workers=1)
    b = p2p.Endpoint(gpu=0, num_workers=1)

    ids = {}
    th = threading.Thread(target=lambda: ids.setdefault("b", b.accept()))
    th.start()
    cid_a = a.connect(b.metadata())
    th.join()

    x = torch.randn(4096, 7168).bfloat16()  # a DeepSeek-shaped activation
    raw = x.numel() * x.element_size()

    got = {}
    rx = threading.Thread(
        target=lambda: got.setdefault("t", p2p.recv_compressed(b, ids["b"])))
    rx.start()
    t0 = time.perf_counter()
    wire = p2p.send_compressed(a, cid_a, x)
    rx.join()
    dt = time.perf_counter() - t0

    assert torch.equal(got["t"], x)
    print(f"raw {raw / 1e6:.1f} MB -> wire {wire / 1e6:.1f} MB "
          f"(ratio {raw / wire:.2f}x) in {dt * 1e3:.1f} ms")


if __name__ == "__main__":
    main()
