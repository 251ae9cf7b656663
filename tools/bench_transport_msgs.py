#!/usr/bin/env python3
"""Transport throughput microbench: small-message rate + bulk bandwidth.

Two endpoints over loopback in one process, F concurrent flows each
driven by its own sender/receiver thread pair (blocking send/recv per
message, so the rate measures the full protocol round trip including
ack handling). Reports msg/s (512 B messages) and MB/s (4 MiB
messages). Used to quantify the engine-threaded transport rework
(>=5x msg/s target over the round-1
single-progress-thread design).

Usage: python tools/bench_transport_msgs.py [--flows 8] [--secs 3]
"""

import argparse
import os
import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402


def run_case(C, flows, secs, msg_bytes, paths=8, chunk=8192):
    os.environ.setdefault("UCCL_TP_CWND_MAX", "1024")
    a = C.TransportEndpoint(num_paths=paths, chunk_bytes=chunk)
    b = C.TransportEndpoint(num_paths=paths, chunk_bytes=chunk)
    fa, fb = [], []
    accepted = []

    def acc():
        for _ in range(flows):
            accepted.append(b.accept())

    t = threading.Thread(target=acc)
    t.start()
    for _ in range(flows):
        fa.append(a.connect(b.metadata()))
    t.join(30)
    fb[:] = accepted

    stop = threading.Event()
    counts = [0] * flows
    src = torch.randint(0, 256, (msg_bytes,), dtype=torch.uint8)
    dsts = [torch.zeros(msg_bytes, dtype=torch.uint8) for _ in range(flows)]

    def sender(i):
        while not stop.is_set():
            a.send(fa[i], src)
            counts[i] += 1
        # one final message so the receiver's pending recv completes
        a.send(fa[i], src)

    def receiver(i):
        while True:
            b.recv(fb[i], dsts[i])
            if stop.is_set():
                return

    ths = []
    for i in range(flows):
        ths.append(threading.Thread(target=receiver, args=(i,)))
        ths.append(threading.Thread(target=sender, args=(i,)))
    t0 = time.perf_counter()
    for th in ths:
        th.start()
    time.sleep(secs)
    stop.set()
    for th in ths:
        th.join(timeout=30)
    dt = time.perf_counter() - t0
    total = sum(counts)
    assert torch.equal(dsts[0], src)
    return total / dt, total * msg_bytes / dt / 1e6


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--flows", type=int, default=8)
    ap.add_argument("--secs", type=float, default=3.0)
    args = ap.parse_args()
    from uccl_amd import _load_native

    C = _load_native(required=False)
    assert C is not None

    rate, _ = run_case(C, args.flows, args.secs, 512)
    _, bw = run_case(C, max(2, args.flows // 2), args.secs, 4 << 20)
    print(f"flows={args.flows} small(512B): {rate:,.0f} msg/s")
    print(f"bulk(4MiB): {bw:,.0f} MB/s")
    print(f'{{"msg_per_s": {rate:.0f}, "bulk_mb_s": {bw:.0f}}}')


if __name__ == "__main__":
    main()
