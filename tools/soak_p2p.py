"""Randomized p2p endpoint soak: connection churn, every op family
(two-sided, one-sided, async batches, compressed, object), both data
planes, with content verification each round.

    PYTHONPATH=. python tools/soak_p2p.py [rounds]
"""

import os
import random
import sys
import threading
import time

import torch


def one_round(rnd, plane):
    os.environ["UCCL_P2P_TRANSPORT"] = plane
    from uccl_amd import p2p

    a = p2p.Endpoint(gpu=0, num_workers=rnd.choice([1, 2]))
    b = p2p.Endpoint(gpu=0, num_workers=rnd.choice([1, 2]))
    ids = {}
    th = threading.Thread(target=lambda: ids.setdefault("b", b.accept()))
    th.start()
    cid_a = a.connect(b.metadata())
    th.join(timeout=30)
    cid_b = ids["b"]
    ops = []

    for _ in range(rnd.randint(2, 6)):
        kind = rnd.choice(["sendrecv", "onesided", "async", "compressed",
                           "object"])
        n = rnd.choice([0, 1, 777, 65536, 1 << 20])
        if kind == "sendrecv":
            src = torch.randint(0, 255, (max(n, 1),),
                                dtype=torch.uint8)[:n]
            dst = torch.zeros(n, dtype=torch.uint8)
            t = threading.Thread(target=lambda: b.recv(cid_b, dst))
            t.start()
            a.send(cid_a, src)
            t.join(timeout=60)
            assert torch.equal(src, dst), "sendrecv mismatch"
        elif kind == "onesided":
            n2 = max(n, 4)
            target = torch.zeros(n2, dtype=torch.uint8)
            mr = b.reg(target)
            ad = b.advertise(mr, 0, n2)
            src = torch.randint(0, 255, (n2,), dtype=torch.uint8)
            a.write(cid_a, src, ad)
            for _ in range(200):
                if torch.equal(target, src):
                    break
                time.sleep(0.01)
            assert torch.equal(target, src), "one-sided write mismatch"
            back = torch.zeros(n2, dtype=torch.uint8)
            a.read(cid_a, back, ad)
            assert torch.equal(back, src), "one-sided read mismatch"
            b.dereg(mr)
        elif kind == "async":
            k = rnd.randint(1, 4)
            srcs = [torch.randint(0, 255, (max(n, 1),),
                                  dtype=torch.uint8)[:n] for _ in range(k)]
            dsts = [torch.zeros(n, dtype=torch.uint8) for _ in range(k)]
            rids = [b.recv_async(cid_b, d) for d in dsts]
            sids = [a.send_async(cid_a, s) for s in srcs]
            deadline = time.time() + 60
            for x, ep in [(i, a) for i in sids] + [(i, b) for i in rids]:
                while not ep.poll_async(x):
                    assert time.time() < deadline, "async timeout"
                    time.sleep(0.001)
            for s, d in zip(srcs, dsts):
                assert torch.equal(s, d), "async mismatch"
        elif kind == "compressed":
            src = torch.randn(max(n // 2, 1)).bfloat16()
            got = {}
            t = threading.Thread(target=lambda: got.setdefault(
                "t", p2p.recv_compressed(b, cid_b)))
            t.start()
            p2p.send_compressed(a, cid_a, src)
            t.join(timeout=60)
            assert torch.equal(got["t"], src), "compressed mismatch"
        else:
            obj = {"i": rnd.randint(0, 1000),
                   "t": torch.randn(max(n // 4, 1)),
                   "nested": [1, {"w": torch.arange(7)}]}
            got = {}
            t = threading.Thread(target=lambda: got.setdefault(
                "o", p2p.recv_object(b, cid_b)))
            t.start()
            p2p.send_object(a, cid_a, obj)
            t.join(timeout=60)
            assert got["o"]["i"] == obj["i"]
            assert torch.equal(got["o"]["t"], obj["t"])
            assert torch.equal(got["o"]["nested"][1]["w"], obj["nested"][1]["w"])
        ops.append(kind)
    return ops


def main():
    rounds = int(sys.argv[1]) if len(sys.argv) > 1 else 20
    rnd = random.Random(os.environ.get("UCCL_SOAK_SEED", time.time_ns()))
    for i in range(rounds):
        plane = rnd.choice(["tcp", "multipath"])
        ops = one_round(rnd, plane)
        print(f"round {i:3d} ok plane={plane} ops={ops}", flush=True)
    print("P2P SOAK OK")


if __name__ == "__main__":
    main()
