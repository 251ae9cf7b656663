#!/usr/bin/env python3
"""Long-running transport soak: random sizes, loss rates, CC modes,
bidirectional traffic. Run manually: python tools/soak_transport.py [iters]."""
import os
import random
import sys
import threading

import torch


def one_round(rnd: random.Random, idx: int):
    os.environ["UCCL_TP_LOSS_PCT"] = str(rnd.choice([0, 0, 2, 5, 10, 20]))
    os.environ["UCCL_TP_ACK_LOSS_PCT"] = str(rnd.choice([0, 0, 10]))
    os.environ["UCCL_TP_CC"] = rnd.choice(["timely", "swift", "eqds"])
    # paced pull quanta occasionally (high rate so soak stays fast)
    os.environ["UCCL_TP_EQDS_MBPS"] = str(
        rnd.choice([0, 0, 0, 400]) if os.environ["UCCL_TP_CC"] == "eqds"
        else 0)
    os.environ["UCCL_TP_CWND_MAX"] = str(rnd.choice([32, 256, 1024]))
    from uccl_amd import _load_native

    C = _load_native(required=False)
    a = C.TransportEndpoint(num_paths=rnd.choice([1, 2, 4, 8]),
                            chunk_bytes=rnd.choice([1024, 4096, 16384]))
    b = C.TransportEndpoint(num_paths=4, chunk_bytes=4096)
    res = {}
    t = threading.Thread(target=lambda: res.__setitem__("f", b.accept()),
                         daemon=True)
    t.start()
    fa = a.connect(b.metadata())
    t.join(20)
    fb = res["f"]

    msgs = [rnd.randrange(0, 1 << rnd.randrange(0, 22)) for _ in range(6)]
    fwd = [torch.randint(0, 256, (max(n, 1),), dtype=torch.uint8)[:n]
           for n in msgs]
    rev = [torch.randint(0, 256, (max(n, 1),), dtype=torch.uint8)[:n]
           for n in msgs]
    got_f = [torch.zeros(n, dtype=torch.uint8) for n in msgs]
    got_r = [torch.zeros(n, dtype=torch.uint8) for n in msgs]

    def b_side():
        for i, n in enumerate(msgs):
            b.recv(fb, got_f[i])
            b.send(fb, rev[i])

    th = threading.Thread(target=b_side, daemon=True)
    th.start()
    for i, n in enumerate(msgs):
        a.send(fa, fwd[i])
        a.recv(fa, got_r[i])
    th.join(120)
    assert not th.is_alive(), f"round {idx}: b side stuck"
    for i in range(len(msgs)):
        assert torch.equal(fwd[i], got_f[i]), f"round {idx} fwd msg {i}"
        assert torch.equal(rev[i], got_r[i]), f"round {idx} rev msg {i}"
    st = a.stats()
    print(f"round {idx:3d} ok  sizes={msgs} loss={os.environ['UCCL_TP_LOSS_PCT']}%"
          f" cc={os.environ['UCCL_TP_CC']} rtx={st.retransmits}+{st.rto_retransmits}",
          flush=True)


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 30
    rnd = random.Random(12345)
    for i in range(iters):
        one_round(rnd, i)
    print("SOAK OK")


if __name__ == "__main__":
    main()
