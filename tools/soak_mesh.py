"""Full-mesh transport soak: N endpoints, all-pairs flows, concurrent
bidirectional random traffic with integrity checks per round.

    PYTHONPATH=. python tools/soak_mesh.py [rounds] [N]
"""

import os
import random
import sys
import threading
import time

import torch


def one_round(rnd, n):
    os.environ["UCCL_TP_CC"] = rnd.choice(["timely", "swift", "eqds"])
    os.environ["UCCL_TP_LOSS_PCT"] = str(rnd.choice([0, 0, 3]))
    os.environ.setdefault("UCCL_TP_CWND_MAX", "256")
    from uccl_amd import _load_native

    C = _load_native(required=False)
    eps = [C.TransportEndpoint(num_paths=2, chunk_bytes=4096)
           for _ in range(n)]
    mds = [e.metadata() for e in eps]
    # all-pairs flows: i dials j for i<j; j accepts with i's tag
    flows = {}
    lock = threading.Lock()

    def acceptor(j, count):
        for _ in range(count):
            f = eps[j].accept()
            with lock:
                flows.setdefault(("acc", j), []).append(f)

    acc_threads = []
    for j in range(n):
        inbound = j  # ranks i<j dial j
        if inbound:
            t = threading.Thread(target=acceptor, args=(j, inbound))
            t.start()
            acc_threads.append(t)
    for i in range(n):
        for j in range(i + 1, n):
            flows[(i, j)] = eps[i].connect(mds[j], tag=i)
    for t in acc_threads:
        t.join(timeout=60)
    # acceptor-side flows need no identity matching: each just echoes

    results = []
    threads = []

    # protocol: dialer sends u32 size then payload; acceptor echoes both
    def echo_flow(j, f):
        sz = torch.zeros(4, dtype=torch.uint8)
        eps[j].recv(f, sz)
        n_ = int.from_bytes(bytes(sz.tolist()), "little")
        body = torch.zeros(n_, dtype=torch.uint8)
        eps[j].recv(f, body)
        eps[j].send(f, sz)
        eps[j].send(f, body)

    def dial_flow(i, j, f):
        nbytes = rnd.randrange(1, 200000)
        payload = torch.randint(0, 255, (nbytes,), dtype=torch.uint8)
        sz = torch.frombuffer(bytearray(nbytes.to_bytes(4, "little")),
                              dtype=torch.uint8)
        eps[i].send(f, sz)
        eps[i].send(f, payload)
        rsz = torch.zeros(4, dtype=torch.uint8)
        eps[i].recv(f, rsz)
        back = torch.zeros(nbytes, dtype=torch.uint8)
        eps[i].recv(f, back)
        results.append(torch.equal(back, payload))

    for j in range(n):
        for f in flows.get(("acc", j), []):
            threads.append(threading.Thread(target=echo_flow, args=(j, f)))
    for i in range(n):
        for j in range(i + 1, n):
            threads.append(threading.Thread(target=dial_flow,
                                            args=(i, j, flows[(i, j)])))
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    npairs = n * (n - 1) // 2
    assert len(results) == npairs and all(results), \
        f"{sum(results)}/{len(results)} pairs ok"
    for e in eps:
        del e


def main():
    rounds = int(sys.argv[1]) if len(sys.argv) > 1 else 10
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 4
    rnd = random.Random(os.environ.get("UCCL_SOAK_SEED", time.time_ns()))
    for i in range(rounds):
        one_round(rnd, n)
        print(f"mesh round {i} ok (n={n}, cc={os.environ['UCCL_TP_CC']}, "
              f"loss={os.environ['UCCL_TP_LOSS_PCT']}%)", flush=True)
    print("MESH SOAK OK")


if __name__ == "__main__":
    main()
