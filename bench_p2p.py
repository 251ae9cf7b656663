#!/usr/bin/env python3
"""P2P engine bandwidth bench (BASELINE config 3: send/recv 4KB-1GB between
2 MI355X, NIXL-style API). Spawns a server/client pair; on a 1-GPU box both
sides share the device (IPC one-copy path = device-local DtoD; on 2 GPUs it
rides xGMI). Prints GB/s per size.
"""

from __future__ import annotations

import json
import os
import subprocess
import sys
import tempfile
import time
import uuid

SIZES = [4 << 10, 64 << 10, 1 << 20, 16 << 20, 256 << 20, 1 << 30]
ITERS = {4 << 10: 200, 64 << 10: 200, 1 << 20: 50, 16 << 20: 20,
         256 << 20: 5, 1 << 30: 3}


def worker(role: str, meta_path: str):
    import torch

    torch.cuda.set_device(int(os.environ.get("UCCL_P2P_BENCH_GPU", "0")))
    from uccl_amd.p2p import Endpoint

    ep = Endpoint(gpu=torch.cuda.current_device(), num_workers=2)
    if role == "server":
        with open(meta_path + ".tmp", "wb") as f:
            f.write(ep.metadata())
        os.rename(meta_path + ".tmp", meta_path)
        cid = ep.accept()
        for size in SIZES:
            t = torch.empty(size, dtype=torch.uint8, device="cuda")
            for _ in range(ITERS[size] + 2):
                ep.recv(cid, t)
            ep.send(cid, torch.ones(1, dtype=torch.uint8, device="cuda"))
    else:
        while not os.path.exists(meta_path):
            time.sleep(0.05)
        with open(meta_path, "rb") as f:
            md = f.read()
        cid = ep.connect(md)
        rows = []
        for size in SIZES:
            t = torch.randint(0, 255, (size,), dtype=torch.uint8,
                              device="cuda")
            for _ in range(2):  # warmup
                ep.send(cid, t)
            t0 = time.perf_counter()
            for _ in range(ITERS[size]):
                ep.send(cid, t)
            dt = (time.perf_counter() - t0) / ITERS[size]
            ack = torch.zeros(1, dtype=torch.uint8, device="cuda")
            ep.recv(cid, ack)
            rows.append({"bytes": size, "us": round(dt * 1e6, 1),
                         "GBps": round(size / dt / 1e9, 2)})
            print(f"{size:>12}  {dt*1e6:>10.1f} us  {size/dt/1e9:>8.2f} GB/s",
                  flush=True)
        print(json.dumps({"metric": "p2p_send_bw", "rows": rows}))


def main():
    if len(sys.argv) > 1:
        worker(sys.argv[1], sys.argv[2])
        return
    meta = os.path.join(tempfile.gettempdir(),
                        f"uccl_p2p_bench_{uuid.uuid4().hex}.meta")
    env = dict(os.environ)
    env.setdefault("PYTHONPATH", os.path.dirname(os.path.abspath(__file__)))
    ps = [subprocess.Popen([sys.executable, __file__, role, meta], env=env)
          for role in ("server", "client")]
    rc = 0
    for p in ps:
        rc |= p.wait(timeout=600)
    sys.exit(rc)


if __name__ == "__main__":
    main()
