"""Worker for the 2-process P2P engine battery (CPU tensors -> TCP path;
GPU tensors -> IPC path when available). Role (server/client) from argv."""

from __future__ import annotations

import os
import signal
import sys
import time

signal.alarm(int(os.environ.get("UCCL_TEST_ALARM", "180")))

import torch


def main():
    role = sys.argv[1]
    meta_path = sys.argv[2]
    use_gpu = os.environ.get("UCCL_P2P_TEST_GPU", "0") == "1"
    dev = "cuda" if use_gpu else "cpu"
    gpu = 0 if use_gpu else -1
    if use_gpu:
        torch.cuda.set_device(0)

    from uccl_amd.p2p import Endpoint

    ep = Endpoint(gpu=gpu, num_workers=2)

    if role == "server":
        with open(meta_path + ".tmp", "wb") as f:
            f.write(ep.metadata())
        os.rename(meta_path + ".tmp", meta_path)
        cid = ep.accept()

        # two-sided: echo test
        t = torch.zeros(1000, dtype=torch.float32, device=dev)
        ep.recv(cid, t)
        assert torch.allclose(t.cpu(), torch.arange(1000, dtype=torch.float32))
        ep.send(cid, t * 2)

        # one-sided: advertise a window, client writes into it then reads it
        win = torch.zeros(4096, dtype=torch.float32, device=dev)
        mr = ep.reg(win)
        ad = ep.advertise(mr, 0, win.numel() * 4)
        with open(meta_path + ".ad.tmp", "wb") as f:
            f.write(ad)
        os.rename(meta_path + ".ad.tmp", meta_path + ".ad")
        # wait for client's done marker
        while not os.path.exists(meta_path + ".done"):
            time.sleep(0.05)
        if use_gpu:
            torch.cuda.synchronize()
        expect = torch.full((4096,), 7.0)
        assert torch.allclose(win.cpu(), expect), win.cpu()[:8]

        # large transfer (spans staging chunks on TCP path)
        big = torch.empty(3 << 20, dtype=torch.float32, device=dev)
        ep.recv(cid, big)
        g = torch.Generator().manual_seed(99)
        want = torch.randn(3 << 20, generator=g)
        assert torch.allclose(big.cpu(), want)
        print("SERVER OK", flush=True)
    else:
        while not os.path.exists(meta_path):
            time.sleep(0.05)
        with open(meta_path, "rb") as f:
            md = f.read()
        cid = ep.connect(md)

        ep.send(cid, torch.arange(1000, dtype=torch.float32, device=dev))
        t = torch.zeros(1000, dtype=torch.float32, device=dev)
        ep.recv(cid, t)
        assert torch.allclose(t.cpu(),
                              torch.arange(1000, dtype=torch.float32) * 2)

        while not os.path.exists(meta_path + ".ad"):
            time.sleep(0.05)
        with open(meta_path + ".ad", "rb") as f:
            ad = f.read()
        src = torch.full((4096,), 7.0, device=dev)
        ep.write(cid, src, ad)
        # one-sided read back
        back = torch.zeros(4096, dtype=torch.float32, device=dev)
        ep.read(cid, back, ad)
        assert torch.allclose(back.cpu(), src.cpu())
        with open(meta_path + ".done.tmp", "w") as f:
            f.write("x")
        os.rename(meta_path + ".done.tmp", meta_path + ".done")

        # async large transfer
        g = torch.Generator().manual_seed(99)
        big = torch.randn(3 << 20, generator=g).to(dev)
        xid = ep.send_async(cid, big)
        while not ep.poll_async(xid):
            time.sleep(0.01)
        print("CLIENT OK", flush=True)


if __name__ == "__main__":
    main()
