"""P2P engine tests. The TCP data plane runs everywhere (CPU tier); the
HIP-IPC one-copy path needs a GPU (gpu tier, 2 processes on one device)."""

import os
import subprocess
import sys
import tempfile
import uuid
from pathlib import Path

import pytest
import torch

REPO = Path(__file__).resolve().parent.parent
WORKER = REPO / "tests" / "workers" / "p2p_worker.py"


def _run_pair(extra_env):
    meta = os.path.join(tempfile.gettempdir(),
                        f"uccl_p2p_{uuid.uuid4().hex}.meta")
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env.update(extra_env)
    ps = [subprocess.Popen([sys.executable, str(WORKER), role, meta],
                           env=env, stdout=subprocess.PIPE,
                           stderr=subprocess.STDOUT)
          for role in ("server", "client")]
    outs, ok = [], True
    for p in ps:
        try:
            out, _ = p.communicate(timeout=200)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
            ok = False
        outs.append(out.decode())
        ok = ok and p.returncode == 0
    joined = "\n=====\n".join(outs)
    assert ok, joined
    assert "SERVER OK" in joined and "CLIENT OK" in joined, joined


def test_p2p_tcp_cpu():
    _run_pair({"UCCL_P2P_TEST_GPU": "0"})


def test_p2p_multipath_cpu():
    # cross-host-style data plane: p2p bytes ride the reliable multipath
    # transport (reference architecture: p2p engine over its own engine)
    _run_pair({"UCCL_P2P_TEST_GPU": "0", "UCCL_P2P_TRANSPORT": "multipath"})


def test_p2p_multipath_lossy_cpu():
    _run_pair({"UCCL_P2P_TEST_GPU": "0", "UCCL_P2P_TRANSPORT": "multipath",
               "UCCL_TP_LOSS_PCT": "5"})


@pytest.mark.gpu
def test_p2p_gpu_ipc():
    if not torch.cuda.is_available():
        pytest.skip("requires GPU")
    _run_pair({"UCCL_P2P_TEST_GPU": "1"})


@pytest.mark.gpu
def test_p2p_gpu_tcp_staging():
    # force the TCP staging path for GPU tensors (IPC disabled)
    if not torch.cuda.is_available():
        pytest.skip("requires GPU")
    _run_pair({"UCCL_P2P_TEST_GPU": "1", "UCCL_P2P_ENABLE_IPC": "0"})


def test_p2p_async_fifo_cpu():
    """Concurrent async sends/recvs on one connection must match in
    submission order even with a multi-worker pool (per-direction ticket
    sequencing in the engine)."""
    import threading
    import time

    import torch

    from uccl_amd import p2p

    a = p2p.Endpoint(gpu=0, num_workers=3)
    b = p2p.Endpoint(gpu=0, num_workers=3)
    ids = {}
    th = threading.Thread(target=lambda: ids.setdefault("b", b.accept()))
    th.start()
    cid_a = a.connect(b.metadata())
    th.join(timeout=30)
    for trial in range(5):
        k = 6
        srcs = [torch.full((4096,), i + trial * 10, dtype=torch.int32)
                for i in range(k)]
        dsts = [torch.zeros(4096, dtype=torch.int32) for _ in range(k)]
        rids = [b.recv_async(ids["b"], d) for d in dsts]
        sids = [a.send_async(cid_a, s) for s in srcs]
        deadline = time.time() + 60
        for i in sids:
            while not a.poll_async(i):
                assert time.time() < deadline
                time.sleep(0.001)
        for i in rids:
            while not b.poll_async(i):
                assert time.time() < deadline
                time.sleep(0.001)
        for s, d in zip(srcs, dsts):
            assert torch.equal(s, d)


def test_p2p_stats_cpu():
    import threading

    import torch

    from uccl_amd import p2p

    a = p2p.Endpoint(gpu=0, num_workers=1)
    b = p2p.Endpoint(gpu=0, num_workers=1)
    ids = {}
    th = threading.Thread(target=lambda: ids.setdefault("b", b.accept()))
    th.start()
    cid = a.connect(b.metadata())
    th.join(timeout=30)
    src = torch.arange(10000, dtype=torch.float32)
    dst = torch.zeros_like(src)
    t = threading.Thread(target=lambda: b.recv(ids["b"], dst))
    t.start()
    a.send(cid, src)
    t.join(timeout=30)
    st = a.stats()
    assert st["send"]["calls"] == 1
    assert st["send"]["bytes"] == 40000
    assert st["send"]["p50_us"] > 0
    assert b.stats()["recv"]["calls"] == 1


def test_rccl_plane_graceful_fallback(monkeypatch):
    """UCCL_P2P_TRANSPORT=rccl on a GPU-less box: both sides negotiate,
    detect the plane is unusable, and fall back to the TCP plane — the
    transfer still completes (reference parity: the p2p NCCL backend is
    the portability fallback, p2p/nccl/nccl_endpoint.h:58)."""
    import torch

    monkeypatch.setenv("UCCL_P2P_TRANSPORT", "rccl")
    from uccl_amd import p2p

    a = p2p.Endpoint(gpu=0, num_workers=1)
    b = p2p.Endpoint(gpu=0, num_workers=1)
    import threading

    ids = {}
    t = threading.Thread(target=lambda: ids.__setitem__("b", b.accept()))
    t.start()
    ids["a"] = a.connect(b.metadata())
    t.join(timeout=20)
    src = torch.arange(4096, dtype=torch.uint8)
    dst = torch.zeros(4096, dtype=torch.uint8)
    exc = []

    def rx():
        try:
            b.recv(ids["b"], dst)
        except Exception as e:
            exc.append(e)

    t2 = threading.Thread(target=rx)
    t2.start()
    a.send(ids["a"], src)
    t2.join(timeout=30)
    assert not exc and torch.equal(src, dst)
