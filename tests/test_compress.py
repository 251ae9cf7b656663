"""Lossless float codec tests (CPU). Reference analog: the DietGPU
compression layer on p2p transfers (p2p/rdma/compression.cc)."""

import threading

import pytest
import torch

from uccl_amd import p2p


def _roundtrip(t, strategy=p2p.STRATEGY_SPLIT_DEFLATE):
    frame = p2p.compress(t, strategy)
    back = p2p.decompress(frame)
    assert back.dtype == t.dtype if t.dtype in (
        torch.float32, torch.float16, torch.bfloat16) else torch.uint8
    return frame, back


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16,
                                   torch.bfloat16])
@pytest.mark.parametrize("n", [0, 1, 511, 4096, 1 << 18])
def test_roundtrip_exact(dtype, n):
    t = torch.randn(n, dtype=torch.float32).to(dtype)
    _, back = _roundtrip(t)
    assert torch.equal(back.view(t.shape), t)


def test_roundtrip_special_values():
    t = torch.tensor([0.0, -0.0, float("inf"), -float("inf"),
                      float("nan"), 1e-38, -1e38], dtype=torch.float32)
    frame, back = _roundtrip(t)
    assert back.shape == t.shape
    assert torch.equal(back.isnan(), t.isnan())
    assert torch.equal(back[~t.isnan()], t[~t.isnan()])


def test_ratio_on_model_like_data():
    # normally-distributed bf16: exponent byte is low-entropy
    t = torch.randn(1 << 20, dtype=torch.float32).bfloat16()
    frame = p2p.compress(t)
    ratio = (t.numel() * 2) / frame.numel()
    assert ratio > 1.25, f"ratio {ratio:.3f}"


def test_incompressible_falls_back_to_raw_planes():
    # uniform random BYTES: deflate cannot shrink the planes -> codec must
    # store them raw (overhead = header only, well under 1%)
    t = torch.randint(0, 256, (1 << 18, ), dtype=torch.uint8) \
        .view(torch.uint8)
    raw = torch.empty(1 << 17, dtype=torch.bfloat16)
    raw.view(torch.uint8).copy_(t[:raw.numel() * 2].view(torch.uint8))
    frame = p2p.compress(raw)
    assert frame.numel() < raw.numel() * 2 * 1.01


def test_strategies():
    t = torch.randn(1 << 16).bfloat16()
    for strat in (p2p.STRATEGY_NONE, p2p.STRATEGY_SPLIT_ONLY,
                  p2p.STRATEGY_SPLIT_DEFLATE):
        frame = p2p.compress(t, strat)
        assert torch.equal(p2p.decompress(frame).view(t.shape), t)


def test_non_float_passthrough():
    t = torch.arange(1000, dtype=torch.int64)
    frame = p2p.compress(t)
    back = p2p.decompress(frame)
    assert torch.equal(back.view(torch.int64), t)


def test_corrupt_frame_raises():
    t = torch.randn(4096).bfloat16()
    frame = p2p.compress(t)
    bad = frame.clone()
    bad[0] = 0  # break magic
    with pytest.raises(Exception):
        p2p.decompress(bad)
    trunc = frame[:frame.numel() // 2].clone()
    with pytest.raises(Exception):
        p2p.decompress(trunc)


def test_endpoint_compressed_transfer():
    # two endpoints in one process over the TCP plane, codec on the wire
    a = p2p.Endpoint(gpu=0, num_workers=1)
    b = p2p.Endpoint(gpu=0, num_workers=1)
    cid_b = {}

    def acceptor():
        cid_b["id"] = b.accept()

    th = threading.Thread(target=acceptor)
    th.start()
    cid_a = a.connect(b.metadata())
    th.join()

    src = torch.randn(257, 1024, dtype=torch.float32).bfloat16()
    got = {}

    def receiver():
        got["t"] = p2p.recv_compressed(b, cid_b["id"])

    rth = threading.Thread(target=receiver)
    rth.start()
    wire_bytes = p2p.send_compressed(a, cid_a, src)
    rth.join()
    assert torch.equal(got["t"], src)
    assert wire_bytes < src.numel() * 2  # actually compressed on the wire


def test_object_transfer():
    # Ray-API-analog object send: nested state dict with tensors
    a = p2p.Endpoint(gpu=0, num_workers=1)
    b = p2p.Endpoint(gpu=0, num_workers=1)
    ids = {}
    th = threading.Thread(target=lambda: ids.setdefault("b", b.accept()))
    th.start()
    cid_a = a.connect(b.metadata())
    th.join()

    obj = {
        "step": 1234,
        "lr": 3e-4,
        "weights": {"w1": torch.randn(64, 32).bfloat16(),
                    "ids": torch.arange(100, dtype=torch.int64)},
        "shapes": [(1, 2), [3, torch.zeros(5)]],
    }
    got = {}
    rx = threading.Thread(
        target=lambda: got.setdefault("o", p2p.recv_object(b, ids["b"])))
    rx.start()
    p2p.send_object(a, cid_a, obj)
    rx.join()

    o = got["o"]
    assert o["step"] == 1234 and o["lr"] == 3e-4
    assert torch.equal(o["weights"]["w1"], obj["weights"]["w1"])
    assert o["weights"]["ids"].dtype == torch.int64
    assert torch.equal(o["weights"]["ids"], obj["weights"]["ids"])
    assert o["shapes"][0] == (1, 2)
    assert torch.equal(o["shapes"][1][1], torch.zeros(5))


def test_endpoint_compressed_transfer_multipath():
    # codec over the multipath reliable plane (env must be set before the
    # endpoints exist; restore afterwards to keep test isolation)
    import os

    old = os.environ.get("UCCL_P2P_TRANSPORT")
    os.environ["UCCL_P2P_TRANSPORT"] = "multipath"
    try:
        a = p2p.Endpoint(gpu=0, num_workers=1)
        b = p2p.Endpoint(gpu=0, num_workers=1)
        ids = {}
        th = threading.Thread(target=lambda: ids.setdefault("b", b.accept()))
        th.start()
        cid_a = a.connect(b.metadata())
        th.join()
        src = torch.randn(1 << 20).bfloat16()
        got = {}
        rx = threading.Thread(
            target=lambda: got.setdefault("t",
                                          p2p.recv_compressed(b, ids["b"])))
        rx.start()
        p2p.send_compressed(a, cid_a, src)
        rx.join(timeout=120)
        assert torch.equal(got["t"], src)
    finally:
        if old is None:
            os.environ.pop("UCCL_P2P_TRANSPORT", None)
        else:
            os.environ["UCCL_P2P_TRANSPORT"] = old


def test_decompress_fuzz_never_crashes():
    # random and mutated frames must raise cleanly, never corrupt memory
    import random

    rnd = random.Random(7)
    # pure garbage
    for n in (0, 1, 8, 24, 25, 200):
        frame = torch.randint(0, 256, (n,), dtype=torch.uint8)
        with pytest.raises(Exception):
            p2p.decompress(frame)
    # valid frame with random single-byte mutations: either raises or
    # returns (possibly wrong bytes if a payload byte flipped) — what it
    # must never do is crash or over-read
    base = p2p.compress(torch.randn(4096).bfloat16())
    for _ in range(200):
        bad = base.clone()
        i = rnd.randrange(bad.numel())
        bad[i] = rnd.randrange(256)
        try:
            p2p.decompress(bad)
        except Exception:
            pass
    # truncations at every header-ish boundary
    for cut in (1, 4, 7, 8, 16, 24, 25, 33, base.numel() - 1):
        with pytest.raises(Exception):
            p2p.decompress(base[:cut].clone())


def test_decompress_overflow_sizes_rejected():
    # craft frames whose plane/block size fields would wrap u64 bounds
    import struct

    base = p2p.compress(torch.randn(4096).bfloat16())
    raw = bytearray(base.tolist())
    # plane table starts at byte 25: [encoding u8][stored u64] per plane
    evil = raw.copy()
    struct.pack_into("<Q", evil, 26, (1 << 64) - 8)  # plane 0 stored size
    with pytest.raises(Exception):
        p2p.decompress(torch.frombuffer(bytearray(evil), dtype=torch.uint8))
    evil2 = raw.copy()
    struct.pack_into("<Q", evil2, 26 + 9, (1 << 63))  # plane 1 stored size
    with pytest.raises(Exception):
        p2p.decompress(torch.frombuffer(bytearray(evil2), dtype=torch.uint8))
