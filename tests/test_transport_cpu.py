"""Software multipath reliable transport tests (CPU, loopback UDP).

Covers: multi-path spraying, SACK selective repeat under injected loss,
RTO recovery, Timely window adaptation, message ordering, zero-byte and
multi-MB messages. Mirrors the reference's loss-recovery experiment
methodology (collective/utran_osdi26ae.md:212-231) with deterministic
drops instead of WQE manipulation.
"""

import os
import threading

import pytest
import torch


def make_pair(**env):
    old = {}
    for k, v in env.items():
        old[k] = os.environ.get(k)
        os.environ[k] = str(v)
    try:
        from uccl_amd import _load_native

        C = _load_native(required=False)
        assert C is not None
        os.environ.setdefault("UCCL_TP_CWND_MAX", "256")
        a = C.TransportEndpoint(num_paths=4, chunk_bytes=4096)
        b = C.TransportEndpoint(num_paths=4, chunk_bytes=4096)
        flows = {}

        def acc():
            flows["b"] = b.accept()

        t = threading.Thread(target=acc)
        t.start()
        flows["a"] = a.connect(b.metadata())
        t.join(timeout=30)
        return C, a, b, flows["a"], flows["b"]
    finally:
        for k, v in old.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v


def xfer(a, b, fa, fb, nbytes, seed):
    g = torch.Generator().manual_seed(seed)
    src = torch.randint(0, 256, (max(nbytes, 1),), generator=g,
                        dtype=torch.uint8)[:nbytes]
    dst = torch.zeros(nbytes, dtype=torch.uint8)
    done = {}

    def rx():
        b.recv(fb, dst)
        done["rx"] = True

    t = threading.Thread(target=rx, daemon=True)
    t.start()
    a.send(fa, src)
    t.join(timeout=60)
    assert done.get("rx"), "recv did not complete"
    assert torch.equal(src, dst), f"payload mismatch at {nbytes} bytes"


def test_transport_clean_path():
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0)
    for n, s in [(0, 1), (1, 2), (100, 3), (4096, 4), (65536, 5),
                 (4 << 20, 6)]:
        xfer(a, b, fa, fb, n, s)
    st = a.stats()
    assert st.msgs_sent == 6
    assert st.data_sent >= (4 << 20) // 4096
    # loopback UDP may still drop under buffer pressure; require the
    # retransmit volume to be a small fraction of traffic, not zero
    assert st.retransmits + st.rto_retransmits <= st.data_sent * 0.10


def test_transport_ordering_and_bidirectional():
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0)
    # several queued messages must arrive in order
    msgs = [torch.full((1000 * (i + 1),), i, dtype=torch.uint8)
            for i in range(5)]
    outs = [torch.zeros_like(m) for m in msgs]

    def rx():
        for o in outs:
            b.recv(fb, o)

    t = threading.Thread(target=rx, daemon=True)
    t.start()
    for m in msgs:
        a.send(fa, m)
    t.join(timeout=60)
    for m, o in zip(msgs, outs):
        assert torch.equal(m, o)
    # reverse direction on the same flow
    back = torch.arange(256, dtype=torch.uint8)
    got = torch.zeros_like(back)

    def rx2():
        a.recv(fa, got)

    t = threading.Thread(target=rx2, daemon=True)
    t.start()
    b.send(fb, back)
    t.join(timeout=60)
    assert torch.equal(back, got)


@pytest.mark.parametrize("loss", [5, 20])
def test_transport_loss_recovery(loss):
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=loss)
    for n, s in [(100000, 11), (1 << 20, 12)]:
        xfer(a, b, fa, fb, n, s)
    st = a.stats()
    assert st.injected_drops > 0, "loss injection did not fire"
    assert st.retransmits + st.rto_retransmits > 0, "no recovery happened"
    assert st.msgs_sent == 2


def test_transport_ack_loss_recovery():
    # dropped ACKs are mostly covered by later acks' cumulative edge (the
    # protocol working as designed); the message must still complete with
    # full integrity, with RTO as the backstop for tail-ack loss
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0, UCCL_TP_ACK_LOSS_PCT=40)
    for i in range(3):
        xfer(a, b, fa, fb, 500000, 41 + i)
    st_b = b.stats()
    assert st_b.injected_drops > 0, "ack-loss injection did not fire"
    assert a.stats().msgs_sent == 3


def test_transport_stats_cc():
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0)
    xfer(a, b, fa, fb, 8 << 20, 21)
    st = a.stats()
    assert st.srtt_us > 0
    assert 2.0 <= st.cwnd <= 4096.0


def test_transport_eqds_credit():
    # receiver-driven credit mode: sender must stall when the granted
    # window is exhausted and resume on ack grants; payload intact
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0, UCCL_TP_CC="eqds",
                                UCCL_TP_RWND_KB=64)
    for n, sd in [(1 << 20, 51), (4 << 20, 52)]:
        xfer(a, b, fa, fb, n, sd)
    st = a.stats()
    assert st.msgs_sent == 2


def test_transport_swift_cc():
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0, UCCL_TP_CC="swift")
    xfer(a, b, fa, fb, 2 << 20, 31)
    st = a.stats()
    assert st.msgs_sent == 1 and st.srtt_us > 0


def test_transport_star_multi_flow():
    """One hub endpoint with flows to 3 spokes; concurrent bidirectional
    traffic on every flow."""
    from uccl_amd import _load_native

    C = _load_native(required=False)
    hub = C.TransportEndpoint(num_paths=2, chunk_bytes=4096)
    spokes = [C.TransportEndpoint(num_paths=2, chunk_bytes=4096)
              for _ in range(3)]
    hub_flows = []

    def acceptor():
        for _ in range(3):
            hub_flows.append(hub.accept())

    t = threading.Thread(target=acceptor, daemon=True)
    t.start()
    spoke_flows = [sp.connect(hub.metadata()) for sp in spokes]
    t.join(timeout=30)
    assert len(hub_flows) == 3

    n = 150000
    to_hub = [torch.randint(0, 255, (n,), dtype=torch.uint8)
              for _ in range(3)]
    from_hub = [torch.randint(0, 255, (n,), dtype=torch.uint8)
                for _ in range(3)]
    got_hub = [torch.zeros(n, dtype=torch.uint8) for _ in range(3)]
    got_spoke = [torch.zeros(n, dtype=torch.uint8) for _ in range(3)]

    threads = []
    # hub: recv+send per accepted flow (order of hub_flows is accept order;
    # content is matched by summing later, not by pairing)
    for idx, f in enumerate(hub_flows):
        def hub_io(i=idx, fl=f):
            hub.recv(fl, got_hub[i])
            hub.send(fl, from_hub[i])
        threads.append(threading.Thread(target=hub_io, daemon=True))
    for i, (sp, f) in enumerate(zip(spokes, spoke_flows)):
        def spoke_io(i=i, sp=sp, fl=f):
            sp.send(fl, to_hub[i])
            sp.recv(fl, got_spoke[i])
        threads.append(threading.Thread(target=spoke_io, daemon=True))
    for th in threads:
        th.start()
    for th in threads:
        th.join(timeout=90)
    # every to_hub payload arrived exactly once (any flow order)
    assert sorted(g.long().sum().item() for g in got_hub) ==         sorted(p.long().sum().item() for p in to_hub)
    for g in got_spoke:
        assert any(torch.equal(g, fh) for fh in from_hub)


def test_transport_pacing():
    """UCCL_TP_PACE_MBPS throttles chunk emission: a 1MB message at a
    10 MB/s pace must take >= ~80ms (unpaced loopback: ~1ms)."""
    import time

    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0, UCCL_TP_PACE_MBPS=10)
    t0 = time.perf_counter()
    xfer(a, b, fa, fb, 1 << 20, 61)
    dt = time.perf_counter() - t0
    assert dt >= 0.08, f"pacing had no effect ({dt*1e3:.1f} ms)"


def test_transport_eqds_paced_pull():
    """Paced pull quanta (UCCL_TP_EQDS_MBPS): the receiver doles credit
    at the configured rate, so a 2MB transfer at 20 MB/s (with a 64KB
    initial window) must take >= ~70ms."""
    import time

    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0, UCCL_TP_CC="eqds",
                                UCCL_TP_EQDS_MBPS=20, UCCL_TP_RWND_KB=64)
    t0 = time.perf_counter()
    xfer(a, b, fa, fb, 2 << 20, 71)
    dt = time.perf_counter() - t0
    assert dt >= 0.07, f"pull pacing had no effect ({dt*1e3:.1f} ms)"
    st = a.stats()
    assert st.msgs_sent == 1


def test_transport_eqds_incast_sharing():
    """Two senders incast into one receiver with a paced aggregate pull
    rate: both must complete, and the total must take at least as long
    as the aggregate rate allows (rate is split across active flows)."""
    import time

    import os

    old = {}
    env = {"UCCL_TP_CC": "eqds", "UCCL_TP_EQDS_MBPS": "40",
           "UCCL_TP_RWND_KB": "64", "UCCL_TP_CWND_MAX": "256"}
    for k, v in env.items():
        old[k] = os.environ.get(k)
        os.environ[k] = v
    try:
        from uccl_amd import _load_native

        C = _load_native(required=False)
        sink = C.TransportEndpoint(num_paths=2, chunk_bytes=4096)
        srcs = [C.TransportEndpoint(num_paths=2, chunk_bytes=4096)
                for _ in range(2)]
        sink_flows = []

        def acceptor():
            for _ in range(2):
                sink_flows.append(sink.accept())

        t = threading.Thread(target=acceptor, daemon=True)
        t.start()
        src_flows = [sp.connect(sink.metadata()) for sp in srcs]
        t.join(timeout=30)

        n = 1 << 20
        payloads = [torch.randint(0, 255, (n,), dtype=torch.uint8)
                    for _ in range(2)]
        outs = [torch.zeros(n, dtype=torch.uint8) for _ in range(2)]
        t0 = time.perf_counter()
        rx = []
        for i, f in enumerate(sink_flows):
            th = threading.Thread(target=lambda i=i, f=f:
                                  sink.recv(f, outs[i]), daemon=True)
            th.start()
            rx.append(th)
        tx = []
        for i, (sp, f) in enumerate(zip(srcs, src_flows)):
            th = threading.Thread(target=lambda i=i, sp=sp, f=f:
                                  sp.send(f, payloads[i]), daemon=True)
            th.start()
            tx.append(th)
        for th in tx + rx:
            th.join(timeout=60)
        dt = time.perf_counter() - t0
        # 2MB total at 40MB/s aggregate minus 2x64KB initial windows
        assert dt >= 0.035, f"incast pull pacing absent ({dt*1e3:.1f} ms)"
        got = sorted(o.long().sum().item() for o in outs)
        want = sorted(p.long().sum().item() for p in payloads)
        assert got == want
    finally:
        for k, v in old.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v


def test_transport_peer_death_aborts_cleanly():
    """Failure detection: when the peer vanishes, the RTO
    abort threshold marks the flow failed and pending sends raise
    instead of hanging."""
    import pytest

    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0, UCCL_TP_RTO_US=2000,
                                UCCL_TP_RTO_ABORT=8)
    # handshake sanity first
    xfer(a, b, fa, fb, 4096, 81)
    del b  # peer endpoint (and its sockets) die
    big = torch.zeros(4 << 20, dtype=torch.uint8)
    with pytest.raises(RuntimeError):
        a.send(fa, big)


def test_transport_garbage_udp_ignored():
    """Stray/garbage UDP datagrams at the data ports (wrong magic, bogus
    flow ids, truncated or oversized frames) must be ignored without
    crashing the endpoint or corrupting live transfers."""
    import random
    import socket
    import struct

    import os
    import re

    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0)
    # find this process's bound UDP ports (the endpoints' path sockets)
    inodes = set()
    for fd in os.listdir("/proc/self/fd"):
        try:
            tgt = os.readlink(f"/proc/self/fd/{fd}")
        except OSError:
            continue
        m = re.match(r"socket:\[(\d+)\]", tgt)
        if m:
            inodes.add(m.group(1))
    ports = []
    with open("/proc/net/udp") as f:
        next(f)
        for line in f:
            parts = line.split()
            if parts[9] in inodes:
                ports.append(int(parts[1].split(":")[1], 16))
    assert ports, "no UDP path sockets found"

    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    rnd = random.Random(3)
    for _ in range(300):
        port = rnd.choice(ports)
        kind = rnd.randrange(4)
        if kind == 0:
            pkt = bytes(rnd.randrange(256) for _ in range(rnd.randrange(90)))
        elif kind == 1:  # valid-looking data hdr, unknown flow
            pkt = struct.pack("<IIQQQIIQ", 0x7563636c, 1, rnd.getrandbits(63),
                              0, 0, rnd.randrange(1 << 20), 4096,
                              rnd.getrandbits(60)) + b"x" * 100
        elif kind == 2:  # ack-ish
            pkt = struct.pack("<IIQI", 0x7563636c, 2, rnd.getrandbits(63),
                              rnd.randrange(1 << 16)) + b"\x00" * 48
        else:
            pkt = b"\xff" * rnd.choice([1, 7, 65, 1400])
        s.sendto(pkt, ("127.0.0.1", port))
    # live transfer still works afterwards
    xfer(a, b, fa, fb, 1 << 20, 91)
    st = a.stats()
    assert st.msgs_sent >= 1


def test_transport_eqds_paced_under_loss():
    """Pull pacing composed with 5% data loss and 5% ack loss: grants,
    SACK recovery, and the credit-refresh path must cooperate."""
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=5, UCCL_TP_ACK_LOSS_PCT=5,
                                UCCL_TP_CC="eqds", UCCL_TP_EQDS_MBPS=200,
                                UCCL_TP_RWND_KB=256)
    for n, sd in [(1 << 20, 95), (3 << 20, 96), (0, 97), (777, 98)]:
        xfer(a, b, fa, fb, n, sd)
    st = a.stats()
    assert st.msgs_sent == 4
    assert st.retransmits + st.rto_retransmits > 0  # loss actually hit


def test_transport_async_post_flush():
    """post_send enqueues without waiting; flush blocks until everything
    posted on the flow is acked; receivers see posting order."""
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0)
    msgs = [torch.randint(0, 256, (n,), dtype=torch.uint8)
            for n in (64, 4096, 1 << 20, 32, 0, 123456)]
    outs = [torch.zeros_like(m) for m in msgs]
    done = {}

    def rx():
        for o in outs:
            b.recv(fb, o)
        done["rx"] = True

    t = threading.Thread(target=rx, daemon=True)
    t.start()
    for m in msgs:
        a.post_send(fa, m)  # returns without waiting for acks
    a.flush(fa)  # all acked from here
    t.join(timeout=60)
    assert done.get("rx")
    for m, o in zip(msgs, outs):
        assert torch.equal(m, o)
    assert a.stats().msgs_sent == len(msgs)


def test_transport_async_flush_survives_loss():
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=15)
    msgs = [torch.randint(0, 256, (8192,), dtype=torch.uint8)
            for _ in range(8)]
    outs = [torch.zeros_like(m) for m in msgs]
    done = {}

    def rx():
        for o in outs:
            b.recv(fb, o)
        done["rx"] = True

    t = threading.Thread(target=rx, daemon=True)
    t.start()
    for m in msgs:
        a.post_send(fa, m)
    a.flush(fa)
    t.join(timeout=120)
    assert done.get("rx")
    for m, o in zip(msgs, outs):
        assert torch.equal(m, o)


def test_transport_flush_after_close_raises():
    """Posting then closing the flow must fail the flush (and any
    blocked senders) instead of hanging."""
    C, a, b, fa, fb = make_pair(UCCL_TP_LOSS_PCT=0)
    t = torch.randint(0, 256, (4096,), dtype=torch.uint8)
    o = torch.zeros_like(t)
    a.post_send(fa, t)
    b.recv(fb, o)  # drain so the flow is idle
    a.flush(fa)
    assert torch.equal(t, o)
    a.close_flow(fa)
    import pytest as _pytest
    with _pytest.raises(RuntimeError):
        a.post_send(fa, t)
        a.flush(fa)
