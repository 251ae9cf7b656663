"""Observability tests: chrome-trace event recorder + latency percentile
histogram (CPU). Reference analogs: NPKit event tracing
(lite-collective core/npkit.cc) and include/util/latency.h."""

import json
import threading

import torch

from uccl_amd import _load_native

C = _load_native(required=False)


def _pair():
    a = C.TransportEndpoint(num_paths=2, chunk_bytes=4096)
    b = C.TransportEndpoint(num_paths=2, chunk_bytes=4096)
    out = {}

    def acc():
        out["fb"] = b.accept()

    t = threading.Thread(target=acc)
    t.start()
    fa = a.connect(b.metadata())
    t.join(timeout=30)
    return a, b, fa, out["fb"]


def test_latency_hist_percentiles():
    h = C.LatencyHist()
    assert h.count() == 0 and h.percentile_us(50) == 0
    for us in range(1, 1001):
        h.record_us(float(us))
    assert h.count() == 1000
    p50 = h.percentile_us(50)
    p99 = h.percentile_us(99)
    # log-bucketed: ~41% resolution upper bounds
    assert 400 <= p50 <= 800, p50
    assert 900 <= p99 <= 2000, p99
    assert p99 >= p50
    h.reset()
    assert h.count() == 0


def test_latency_hist_wide_range():
    h = C.LatencyHist()
    h.record_ns(5)
    h.record_us(3.0)
    h.record_us(30000.0)  # 30 ms
    assert h.count() == 3
    assert h.percentile_us(100) >= 20000


def test_trace_spans_from_transport():
    C.trace_clear()
    C.trace_set_enabled(True)
    try:
        a, b, fa, fb = _pair()
        src = torch.arange(256, dtype=torch.uint8)
        dst = torch.zeros(256, dtype=torch.uint8)
        t = threading.Thread(target=lambda: b.recv(fb, dst))
        t.start()
        a.send(fa, src)
        t.join(timeout=30)
        assert torch.equal(src, dst)
    finally:
        C.trace_set_enabled(False)
    assert C.trace_num_events() >= 4  # send_msg + recv_msg begin/end
    doc = json.loads(C.trace_dump_json())
    evs = doc["traceEvents"]
    names = {(e["name"], e["ph"]) for e in evs}
    assert ("send_msg", "B") in names and ("send_msg", "E") in names
    assert ("recv_msg", "B") in names and ("recv_msg", "E") in names
    for e in evs:
        assert e["cat"] == "transport"
        assert isinstance(e["ts"], int)
    C.trace_clear()
    assert C.trace_num_events() == 0


def test_trace_disabled_records_nothing():
    C.trace_clear()
    assert not C.trace_enabled()
    a, b, fa, fb = _pair()
    src = torch.zeros(64, dtype=torch.uint8)
    dst = torch.zeros(64, dtype=torch.uint8)
    t = threading.Thread(target=lambda: b.recv(fb, dst))
    t.start()
    a.send(fa, src)
    t.join(timeout=30)
    assert C.trace_num_events() == 0


def test_transport_stats_rtt_percentiles():
    a, b, fa, fb = _pair()
    src = torch.zeros(1 << 20, dtype=torch.uint8)
    dst = torch.zeros(1 << 20, dtype=torch.uint8)
    t = threading.Thread(target=lambda: b.recv(fb, dst))
    t.start()
    a.send(fa, src)
    t.join(timeout=30)
    st = a.stats()
    assert st.rtt_p50_us > 0
    assert st.rtt_p99_us >= st.rtt_p50_us


# --- lockless rings (reference jring analog, core/ring.h) -----------------

def test_spsc_ring_order_and_capacity():
    r = C.SpscRingU64(8)
    for i in range(8):
        assert r.push(i)
    assert not r.push(99)  # full
    assert r.size() == 8
    for i in range(8):
        assert r.pop() == i  # FIFO
    assert r.pop() is None


def test_mpmc_ring_threads():
    r = C.MpmcRingU64(1 << 12)
    nprod, nitems = 4, 20000
    popped = []
    lock = threading.Lock()
    stop = threading.Event()

    def producer(base):
        for i in range(nitems):
            v = base * nitems + i
            while not r.push(v):
                pass

    def consumer():
        local = []
        while not stop.is_set() or r.size_approx():
            v = r.pop()
            if v is not None:
                local.append(v)
        with lock:
            popped.extend(local)

    cons = [threading.Thread(target=consumer) for _ in range(3)]
    prods = [threading.Thread(target=producer, args=(b,))
             for b in range(nprod)]
    for t in cons + prods:
        t.start()
    for t in prods:
        t.join(timeout=60)
    stop.set()
    for t in cons:
        t.join(timeout=60)
    assert sorted(popped) == list(range(nprod * nitems))


def test_ring_capacity_validation():
    import pytest

    with pytest.raises(Exception):
        C.MpmcRingU64(100)  # not a power of two


def test_trace_ukernel_execution():
    import json

    import torch

    from uccl_amd import ukernel as uk

    C.trace_clear()
    C.trace_set_enabled(True)
    try:
        topo = uk.Topology(2)
        g = uk.lower(uk.plan_allreduce_oneshot(topo, 1024))
        ins = [torch.ones(256), torch.ones(256) * 2]
        outs, _ = uk.execute_host(g, ins)
        assert torch.equal(outs[0], torch.full((256,), 3.0))
    finally:
        C.trace_set_enabled(False)
    doc = json.loads(C.trace_dump_json())
    names = {e["name"] for e in doc["traceEvents"]
             if e["cat"] == "ukernel"}
    assert {"put", "reduce", "copy", "signal"} <= names
    C.trace_clear()


def test_spsc_ring_two_threads():
    r = C.SpscRingU64(1 << 10)
    n = 200000
    got = []

    def consumer():
        expect = 0
        while expect < n:
            v = r.pop()
            if v is not None:
                assert v == expect  # strict FIFO
                expect += 1
        got.append(expect)

    t = threading.Thread(target=consumer)
    t.start()
    for i in range(n):
        while not r.push(i):
            pass
    t.join(timeout=60)
    assert got == [n]
