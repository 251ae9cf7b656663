"""Topology utils degrade gracefully without GPUs."""


def test_topology_summary_no_gpu():
    from uccl_amd.utils import topology_summary

    out = topology_summary()
    assert "GPU(s)" in out


# --- p2p helper utils (MR interval map + XferDesc; reference p2p/utils.py
# interval tree and engine_api.cc XferDesc) --------------------------------

def test_mrmap_lookup():
    from uccl_amd.p2p.utils import MRMap

    m = MRMap()
    m.add(1, 0x1000, 0x100)
    m.add(2, 0x3000, 0x1000)
    assert m.find(0x1000) == (1, 0)
    assert m.find(0x10ff) == (1, 0xff)
    assert m.find(0x1100) is None          # one past the end
    assert m.find(0x3800, 0x800) == (2, 0x800)
    assert m.find(0x3800, 0x801) is None   # crosses the end
    assert m.find(0x500) is None
    assert len(m) == 2
    assert m.remove(1) and not m.remove(1)
    assert m.find(0x1000) is None


def test_mrmap_rejects_overlap():
    import pytest

    from uccl_amd.p2p.utils import MRMap

    m = MRMap()
    m.add(1, 100, 50)
    with pytest.raises(ValueError):
        m.add(2, 120, 10)     # inside
    with pytest.raises(ValueError):
        m.add(3, 90, 20)      # straddles the start
    m.add(4, 150, 10)         # adjacent is fine


def test_mrmap_tensor():
    import torch

    from uccl_amd.p2p.utils import MRMap

    t = torch.zeros(1024, dtype=torch.float32)
    m = MRMap()
    m.add(7, t.data_ptr(), t.numel() * 4)
    assert m.find_tensor(t) == (7, 0)
    assert m.find_tensor(t[256:512]) == (7, 1024)


def test_xferdesc_roundtrip_and_split():
    from uccl_amd.p2p.utils import XferDesc

    d = XferDesc(mr_id=3, offset=4096, bytes=10_000, tag=9,
                 meta={"dtype": "bf16"})
    back = XferDesc.deserialize(d.serialize())
    assert back == d
    parts = d.split(4096)
    assert [p.bytes for p in parts] == [4096, 4096, 1808]
    assert parts[2].offset == 4096 + 8192
    assert all(p.tag == 9 for p in parts)


def test_prometheus_metrics_export():
    import threading

    import torch

    from uccl_amd import _load_native
    from uccl_amd.utils import metrics

    C = _load_native(required=False)
    a = C.TransportEndpoint(num_paths=2, chunk_bytes=4096)
    b = C.TransportEndpoint(num_paths=2, chunk_bytes=4096)
    out = {}
    th = threading.Thread(target=lambda: out.setdefault("fb", b.accept()))
    th.start()
    fa = a.connect(b.metadata())
    th.join(timeout=30)
    src = torch.zeros(65536, dtype=torch.uint8)
    dst = torch.zeros(65536, dtype=torch.uint8)
    t = threading.Thread(target=lambda: b.recv(out["fb"], dst))
    t.start()
    a.send(fa, src)
    t.join(timeout=30)

    metrics.track_transport("tp_a", a)
    text = metrics.render().decode()
    assert 'uccl_transport_msgs_sent{name="tp_a"} 1.0' in text
    assert "uccl_transport_rtt_p50_us" in text
    metrics.untrack("transport", "tp_a")
    assert "tp_a" not in metrics.render().decode()


def test_prometheus_p2p_metrics():
    import threading

    import torch

    from uccl_amd import p2p
    from uccl_amd.utils import metrics

    a = p2p.Endpoint(gpu=0, num_workers=1)
    b = p2p.Endpoint(gpu=0, num_workers=1)
    ids = {}
    th = threading.Thread(target=lambda: ids.setdefault("b", b.accept()))
    th.start()
    cid = a.connect(b.metadata())
    th.join(timeout=30)
    src = torch.zeros(1000, dtype=torch.uint8)
    dst = torch.zeros(1000, dtype=torch.uint8)
    t = threading.Thread(target=lambda: b.recv(ids["b"], dst))
    t.start()
    a.send(cid, src)
    t.join(timeout=30)
    metrics.track_p2p("ep_a", a)
    text = metrics.render().decode()
    assert 'uccl_p2p_send_calls{name="ep_a"} 1.0' in text
    assert 'uccl_p2p_send_bytes{name="ep_a"} 1000.0' in text
    metrics.untrack("p2p", "ep_a")
