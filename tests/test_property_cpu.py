"""Property-based tests (hypothesis) for the pure-function layers:
codec roundtrips over arbitrary payloads and MRMap interval invariants."""

import bisect

import pytest
import torch
from hypothesis import given, settings, strategies as st

from uccl_amd import p2p
from uccl_amd.p2p.utils import MRMap


@settings(max_examples=60, deadline=None)
@given(data=st.binary(min_size=0, max_size=20000),
       dtype=st.sampled_from(["float32", "float16", "bfloat16", "uint8"]),
       strategy=st.sampled_from([0, 1, 2]))
def test_codec_roundtrip_any_bytes(data, dtype, strategy):
    dt = getattr(torch, dtype)
    es = torch.tensor([], dtype=dt).element_size()
    n = len(data) - len(data) % es
    t = torch.frombuffer(bytearray(data[:n]), dtype=dt) if n else \
        torch.empty(0, dtype=dt)
    frame = p2p.compress(t, strategy)
    back = p2p.decompress(frame)
    assert bool((back.view(torch.uint8) ==
                 t.contiguous().view(torch.uint8).view(-1)).all())


@settings(max_examples=60, deadline=None)
@given(st.lists(st.tuples(st.integers(0, 1 << 40), st.integers(1, 1 << 20)),
                max_size=30))
def test_mrmap_invariants(regions):
    m = MRMap()
    accepted = []  # (base, len, id)
    for i, (base, length) in enumerate(regions):
        try:
            m.add(i, base, length)
            accepted.append((base, length, i))
        except ValueError:
            # must genuinely overlap something already accepted
            assert any(base < ab + al and ab < base + length
                       for ab, al, _ in accepted)
    assert len(m) == len(accepted)
    for ab, al, aid in accepted:
        assert m.find(ab) == (aid, 0)
        assert m.find(ab + al - 1) == (aid, al - 1)
        assert m.find(ab, al) == (aid, 0)
        assert m.find(ab, al + 1) is None
    # probes just outside each region must not return that region
    for ab, al, aid in accepted:
        hit = m.find(ab + al)
        assert hit is None or hit[0] != aid


@settings(max_examples=60, deadline=None)
@given(st.integers(0, 1 << 40), st.integers(1, 100000),
       st.integers(1, 1 << 22))
def test_xferdesc_split_covers_exactly(offset, nbytes, chunk):
    from uccl_amd.p2p.utils import XferDesc

    d = XferDesc(mr_id=1, offset=offset, bytes=nbytes, tag=5)
    parts = d.split(chunk)
    assert sum(p.bytes for p in parts) == nbytes
    pos = offset
    for p in parts:
        assert p.offset == pos and 0 < p.bytes <= chunk
        pos += p.bytes
    rt = XferDesc.deserialize(d.serialize())
    assert rt == d


@settings(max_examples=40, deadline=None)
@given(st.lists(st.floats(min_value=0.001, max_value=1e7,
                          allow_nan=False), min_size=1, max_size=300))
def test_latency_hist_percentile_monotone(samples):
    from uccl_amd import _load_native

    C = _load_native(required=False)
    h = C.LatencyHist()
    for us in samples:
        h.record_us(us)
    assert h.count() == len(samples)
    last = 0.0
    for p in (0, 25, 50, 75, 90, 99, 100):
        v = h.percentile_us(p)
        assert v >= last
        last = v
    # upper bound respects bucket resolution (~41% + rounding)
    assert h.percentile_us(100) <= max(samples) * 2 + 1e-3
    assert h.percentile_us(0) >= min(samples) / 2 - 1e-3
