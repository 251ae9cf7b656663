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
