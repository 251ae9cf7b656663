"""NIXL-style point-to-point transfer engine (KV-cache / weight transfer).

API parity with the reference's `uccl.p2p` Endpoint (p2p/engine_api.cc):
connect/accept, tensor MR registration, blocking + async send/recv, and
one-sided read/write against receiver-advertised windows, with a same-host
HIP-IPC one-copy fast path and TCP (pinned-staged) for everything else.

    ep  = uccl_amd.p2p.Endpoint(gpu=0)
    md  = ep.metadata()                    # ship out-of-band
    cid = ep.connect(remote_md)            # or ep.accept()
    mr  = ep.reg(tensor)
    ad  = ep.advertise(mr, 0, nbytes)      # receiver side
    ep.write(cid, src_tensor, ad)          # writer side (one-sided)
"""

from __future__ import annotations


def __getattr__(name):
    if name == "Endpoint":
        from uccl_amd import _load_native

        return _load_native(required=True).Endpoint
    raise AttributeError(name)
