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


class _EndpointExtras:
    """Vectored helpers layered over the native Endpoint (parity with the
    reference's readv/writev, p2p/engine.h:243)."""

    @staticmethod
    def writev(ep, conn_id, tensors, adverts):
        ids = [ep.write_async(conn_id, t, ad)
               for t, ad in zip(tensors, adverts)]
        import time

        for i in ids:
            while not ep.poll_async(i):
                time.sleep(0.0005)

    @staticmethod
    def readv(ep, conn_id, tensors, adverts):
        ids = [ep.read_async(conn_id, t, ad)
               for t, ad in zip(tensors, adverts)]
        import time

        for i in ids:
            while not ep.poll_async(i):
                time.sleep(0.0005)


def __getattr__(name):
    if name == "Endpoint":
        from uccl_amd import _load_native

        return _load_native(required=True).Endpoint
    if name in ("writev", "readv"):
        return getattr(_EndpointExtras, name)
    raise AttributeError(name)
