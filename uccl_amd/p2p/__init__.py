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


# --- lossless float compression (reference parity: DietGPU layer,
# p2p/rdma/compression.cc; strategies via UCCL_P2P_COMPRESS_STRATEGY) ------

STRATEGY_NONE = 0
STRATEGY_SPLIT_ONLY = 1
STRATEGY_SPLIT_DEFLATE = 2

_HDR = "<IQi8q"  # magic, frame_bytes, ndim, dims[8]
_HDR_MAGIC = 0x50435A46


def _native():
    from uccl_amd import _load_native

    return _load_native(required=True)


def compress(tensor, strategy: int = STRATEGY_SPLIT_DEFLATE):
    """Bitwise-lossless codec: plane split + deflate for f32/f16/bf16 host
    tensors (anything else passes through framed). Returns a uint8 frame."""
    return _native().comp_compress(tensor.contiguous().view(-1), strategy)


def decompress(frame):
    """Inverse of compress(); returns a flat tensor of the original dtype."""
    return _native().comp_decompress(frame)


def default_strategy():
    import os

    return int(os.environ.get("UCCL_P2P_COMPRESS_STRATEGY",
                              str(STRATEGY_SPLIT_DEFLATE)))


def send_compressed(ep, conn_id, tensor, strategy=None):
    """Send a host tensor through `ep` with the codec applied (128-byte
    shape header + frame). Pair with recv_compressed on the other side."""
    import struct

    import torch

    t = tensor.contiguous()
    frame = compress(t.view(-1), default_strategy() if strategy is None
                     else strategy)
    dims = list(t.shape)[:8]
    dims += [0] * (8 - len(dims))
    hdr = struct.pack(_HDR, _HDR_MAGIC, frame.numel(), t.dim(), *dims)
    ep.send(conn_id, torch.frombuffer(bytearray(hdr), dtype=torch.uint8))
    ep.send(conn_id, frame)
    return frame.numel()


def recv_compressed(ep, conn_id):
    import struct

    import torch

    hdr_t = torch.empty(struct.calcsize(_HDR), dtype=torch.uint8)
    ep.recv(conn_id, hdr_t)
    magic, fbytes, ndim, *dims = struct.unpack(_HDR, bytes(hdr_t.tolist()))
    assert magic == _HDR_MAGIC, "recv_compressed: bad header"
    frame = torch.empty(fbytes, dtype=torch.uint8)
    ep.recv(conn_id, frame)
    out = decompress(frame)
    return out.view(*dims[:ndim]) if ndim else out


def __getattr__(name):
    if name == "Endpoint":
        from uccl_amd import _load_native

        return _load_native(required=True).Endpoint
    if name in ("writev", "readv"):
        return getattr(_EndpointExtras, name)
    raise AttributeError(name)