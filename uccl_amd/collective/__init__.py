"""Python API for the xGMI collective engine.

Bootstrap uses torch.distributed (any backend — gloo suffices) purely to
exchange the 64-byte HIP IPC handles, mirroring the reference's split
between TCP out-of-band rendezvous and the data plane;
all data-plane traffic then flows through uccl_amd's own CDNA4 kernels
over xGMI peer-HBM access.

Usage (one process per GPU):
    import torch.distributed as dist
    dist.init_process_group("gloo")          # rendezvous only
    comm = uccl_amd.collective.init()        # data plane: uccl_amd
    comm.all_reduce(tensor)                  # async on current stream
"""

from __future__ import annotations

import os
from typing import Optional

import torch


class Communicator:
    """Thin wrapper over the native uccl::Communicator."""

    def __init__(self, rank: int, world: int, device: Optional[int] = None,
                 heap_bytes: int = 0):
        from uccl_amd import _load_native

        C = _load_native(required=True)
        if device is None:
            device = torch.cuda.current_device()
        self._c = C.Communicator(rank, world, device, heap_bytes)
        self.rank = rank
        self.world = world
        self.device = device

    # -- bootstrap -----------------------------------------------------------
    def handle_bytes(self) -> bytes:
        return self._c.handle_bytes()

    def connect(self, all_handles: list[bytes]) -> None:
        self._c.connect(list(all_handles))

    # -- collectives (async on the current torch stream) ---------------------
    def all_reduce(self, t: torch.Tensor, op: str = "sum") -> None:
        """op: sum | prod | min | max (fp32 accumulation for 16/8-bit floats)."""
        self._c.all_reduce(t, op)

    def all_gather(self, out: torch.Tensor, inp: torch.Tensor) -> None:
        self._c.all_gather(out, inp)

    def reduce_scatter(self, out: torch.Tensor, inp: torch.Tensor,
                       op: str = "sum") -> None:
        self._c.reduce_scatter(out, inp, op)

    def broadcast(self, t: torch.Tensor, root: int) -> None:
        self._c.broadcast(t, root)

    def all_to_all(self, out: torch.Tensor, inp: torch.Tensor) -> None:
        self._c.all_to_all(out, inp)

    def send(self, t: torch.Tensor, dst: int) -> None:
        self._c.send(t, dst)

    def recv(self, t: torch.Tensor, src: int) -> None:
        self._c.recv(t, src)

    def barrier(self) -> None:
        self._c.barrier()

    # -- symmetric (zero-copy) tensors --------------------------------------
    def symmetric_tensor(self, sizes, dtype=torch.bfloat16) -> torch.Tensor:
        """Allocate a tensor in the registered symmetric region. Every rank
        must make the same symmetric allocations in the same order.
        Collectives on these tensors skip all staging copies (the kernels
        read/write every rank's buffer directly over xGMI)."""
        if isinstance(sizes, int):
            sizes = [sizes]
        return self._c.symmetric_tensor(list(sizes), dtype)

    def is_symmetric(self, t: torch.Tensor) -> bool:
        return self._c.is_symmetric(t)

    def stats(self) -> dict:
        """Per-op host-side call/byte tallies (observability parity with
        the reference's stats counters)."""
        return self._c.stats()


def init(group=None, device: Optional[int] = None,
         heap_bytes: int = 0) -> Communicator:
    """Create a Communicator using torch.distributed for rendezvous."""
    import torch.distributed as dist

    if dist.is_available() and dist.is_initialized():
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
    else:
        rank, world = 0, 1
    if device is None:
        local = int(os.environ.get("LOCAL_RANK", rank))
        ndev = torch.cuda.device_count()
        device = local % max(ndev, 1)
        torch.cuda.set_device(device)
    comm = Communicator(rank, world, device, heap_bytes)
    if world > 1:
        handles = [None] * world
        dist.all_gather_object(handles, comm.handle_bytes(), group=group)
        comm.connect(handles)
    return comm


_backend_registered = False


def register_torch_backend() -> None:
    """Register the native "uccl" c10d backend so that
    torch.distributed.init_process_group(backend="uccl") routes collectives
    through the xGMI engine directly (drop-in, no code changes)."""
    global _backend_registered
    if _backend_registered:
        return
    import torch.distributed as dist

    from uccl_amd import _load_native

    C = _load_native(required=True)

    def _creator(store, rank, size, timeout):
        return C._create_uccl_backend(store, rank, size, timeout)

    dist.Backend.register_backend("uccl", _creator, devices=["cuda"])
    _backend_registered = True
