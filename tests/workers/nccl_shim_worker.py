"""Worker driving libuccl_nccl.so (the NCCL C-ABI drop-in) via ctypes:
GetUniqueId / CommInitRank bootstrap, then allreduce / broadcast /
allgather / reducescatter on torch GPU tensors, checked against torch."""

from __future__ import annotations

import ctypes
import os
import signal
import sys
import time

signal.alarm(int(os.environ.get("UCCL_TEST_ALARM", "200")))

import torch


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    id_path = sys.argv[1]
    torch.cuda.set_device(0)
    torch.cuda.init()

    from uccl_amd._build import PKG_DIR

    lib = ctypes.CDLL(str(PKG_DIR / "lib" / "libuccl_nccl.so"))
    class NcclId(ctypes.Structure):
        _fields_ = [("b", ctypes.c_byte * 128)]  # passed BY VALUE

    NCCL_ID = NcclId
    lib.ncclGetUniqueId.argtypes = [ctypes.POINTER(NcclId)]
    lib.ncclCommInitRank.argtypes = [ctypes.POINTER(ctypes.c_void_p),
                                     ctypes.c_int, NcclId, ctypes.c_int]
    for f, extra in (("ncclAllReduce", [ctypes.c_int]),
                     ("ncclReduceScatter", [ctypes.c_int])):
        fn = getattr(lib, f)
        fn.restype = ctypes.c_int
        fn.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t,
                       ctypes.c_int] + extra + [ctypes.c_void_p,
                                                ctypes.c_void_p]
    lib.ncclAllToAll.restype = ctypes.c_int
    lib.ncclAllToAll.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                 ctypes.c_size_t, ctypes.c_int,
                                 ctypes.c_void_p, ctypes.c_void_p]
    lib.ncclAllGather.restype = ctypes.c_int
    lib.ncclAllGather.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                  ctypes.c_size_t, ctypes.c_int,
                                  ctypes.c_void_p, ctypes.c_void_p]
    lib.ncclBroadcast.restype = ctypes.c_int
    lib.ncclBroadcast.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                  ctypes.c_size_t, ctypes.c_int,
                                  ctypes.c_int, ctypes.c_void_p,
                                  ctypes.c_void_p]
    lib.ncclCommDestroy.argtypes = [ctypes.c_void_p]

    ncclFloat32, ncclSum = 7, 0

    uid = NCCL_ID()
    if rank == 0:
        lib.ncclGetUniqueId(ctypes.byref(uid))
        with open(id_path + ".tmp", "wb") as f:
            f.write(bytes(uid.b))
        os.rename(id_path + ".tmp", id_path)
    else:
        while not os.path.exists(id_path):
            time.sleep(0.05)
        raw = open(id_path, "rb").read()
        uid = NCCL_ID()
        ctypes.memmove(uid.b, raw, 128)

    comm = ctypes.c_void_p()
    rc = lib.ncclCommInitRank(ctypes.byref(comm), world, uid, rank)
    assert rc == 0, f"CommInitRank rc={rc}"
    print(f"[rank {rank}] init OK", flush=True)

    stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
    want = sum(r + 1.0 for r in range(world))

    # in-place allreduce
    t = torch.full((65536,), float(rank + 1), device="cuda")
    rc = lib.ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                           ncclFloat32, ncclSum, comm, stream)
    torch.cuda.synchronize()
    assert rc == 0 and torch.allclose(t, torch.full_like(t, want)), t[:3]
    print(f"[rank {rank}] allreduce OK", flush=True)

    # out-of-place allreduce
    src = torch.full((4096,), float(rank + 1), device="cuda")
    dst = torch.zeros(4096, device="cuda")
    rc = lib.ncclAllReduce(src.data_ptr(), dst.data_ptr(), 4096,
                           ncclFloat32, ncclSum, comm, stream)
    torch.cuda.synchronize()
    assert rc == 0 and torch.allclose(dst, torch.full_like(dst, want))

    print(f"[rank {rank}] oop allreduce OK", flush=True)

    # broadcast
    b = torch.full((1024,), float(rank * 5), device="cuda")
    rc = lib.ncclBroadcast(b.data_ptr(), b.data_ptr(), 1024, ncclFloat32,
                           0, comm, stream)
    torch.cuda.synchronize()
    assert rc == 0 and torch.allclose(b, torch.zeros_like(b))

    print(f"[rank {rank}] broadcast OK", flush=True)

    # allgather
    ag_in = torch.full((256,), float(rank), device="cuda")
    ag_out = torch.empty(256 * world, device="cuda")
    rc = lib.ncclAllGather(ag_in.data_ptr(), ag_out.data_ptr(), 256,
                           ncclFloat32, comm, stream)
    torch.cuda.synchronize()
    assert rc == 0
    for r in range(world):
        assert torch.allclose(ag_out[r * 256:(r + 1) * 256],
                              torch.full((256,), float(r), device="cuda"))

    print(f"[rank {rank}] allgather OK", flush=True)

    # reduce_scatter
    rs_in = torch.full((512 * world,), float(rank + 1), device="cuda")
    rs_out = torch.empty(512, device="cuda")
    rc = lib.ncclReduceScatter(rs_in.data_ptr(), rs_out.data_ptr(), 512,
                               ncclFloat32, ncclSum, comm, stream)
    torch.cuda.synchronize()
    assert rc == 0 and torch.allclose(rs_out, torch.full_like(rs_out, want))

    print(f"[rank {rank}] reduce_scatter OK", flush=True)

    # all_to_all (RCCL extension): segment j of rank i's input lands as
    # segment i of rank j's output
    a2a_in = torch.cat([torch.full((128,), float(rank * 10 + j),
                                   device="cuda") for j in range(world)])
    a2a_out = torch.empty_like(a2a_in)
    rc = lib.ncclAllToAll(a2a_in.data_ptr(), a2a_out.data_ptr(), 128,
                          ncclFloat32, comm, stream)
    torch.cuda.synchronize()
    assert rc == 0
    for src in range(world):
        seg = a2a_out[src * 128:(src + 1) * 128]
        assert torch.allclose(
            seg, torch.full_like(seg, float(src * 10 + rank))), (src, seg[0])

    print(f"[rank {rank}] alltoall OK", flush=True)

    # --- op matrix: prod / min / max / avg (rccl-tests patterns) ------------
    ncclProd, ncclMax, ncclMin, ncclAvg = 1, 2, 3, 4
    prod_want = 1.0
    for r in range(world):
        prod_want *= (r + 1.0)
    for op, wantv in ((ncclProd, prod_want), (ncclMin, 1.0),
                      (ncclMax, float(world)),
                      (ncclAvg, want / world)):
        t = torch.full((8192,), float(rank + 1), device="cuda")
        rc = lib.ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                               ncclFloat32, op, comm, stream)
        torch.cuda.synchronize()
        assert rc == 0 and torch.allclose(
            t, torch.full_like(t, wantv)), (op, float(t[0]), wantv)
        # reduce_scatter with the same op
        rs_in = torch.full((256 * world,), float(rank + 1), device="cuda")
        rs_out = torch.empty(256, device="cuda")
        rc = lib.ncclReduceScatter(rs_in.data_ptr(), rs_out.data_ptr(),
                                   256, ncclFloat32, op, comm, stream)
        torch.cuda.synchronize()
        assert rc == 0 and torch.allclose(
            rs_out, torch.full_like(rs_out, wantv)), (op, float(rs_out[0]))

    print(f"[rank {rank}] op matrix OK", flush=True)

    # --- grouped send/recv (the advisor-r1 deadlock pattern): paired
    # >2MB sendrecv inside ncclGroupStart/End must complete ---------------
    if world >= 2:
        lib.ncclGroupStart.restype = ctypes.c_int
        lib.ncclGroupEnd.restype = ctypes.c_int
        lib.ncclSend.restype = ctypes.c_int
        lib.ncclSend.argtypes = [ctypes.c_void_p, ctypes.c_size_t,
                                 ctypes.c_int, ctypes.c_int,
                                 ctypes.c_void_p, ctypes.c_void_p]
        lib.ncclRecv.restype = ctypes.c_int
        lib.ncclRecv.argtypes = [ctypes.c_void_p, ctypes.c_size_t,
                                 ctypes.c_int, ctypes.c_int,
                                 ctypes.c_void_p, ctypes.c_void_p]
        peer = rank ^ 1
        n = 3 << 20  # 12 MB of fp32: multiple 2MB slot credits deep
        sbuf = torch.full((n,), float(rank + 3), device="cuda")
        rbuf = torch.zeros(n, device="cuda")
        assert lib.ncclGroupStart() == 0
        assert lib.ncclSend(sbuf.data_ptr(), n, ncclFloat32, peer, comm,
                            stream) == 0
        assert lib.ncclRecv(rbuf.data_ptr(), n, ncclFloat32, peer, comm,
                            stream) == 0
        assert lib.ncclGroupEnd() == 0
        torch.cuda.synchronize()
        assert torch.allclose(rbuf, torch.full_like(rbuf, float(peer + 3)))
        print(f"[rank {rank}] grouped sendrecv OK", flush=True)

    lib.ncclCommDestroy(comm)
    print(f"[rank {rank}] NCCL SHIM OK", flush=True)


if __name__ == "__main__":
    main()
