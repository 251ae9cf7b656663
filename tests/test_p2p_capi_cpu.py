"""Flat C API (libuccl_p2p.so) loopback test via ctypes — host memory, no
GPU needed. Parity evidence for the reference's uccl_engine_* C API."""

import ctypes

from uccl_amd._build import build_plugin, PKG_DIR


def test_c_api_loopback():
    build_plugin()
    lib = ctypes.CDLL(str(PKG_DIR / "lib" / "libuccl_p2p.so"))
    lib.uccl_engine_create.restype = ctypes.c_void_p
    lib.uccl_engine_create.argtypes = [ctypes.c_int, ctypes.c_int]
    lib.uccl_engine_metadata.restype = ctypes.c_int
    lib.uccl_engine_metadata.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                         ctypes.c_size_t]
    lib.uccl_engine_connect.restype = ctypes.c_uint64
    lib.uccl_engine_connect.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                        ctypes.c_size_t]
    lib.uccl_engine_accept.restype = ctypes.c_uint64
    lib.uccl_engine_accept.argtypes = [ctypes.c_void_p]
    lib.uccl_engine_reg.restype = ctypes.c_uint64
    lib.uccl_engine_reg.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                    ctypes.c_size_t, ctypes.c_int]
    lib.uccl_engine_advertise.restype = ctypes.c_int
    lib.uccl_engine_advertise.argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                          ctypes.c_uint64, ctypes.c_uint64,
                                          ctypes.c_void_p, ctypes.c_size_t]
    for fn in ("send", "recv"):
        f = getattr(lib, f"uccl_engine_{fn}")
        f.restype = ctypes.c_int
        f.argtypes = [ctypes.c_void_p, ctypes.c_uint64, ctypes.c_void_p,
                      ctypes.c_size_t, ctypes.c_int]
    for fn in ("write", "read"):
        f = getattr(lib, f"uccl_engine_{fn}")
        f.restype = ctypes.c_int
        f.argtypes = [ctypes.c_void_p, ctypes.c_uint64, ctypes.c_void_p,
                      ctypes.c_size_t, ctypes.c_int, ctypes.c_void_p,
                      ctypes.c_size_t]
    lib.uccl_engine_destroy.argtypes = [ctypes.c_void_p]

    a = lib.uccl_engine_create(-1, 1)
    b = lib.uccl_engine_create(-1, 1)
    assert a and b

    md = ctypes.create_string_buffer(256)
    n = lib.uccl_engine_metadata(b, md, 256)
    assert n > 0

    import threading

    got = {}

    def acc():
        got["cb"] = lib.uccl_engine_accept(b)

    t = threading.Thread(target=acc)
    t.start()
    ca = lib.uccl_engine_connect(a, md, n)
    t.join(timeout=30)
    assert ca and got["cb"]

    # send/recv
    src = (ctypes.c_ubyte * 1000)(*range(250)) ; dst = (ctypes.c_ubyte * 1000)()
    def rx():
        got["rc"] = lib.uccl_engine_recv(b, got["cb"], dst, 1000, -1)
    t = threading.Thread(target=rx)
    t.start()
    assert lib.uccl_engine_send(a, ca, src, 1000, -1) == 0
    t.join(timeout=30)
    assert got["rc"] == 0
    assert bytes(dst[:250]) == bytes(src[:250])

    # one-sided write into an advertised window
    win = (ctypes.c_ubyte * 4096)()
    mr = lib.uccl_engine_reg(b, win, 4096, -1)
    ad = ctypes.create_string_buffer(64)
    adn = lib.uccl_engine_advertise(b, mr, 0, 4096, ad, 64)
    assert adn > 0
    payload = (ctypes.c_ubyte * 4096)(*([7] * 4096))
    assert lib.uccl_engine_write(a, ca, payload, 4096, -1, ad, adn) == 0
    assert bytes(win[:16]) == b"\x07" * 16

    back = (ctypes.c_ubyte * 4096)()
    assert lib.uccl_engine_read(a, ca, back, 4096, -1, ad, adn) == 0
    assert bytes(back[:16]) == b"\x07" * 16

    lib.uccl_engine_destroy(a)
    lib.uccl_engine_destroy(b)
