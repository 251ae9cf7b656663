"""In-tree build driver for the uccl_amd native extension.

Compiles the HIP/C++ sources directly with hipcc for gfx950 (no hipify, no
CUDA compatibility layer) and links against the torch-ROCm libraries. The
resulting .so lives inside the package tree so it travels with repo
snapshots (gpurun) and is importable without installation.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
ROCM = Path(os.environ.get("ROCM_PATH", "/opt/rocm"))
HIPCC = str(ROCM / "bin" / "hipcc")
GPU_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

SOURCES = [
    CSRC / "collective" / "kernels.hip",
    CSRC / "collective" / "communicator.cpp",
    CSRC / "collective" / "pg_backend.cpp",
    CSRC / "p2p" / "endpoint.cpp",
    CSRC / "p2p" / "rccl_plane.cpp",
    CSRC / "transport" / "reliable.cpp",
    CSRC / "transport" / "udp_fabric.cpp",
    CSRC / "transport" / "verbs_fabric.cpp",
    CSRC / "ep" / "ep_kernels.hip",
    CSRC / "ep" / "ep_buffer.cpp",
    CSRC / "ep" / "ep_proxy.cpp",
    CSRC / "ukernel" / "ukernel.cpp",
    CSRC / "ukernel" / "uk_device.hip",
    CSRC / "p2p" / "compress.cpp",
    CSRC / "p2p" / "gpu_codec.hip",
    CSRC / "core" / "trace.cpp",
    CSRC / "bindings" / "module.cpp",
]

EXT_NAME = "_C" + (sysconfig.get_config_var("EXT_SUFFIX") or ".so")


def _torch_paths():
    import torch

    troot = Path(torch.__file__).parent
    return [troot / "include", troot / "include/torch/csrc/api/include"], troot / "lib"


def _stale(target: Path, deps) -> bool:
    if not target.exists():
        return True
    t = target.stat().st_mtime
    return any(d.stat().st_mtime > t for d in deps)


def build_plugin(verbose: bool = False) -> Path:
    """Build librccl-net-uccl.so (RCCL net plugin, pure sockets, no HIP)
    and the dlopen test harness."""
    plugdir = PKG_DIR / "lib"
    plugdir.mkdir(exist_ok=True)
    target = plugdir / "librccl-net-uccl.so"
    src = CSRC / "plugin" / "tcp_plugin.cpp"
    harness_src = CSRC / "plugin" / "plugin_test_main.cpp"
    harness = plugdir / "plugin_test"
    deps = [src, harness_src] + list((CSRC / "plugin").glob("*.h")) + [
        CSRC / "core" / "net.h", CSRC / "core" / "log.h",
        CSRC / "core" / "env.h", CSRC / "transport" / "reliable.h",
        CSRC / "transport" / "reliable.cpp"]
    capi = plugdir / "libuccl_p2p.so"
    capi_srcs = [CSRC / "p2p" / "c_api.cpp", CSRC / "p2p" / "endpoint.cpp",
                 CSRC / "p2p" / "rccl_plane.cpp",
                 CSRC / "transport" / "reliable.cpp",
                 CSRC / "transport" / "udp_fabric.cpp",
                 CSRC / "transport" / "verbs_fabric.cpp",
                 CSRC / "core" / "trace.cpp",
                 CSRC / "p2p" / "c_api.h", CSRC / "p2p" / "endpoint.h",
                 CSRC / "transport" / "reliable.h",
                 CSRC / "transport" / "fabric.h"]
    shim_check = plugdir / "libuccl_nccl.so"
    shim_deps = [CSRC / "nccl_shim" / "nccl_shim.cpp",
                 CSRC / "collective" / "kernels.hip",
                 CSRC / "collective" / "communicator.cpp"]
    if (not _stale(target, deps) and not _stale(harness, deps)
            and not _stale(capi, capi_srcs + deps)
            and not _stale(shim_check, shim_deps)):
        return target
    import subprocess as sp

    reliable = CSRC / "transport" / "reliable.cpp"
    udpfab = CSRC / "transport" / "udp_fabric.cpp"
    verbsfab = CSRC / "transport" / "verbs_fabric.cpp"
    tracecc = CSRC / "core" / "trace.cpp"
    for cmd in (
        # hipcc + UCCL_NET_HIP: NCCL_PTR_CUDA support (device MRs staged
        # through pinned bounce buffers). Loading on a GPU-less box still
        # works (amdhip64 resolves; hip calls only run for device MRs).
        [HIPCC, "-O2", "-g", "-std=c++17", "-fPIC", "-shared",
         f"--offload-arch={GPU_ARCH}", "-DUCCL_NET_HIP=1", str(src),
         str(reliable), str(udpfab), str(verbsfab), str(tracecc),
         "-o", str(target), "-pthread", "-ldl",
         f"-L{ROCM}/lib", "-lamdhip64", f"-Wl,-rpath,{ROCM}/lib"],
        # mock verbs provider: software-loopback RDMA for the CPU test
        # tier (drives the whole verbs fabric without a NIC)
        ["g++", "-O2", "-std=c++17", "-fPIC", "-shared",
         str(CSRC / "transport" / "mock_verbs_provider.cpp"),
         "-o", str(plugdir / "libuccl_verbs_mock.so"), "-pthread"],
        [HIPCC, "-O2", "-std=c++17", "-DUCCL_NET_HIP_TEST=1",
         f"--offload-arch={GPU_ARCH}", str(harness_src), "-o", str(harness),
         "-ldl", "-pthread", f"-L{ROCM}/lib", "-lamdhip64",
         f"-Wl,-rpath,{ROCM}/lib"],
    ):
        if verbose:
            print("[uccl_amd build]", " ".join(cmd), file=sys.stderr)
        r = sp.run(cmd, stdout=sp.PIPE, stderr=sp.STDOUT)
        if r.returncode != 0:
            raise RuntimeError(f"plugin build failed:\n{r.stdout.decode()}")
    # NCCL-compatible alias
    alias = plugdir / "libnccl-net-uccl.so"
    if not alias.exists():
        try:
            alias.symlink_to(target.name)
        except FileExistsError:
            pass
    # NCCL C-ABI drop-in over the collective engine (lite-collective role)
    shim = plugdir / "libuccl_nccl.so"
    shim_srcs = [CSRC / "nccl_shim" / "nccl_shim.cpp",
                 CSRC / "collective" / "kernels.hip",
                 CSRC / "collective" / "communicator.cpp"]
    shim_hdrs = list((CSRC / "collective").glob("*.h")) + [
        CSRC / "device" / "primitives.h"]
    if _stale(shim, shim_srcs + shim_hdrs + deps):
        cmd = [HIPCC, "-O3", "-std=c++17", "-fPIC", "-shared",
               f"--offload-arch={GPU_ARCH}"] + [str(x) for x in shim_srcs] + [
               "-o", str(shim), "-pthread", f"-L{ROCM}/lib", "-lamdhip64",
               f"-Wl,-rpath,{ROCM}/lib"]
        if verbose:
            print("[uccl_amd build]", " ".join(cmd), file=sys.stderr)
        r = sp.run(cmd, stdout=sp.PIPE, stderr=sp.STDOUT)
        if r.returncode != 0:
            raise RuntimeError(f"nccl shim build failed:\n{r.stdout.decode()}")

    # real-ibverbs provider: only when rdma-core headers are installed
    # (the dev/CI image has none; the mock provider covers the fabric
    # logic there and this file stays compile-checked by inspection)
    if Path("/usr/include/infiniband/verbs.h").exists():
        cmd = ["gcc", "-O2", "-fPIC", "-shared",
               str(CSRC / "transport" / "verbs_adapter.c"),
               "-o", str(plugdir / "libuccl_verbs_ib.so"), "-libverbs"]
        if verbose:
            print("[uccl_amd build]", " ".join(cmd), file=sys.stderr)
        r = sp.run(cmd, stdout=sp.PIPE, stderr=sp.STDOUT)
        if r.returncode != 0:
            raise RuntimeError(f"verbs adapter build failed:\n{r.stdout.decode()}")

    # flat C API lib for NIXL-style integrators (reference: p2p/uccl_engine.h)
    if _stale(capi, capi_srcs + deps):
        cc_srcs = [x for x in capi_srcs if x.suffix == ".cpp"]
        cmd = [HIPCC, "-O2", "-std=c++17", "-fPIC", "-shared",
               f"--offload-arch={GPU_ARCH}"] + [str(x) for x in cc_srcs] + [
               "-o", str(capi), "-pthread", "-ldl", f"-L{ROCM}/lib",
               "-lamdhip64", f"-Wl,-rpath,{ROCM}/lib"]
        if verbose:
            print("[uccl_amd build]", " ".join(cmd), file=sys.stderr)
        r = sp.run(cmd, stdout=sp.PIPE, stderr=sp.STDOUT)
        if r.returncode != 0:
            raise RuntimeError(f"c_api build failed:\n{r.stdout.decode()}")
    return target


def build(verbose: bool = False, force: bool = False) -> Path:
    """Compile the extension in-tree. Returns the path to the built .so."""
    target = PKG_DIR / EXT_NAME
    incs, libdir = _torch_paths()
    headers = list(CSRC.rglob("*.h"))
    sources = [s for s in SOURCES if s.exists()]
    if not force and not _stale(target, sources + headers):
        return target

    objdir = PKG_DIR / ".build"
    objdir.mkdir(exist_ok=True)

    py_inc = sysconfig.get_paths()["include"]
    common = [
        f"--offload-arch={GPU_ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DUSE_ROCM=1",
        "-DTORCH_EXTENSION_NAME=_C",
        "-D_GLIBCXX_USE_CXX11_ABI=1",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
        f"-I{py_inc}",
    ] + [f"-I{i}" for i in incs]

    objs = []
    procs = []
    for src in sources:
        obj = objdir / (src.stem + ".o")
        objs.append(obj)
        if not force and not _stale(obj, [src] + headers):
            continue
        cmd = [HIPCC, "-c", str(src), "-o", str(obj)] + common
        if verbose:
            print("[uccl_amd build]", " ".join(cmd), file=sys.stderr)
        procs.append((src, subprocess.Popen(cmd, stdout=subprocess.PIPE,
                                            stderr=subprocess.STDOUT)))
    failed = False
    for src, p in procs:
        out, _ = p.communicate()
        if p.returncode != 0:
            failed = True
            print(f"[uccl_amd build] FAILED {src}:\n{out.decode()}",
                  file=sys.stderr)
        elif verbose and out:
            print(out.decode(), file=sys.stderr)
    if failed:
        raise RuntimeError("uccl_amd native build failed")

    link = (
        [HIPCC, "-shared", "-o", str(target)]
        + [str(o) for o in objs]
        + [
            f"-L{libdir}",
            "-ltorch",
            "-ltorch_cpu",
            "-ltorch_hip",
            "-ltorch_python",
            "-lc10",
            "-lc10_hip",
            "-lamdhip64",
            "-lz",
            f"-Wl,-rpath,{libdir}",
            f"-L{ROCM}/lib",
            f"-Wl,-rpath,{ROCM}/lib",
        ]
    )
    if verbose:
        print("[uccl_amd build]", " ".join(link), file=sys.stderr)
    r = subprocess.run(link, stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    if r.returncode != 0:
        raise RuntimeError(f"uccl_amd link failed:\n{r.stdout.decode()}")
    build_plugin(verbose=verbose)
    return target


if __name__ == "__main__":
    build(verbose=True, force="--force" in sys.argv)
    print("built:", PKG_DIR / EXT_NAME)
