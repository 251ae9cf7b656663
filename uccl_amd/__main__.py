"""Command-line entry: `python -m uccl_amd [info|env|trace-demo]`."""

from __future__ import annotations

import sys


def info():
    import torch

    import uccl_amd

    native = uccl_amd.native_available()
    print(f"uccl_amd {uccl_amd.__version__}")
    print(f"  torch        : {torch.__version__}")
    print(f"  hip          : {torch.version.hip}")
    print(f"  gpu          : {torch.cuda.is_available()} "
          f"({torch.cuda.device_count()} visible)" if native else
          f"  gpu          : {torch.cuda.is_available()}")
    print(f"  native ext   : {'loaded' if native else 'NOT built'}")
    if native and torch.cuda.is_available():
        from uccl_amd.utils import topology_summary

        print(topology_summary())
    libs = []
    from pathlib import Path

    libdir = Path(uccl_amd.__file__).parent / "lib"
    if libdir.exists():
        libs = sorted(p.name for p in libdir.glob("*.so*"))
    print(f"  drop-in libs : {', '.join(libs) if libs else '(none built)'}")


def env():
    import os

    groups = {
        "core": ["UCCL_LOG_LEVEL", "UCCL_TRACE", "UCCL_SOCKET_IFNAME"],
        "collective": ["UCCL_SYM_HEAP_MB", "UCCL_SYM_USER_MB",
                       "UCCL_LL_THRESHOLD", "UCCL_ONESHOT_THRESHOLD",
                       "UCCL_WORLD1_STAGED", "UCCL_BLOCKING_WAIT",
                       "UCCL_ENGINE_STATS"],
        "p2p": ["UCCL_P2P_TRANSPORT", "UCCL_P2P_ENABLE_IPC",
                "UCCL_P2P_COMPRESS_STRATEGY"],
        "transport": ["UCCL_TP_CC", "UCCL_TP_EQDS_MBPS", "UCCL_TP_CWND_MAX",
                      "UCCL_TP_RWND_KB", "UCCL_TP_DUPACK_THRES",
                      "UCCL_TP_RTO_US", "UCCL_TP_RTO_ABORT",
                      "UCCL_TP_PACE_MBPS", "UCCL_TP_LOSS_PCT",
                      "UCCL_TP_ACK_LOSS_PCT"],
        "plugin": ["UCCL_NET_TRANSPORT", "UCCL_NET_PATHS", "UCCL_NET_CHUNK",
                   "UCCL_NET_DEBUG"],
        "ep": ["UCCL_EP_FORCE_PROXY", "UCCL_EP_PROXY_PATHS",
               "UCCL_EP_PROXY_CHUNK"],
    }
    for grp, keys in groups.items():
        print(f"[{grp}]")
        for k in keys:
            v = os.environ.get(k)
            print(f"  {k} = {v if v is not None else '(default)'}")


def main():
    cmd = sys.argv[1] if len(sys.argv) > 1 else "info"
    if cmd == "info":
        info()
    elif cmd == "env":
        env()
    else:
        print(__doc__)
        sys.exit(2)


if __name__ == "__main__":
    main()
