"""uccl_amd — an MI355X-native GPU communication framework.

Capabilities (mirroring the uccl-project/uccl reference, re-designed for
CDNA4 + xGMI):
  - uccl_amd.collective : xGMI intra-node collective engine (hand-written
    gfx950 HIP kernels: LL packet / one-shot fullmesh / two-shot RS+AG)
  - uccl_amd.p2p        : NIXL-style point-to-point transfer engine
  - uccl_amd.ep         : DeepEP-compatible expert-parallel communication
  - uccl_amd.transport  : software multipath reliable transport (chunking,
    path spraying, SACK selective repeat, Timely/Swift/paced-EQDS CC)
  - uccl_amd.ukernel    : chunk-graph planner / lowering / spray executor
    (relay spraying over xGMI paths, cost-model algorithm choice)
  - uccl_amd.p2p        : also lossless float compression
    (send_compressed / send_object) and MR/XferDesc helpers
  - observability       : chrome-trace recorder (UCCL_TRACE=1), latency
    percentiles in transport/p2p stats
"""

from __future__ import annotations


__version__ = "0.1.0"

_ext = None
_IMPORT_ERROR: Exception | None = None


def _load_native(required: bool | None = None):
    """Import the native extension, building it on demand.

    On a GPU machine the native engine is mandatory — a silent CPU/eager
    fallback would invalidate every benchmark — so failure raises. On
    CPU-only machines (CI) callers may probe with required=False.
    """
    global _ext, _IMPORT_ERROR
    if _ext is not None:
        return _ext
    import importlib

    import torch  # noqa: F401  (the extension links against libtorch)

    try:
        _ext = importlib.import_module("uccl_amd._C")
        return _ext
    except ImportError as e:
        _IMPORT_ERROR = e
    try:
        from uccl_amd._build import build

        build()
        _ext = importlib.import_module("uccl_amd._C")
        return _ext
    except Exception as e:  # noqa: BLE001
        _IMPORT_ERROR = e
    if required is None:
        import torch

        required = torch.cuda.is_available()
    if required:
        raise RuntimeError(
            "uccl_amd native extension is required on a GPU machine but "
            f"could not be loaded: {_IMPORT_ERROR}"
        )
    return None


def native_available() -> bool:
    return _load_native(required=False) is not None
