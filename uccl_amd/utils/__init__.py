"""Diagnostics and topology helpers."""

from __future__ import annotations


def peer_matrix():
    """NxN matrix of hipDeviceCanAccessPeer (1 = xGMI peer-accessible)."""
    from uccl_amd import _load_native

    return _load_native(required=True).peer_matrix()


def topology_summary() -> str:
    import torch

    n = torch.cuda.device_count()
    lines = [f"{n} GPU(s)"]
    if n:
        m = peer_matrix()
        for a in range(n):
            row = " ".join("X" if m[a][b] else "." for b in range(n))
            lines.append(f"  gpu{a}: {row}  ({torch.cuda.get_device_name(a)})")
    return "\n".join(lines)
