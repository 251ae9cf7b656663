"""Python surface of the software multipath reliable transport."""
from __future__ import annotations


def __getattr__(name):
    if name in ("TransportEndpoint", "TransportStats"):
        from uccl_amd import _load_native

        return getattr(_load_native(required=True), name)
    raise AttributeError(name)
