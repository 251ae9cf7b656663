"""P2P helper utilities.

Reference analogs: the interval-tree MR lookup in p2p/utils.py:115-160
(find which registered memory region covers a pointer range) and the
XferDesc/XferHandle descriptor (de)serialization of p2p/engine_api.cc
used in NIXL-style workflows (ship a descriptor out-of-band, transfer
against it later).
"""

from __future__ import annotations

import bisect
import json
from dataclasses import dataclass, field
from typing import List, Optional, Tuple


class MRMap:
    """Sorted-interval map from [base, base+len) address ranges to MR ids.

    bisect-based (O(log n) lookup); rejects overlapping registrations,
    which on this engine indicate a double-reg bug.
    """

    def __init__(self):
        self._bases: List[int] = []
        self._entries: List[Tuple[int, int, int]] = []  # (base, len, mr_id)

    def add(self, mr_id: int, base: int, length: int) -> None:
        if length <= 0:
            raise ValueError("empty MR")
        i = bisect.bisect_right(self._bases, base)
        if i > 0:
            pb, pl, _ = self._entries[i - 1]
            if pb + pl > base:
                raise ValueError("overlapping MR registration")
        if i < len(self._entries) and base + length > self._entries[i][0]:
            raise ValueError("overlapping MR registration")
        self._bases.insert(i, base)
        self._entries.insert(i, (base, length, mr_id))

    def remove(self, mr_id: int) -> bool:
        for i, (_, _, mid) in enumerate(self._entries):
            if mid == mr_id:
                del self._bases[i]
                del self._entries[i]
                return True
        return False

    def find(self, addr: int, length: int = 1) -> Optional[Tuple[int, int]]:
        """(mr_id, offset) of the MR fully covering [addr, addr+length),
        or None."""
        i = bisect.bisect_right(self._bases, addr)
        if i == 0:
            return None
        base, mlen, mr_id = self._entries[i - 1]
        if addr + length <= base + mlen:
            return mr_id, addr - base
        return None

    def find_tensor(self, tensor) -> Optional[Tuple[int, int]]:
        return self.find(tensor.data_ptr(),
                         tensor.numel() * tensor.element_size())

    def __len__(self):
        return len(self._entries)


@dataclass
class XferDesc:
    """Serializable transfer descriptor (reference XferDesc,
    p2p/engine_api.cc): which MR window a peer may read/write, shipped
    out-of-band (e.g. over torch.distributed) as a compact string."""

    mr_id: int
    offset: int
    bytes: int
    tag: int = 0
    meta: dict = field(default_factory=dict)

    def serialize(self) -> str:
        return json.dumps({"m": self.mr_id, "o": self.offset,
                           "b": self.bytes, "t": self.tag,
                           "x": self.meta}, separators=(",", ":"))

    @staticmethod
    def deserialize(s: str) -> "XferDesc":
        d = json.loads(s)
        return XferDesc(mr_id=d["m"], offset=d["o"], bytes=d["b"],
                        tag=d.get("t", 0), meta=d.get("x", {}))

    def split(self, chunk_bytes: int) -> List["XferDesc"]:
        """Chop into ≤chunk_bytes sub-descriptors (vectored transfers)."""
        out = []
        off = 0
        while off < self.bytes:
            n = min(chunk_bytes, self.bytes - off)
            out.append(XferDesc(self.mr_id, self.offset + off, n, self.tag,
                                dict(self.meta)))
            off += n
        return out
