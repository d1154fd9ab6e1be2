"""CU-mask computation and per-device mask allocation for the 256-CU MI355X.

This is the hard-partition half of core limiting (SURVEY.md §7 hard part 2):
the plugin computes a disjoint CU set per container and exports it as
``HSA_CU_MASK`` so ROCr pins every queue of the container to those CUs —
the CDNA4-native equivalent of the reference DCU plugin's vdev ``cu_mask``
hex strings (corealloc.go:8-77), enforced by the runtime rather than by
intercepting launches.

MI355X-first layout policy: the chip is 8 XCDs x 32 CUs, and each XCD has
its own (non-coherent) 4 MiB L2 — so masks are packed XCD-first: a 25%
slice gets 2 whole XCDs (own L2s, minimal cross-tenant interference), and
only the remainder shares an XCD.  The first-fit bit allocator mirrors the
DCU allocator's behavior (init/add/alloc round-trips are table-tested like
corealloc_test.go:10-36).

HSA_CU_MASK syntax: "<dev>:<lo>-<hi>[,<lo>-<hi>...]" groups separated by
';' (ROCr runtime env spec).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

MI355X_CUS = 256
CUS_PER_XCD = 32
NUM_XCDS = 8


def cus_for_percent(percent: int, total_cus: int = MI355X_CUS) -> int:
    """CU count for a percent request, minimum 1 XCD-quarter granularity.

    Rounded up so "10%" on 256 CUs gives 26 CUs; a 0 percent request means
    "no core limit" and gets no mask.
    """
    if percent <= 0:
        return 0
    if percent >= 100:
        return total_cus
    n = (total_cus * percent + 99) // 100
    return max(1, n)


def mask_to_ranges(mask: int) -> List[Tuple[int, int]]:
    ranges: List[Tuple[int, int]] = []
    start = None
    for cu in range(MI355X_CUS + 1):
        if cu < MI355X_CUS and (mask >> cu) & 1:
            if start is None:
                start = cu
        else:
            if start is not None:
                ranges.append((start, cu - 1))
                start = None
    return ranges


def ranges_to_env(dev_index: int, mask: int) -> str:
    ranges = mask_to_ranges(mask)
    if not ranges:
        return ""
    return f"{dev_index}:" + ",".join(
        f"{lo}-{hi}" if lo != hi else str(lo) for lo, hi in ranges
    )


class CoreMaskAllocator:
    """First-fit XCD-aware CU allocation per physical device.

    Used by the plugin's Allocate to hand each container a disjoint CU set;
    released when the container's pod dies (reconciliation).  In-memory
    state, rebuilt from live pods on restart (the reference DCU plugin
    rebuilds from its vdev dirs, server.go:274-316).
    """

    def __init__(self, total_cus: int = MI355X_CUS):
        self.total_cus = total_cus
        self.used: Dict[str, int] = {}  # device uuid -> busy bitmask

    def alloc(self, device_uuid: str, percent: int) -> Optional[int]:
        """Allocate a CU bitmask for percent cores; None if percent<=0 or
        device fully booked for the request."""
        n = cus_for_percent(percent, self.total_cus)
        if n == 0:
            return None
        busy = self.used.get(device_uuid, 0)
        mask = 0
        remaining = n
        # pass 1: whole free XCDs (L2 isolation first)
        for x in range(NUM_XCDS):
            if remaining < CUS_PER_XCD:
                break
            xmask = ((1 << CUS_PER_XCD) - 1) << (x * CUS_PER_XCD)
            if busy & xmask == 0 and mask & xmask == 0:
                mask |= xmask
                remaining -= CUS_PER_XCD
        # pass 2: first-fit free CUs, lowest index first
        if remaining > 0:
            for cu in range(self.total_cus):
                if remaining == 0:
                    break
                bit = 1 << cu
                if (busy | mask) & bit:
                    continue
                mask |= bit
                remaining -= 1
        if remaining > 0:
            return None  # over-committed
        self.used[device_uuid] = busy | mask
        return mask

    def adopt(self, device_uuid: str, mask: int) -> bool:
        """Re-register a mask handed out before a plugin restart
        (reconciliation, the DCU RefreshContainerDevices analog,
        dcu/server.go:274-316).  False if it conflicts with live state."""
        busy = self.used.get(device_uuid, 0)
        if busy & mask:
            return False
        self.used[device_uuid] = busy | mask
        return True

    def free(self, device_uuid: str, mask: int) -> None:
        self.used[device_uuid] = self.used.get(device_uuid, 0) & ~mask

    def used_count(self, device_uuid: str) -> int:
        return bin(self.used.get(device_uuid, 0)).count("1")


def hsa_cu_mask_env(assignments: List[Tuple[int, int]]) -> str:
    """assignments: [(visible_device_index, mask)] -> HSA_CU_MASK value."""
    parts = [ranges_to_env(i, m) for i, m in assignments if m]
    return ";".join(p for p in parts if p)
