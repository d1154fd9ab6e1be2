"""Python mirror of the C shared region (the L1 <-> L0' ABI).

Offsets are NOT hard-coded: they are read from the authoritative
``vgpu_region_layout_json`` export of libvgpu-hip.so, so the monitor can
never drift from the C struct (the reference keeps a hand-mirrored Go
struct instead, cmd/vGPUmonitor/cudevshr.go:15-58 — this is safer).

The monitor maps the region read-write and does word-sized reads/writes
only (usage snapshots; recent_kernel / utilization_switch / priority
feedback), which need no cross-process lock.
"""
from __future__ import annotations

import ctypes
import json
import mmap
import os
import struct
from dataclasses import dataclass
from pathlib import Path
from typing import List, Optional

_DEFAULT_LIB = Path(__file__).resolve().parent.parent / "csrc" / "libvgpu-hip.so"

_layout_cache: Optional[dict] = None

VGPU_MAGIC = 0x4D495655


def region_layout(lib_path: Optional[str] = None) -> dict:
    global _layout_cache
    if _layout_cache is None or lib_path is not None:
        lib = ctypes.CDLL(str(lib_path or _DEFAULT_LIB))
        buf = ctypes.create_string_buffer(4096)
        n = lib.vgpu_region_layout_json(buf, 4096)
        if n <= 0:
            raise RuntimeError("vgpu_region_layout_json failed")
        layout = json.loads(buf.value.decode())
        if lib_path is None:
            _layout_cache = layout
        return layout
    return _layout_cache


@dataclass
class ProcUsage:
    pid: int
    host_pid: int
    used_bytes: List[int]          # per device, .total
    monitor_used: List[int]


@dataclass
class RegionSnapshot:
    num_devices: int
    uuids: List[str]
    limit: List[int]
    sm_limit: List[int]
    procs: List[ProcUsage]
    utilization_switch: int
    recent_kernel: int
    priority: int
    oversubscribe: int

    def device_usage(self, dev: int) -> int:
        return sum(p.used_bytes[dev] for p in self.procs)


class SharedRegion:
    def __init__(self, path: str, lib_path: Optional[str] = None):
        self.path = path
        self.layout = region_layout(lib_path)
        size = self.layout["_size"]
        self._f = open(path, "r+b")
        self._mm = mmap.mmap(self._f.fileno(), size)

    def close(self):
        self._mm.close()
        self._f.close()

    # -- primitive accessors ------------------------------------------------
    def _u32(self, off: int) -> int:
        return struct.unpack_from("<I", self._mm, off)[0]

    def _i32(self, off: int) -> int:
        return struct.unpack_from("<i", self._mm, off)[0]

    def _u64(self, off: int) -> int:
        return struct.unpack_from("<Q", self._mm, off)[0]

    def _w_i32(self, off: int, val: int) -> None:
        struct.pack_into("<i", self._mm, off, val)

    @property
    def valid(self) -> bool:
        return self._u32(self.layout["magic"]) == VGPU_MAGIC

    # -- feedback writes (monitor -> interceptor) ---------------------------
    def set_recent_kernel(self, val: int) -> None:
        self._w_i32(self.layout["recent_kernel"], val)

    def get_recent_kernel(self) -> int:
        return self._i32(self.layout["recent_kernel"])

    def set_utilization_switch(self, val: int) -> None:
        self._w_i32(self.layout["utilization_switch"], val)

    def get_priority(self) -> int:
        return self._i32(self.layout["priority"])

    def set_host_pid(self, slot: int, host_pid: int) -> None:
        off = (self.layout["procs"] + slot * self.layout["_proc_slot_size"]
               + self.layout["_proc_host_pid"])
        self._w_i32(off, host_pid)

    def set_monitor_scale(self, dev: int, scale: float, now_ns: int) -> None:
        """Write the node-arbitrated throttle scale (fixed-point x1e6) and
        freshness timestamp; the limiter honors it while younger than
        2.5x the monitor interval (set_monitor_interval)."""
        struct.pack_into("<q", self._mm,
                         self.layout["monitor_scale_fp"] + 8 * dev,
                         int(scale * 1e6))
        struct.pack_into("<Q", self._mm,
                         self.layout["monitor_scale_ts_ns"], now_ns)

    def set_monitor_interval(self, interval_s: float) -> None:
        """Publish the monitor's feedback period so the limiter can size
        its freshness window (2.5x interval); without this a 5 s monitor
        would expire a 2 s window for 3 of every 5 seconds and the limiter
        would oscillate between arbitrated and local control."""
        struct.pack_into("<Q", self._mm, self.layout["monitor_interval_ns"],
                         int(interval_s * 1e9))

    def get_monitor_interval(self) -> float:
        return struct.unpack_from(
            "<Q", self._mm, self.layout["monitor_interval_ns"])[0] / 1e9

    # -- limiter introspection (debug/observability) ------------------------
    def get_token_fill_rate(self, dev: int) -> int:
        """tokens/s the limiter last applied (post-scale) — written by
        refill(); reveals which control branch (arbitrated vs local EMA)
        a container is actually running under."""
        return struct.unpack_from(
            "<q", self._mm, self.layout["token_fill_rate"] + 8 * dev)[0]

    def get_core_tokens(self, dev: int) -> int:
        return struct.unpack_from(
            "<q", self._mm, self.layout["core_tokens"] + 8 * dev)[0]

    def get_monitor_scale(self, dev: int) -> float:
        v = struct.unpack_from(
            "<q", self._mm, self.layout["monitor_scale_fp"] + 8 * dev)[0]
        return v / 1e6

    # -- snapshot ------------------------------------------------------------
    def snapshot(self) -> RegionSnapshot:
        L = self.layout
        nd = min(self._u64(L["num_devices"]), L["_max_devices"]) or L["_max_devices"]
        nd = int(nd)
        uuids = []
        for i in range(nd):
            off = L["uuids"] + i * L["_uuid_len"]
            raw = self._mm[off:off + L["_uuid_len"]]
            uuids.append(raw.split(b"\0", 1)[0].decode(errors="replace"))
        limit = [self._u64(L["limit"] + 8 * i) for i in range(nd)]
        sm_limit = [self._u64(L["sm_limit"] + 8 * i) for i in range(nd)]
        procs: List[ProcUsage] = []
        slot_size = L["_proc_slot_size"]
        devmem_size = L["_devmem_size"]
        total_off_in_devmem = 4 * 8  # context, module, buffer, offset, then total
        for s in range(L["_max_procs"]):
            base = L["procs"] + s * slot_size
            pid = self._i32(base + L["_proc_pid"])
            if pid == 0:
                continue
            used = []
            monitor_used = []
            for d in range(nd):
                u_off = base + L["_proc_used"] + d * devmem_size + total_off_in_devmem
                used.append(self._u64(u_off))
                m_off = base + L["_proc_monitor_used"] + 8 * d
                monitor_used.append(self._u64(m_off))
            procs.append(ProcUsage(
                pid=pid,
                host_pid=self._i32(base + L["_proc_host_pid"]),
                used_bytes=used,
                monitor_used=monitor_used,
            ))
        return RegionSnapshot(
            num_devices=nd,
            uuids=uuids,
            limit=limit,
            sm_limit=sm_limit,
            procs=procs,
            utilization_switch=self._i32(L["utilization_switch"]),
            recent_kernel=self._i32(L["recent_kernel"]),
            priority=self._i32(L["priority"]),
            oversubscribe=int(self._u64(L["oversubscribe"])),
        )
