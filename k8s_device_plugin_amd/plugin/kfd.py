"""MI355X enumeration from KFD sysfs.

No cgo, no vendor CLI parsing: everything the plugin needs is in
``/sys/class/kfd/kfd/topology/nodes/*`` (the reference's DCU plugin shells
out to hy-smi/hdmcli and parses with Sscanf — server.go:50-175; the NVIDIA
path uses NVML.  KFD sysfs is the MI355X-native source):

- ``properties``: simd_count / simd_per_cu -> CU count (MI355X: 1024/4=256),
  unique_id -> stable UUID, location_id+domain -> PCI BDF, drm_render_minor
  -> /dev/dri node mapping, gfx_target_version.
- ``mem_banks/*/properties``: size_in_bytes -> HBM capacity (288 GB).
- NUMA: /sys/bus/pci/devices/<bdf>/numa_node (no hwloc needed — the
  reference itself reads sysfs on the NVIDIA path, rm/nvml_devices.go:134).

The sysfs roots are constructor parameters so tests run against fixture
trees (the cndev-mock testing pattern).
"""
from __future__ import annotations

import glob
import logging
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional

log = logging.getLogger(__name__)

KFD_TOPOLOGY = "/sys/class/kfd/kfd/topology"
PCI_DEVICES = "/sys/bus/pci/devices"
DRM_CLASS = "/sys/class/drm"
KFD_DEV = "/dev/kfd"


@dataclass
class PhysicalGPU:
    index: int              # enumeration order (stable: by node id)
    node_id: int            # KFD topology node number
    gpu_id: int             # KFD gpu_id
    uuid: str               # "GPU-<unique_id hex>"
    cu_count: int           # 256 on MI355X
    mem_bytes: int          # HBM3E bank size (288 GB)
    numa_node: int
    pci_bdf: str            # "0000:0c:00.0"
    drm_render_minor: int   # /dev/dri/renderD<minor>
    gfx_target: str         # e.g. "gfx950"
    io_links: Dict[int, int] = field(default_factory=dict)  # node_to -> type
    # MI300+/MI355X compute/memory partitioning (the MIG analog: in CPX each
    # XCD is its own KFD node; amdgpu exposes the mode per drm card)
    compute_partition: str = "SPX"   # SPX / DPX / QPX / CPX
    memory_partition: str = "NPS1"   # NPS1 / NPS2 / NPS4
    # DRM card index resolved via the device's PCI BDF symlink
    # (/sys/bus/pci/devices/<bdf>/drm/card*); -1 if unresolvable.  Card
    # numbering is NOT guaranteed to be render_minor-128 on hosts with
    # other DRM devices — the arithmetic is only a fallback.
    drm_card_no: int = -1
    # CPX bookkeeping: partitions of one physical card share unique_id;
    # partition_index disambiguates (0..7), parent_uuid groups them
    partition_index: int = 0
    partition_count: int = 1     # siblings sharing the physical card
    parent_uuid: str = ""        # == uuid when unpartitioned

    @property
    def drm_card(self) -> int:
        if self.drm_card_no >= 0:
            return self.drm_card_no
        return self.drm_render_minor - 128

    @property
    def device_paths(self) -> List[str]:
        return [
            KFD_DEV,
            f"/dev/dri/card{self.drm_card}",
            f"/dev/dri/renderD{self.drm_render_minor}",
        ]


def _read_properties(path: str) -> Dict[str, int]:
    props: Dict[str, int] = {}
    try:
        with open(path) as f:
            for line in f:
                parts = line.split()
                if len(parts) == 2:
                    try:
                        props[parts[0]] = int(parts[1])
                    except ValueError:
                        pass
    except OSError:
        pass
    return props


def _read_int(path: str, default: int = -1) -> int:
    try:
        with open(path) as f:
            return int(f.read().strip(), 0)
    except (OSError, ValueError):
        return default


def _read_str(path: str, default: str = "") -> str:
    try:
        with open(path) as f:
            return f.read().strip()
    except OSError:
        return default


def _resolve_drm_card(pci_root: str, bdf: str) -> int:
    """DRM card index for a PCI device, from its sysfs drm/ directory —
    authoritative, unlike render_minor-128 arithmetic (other DRM devices
    on the host shift card numbering)."""
    for card in glob.glob(os.path.join(pci_root, bdf, "drm", "card*")):
        name = os.path.basename(card)
        if name.startswith("card") and name[4:].isdigit():
            return int(name[4:])
    return -1


def enumerate_gpus(
    topology_root: str = KFD_TOPOLOGY, pci_root: str = PCI_DEVICES,
    drm_root: str = DRM_CLASS,
) -> List[PhysicalGPU]:
    gpus: List[PhysicalGPU] = []
    nodes_dir = os.path.join(topology_root, "nodes")
    if not os.path.isdir(nodes_dir):
        log.warning("KFD topology not present at %s", nodes_dir)
        return gpus
    for node_path in sorted(
        glob.glob(os.path.join(nodes_dir, "*")),
        key=lambda p: int(os.path.basename(p)) if os.path.basename(p).isdigit() else 1 << 30,
    ):
        node_name = os.path.basename(node_path)
        if not node_name.isdigit():
            continue
        props = _read_properties(os.path.join(node_path, "properties"))
        simd_count = props.get("simd_count", 0)
        if simd_count <= 0:
            continue  # CPU node
        simd_per_cu = props.get("simd_per_cu", 4) or 4
        cu_count = simd_count // simd_per_cu
        gpu_id = _read_int(os.path.join(node_path, "gpu_id"), 0)
        unique_id = props.get("unique_id", 0)
        uuid = f"GPU-{unique_id:016x}" if unique_id else f"GPU-kfd-{gpu_id}"
        location = props.get("location_id", 0)
        domain = props.get("domain", 0)
        bdf = f"{domain:04x}:{(location >> 8) & 0xff:02x}:{(location >> 3) & 0x1f:02x}.{location & 0x7}"
        mem_bytes = 0
        for bank in glob.glob(os.path.join(node_path, "mem_banks", "*", "properties")):
            bprops = _read_properties(bank)
            # heap_type 1/2 = FB public/private (device HBM)
            if bprops.get("heap_type", 0) in (1, 2):
                mem_bytes += bprops.get("size_in_bytes", 0)
        numa = _read_int(os.path.join(pci_root, bdf, "numa_node"), -1)
        if numa < 0:
            numa = 0
        gfx_ver = props.get("gfx_target_version", 0)
        # 90500 -> gfx950 encoding: major*10000 + minor*100 + step
        gfx = f"gfx{gfx_ver // 10000}{(gfx_ver // 100) % 100}{gfx_ver % 100:x}" if gfx_ver else ""
        io_links: Dict[int, int] = {}
        for link in glob.glob(os.path.join(node_path, "io_links", "*", "properties")):
            lprops = _read_properties(link)
            if "node_to" in lprops:
                io_links[lprops["node_to"]] = lprops.get("type", 0)
        render_minor = props.get("drm_render_minor", 128 + len(gpus))
        card_no = _resolve_drm_card(pci_root, bdf)
        if card_no < 0:
            card_no = render_minor - 128  # fallback arithmetic
        card_dev = os.path.join(drm_root, f"card{card_no}", "device")
        compute_part = _read_str(
            os.path.join(card_dev, "current_compute_partition"), "SPX") or "SPX"
        memory_part = _read_str(
            os.path.join(card_dev, "current_memory_partition"), "NPS1") or "NPS1"
        gpus.append(
            PhysicalGPU(
                index=len(gpus),
                node_id=int(node_name),
                gpu_id=gpu_id,
                uuid=uuid,
                cu_count=cu_count,
                mem_bytes=mem_bytes,
                numa_node=numa,
                pci_bdf=bdf,
                drm_render_minor=render_minor,
                gfx_target=gfx,
                io_links=io_links,
                compute_partition=compute_part,
                memory_partition=memory_part,
                drm_card_no=_resolve_drm_card(pci_root, bdf),
            )
        )
    _disambiguate_partitions(gpus)
    return gpus


def _disambiguate_partitions(gpus: List[PhysicalGPU]) -> None:
    """CPX: the 8 XCD partitions of one card share amdgpu's unique_id, so
    the raw UUIDs collide.  Suffix each sibling with its partition index
    (GPU-<id>.<k>) — the MIG-instance-UUID analog — and, when siblings all
    report the SAME memory banks (CPX+NPS1: one shared HBM view per
    partition), divide the advertised capacity so the card's memory is not
    counted 8x."""
    by_uid: Dict[str, List[PhysicalGPU]] = {}
    for g in gpus:
        by_uid.setdefault(g.uuid, []).append(g)
    for uid, group in by_uid.items():
        if len(group) == 1:
            g = group[0]
            g.parent_uuid = g.uuid
            continue
        shared_view = len({g.mem_bytes for g in group}) == 1
        for k, g in enumerate(sorted(group, key=lambda g: g.node_id)):
            g.parent_uuid = uid
            g.partition_index = k
            g.partition_count = len(group)
            g.uuid = f"{uid}.{k}"
            if shared_view and g.mem_bytes:
                g.mem_bytes //= len(group)


def kfd_healthy(kfd_dev: str = KFD_DEV) -> bool:
    """Health = /dev/kfd openable (reference DCU simpleHealthCheck,
    dcu/server.go:225-234); per-device RAS checks layer on top."""
    try:
        fd = os.open(kfd_dev, os.O_RDONLY)
        os.close(fd)
        return True
    except OSError:
        return False
