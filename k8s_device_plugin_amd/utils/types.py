"""Shared types and annotation-protocol constants.

This is the MI355X analogue of the reference's shared layer
(/root/reference/pkg/util/types.go:26-122): the annotation keys, the bind
phases, and the request/assignment structs that the scheduler writes and the
device plugin consumes.  The *schema* (what each annotation carries and the
field order of the string codec) is kept compatible so operators migrating
from the reference find the same shapes; the key prefixes are our own
(``amd.io`` / ``vgpu.amd.com`` instead of ``4pd.io`` / ``hami.sh``).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

# ---------------------------------------------------------------------------
# Annotation keys (pod side).  Reference: pkg/util/types.go:26-35.
# ---------------------------------------------------------------------------
ASSIGNED_TIME_ANNO = "amd.io/vgpu-time"
ASSIGNED_IDS_ANNO = "amd.io/vgpu-ids-new"
ASSIGNED_NODE_ANNO = "amd.io/vgpu-node"
BIND_TIME_ANNO = "amd.io/bind-time"
BIND_PHASE_ANNO = "amd.io/bind-phase"

BIND_PHASE_ALLOCATING = "allocating"
BIND_PHASE_FAILED = "failed"
BIND_PHASE_SUCCESS = "success"

# Per-device-type request/support annotations.  Reference:
# pkg/device/nvidia/device.go:38-39 (hami.sh/vgpu-devices-to-allocate /
# -allocated).  Keyed by device type string.
IN_REQUEST_DEVICES: Dict[str, str] = {}
SUPPORT_DEVICES: Dict[str, str] = {}

# Node lock annotation.  Reference: pkg/util/nodelock/nodelock.go:18.
NODE_LOCK_ANNO = "amd.io/mutex.lock"
XGMI_ANNO = "amd.io/node-xgmi"
SCHEDULER_POLICY_ANNO = "amd.com/gpu-scheduler-policy"
NODE_LOCK_EXPIRE_SECONDS = 300.0  # 5 min auto-expiry (nodelock.go:96-103)

DEVICE_LIMIT = 100  # max devices per request (types.go:41)

# best-effort / restricted / guaranteed NUMA/topology policies (types.go:45-47)
BEST_EFFORT = "best-effort"
RESTRICTED = "restricted"
GUARANTEED = "guaranteed"


# ---------------------------------------------------------------------------
# Device inventory / request / assignment records.
# ---------------------------------------------------------------------------
@dataclass
class DeviceInfo:
    """One physical GPU as advertised in the node register annotation.

    Reference: pkg/api/device_register.go:13-23.  ``devmem`` is MiB,
    ``devcore`` is percent units (100 == one whole GPU's CUs).
    """

    id: str
    count: int          # how many vGPU slots this card is split into
    devmem: int         # MiB (MI355X: 294912 for a whole card, pre-scaling)
    devcore: int        # percent (100 per whole card, pre-scaling)
    type: str
    numa: int
    health: bool
    index: int = 0      # physical index on the node (not serialized)


@dataclass
class ContainerDevice:
    """One vGPU slice assigned to one container.

    Reference: pkg/util/types.go:85-91.
    """

    idx: int = 0        # device index within the node snapshot (not serialized)
    uuid: str = ""
    type: str = ""
    usedmem: int = 0    # MiB
    usedcores: int = 0  # percent


@dataclass
class ContainerDeviceRequest:
    """Decoded resource request of one container.

    Reference: pkg/util/types.go:93-99.  ``mem_percentage_req`` uses 101 as
    the "not set" sentinel, exactly like the reference
    (pkg/device/nvidia/device.go:138-155) so the fallback rules match.
    """

    nums: int = 0
    type: str = ""
    memreq: int = 0                # MiB; 0 = not set
    mem_percentage_req: int = 101  # 101 = not set
    coresreq: int = 0              # percent


# Aliases mirroring the Go shapes (types.go:101-108)
ContainerDevices = List[ContainerDevice]
ContainerDeviceRequests = Dict[str, ContainerDeviceRequest]
PodSingleDevice = List[ContainerDevices]          # per-container
PodDeviceRequests = List[ContainerDeviceRequests]  # per-container
PodDevices = Dict[str, PodSingleDevice]           # per device type


@dataclass
class DeviceUsage:
    """Mutable per-device usage snapshot used by the scheduler.

    Reference: pkg/util/types.go:110-122.
    """

    id: str
    index: int = 0
    used: int = 0
    count: int = 0
    usedmem: int = 0
    totalmem: int = 0
    totalcore: int = 0
    usedcores: int = 0
    numa: int = 0
    type: str = ""
    health: bool = True


# ---------------------------------------------------------------------------
# Minimal pod/node object model.
#
# We do not depend on a kubernetes client package (none is installed and the
# build is offline); components operate on these thin records which both the
# REST client and the in-memory fake produce from/into k8s JSON.
# ---------------------------------------------------------------------------
@dataclass
class ContainerSpec:
    name: str
    # resource limits/requests: resource name -> integer quantity
    limits: Dict[str, int] = field(default_factory=dict)
    requests: Dict[str, int] = field(default_factory=dict)
    env: Dict[str, str] = field(default_factory=dict)
    security_privileged: bool = False


@dataclass
class PodInfo:
    name: str
    namespace: str = "default"
    uid: str = ""
    annotations: Dict[str, str] = field(default_factory=dict)
    labels: Dict[str, str] = field(default_factory=dict)
    containers: List[ContainerSpec] = field(default_factory=list)
    node_name: str = ""
    scheduler_name: str = ""
    phase: str = "Pending"

    @staticmethod
    def from_k8s(obj: dict) -> "PodInfo":
        meta = obj.get("metadata", {}) or {}
        spec = obj.get("spec", {}) or {}
        status = obj.get("status", {}) or {}
        ctrs = []
        for c in spec.get("containers", []) or []:
            res = c.get("resources", {}) or {}
            sc = c.get("securityContext", {}) or {}
            ctrs.append(
                ContainerSpec(
                    name=c.get("name", ""),
                    limits={k: _parse_quantity(v) for k, v in (res.get("limits") or {}).items()},
                    requests={k: _parse_quantity(v) for k, v in (res.get("requests") or {}).items()},
                    env={e.get("name", ""): str(e.get("value", "")) for e in (c.get("env") or [])},
                    security_privileged=bool(sc.get("privileged", False)),
                )
            )
        return PodInfo(
            name=meta.get("name", ""),
            namespace=meta.get("namespace", "default"),
            uid=meta.get("uid", ""),
            annotations=dict(meta.get("annotations") or {}),
            labels=dict(meta.get("labels") or {}),
            containers=ctrs,
            node_name=spec.get("nodeName", "") or "",
            scheduler_name=spec.get("schedulerName", "") or "",
            phase=status.get("phase", "Pending") or "Pending",
        )


@dataclass
class NodeInfo:
    name: str
    annotations: Dict[str, str] = field(default_factory=dict)
    labels: Dict[str, str] = field(default_factory=dict)
    devices: List[DeviceInfo] = field(default_factory=list)

    @staticmethod
    def from_k8s(obj: dict) -> "NodeInfo":
        meta = obj.get("metadata", {}) or {}
        return NodeInfo(
            name=meta.get("name", ""),
            annotations=dict(meta.get("annotations") or {}),
            labels=dict(meta.get("labels") or {}),
        )


def _parse_quantity(v) -> int:
    """Parse a k8s resource quantity into an integer count/MiB.

    Device-plugin resources are plain integers; memory-style quantities may
    carry Ki/Mi/Gi or k/M/G suffixes.  Values are normalized the way the
    reference treats gpumem: plain number == MiB.
    """
    if isinstance(v, (int, float)):
        return int(v)
    s = str(v).strip()
    mult = 1
    suffixes = {
        "Ki": 1.0 / 1024,
        "Mi": 1,
        "Gi": 1024,
        "Ti": 1024 * 1024,
        "k": 1.0 / 1024,
        "M": 1,
        "G": 1024,
        "T": 1024 * 1024,
    }
    for suf, m in suffixes.items():
        if s.endswith(suf):
            return int(float(s[: -len(suf)]) * m)
    try:
        return int(float(s) * mult)
    except ValueError:
        return 0
