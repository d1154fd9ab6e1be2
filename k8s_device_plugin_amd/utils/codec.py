"""String codec for the node/pod annotation wire protocol.

This is the contract both the scheduler and the device plugin parse; the
format matches the reference byte-for-byte so docs/develop/protocol.md of the
reference applies unchanged:

- node register annotation (one entry per physical GPU, ':'-terminated):
  ``uuid,count,devmem,devcore,type,numa,health:...``
  (reference EncodeNodeDevices, /root/reference/pkg/util/util.go:111-118)

- container devices (one entry per assigned vGPU slice, ':'-terminated):
  ``uuid,type,usedmem,usedcores:...``
  (EncodeContainerDevices, util.go:120-128)

- pod devices: container strings joined, each pod-single-device group
  terminated by ';' (EncodePodSingleDevice/EncodePodDevices, util.go:142-160).
"""
from __future__ import annotations

from typing import Dict, List

from .types import (
    ContainerDevice,
    ContainerDevices,
    DeviceInfo,
    PodDevices,
    PodSingleDevice,
)


class CodecError(ValueError):
    pass


def encode_node_devices(devices: List[DeviceInfo]) -> str:
    out = []
    for d in devices:
        out.append(
            f"{d.id},{d.count},{d.devmem},{d.devcore},{d.type},{d.numa},"
            f"{'true' if d.health else 'false'}:"
        )
    return "".join(out)


def decode_node_devices(s: str) -> List[DeviceInfo]:
    # Reference behavior (util.go:78-109): a string without ':' is an error;
    # entries without ',' (e.g. the trailing empty piece) are skipped; a
    # 7-field entry is required.
    if ":" not in s:
        raise CodecError("node annotation does not decode: missing ':'")
    ret: List[DeviceInfo] = []
    for piece in s.split(":"):
        if "," not in piece:
            continue
        items = piece.split(",")
        if len(items) != 7:
            raise CodecError(f"node annotation entry has {len(items)} fields, want 7")
        ret.append(
            DeviceInfo(
                id=items[0],
                count=_to_int(items[1]),
                devmem=_to_int(items[2]),
                devcore=_to_int(items[3]),
                type=items[4],
                numa=_to_int(items[5]),
                health=items[6].strip().lower() == "true",
            )
        )
    return ret


def encode_container_devices(cd: ContainerDevices) -> str:
    return "".join(f"{d.uuid},{d.type},{d.usedmem},{d.usedcores}:" for d in cd)


def decode_container_devices(s: str) -> ContainerDevices:
    if not s:
        return []
    out: ContainerDevices = []
    for piece in s.split(":"):
        if "," not in piece:
            continue
        items = piece.split(",")
        if len(items) < 4:
            raise CodecError("pod annotation format error: information missing")
        out.append(
            ContainerDevice(
                uuid=items[0],
                type=items[1],
                usedmem=_to_int(items[2]),
                usedcores=_to_int(items[3]),
            )
        )
    return out


def encode_pod_single_device(pd: PodSingleDevice) -> str:
    # One container-devices string per container, ';'-terminated per
    # container.  NOTE: the reference snapshot appends ';' once per pod
    # (util.go:142-150), which merges all containers into one group and
    # mis-allocates multi-container pods; later HAMi fixed it to
    # per-container ';' — we implement the fixed format.
    return "".join(encode_container_devices(c) + ";" for c in pd)


def encode_pod_devices(checklist: Dict[str, str], pd: PodDevices) -> Dict[str, str]:
    """checklist maps device type -> annotation key (IN_REQUEST/SUPPORT)."""
    return {checklist[t]: encode_pod_single_device(sd) for t, sd in pd.items()}


def decode_pod_devices(checklist: Dict[str, str], annos: Dict[str, str]) -> PodDevices:
    if not annos:
        return {}
    pd: PodDevices = {}
    for dev_type, anno_key in checklist.items():
        s = annos.get(anno_key)
        if s is None:
            continue
        parts = s.split(";")
        if parts and parts[-1] == "":
            parts.pop()  # trailing ';' terminator, not an empty container
        pd[dev_type] = [decode_container_devices(part) for part in parts]
    return pd


def _to_int(s: str) -> int:
    try:
        return int(s)
    except ValueError:
        return 0


# ---------------------------------------------------------------------------
# xGMI adjacency wire codec (MI355X-native addition: lets the scheduler make
# topology-aware multi-GPU picks; no reference analog — the NVIDIA aligned
# allocator is node-local and dormant, rm/allocate.go:44-64)
# ---------------------------------------------------------------------------
def encode_node_xgmi(adj: "Dict[str, List[str]]") -> str:
    """{uuid: [peer uuids]} -> "uuidA:uuidB|uuidC;uuidD:..;" """
    parts = []
    for uuid in sorted(adj):
        parts.append(f"{uuid}:{'|'.join(sorted(adj[uuid]))}")
    return ";".join(parts) + (";" if parts else "")


def decode_node_xgmi(s: str) -> "Dict[str, List[str]]":
    out: Dict[str, List[str]] = {}
    for part in s.split(";"):
        part = part.strip()
        if not part:
            continue
        uuid, _, peers = part.partition(":")
        out[uuid] = [p for p in peers.split("|") if p]
    return out
