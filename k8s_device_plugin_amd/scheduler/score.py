"""Device fitting and node scoring — pure functions over in-memory state.

Behavioral spec ported from /root/reference/pkg/scheduler/score.go:86-226:

- devices are sorted by (numa, free slot count) ascending and walked from the
  BACK (most free slots within the highest NUMA domain first);
- NUMA binding: when the pod asserts numa-bind and the walk crosses a NUMA
  boundary, the partial multi-GPU pick is thrown away and restarted so all
  chosen GPUs share one NUMA node (score.go:100-105);
- memory request resolution: absolute MiB wins; otherwise percentage of the
  device's total (101 == unset sentinel);
- coresreq > 100 is invalid; coresreq == 100 means exclusive (rejects a card
  with any user); coresreq == 0 cannot land on a core-full card;
- node score = sum over containers of total/free + idle-device bonus, and the
  HIGHEST score wins => bin-packing onto the busiest fitting node.

MI355X note: "cores" are CU percent of the 256-CU chip; the plugin registers
devcore=100 per card so 10 => 26 CUs worth (rounded up to CU granularity by
the enforcement layer, see ops/cumask.py).
"""
from __future__ import annotations

import itertools
import logging
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..device import get_devices
from ..utils.types import (
    BEST_EFFORT,
    GUARANTEED,
    RESTRICTED,
    SCHEDULER_POLICY_ANNO,
    ContainerDevice,
    ContainerDeviceRequest,
    ContainerDeviceRequests,
    DeviceUsage,
    PodDeviceRequests,
    PodDevices,
    PodInfo,
)

log = logging.getLogger(__name__)


@dataclass
class NodeUsage:
    devices: List[DeviceUsage] = field(default_factory=list)
    # uuid -> xGMI peer uuids (empty when the node has not advertised it)
    xgmi: Dict[str, List[str]] = field(default_factory=dict)


@dataclass
class NodeScore:
    node_id: str
    devices: PodDevices = field(default_factory=dict)
    score: float = 0.0


def check_type(
    annos: Dict[str, str], d: DeviceUsage, req: ContainerDeviceRequest
) -> Tuple[bool, bool]:
    """(fits type, numa assert) — score.go:71-84."""
    # general vendor-word containment check (AMD -> AMD-Instinct-MI355X)
    if req.type not in d.type:
        return False, False
    for dev in get_devices().values():
        found, passes, numa_assert = dev.check_type(annos, d, req)
        if found:
            return passes, numa_assert
    return False, False


def _xgmi_select(
    eligible: List[Tuple[int, DeviceUsage, int]],
    nums: int,
    policy: str,
    xgmi: Dict[str, List[str]],
) -> Optional[List[Tuple[int, DeviceUsage, int]]]:
    """Pick ``nums`` of the eligible devices maximizing xGMI edges.

    MI355X-native multi-GPU placement (the MLU ring-allocator analog,
    reference mlu/allocator/spider.go policies): ``guaranteed`` requires a
    full xGMI clique, ``restricted`` one NUMA node, ``best-effort`` takes
    the most-connected subset.  Ties keep the greedy (busiest-first) order.
    """
    peer = {u: set(p) for u, p in xgmi.items()}

    def edges(combo):
        uuids = [e[1].id for e in combo]
        return sum(1 for i in range(len(uuids)) for j in range(i + 1, len(uuids))
                   if uuids[j] in peer.get(uuids[i], ()))

    best = None
    max_edges = nums * (nums - 1) // 2
    for order, combo in enumerate(itertools.combinations(eligible, nums)):
        if policy == GUARANTEED and edges(combo) != max_edges:
            continue
        if policy == RESTRICTED and len({e[1].numa for e in combo}) > 1:
            continue
        key = (-edges(combo), order)
        if best is None or key < best[0]:
            best = (key, list(combo))
    return best[1] if best else None


def fit_in_certain_device(
    node: NodeUsage,
    request: ContainerDeviceRequest,
    annos: Dict[str, str],
) -> Tuple[bool, Dict[str, List[ContainerDevice]]]:
    """score.go:86-157 (+ xGMI-aware multi-GPU subset selection)."""
    nums = request.nums
    policy = annos.get(SCHEDULER_POLICY_ANNO, BEST_EFFORT)
    # topology mode: gather every eligible device, then pick the subset;
    # otherwise stop at the first fit exactly like the reference
    topo_mode = nums > 1 and bool(node.xgmi)
    prevnuma = -1
    eligible: List[Tuple[int, DeviceUsage, int]] = []  # (idx, dev, memreq)
    tmp_devs: Dict[str, List[ContainerDevice]] = {}

    def emit(selection):
        tmp_devs[request.type] = [
            ContainerDevice(idx=i, uuid=d.id, type=request.type,
                            usedmem=memreq, usedcores=request.coresreq)
            for i, d, memreq in selection
        ]
        return True, tmp_devs

    for i in range(len(node.devices) - 1, -1, -1):
        d = node.devices[i]
        found, numa_assert = check_type(annos, d, request)
        if not found:
            continue
        if numa_assert and prevnuma != d.numa:
            # domain boundary: if the previous NUMA domain already holds
            # enough devices, select within it
            if len(eligible) >= nums:
                break
            prevnuma = d.numa
            eligible = []
        if d.count <= d.used:
            continue
        if request.coresreq > 100:
            log.error("core limit can't exceed 100")
            return False, tmp_devs
        memreq = 0
        if request.memreq > 0:
            memreq = request.memreq
        if request.mem_percentage_req != 101 and request.memreq == 0:
            memreq = d.totalmem * request.mem_percentage_req // 100
        if d.totalmem - d.usedmem < memreq:
            continue
        if d.totalcore - d.usedcores < request.coresreq:
            continue
        # exclusive request on a shared card
        if d.totalcore == 100 and request.coresreq == 100 and d.used > 0:
            continue
        # a core=0 job can't land on an already core-full card
        if d.totalcore != 0 and d.usedcores == d.totalcore and request.coresreq == 0:
            continue
        eligible.append((i, d, memreq))
        if not topo_mode and len(eligible) == nums:
            return emit(eligible)

    if len(eligible) < nums:
        return False, tmp_devs
    if topo_mode:
        selection = _xgmi_select(eligible, nums, policy, node.xgmi)
        if selection is None:
            return False, tmp_devs  # policy unsatisfiable on this node
        return emit(selection)
    return emit(eligible[:nums])


def fit_in_devices(
    node: NodeUsage,
    requests: ContainerDeviceRequests,
    annos: Dict[str, str],
    devinput: PodDevices,
) -> Tuple[bool, float]:
    """One container's requests against one node; mutates node usage and
    appends this container's picks to devinput (score.go:159-190)."""
    total = 0
    free = 0
    sums = sum(int(k.nums) for k in requests.values())
    for k in requests.values():
        if int(k.nums) > len(node.devices):
            return False, 0.0
        node.devices.sort(key=lambda d: (d.numa, d.count - d.used))
        fit, tmp_devs = fit_in_certain_device(node, k, annos)
        if not fit:
            return False, 0.0
        picks = tmp_devs.get(k.type, [])
        for val in picks:
            d = node.devices[val.idx]
            total += d.count
            free += d.count - d.used
            d.used += 1
            d.usedcores += val.usedcores
            d.usedmem += val.usedmem
        devinput.setdefault(k.type, []).append(picks)
    score = (total / free if free else 0.0) + (len(node.devices) - sums)
    return True, score


def calc_score(
    nodes: Dict[str, NodeUsage],
    nums: PodDeviceRequests,
    annos: Dict[str, str],
) -> List[NodeScore]:
    """Score every node that fits the whole pod (score.go:192-226).

    Deviation from the reference (documented): containers with no GPU request
    get an explicit empty entry in every device type's per-container list so
    annotation container indices always equal pod container indices.  The
    reference's alignment is implicit and breaks for leading non-GPU
    containers (score.go:205-210).
    """
    res: List[NodeScore] = []
    for node_id, node in nodes.items():
        ns = NodeScore(node_id=node_id)
        fits = True
        for ctr_requests in nums:
            sums = sum(int(k.nums) for k in ctr_requests.values())
            if sums == 0:
                for t in ns.devices:
                    ns.devices[t].append([])
                continue
            fit, score = fit_in_devices(node, ctr_requests, annos, ns.devices)
            if fit:
                ns.score += score
            else:
                fits = False
                break
        if fits and ns.devices:
            res.append(ns)
    return res


def pod_device_requests(pod: PodInfo) -> PodDeviceRequests:
    """Per-container request extraction (reference pkg/k8sutil/pod.go:26-41)."""
    out: PodDeviceRequests = []
    for ctr in pod.containers:
        reqs: ContainerDeviceRequests = {}
        for dev in get_devices().values():
            r = dev.generate_resource_requests(ctr)
            if r.nums > 0:
                reqs[r.type] = r
        out.append(reqs)
    return out
