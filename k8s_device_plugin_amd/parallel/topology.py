"""xGMI / NUMA topology-aware placement.

The MI355X analogue of the reference's interconnect-aware allocators
(MLU-Link ring allocator, pkg/device-plugin/mlu/allocator/{board,spider}.go,
and the dormant NVLink aligned-alloc, rm/allocate.go:27-64):
KFD publishes per-GPU ``io_links`` (type 11 = xGMI, each MI355X has 7
point-to-point links at ~153 GB/s), so multi-GPU picks should land on
xGMI-connected cliques and co-NUMA groups.

Policies mirror the MLU allocator's modes (types.go:45-47):
  best-effort  prefer the best-connected subset, never fail;
  restricted   require same-NUMA;
  guaranteed   require a fully xGMI-connected clique.
"""
from __future__ import annotations

import itertools
import logging
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Set, Tuple

from ..utils.types import BEST_EFFORT, GUARANTEED, RESTRICTED
from ..plugin.kfd import PhysicalGPU

log = logging.getLogger(__name__)

XGMI_LINK_TYPE = 11


@dataclass
class GPUTopology:
    gpus: List[PhysicalGPU]
    # adjacency by gpu index: peers reachable over xGMI
    xgmi_peers: Dict[int, Set[int]]

    @staticmethod
    def from_gpus(gpus: Sequence[PhysicalGPU]) -> "GPUTopology":
        node_to_index = {g.node_id: g.index for g in gpus}
        adj: Dict[int, Set[int]] = {g.index: set() for g in gpus}
        for g in gpus:
            for node_to, link_type in g.io_links.items():
                if link_type == XGMI_LINK_TYPE and node_to in node_to_index:
                    adj[g.index].add(node_to_index[node_to])
        return GPUTopology(gpus=list(gpus), xgmi_peers=adj)

    def xgmi_degree(self, subset: Sequence[int]) -> int:
        """Number of direct xGMI edges inside the subset."""
        s = set(subset)
        return sum(1 for a in s for b in self.xgmi_peers[a] if b in s and a < b)

    def is_clique(self, subset: Sequence[int]) -> bool:
        s = set(subset)
        return all((s - {a}) <= self.xgmi_peers[a] for a in s)

    def same_numa(self, subset: Sequence[int]) -> bool:
        return len({self.gpus[i].numa_node for i in subset}) <= 1


def pick_gpus(
    topo: GPUTopology,
    candidates: Sequence[int],
    count: int,
    policy: str = BEST_EFFORT,
) -> Optional[List[int]]:
    """Choose ``count`` GPUs from candidate indices maximizing xGMI
    connectivity, then NUMA locality.

    Exhaustive for practical sizes (<= 8 GPUs per node); returns None when
    the policy cannot be satisfied.
    """
    cands = list(candidates)
    if count <= 0 or count > len(cands):
        return None
    if count == 1:
        return [cands[0]]
    best: Optional[Tuple[int, int, List[int]]] = None  # (-edges, numa_spread)
    for combo in itertools.combinations(cands, count):
        if policy == GUARANTEED and not topo.is_clique(combo):
            continue
        if policy == RESTRICTED and not topo.same_numa(combo):
            continue
        edges = topo.xgmi_degree(combo)
        numa_spread = len({topo.gpus[i].numa_node for i in combo})
        key = (-edges, numa_spread, list(combo))
        if best is None or key < best:
            best = key
    if best is None:
        return None
    return best[2]


def score_subset(topo: GPUTopology, subset: Sequence[int]) -> float:
    """Connectivity score in [0,1]: achieved xGMI edges / max possible."""
    n = len(subset)
    if n <= 1:
        return 1.0
    max_edges = n * (n - 1) // 2
    return topo.xgmi_degree(subset) / max_edges
