"""xGMI topology-aware selection tests (MLU allocator analog)."""
import pytest

from k8s_device_plugin_amd.parallel.topology import (
    GPUTopology,
    pick_gpus,
    score_subset,
)
from k8s_device_plugin_amd.plugin.kfd import enumerate_gpus
from k8s_device_plugin_amd.utils.types import BEST_EFFORT, GUARANTEED, RESTRICTED
from tests.test_plugin import make_kfd_tree


def full_mesh_topo(tmp_path, n=8):
    topo_root, pci = make_kfd_tree(tmp_path, n_gpus=n)
    return GPUTopology.from_gpus(enumerate_gpus(str(topo_root), str(pci)))


def island_topo(tmp_path):
    """Two xGMI islands of 4: links only inside each half."""
    topo_root, pci = make_kfd_tree(tmp_path, n_gpus=8)
    gpus = enumerate_gpus(str(topo_root), str(pci))
    for g in gpus:
        island = g.index // 4
        g.io_links = {
            peer_node: 11
            for peer_node, t in g.io_links.items()
            if (peer_node - 1) // 4 == island
        }
    return GPUTopology.from_gpus(gpus)


class TestTopology:
    def test_full_mesh_adjacency(self, tmp_path):
        topo = full_mesh_topo(tmp_path)
        assert all(len(p) == 7 for p in topo.xgmi_peers.values())
        assert topo.is_clique([0, 3, 5, 7])
        assert score_subset(topo, [0, 1, 2, 3]) == 1.0

    def test_islands(self, tmp_path):
        topo = island_topo(tmp_path)
        assert topo.is_clique([0, 1, 2, 3])
        assert not topo.is_clique([0, 4])
        assert score_subset(topo, [0, 1, 4, 5]) == pytest.approx(2 / 6)

    def test_pick_prefers_connected(self, tmp_path):
        topo = island_topo(tmp_path)
        pick = pick_gpus(topo, [2, 3, 4, 5], 2, BEST_EFFORT)
        assert sorted(pick) in ([2, 3], [4, 5])  # either island pair, never cross

    def test_guaranteed_fails_across_islands(self, tmp_path):
        topo = island_topo(tmp_path)
        # only 2 left in island0 + 3 in island1: no 4-clique available
        assert pick_gpus(topo, [0, 1, 4, 5, 6], 4, GUARANTEED) is None
        assert pick_gpus(topo, [0, 1, 4, 5, 6], 3, GUARANTEED) == [4, 5, 6]

    def test_restricted_numa(self, tmp_path):
        topo = full_mesh_topo(tmp_path)  # numa 0: gpus 0-3, numa 1: gpus 4-7
        pick = pick_gpus(topo, [2, 3, 4, 5], 2, RESTRICTED)
        assert {topo.gpus[i].numa_node for i in pick} == {0} or \
               {topo.gpus[i].numa_node for i in pick} == {1}
        assert pick_gpus(topo, [3, 4], 2, RESTRICTED) is None

    def test_best_effort_never_fails(self, tmp_path):
        topo = island_topo(tmp_path)
        pick = pick_gpus(topo, list(range(8)), 6, BEST_EFFORT)
        assert len(pick) == 6
