"""CPX compute-partition support (the MI355X hard-isolation path).

KFD ignores per-queue CU masks on multi-XCD gfx9, so the hard partition
on MI355X is amdgpu CPX mode: 8 XCD-GPUs per card, each its own KFD node
and render node.  These tests drive the whole chain on fixture sysfs
trees: enumeration (UUID disambiguation, shared-memory-view division),
advertising (distinct MI355X-CPX type, per-partition memory), Allocate
(only the partition's render node is mounted -> isolation by
construction), and the sysfs mode switch.  Reference analogs: MIG device
maps (rm/device_map.go:37-118) + the DCU vdev cu_mask contract
(hygon/dcu/corealloc.go:8-77).
"""
import grpc
import pytest

from k8s_device_plugin_amd.plugin import partition
from k8s_device_plugin_amd.plugin.config import PluginConfig
from k8s_device_plugin_amd.plugin.kfd import enumerate_gpus
from k8s_device_plugin_amd.plugin.rm import ResourceManager
from k8s_device_plugin_amd.plugin.server import VGPUDevicePlugin
from k8s_device_plugin_amd.proto import deviceplugin as dp
from k8s_device_plugin_amd.utils.codec import encode_pod_single_device
from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
from k8s_device_plugin_amd.utils.types import (
    ASSIGNED_NODE_ANNO,
    BIND_PHASE_ALLOCATING,
    BIND_PHASE_ANNO,
    BIND_PHASE_SUCCESS,
    BIND_TIME_ANNO,
    IN_REQUEST_DEVICES,
    NODE_LOCK_ANNO,
    ContainerDevice,
    ContainerSpec,
    NodeInfo,
    PodInfo,
)

from test_plugin import StubKubelet

GB = 1 << 30


def make_cpx_tree(root, partitions=8, mem_bytes=288 * GB):
    """One physical MI355X in CPX: 8 KFD nodes sharing unique_id and PCI
    BDF, each 32 CUs with its own render node; amdgpu reports the mode in
    the (single, per-card) current_compute_partition file."""
    nodes = root / "topology" / "nodes"
    pci = root / "pci"
    drm = root / "drm"
    cpu = nodes / "0"
    cpu.mkdir(parents=True)
    (cpu / "properties").write_text("simd_count 0\ncpu_cores_count 96\n")
    bdf = "0000:0c:00.0"
    for k in range(partitions):
        nd = nodes / str(k + 1)
        nd.mkdir(parents=True)
        props = [
            "simd_count 128",          # 32 CUs x 4 SIMD
            "simd_per_cu 4",
            f"unique_id {0xABC123}",   # SAME for every partition
            f"location_id {0x0C << 8}",
            "domain 0",
            f"drm_render_minor {128 + k}",
            "gfx_target_version 90500",
        ]
        (nd / "properties").write_text("\n".join(props) + "\n")
        (nd / "gpu_id").write_text(str(20000 + k))
        mb = nd / "mem_banks" / "0"
        mb.mkdir(parents=True)
        # CPX+NPS1: every partition sees the whole HBM
        (mb / "properties").write_text(
            f"heap_type 1\nsize_in_bytes {mem_bytes}\n")
    bd = pci / bdf
    (bd / "drm").mkdir(parents=True)
    (bd / "numa_node").write_text("0")
    (bd / "drm" / "card0").mkdir()
    card_dev = drm / "card0" / "device"
    card_dev.mkdir(parents=True)
    (card_dev / "current_compute_partition").write_text("CPX")
    return root / "topology", pci, drm


class TestCPXEnumeration:
    def test_partitions_disambiguated(self, tmp_path):
        topo, pci, drm = make_cpx_tree(tmp_path)
        gpus = enumerate_gpus(str(topo), str(pci), str(drm))
        assert len(gpus) == 8
        uuids = [g.uuid for g in gpus]
        assert len(set(uuids)) == 8, "partition UUIDs must not collide"
        base = f"GPU-{0xABC123:016x}"
        assert uuids == [f"{base}.{k}" for k in range(8)]
        for k, g in enumerate(gpus):
            assert g.cu_count == 32
            assert g.partition_index == k
            assert g.partition_count == 8
            assert g.parent_uuid == base
            assert g.compute_partition == "CPX"
            assert f"/dev/dri/renderD{128 + k}" in g.device_paths

    def test_shared_memory_view_divided(self, tmp_path):
        """CPX+NPS1 reports the full 288 GB on every partition — the
        advertised capacity must be per-partition, not 8x the card."""
        topo, pci, drm = make_cpx_tree(tmp_path)
        gpus = enumerate_gpus(str(topo), str(pci), str(drm))
        assert all(g.mem_bytes == 288 * GB // 8 for g in gpus)

    def test_spx_unaffected(self, tmp_path):
        from test_plugin import make_kfd_tree

        topo, pci = make_kfd_tree(tmp_path, n_gpus=2)
        gpus = enumerate_gpus(str(topo), str(pci))
        assert [g.partition_count for g in gpus] == [1, 1]
        assert all(g.parent_uuid == g.uuid for g in gpus)
        assert all("." not in g.uuid for g in gpus)


class TestCPXAdvertising:
    def test_distinct_type_and_memory(self, tmp_path):
        topo, pci, drm = make_cpx_tree(tmp_path)
        gpus = enumerate_gpus(str(topo), str(pci), str(drm))
        rm = ResourceManager(gpus, split_count=4)
        infos = rm.api_devices()
        assert len(infos) == 8
        for info in infos:
            assert info.type == "AMD-Instinct-MI355X-CPX"
            assert info.devmem == (288 * GB // 8) >> 20
            assert info.count == 4

    def test_fake_device_roundtrip_with_dotted_uuid(self, tmp_path):
        topo, pci, drm = make_cpx_tree(tmp_path)
        gpus = enumerate_gpus(str(topo), str(pci), str(drm))
        rm = ResourceManager(gpus, split_count=4)
        fakes = rm.fake_devices()
        assert len(fakes) == 32
        for f in fakes:
            assert rm.uuid_of_fake(f.id) == f.uuid
            assert rm.by_uuid(f.uuid) is not None


class TestCPXAllocate:
    def test_only_partition_nodes_mounted(self, tmp_path):
        """Hard isolation by construction: a pod allocated partition 3
        gets /dev/kfd + partition 3's render node and nothing from the
        other 7 XCDs."""
        topo, pci, drm = make_cpx_tree(tmp_path)
        gpus = enumerate_gpus(str(topo), str(pci), str(drm))
        rm = ResourceManager(gpus, split_count=4)
        client = FakeKubeClient()
        client.add_node(NodeInfo(name="node1"))
        cfg = PluginConfig(
            node_name="node1",
            hook_path=str(tmp_path / "hook"),
            plugin_socket_dir=str(tmp_path),
            kubelet_socket=str(tmp_path / "kubelet.sock"),
        )
        plugin = VGPUDevicePlugin(cfg, rm, client)
        plugin.serve()
        try:
            uuid = gpus[3].uuid
            devs = [[ContainerDevice(uuid=uuid, type="AMD",
                                     usedmem=36864, usedcores=100)]]
            pod = PodInfo(
                name="p1", uid="uid-p1",
                containers=[ContainerSpec(name="main",
                                          limits={"amd.com/gpu": 1})],
                annotations={
                    BIND_TIME_ANNO: "123",
                    BIND_PHASE_ANNO: BIND_PHASE_ALLOCATING,
                    ASSIGNED_NODE_ANNO: "node1",
                    IN_REQUEST_DEVICES["AMD"]: encode_pod_single_device(devs),
                },
            )
            client.add_pod(pod)
            client.patch_node_annotations(
                "node1", {NODE_LOCK_ANNO: "2026-01-01T00:00:00Z"})
            with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
                stub = dp.DevicePluginClient(ch)
                req = dp.AllocateRequest()
                req.container_requests.add(devicesIDs=[f"{uuid}-0"])
                resp = stub.Allocate(req)
            ctr = resp.container_responses[0]
            devpaths = {d.container_path for d in ctr.devices}
            assert "/dev/kfd" in devpaths
            assert "/dev/dri/renderD131" in devpaths  # partition 3
            for k in range(8):
                if k == 3:
                    continue
                assert f"/dev/dri/renderD{128 + k}" not in devpaths
            envs = dict(ctr.envs)
            assert envs["ROCR_VISIBLE_DEVICES"] == uuid
            # whole partition allocated (100 cores of a 32-CU device):
            # no HSA_CU_MASK needed, hardware does the fencing
            assert "HSA_CU_MASK" not in envs
            assert client.get_pod("p1").annotations[BIND_PHASE_ANNO] == \
                BIND_PHASE_SUCCESS
        finally:
            plugin.stop()


class TestPartitionSwitch:
    def test_apply_mode_writes_sysfs(self, tmp_path):
        card = tmp_path / "card0" / "device"
        card.mkdir(parents=True)
        f = card / "current_compute_partition"
        f.write_text("SPX")
        assert partition.apply_mode("CPX", str(tmp_path))
        assert f.read_text() == "CPX"
        assert partition.apply_mode("SPX", str(tmp_path))
        assert f.read_text() == "SPX"

    def test_keep_is_noop(self, tmp_path):
        assert partition.apply_mode("keep", str(tmp_path))

    def test_no_cards_fails(self, tmp_path):
        assert not partition.apply_mode("CPX", str(tmp_path))

    def test_invalid_mode_raises(self, tmp_path):
        card = tmp_path / "card0" / "device"
        card.mkdir(parents=True)
        (card / "current_compute_partition").write_text("SPX")
        with pytest.raises(ValueError):
            partition.write_mode(
                str(card / "current_compute_partition"), "XPX")


class TestConfigFlag:
    def test_compute_partition_flag(self):
        from k8s_device_plugin_amd.plugin.config import parse_args

        cfg = parse_args(["--node-name", "n1", "--compute-partition", "CPX",
                          "--config-file", "/nonexistent"])
        assert cfg.compute_partition == "CPX"

    def test_node_json_override(self, tmp_path):
        import json

        from k8s_device_plugin_amd.plugin.config import (
            PluginConfig,
            apply_node_config,
        )

        cfgfile = tmp_path / "config.json"
        cfgfile.write_text(json.dumps({
            "nodeconfig": [{"name": "n1", "computepartition": "CPX"}]}))
        cfg = PluginConfig(node_name="n1", config_file=str(cfgfile))
        assert apply_node_config(cfg).compute_partition == "CPX"


class TestNPSPartitionedMemory:
    def test_distinct_banks_not_divided(self, tmp_path):
        """NPS4/NPS8: partitions report their OWN memory banks (different
        sizes in this fixture) — the shared-view division must NOT fire."""
        import shutil

        topo, pci, drm = make_cpx_tree(tmp_path)
        # rewrite each partition's bank to a distinct size
        for k in range(8):
            nd = tmp_path / "topology" / "nodes" / str(k + 1)
            (nd / "mem_banks" / "0" / "properties").write_text(
                f"heap_type 1\nsize_in_bytes {(36 + k) * GB}\n")
        gpus = enumerate_gpus(str(topo), str(pci), str(drm))
        assert len(gpus) == 8
        # UUIDs still disambiguated, memory untouched
        assert len({g.uuid for g in gpus}) == 8
        assert sorted(g.mem_bytes for g in gpus) == \
            [(36 + k) * GB for k in range(8)]


class TestPartitionSwitchErrors:
    def test_unwritable_path_fails_cleanly(self, tmp_path):
        """gpurun-style containers mount /sys read-only: the write must
        fail cleanly (logged, False) and never raise.  Simulated with a
        directory at the file path (root ignores mode bits, so chmod
        cannot model EROFS)."""
        card = tmp_path / "card0" / "device"
        card.mkdir(parents=True)
        f = card / "current_compute_partition"
        f.mkdir()  # open(.., "w") -> IsADirectoryError (an OSError)
        assert partition.write_mode(str(f), "CPX") is False

    def test_write_already_in_mode_is_noop(self, tmp_path):
        card = tmp_path / "card0" / "device"
        card.mkdir(parents=True)
        f = card / "current_compute_partition"
        f.write_text("CPX")
        assert partition.write_mode(str(f), "CPX")
