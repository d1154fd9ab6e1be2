"""Device plugin tests: KFD enumeration (fixture sysfs tree), CU-mask
allocator, fake-device fan-out, and the full kubelet gRPC path against a
stub kubelet socket — BASELINE config 1, the test the reference never had
(SURVEY.md §4).
"""
import json
import os
import threading
import time
from concurrent import futures
from pathlib import Path

import grpc
import pytest

from k8s_device_plugin_amd.device.amd import HANDSHAKE_ANNO, REGISTER_ANNO
from k8s_device_plugin_amd.ops.cumask import (
    CoreMaskAllocator,
    cus_for_percent,
    hsa_cu_mask_env,
    mask_to_ranges,
)
from k8s_device_plugin_amd.plugin.config import PluginConfig
from k8s_device_plugin_amd.plugin.health import device_healthy
from k8s_device_plugin_amd.plugin.kfd import enumerate_gpus
from k8s_device_plugin_amd.plugin.register import register_once
from k8s_device_plugin_amd.plugin.rm import ResourceManager
from k8s_device_plugin_amd.plugin.server import VGPUDevicePlugin
from k8s_device_plugin_amd.proto import deviceplugin as dp
from k8s_device_plugin_amd.utils.codec import (
    decode_node_devices,
    encode_pod_single_device,
)
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

GB = 1 << 30


def make_kfd_tree(root: Path, n_gpus=8, mem_bytes=288 * GB, cus=256):
    """Synthetic MI355X node: 1 CPU node + n GPU nodes with xGMI io_links."""
    nodes = root / "topology" / "nodes"
    pci = root / "pci"
    cpu = nodes / "0"
    cpu.mkdir(parents=True)
    (cpu / "properties").write_text("simd_count 0\ncpu_cores_count 96\n")
    for i in range(n_gpus):
        nd = nodes / str(i + 1)
        nd.mkdir(parents=True)
        loc = (0x0C + i) << 8
        props = [
            f"simd_count {cus * 4}",
            "simd_per_cu 4",
            f"unique_id {0xABC000 + i}",
            f"location_id {loc}",
            "domain 0",
            f"drm_render_minor {128 + i}",
            "gfx_target_version 90500",
        ]
        (nd / "properties").write_text("\n".join(props) + "\n")
        (nd / "gpu_id").write_text(str(10000 + i))
        mb = nd / "mem_banks" / "0"
        mb.mkdir(parents=True)
        (mb / "properties").write_text(
            f"heap_type 1\nsize_in_bytes {mem_bytes}\n")
        links = nd / "io_links"
        for j in range(n_gpus):
            if j == i:
                continue
            ld = links / str(j)
            ld.mkdir(parents=True)
            (ld / "properties").write_text(
                f"type 11\nnode_from {i + 1}\nnode_to {j + 1}\nweight 15\n")
        bdf = f"0000:{0x0c + i:02x}:00.0"
        bd = pci / bdf
        bd.mkdir(parents=True)
        (bd / "numa_node").write_text(str(i // (n_gpus // 2 or 1)))
    return root / "topology", pci


class TestKFDEnumeration:
    def test_mi355x_node(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path)
        gpus = enumerate_gpus(str(topo), str(pci))
        assert len(gpus) == 8
        g = gpus[0]
        assert g.cu_count == 256
        assert g.mem_bytes == 288 * GB
        assert g.uuid == f"GPU-{0xABC000:016x}"
        assert g.gfx_target == "gfx950"
        assert g.drm_card == 0
        assert g.numa_node == 0 and gpus[7].numa_node == 1
        assert "/dev/dri/renderD128" in g.device_paths
        # all peers xGMI-linked
        assert len(g.io_links) == 7
        assert all(t == 11 for t in g.io_links.values())

    def test_empty_root(self, tmp_path):
        assert enumerate_gpus(str(tmp_path), str(tmp_path)) == []


class TestCUMask:
    def test_cus_for_percent(self):
        assert cus_for_percent(10) == 26   # 10% of 256, rounded up
        assert cus_for_percent(25) == 64
        assert cus_for_percent(100) == 256
        assert cus_for_percent(0) == 0

    def test_whole_xcds_first(self):
        a = CoreMaskAllocator()
        m = a.alloc("gpu0", 25)  # 64 CUs = exactly 2 XCDs
        assert bin(m).count("1") == 64
        ranges = mask_to_ranges(m)
        assert ranges == [(0, 63)]  # XCD0 + XCD1, contiguous

    def test_disjoint_allocations(self):
        a = CoreMaskAllocator()
        masks = [a.alloc("gpu0", 10) for _ in range(10)]  # 10 x 26 CUs = 260 > 256
        assert all(m is not None for m in masks[:9])
        assert masks[9] is None  # over-committed
        combined = 0
        for m in masks[:9]:
            assert combined & m == 0
            combined |= m

    def test_free_then_realloc(self):
        a = CoreMaskAllocator()
        m1 = a.alloc("gpu0", 50)
        a.free("gpu0", m1)
        m2 = a.alloc("gpu0", 100)
        assert bin(m2).count("1") == 256

    def test_env_format(self):
        a = CoreMaskAllocator()
        m = a.alloc("gpu0", 25)
        env = hsa_cu_mask_env([(0, m)])
        assert env == "0:0-63"


class TestResourceManager:
    def test_fanout(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path, n_gpus=2)
        rm = ResourceManager(enumerate_gpus(str(topo), str(pci)), split_count=10)
        fakes = rm.fake_devices()
        assert len(fakes) == 20
        assert fakes[0].id.endswith("-0") and fakes[9].id.endswith("-9")
        assert ResourceManager.uuid_of_fake(fakes[3].id) == fakes[3].uuid

    def test_api_devices_scaling(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path, n_gpus=1)
        rm = ResourceManager(enumerate_gpus(str(topo), str(pci)),
                             split_count=4, memory_scaling=1.5)
        d = rm.api_devices()[0]
        assert d.count == 4
        assert d.devmem == int(288 * 1024 * 1.5)  # MiB, scaled (oversubscribe)
        assert d.devcore == 100

    def test_register_annotation_roundtrip(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path, n_gpus=2)
        rm = ResourceManager(enumerate_gpus(str(topo), str(pci)), split_count=10)
        client = FakeKubeClient()
        client.add_node(NodeInfo(name="node1"))
        register_once(client, "node1", rm)
        annos = client.get_node("node1").annotations
        assert annos[HANDSHAKE_ANNO].startswith("Reported ")
        devs = decode_node_devices(annos[REGISTER_ANNO])
        assert len(devs) == 2 and devs[0].devmem == 288 * 1024


class TestHealth:
    def test_ras_ue_marks_unhealthy(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path, n_gpus=1)
        gpus = enumerate_gpus(str(topo), str(pci))
        kfd_dev = tmp_path / "kfd"
        kfd_dev.write_text("")
        drm = tmp_path / "drm"
        ras = drm / "card0" / "device" / "ras"
        ras.mkdir(parents=True)
        (ras / "ue_count").write_text("0\n")
        assert device_healthy(gpus[0], str(drm), str(kfd_dev), str(topo))
        (ras / "ue_count").write_text("3\n")
        assert not device_healthy(gpus[0], str(drm), str(kfd_dev), str(topo))

    def test_missing_kfd_unhealthy(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path, n_gpus=1)
        gpus = enumerate_gpus(str(topo), str(pci))
        assert not device_healthy(gpus[0], str(tmp_path), str(tmp_path / "nokfd"), str(topo))


class StubKubelet:
    """Hosts the Registration service on a unix socket (BASELINE config 1)."""

    def __init__(self, socket_path):
        self.socket_path = socket_path
        self.registrations = []
        self.server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
        self.server.add_generic_rpc_handlers((dp.registration_service(self),))
        self.server.add_insecure_port(f"unix://{socket_path}")

    def Register(self, request, context):
        self.registrations.append(
            (request.version, request.endpoint, request.resource_name))
        return dp.Empty()


@pytest.fixture
def plugin_env(tmp_path):
    topo, pci = make_kfd_tree(tmp_path, n_gpus=2)
    gpus = enumerate_gpus(str(topo), str(pci))
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
    kubelet = StubKubelet(cfg.kubelet_socket)
    kubelet.server.start()
    yield plugin, kubelet, client, rm, cfg
    plugin.stop()
    kubelet.server.stop(grace=0)


class TestGRPCPath:
    def test_register_and_listandwatch(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        plugin.register_with_kubelet()
        assert kubelet.registrations == [("v1beta1", "amd-gpu.sock", "amd.com/gpu")]
        with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
            stub = dp.DevicePluginClient(ch)
            opts = stub.GetDevicePluginOptions(dp.Empty())
            assert not opts.pre_start_required
            stream = stub.ListAndWatch(dp.Empty())
            first = next(stream)
            assert len(first.devices) == 8  # 2 GPUs x 4 splits
            assert all(d.health == dp.HEALTHY for d in first.devices)
            assert first.devices[0].topology.nodes[0].ID == 0
            # flip health -> stream must re-send with Unhealthy
            rm.set_health(rm.gpus[0].uuid, False)
            plugin.notify_update()
            second = next(stream)
            unhealthy = [d for d in second.devices if d.health == dp.UNHEALTHY]
            assert len(unhealthy) == 4
            stream.cancel()

    def _bind_pod(self, client, rm, mem=73728, cores=25):
        uuid = rm.gpus[0].uuid
        devs = [[ContainerDevice(uuid=uuid, type="AMD", usedmem=mem, usedcores=cores)]]
        pod = PodInfo(
            name="p1", uid="uid-p1",
            containers=[ContainerSpec(name="main", limits={"amd.com/gpu": 1})],
            annotations={
                BIND_TIME_ANNO: "123",
                BIND_PHASE_ANNO: BIND_PHASE_ALLOCATING,
                ASSIGNED_NODE_ANNO: "node1",
                IN_REQUEST_DEVICES["AMD"]: encode_pod_single_device(devs),
            },
        )
        client.add_pod(pod)
        client.patch_node_annotations("node1", {NODE_LOCK_ANNO: "2026-01-01T00:00:00Z"})
        return pod, uuid

    def test_allocate_injects_enforcement(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        pod, uuid = self._bind_pod(client, rm)
        with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
            stub = dp.DevicePluginClient(ch)
            req = dp.AllocateRequest()
            req.container_requests.add(devicesIDs=[f"{uuid}-0"])
            resp = stub.Allocate(req)
        assert len(resp.container_responses) == 1
        envs = dict(resp.container_responses[0].envs)
        assert envs["VGPU_DEVICE_MEMORY_LIMIT_0"] == "73728m"
        assert envs["VGPU_DEVICE_CU_LIMIT"] == "25"
        assert envs["ROCR_VISIBLE_DEVICES"] == uuid
        assert envs["HSA_CU_MASK"] == "0:0-63"  # 25% -> 2 whole XCDs
        assert envs["VGPU_DEVICE_MEMORY_SHARED_CACHE"].endswith(".cache")
        mounts = {m.container_path: m for m in resp.container_responses[0].mounts}
        assert f"{cfg.hook_path}/vgpu/libvgpu-hip.so" in mounts
        assert "/etc/ld.so.preload" in mounts
        assert mounts["/etc/ld.so.preload"].read_only
        devs = {d.container_path for d in resp.container_responses[0].devices}
        assert "/dev/kfd" in devs and "/dev/dri/renderD128" in devs
        # annotation consumed + success + lock released
        stored = client.get_pod("p1")
        assert stored.annotations[BIND_PHASE_ANNO] == BIND_PHASE_SUCCESS
        assert NODE_LOCK_ANNO not in client.get_node("node1").annotations

    def test_allocate_disable_control_skips_preload(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        pod, uuid = self._bind_pod(client, rm)
        pod.containers[0].env["VGPU_DISABLE_CONTROL"] = "1"
        with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
            stub = dp.DevicePluginClient(ch)
            req = dp.AllocateRequest()
            req.container_requests.add(devicesIDs=[f"{uuid}-0"])
            resp = stub.Allocate(req)
        mounts = {m.container_path for m in resp.container_responses[0].mounts}
        assert "/etc/ld.so.preload" not in mounts

    def test_allocate_count_mismatch_fails_pod(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        pod, uuid = self._bind_pod(client, rm)
        with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
            stub = dp.DevicePluginClient(ch)
            req = dp.AllocateRequest()
            req.container_requests.add(devicesIDs=[f"{uuid}-0", f"{uuid}-1"])
            with pytest.raises(grpc.RpcError):
                stub.Allocate(req)
        assert client.get_pod("p1").annotations[BIND_PHASE_ANNO] == "failed"
        assert NODE_LOCK_ANNO not in client.get_node("node1").annotations

    def test_allocate_no_pending_pod(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
            stub = dp.DevicePluginClient(ch)
            req = dp.AllocateRequest()
            req.container_requests.add(devicesIDs=["x-0"])
            with pytest.raises(grpc.RpcError):
                stub.Allocate(req)

    def test_oversubscribe_env(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        cfg.device_memory_scaling = 1.5
        pod, uuid = self._bind_pod(client, rm, mem=409600)
        with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
            stub = dp.DevicePluginClient(ch)
            req = dp.AllocateRequest()
            req.container_requests.add(devicesIDs=[f"{uuid}-0"])
            resp = stub.Allocate(req)
        envs = dict(resp.container_responses[0].envs)
        assert envs["VGPU_OVERSUBSCRIBE"] == "true"
        assert envs["HSA_XNACK"] == "1"
        assert envs["VGPU_DEVICE_MEMORY_LIMIT_0"] == "409600m"


class TestComputePartition:
    """MI355X SPX/CPX partition modes (the MIG analog: amdgpu exposes
    current_compute_partition per drm card; in CPX each XCD is its own KFD
    node).  Partitioned cards advertise a distinct device type so gputype
    white/blacklists can target them."""

    def _drm_tree(self, root: Path, n, mode="CPX", mem_mode="NPS2"):
        drm = root / "drm"
        for i in range(n):
            d = drm / f"card{i}" / "device"
            d.mkdir(parents=True)
            (d / "current_compute_partition").write_text(mode + "\n")
            (d / "current_memory_partition").write_text(mem_mode + "\n")
        return drm

    def test_partition_mode_read(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path, n_gpus=2)
        drm = self._drm_tree(tmp_path, 2, "CPX", "NPS2")
        gpus = enumerate_gpus(str(topo), str(pci), str(drm))
        assert all(g.compute_partition == "CPX" for g in gpus)
        assert all(g.memory_partition == "NPS2" for g in gpus)

    def test_default_spx_when_absent(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path, n_gpus=1)
        gpus = enumerate_gpus(str(topo), str(pci), str(tmp_path / "nodrm"))
        assert gpus[0].compute_partition == "SPX"
        assert gpus[0].memory_partition == "NPS1"

    def test_partitioned_type_in_registration(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path, n_gpus=2)
        drm = self._drm_tree(tmp_path, 2, "CPX")
        rm = ResourceManager(enumerate_gpus(str(topo), str(pci), str(drm)),
                             split_count=4)
        devs = rm.api_devices()
        assert all(d.type == "AMD-Instinct-MI355X-CPX" for d in devs)

    def test_spx_type_unchanged(self, tmp_path):
        topo, pci = make_kfd_tree(tmp_path, n_gpus=1)
        rm = ResourceManager(enumerate_gpus(str(topo), str(pci),
                                            str(tmp_path / "nodrm")),
                             split_count=4)
        assert rm.api_devices()[0].type == "AMD-Instinct-MI355X"


class TestReconciliation:
    """Restart reconciliation: allocation records persist under
    <hook>/vgpu/containers/<uid>_<ctr>/vgpu.json and a fresh plugin adopts
    the CU masks (reference DCU RefreshContainerDevices,
    dcu/server.go:274-316)."""

    def _allocate(self, plugin, client, rm):
        uuid = rm.gpus[0].uuid
        devs = [[ContainerDevice(uuid=uuid, type="AMD", usedmem=73728,
                                 usedcores=25)]]
        pod = PodInfo(
            name="p1", uid="uid-p1",
            containers=[ContainerSpec(name="main", limits={"amd.com/gpu": 1})],
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
            stub.Allocate(req)
        return uuid

    def test_record_written_and_adopted_after_restart(self, plugin_env, tmp_path):
        plugin, kubelet, client, rm, cfg = plugin_env
        uuid = self._allocate(plugin, client, rm)
        rec = Path(cfg.hook_path) / "vgpu" / "containers" / "uid-p1_main" / "vgpu.json"
        assert rec.is_file()
        data = json.loads(rec.read_text())
        assert data["devices"][0]["uuid"] == uuid
        mask = int(data["devices"][0]["cu_mask"], 16)
        assert bin(mask).count("1") == 64  # 25% of 256

        # "restart": a fresh plugin instance over the same hook dir
        plugin2 = VGPUDevicePlugin(cfg, rm, client)
        assert plugin2.cumask.used_count(uuid) == 0
        plugin2.reconcile({"uid-p1"})
        assert plugin2.cumask.used_count(uuid) == 64
        assert ("uuid" and uuid) and plugin2.pod_masks["uid-p1"] == [(uuid, mask)]
        # idempotent
        plugin2.reconcile({"uid-p1"})
        assert plugin2.cumask.used_count(uuid) == 64

    def test_orphan_record_removed_and_mask_freed(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        uuid = self._allocate(plugin, client, rm)
        d = Path(cfg.hook_path) / "vgpu" / "containers" / "uid-p1_main"
        assert d.is_dir()
        assert plugin.cumask.used_count(uuid) == 64
        plugin.reconcile(set())  # pod gone
        assert not d.exists()
        assert plugin.cumask.used_count(uuid) == 0


class TestPreferredAllocation:
    """xGMI-aligned GetPreferredAllocation (live, unlike the reference's
    commented-out verb, server.go:270-285 / rm/allocate.go:44-121)."""

    def _req(self, avail, size, must=()):
        req = dp.PreferredAllocationRequest()
        req.container_requests.add(
            available_deviceIDs=avail, must_include_deviceIDs=list(must),
            allocation_size=size)
        return req

    def test_fractional_packs_one_gpu(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        u0, u1 = rm.gpus[0].uuid, rm.gpus[1].uuid
        avail = [f"{u0}-0", f"{u1}-0", f"{u0}-1", f"{u1}-1"]
        resp = plugin.GetPreferredAllocation(self._req(avail, 2), None)
        got = list(resp.container_responses[0].deviceIDs)
        assert len(got) == 2
        assert len({ResourceManager.uuid_of_fake(f) for f in got}) == 1

    def test_must_include_respected_and_packed(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        u0, u1 = rm.gpus[0].uuid, rm.gpus[1].uuid
        avail = [f"{u0}-1", f"{u1}-0", f"{u1}-1"]
        resp = plugin.GetPreferredAllocation(
            self._req(avail, 2, must=[f"{u1}-2"]), None)
        got = list(resp.container_responses[0].deviceIDs)
        assert got[0] == f"{u1}-2"
        # packs onto u1 rather than spreading to u0
        assert set(got) == {f"{u1}-2", f"{u1}-0"} or set(got) == {f"{u1}-2", f"{u1}-1"}

    def test_options_advertise_preferred(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        opts = plugin.GetDevicePluginOptions(dp.Empty(), None)
        assert opts.get_preferred_allocation_available


class TestMultiContainerAllocate:
    """Two GPU containers in one pod: each Allocate call consumes ONE
    container's assignment (reference util.go:216-271 erase semantics)."""

    def test_two_containers_two_calls(self, plugin_env):
        plugin, kubelet, client, rm, cfg = plugin_env
        u0, u1 = rm.gpus[0].uuid, rm.gpus[1].uuid
        devs = [
            [ContainerDevice(uuid=u0, type="AMD", usedmem=1024, usedcores=10)],
            [ContainerDevice(uuid=u1, type="AMD", usedmem=2048, usedcores=20)],
        ]
        pod = PodInfo(
            name="p2", uid="uid-p2",
            containers=[
                ContainerSpec(name="c0", limits={"amd.com/gpu": 1}),
                ContainerSpec(name="c1", limits={"amd.com/gpu": 1}),
            ],
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
            # kubelet allocates container by container (one request each)
            req = dp.AllocateRequest()
            req.container_requests.add(devicesIDs=[f"{u0}-0"])
            r1 = stub.Allocate(req)
            e1 = dict(r1.container_responses[0].envs)
            assert e1["VGPU_DEVICE_MEMORY_LIMIT_0"] == "1024m"
            # pod not yet fully consumed -> still allocating
            assert client.get_pod("p2").annotations[BIND_PHASE_ANNO] == BIND_PHASE_ALLOCATING
            req = dp.AllocateRequest()
            req.container_requests.add(devicesIDs=[f"{u1}-0"])
            r2 = stub.Allocate(req)
            e2 = dict(r2.container_responses[0].envs)
            assert e2["VGPU_DEVICE_MEMORY_LIMIT_0"] == "2048m"
            assert e2["ROCR_VISIBLE_DEVICES"] == u1
        # both consumed -> success + lock released
        assert client.get_pod("p2").annotations[BIND_PHASE_ANNO] == BIND_PHASE_SUCCESS
        assert NODE_LOCK_ANNO not in client.get_node("node1").annotations


class TestNodeConfigOverride:
    """Per-node JSON ConfigMap precedence: node JSON > CLI > defaults
    (reference vgpucfg.go:81-107)."""

    def test_matching_node_overridden(self, tmp_path):
        from k8s_device_plugin_amd.plugin.config import (
            PluginConfig,
            apply_node_config,
        )

        cfgfile = tmp_path / "config.json"
        cfgfile.write_text(json.dumps({"nodeconfig": [
            {"name": "other", "devicesplitcount": 2},
            {"name": "n1", "devicesplitcount": 4,
             "devicememoryscaling": 1.5, "devicecorescaling": 0.5},
        ]}))
        cfg = PluginConfig(node_name="n1", device_split_count=10,
                           config_file=str(cfgfile))
        out = apply_node_config(cfg)
        assert out.device_split_count == 4
        assert out.device_memory_scaling == 1.5
        assert out.device_cores_scaling == 0.5

    def test_non_matching_node_untouched(self, tmp_path):
        from k8s_device_plugin_amd.plugin.config import (
            PluginConfig,
            apply_node_config,
        )

        cfgfile = tmp_path / "config.json"
        cfgfile.write_text(json.dumps({"nodeconfig": [
            {"name": "other", "devicesplitcount": 2}]}))
        cfg = PluginConfig(node_name="n1", device_split_count=10,
                           config_file=str(cfgfile))
        assert apply_node_config(cfg).device_split_count == 10

    def test_bad_json_ignored(self, tmp_path):
        from k8s_device_plugin_amd.plugin.config import (
            PluginConfig,
            apply_node_config,
        )

        cfgfile = tmp_path / "config.json"
        cfgfile.write_text("{nope")
        cfg = PluginConfig(node_name="n1", config_file=str(cfgfile))
        assert apply_node_config(cfg).device_split_count == 10
