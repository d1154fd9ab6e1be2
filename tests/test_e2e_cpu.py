"""Full-stack CPU integration: the complete pod lifecycle across all three
protocols (SURVEY.md §1 data flows) with zero hardware —

  plugin registers node annos -> scheduler ingests -> webhook mutates pod ->
  /filter picks + pre-patches -> /bind locks + allocating -> kubelet
  Allocate (gRPC over unix socket) injects env/mounts -> a PLT-linked HIP
  consumer under LD_PRELOAD enforces the injected quota in the shared
  region -> the monitor scans the region and exports the pod's usage.

The reference has no test like this at any level (no e2e).
"""
import json
import os
import subprocess
import time
from concurrent import futures
from pathlib import Path

import grpc
import pytest

from k8s_device_plugin_amd.monitor.metrics import MonitorCollector, metrics_text
from k8s_device_plugin_amd.monitor.pathmon import PathMonitor
from k8s_device_plugin_amd.plugin.config import PluginConfig
from k8s_device_plugin_amd.plugin.kfd import enumerate_gpus
from k8s_device_plugin_amd.plugin.register import register_once
from k8s_device_plugin_amd.plugin.rm import ResourceManager
from k8s_device_plugin_amd.plugin.server import VGPUDevicePlugin
from k8s_device_plugin_amd.proto import deviceplugin as dp
from k8s_device_plugin_amd.scheduler.core import Scheduler
from k8s_device_plugin_amd.scheduler.webhook import handle_admission_review
from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
from k8s_device_plugin_amd.utils.types import (
    BIND_PHASE_ANNO,
    BIND_PHASE_SUCCESS,
    ASSIGNED_NODE_ANNO,
    NodeInfo,
    PodInfo,
)
from tests.test_plugin import make_kfd_tree

CSRC = Path(__file__).resolve().parent.parent / "k8s_device_plugin_amd" / "csrc"
MIB = 1 << 20


def test_pod_lifecycle_end_to_end(tmp_path):
    client = FakeKubeClient()
    client.add_node(NodeInfo(name="n1"))

    # ---- node agent comes up and registers --------------------------------
    topo, pci = make_kfd_tree(tmp_path, n_gpus=2)
    rm = ResourceManager(enumerate_gpus(str(topo), str(pci)), split_count=4)
    cfg = PluginConfig(
        node_name="n1",
        hook_path=str(tmp_path / "hook"),
        plugin_socket_dir=str(tmp_path),
        kubelet_socket=str(tmp_path / "kubelet.sock"),
    )
    plugin = VGPUDevicePlugin(cfg, rm, client)
    plugin.serve()
    register_once(client, "n1", rm)

    # ---- scheduler ingests the node ---------------------------------------
    sched = Scheduler(client)
    sched.register_from_node_annotations_once()

    # ---- user creates a pod; webhook mutates it ---------------------------
    pod_obj = {
        "kind": "Pod",
        "metadata": {"name": "train-1", "namespace": "default",
                     "uid": "uid-train-1"},
        "spec": {"containers": [{
            "name": "main",
            "resources": {"limits": {
                "amd.com/gpu": "1",
                "amd.com/gpumem": "73728",
                "amd.com/gpucores": "25",
            }},
        }]},
    }
    review = handle_admission_review(
        {"apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
         "request": {"uid": "rev-1", "object": pod_obj,
                     "kind": {"kind": "Pod"}}})
    assert review["response"]["allowed"]
    import base64

    patches = json.loads(base64.b64decode(review["response"]["patch"]))
    assert any(p["path"] == "/spec/schedulerName" for p in patches)

    pod = PodInfo.from_k8s(pod_obj)
    client.add_pod(pod)

    # ---- /filter then /bind ------------------------------------------------
    fr = sched.filter(pod, ["n1"])
    assert fr.node_names == ["n1"], fr
    assert pod.annotations[ASSIGNED_NODE_ANNO] == "n1"
    br = sched.bind("train-1", "default", "n1")
    assert not br.error

    # ---- kubelet calls Allocate -------------------------------------------
    with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
        stub = dp.DevicePluginClient(ch)
        req = dp.AllocateRequest()
        # kubelet hands over one fake-device id; the assignment in the
        # annotations is authoritative for WHICH physical GPU
        req.container_requests.add(devicesIDs=[f"{rm.gpus[0].uuid}-0"])
        resp = stub.Allocate(req)
    envs = dict(resp.container_responses[0].envs)
    assert envs["VGPU_DEVICE_MEMORY_LIMIT_0"] == "73728m"
    assert envs["VGPU_DEVICE_CU_LIMIT"] == "25"
    assert "HSA_CU_MASK" in envs
    assert client.get_pod("train-1").annotations[BIND_PHASE_ANNO] == BIND_PHASE_SUCCESS

    # ---- the container runs: enforcement via the injected env -------------
    ctr_dir = Path(cfg.hook_path) / "vgpu" / "containers" / "uid-train-1_main"
    assert ctr_dir.is_dir()  # created at Allocate
    cache = ctr_dir / "region.cache"
    run_env = dict(os.environ)
    run_env.update({
        "LD_LIBRARY_PATH": str(CSRC / "fakehip"),
        "LD_PRELOAD": str(CSRC / "libvgpu-hip.so"),
        "VGPU_REAL_HIP_PATH": str(CSRC / "fakehip" / "libamdhip64.so"),
        "VGPU_DEVICE_MEMORY_LIMIT_0": envs["VGPU_DEVICE_MEMORY_LIMIT_0"],
        "VGPU_DEVICE_MEMORY_LIMIT": envs["VGPU_DEVICE_MEMORY_LIMIT_0"],
        "VGPU_DEVICE_CU_LIMIT": envs["VGPU_DEVICE_CU_LIMIT"],
        "VGPU_DEVICE_UUIDS": envs["VGPU_DEVICE_UUIDS"],
        # the real mount maps the host cache dir; point the region there
        "VGPU_DEVICE_MEMORY_SHARED_CACHE": str(cache),
    })
    out = subprocess.run(
        [str(CSRC / "test" / "hip_consumer"),
         "meminfo", "alloc", str(1024 * MIB),
         "alloc", str(80000 * MIB), "meminfo", "sleep", "1000"],
        env=run_env, capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    lines = [json.loads(l) for l in out.stdout.splitlines()]
    assert lines[0]["total"] == 73728 * MIB     # quota view
    assert lines[1]["err"] == 0                 # within quota
    assert lines[2]["err"] == 2                 # beyond quota -> OOM
    assert lines[3]["free"] == (73728 - 1024) * MIB

    # ---- monitor sees the container ----------------------------------------
    pathmon = PathMonitor(str(Path(cfg.hook_path) / "vgpu"))
    pathmon.scan({"uid-train-1"})
    # the consumer exited, but the region file persists with its ledger
    entries = pathmon.live_regions()
    assert len(entries) == 1
    snap = entries[0].region.snapshot()
    assert snap.limit[0] == 73728 * MIB
    mtext = metrics_text(MonitorCollector(pathmon, rm.gpus)).decode()
    assert "vGPU_device_memory_limit_in_bytes" in mtext

    plugin.stop()


def test_oversubscription_end_to_end(tmp_path):
    """BASELINE config 5 acceptance on CPU: deviceMemoryScaling=1.4 -> the
    node advertises ~400 GB on a 288 GB card, a pod requests beyond
    physical, Allocate injects VGPU_OVERSUBSCRIBE, and the container's
    device allocations become managed (fake-runtime managed counter)."""
    client = FakeKubeClient()
    client.add_node(NodeInfo(name="n1"))
    topo, pci = make_kfd_tree(tmp_path, n_gpus=1)
    rm = ResourceManager(enumerate_gpus(str(topo), str(pci)), split_count=4,
                         memory_scaling=1.4)
    cfg = PluginConfig(
        node_name="n1",
        hook_path=str(tmp_path / "hook"),
        plugin_socket_dir=str(tmp_path),
        kubelet_socket=str(tmp_path / "kubelet.sock"),
        device_memory_scaling=1.4,
    )
    plugin = VGPUDevicePlugin(cfg, rm, client)
    plugin.serve()
    register_once(client, "n1", rm)
    sched = Scheduler(client)
    sched.register_from_node_annotations_once()

    # the advertised capacity is scaled past physical
    adv = rm.api_devices()[0].devmem
    assert adv > 288 * 1024  # MiB

    pod_obj = {
        "kind": "Pod",
        "metadata": {"name": "big-1", "namespace": "default", "uid": "uid-big-1"},
        "spec": {"containers": [{
            "name": "main",
            "resources": {"limits": {
                "amd.com/gpu": "1",
                "amd.com/gpumem": "409600",   # 400 GiB on a 288 GiB card
            }},
        }]},
    }
    pod = PodInfo.from_k8s(pod_obj)
    client.add_pod(pod)
    fr = sched.filter(pod, ["n1"])
    assert fr.node_names == ["n1"], fr
    assert not sched.bind("big-1", "default", "n1").error

    with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
        stub = dp.DevicePluginClient(ch)
        req = dp.AllocateRequest()
        req.container_requests.add(devicesIDs=[f"{rm.gpus[0].uuid}-0"])
        resp = stub.Allocate(req)
    envs = dict(resp.container_responses[0].envs)
    assert envs["VGPU_OVERSUBSCRIBE"] == "true"
    assert envs["HSA_XNACK"] == "1"
    assert envs["VGPU_DEVICE_MEMORY_LIMIT_0"] == "409600m"

    # inside the "container": allocations route to managed memory
    ctr_dir = Path(cfg.hook_path) / "vgpu" / "containers" / "uid-big-1_main"
    run_env = dict(os.environ)
    run_env.update({
        "LD_LIBRARY_PATH": str(CSRC / "fakehip"),
        "LD_PRELOAD": str(CSRC / "libvgpu-hip.so"),
        "VGPU_REAL_HIP_PATH": str(CSRC / "fakehip" / "libamdhip64.so"),
        "VGPU_DEVICE_MEMORY_LIMIT": envs["VGPU_DEVICE_MEMORY_LIMIT_0"],
        "VGPU_OVERSUBSCRIBE": envs["VGPU_OVERSUBSCRIBE"],
        "VGPU_DEVICE_MEMORY_SHARED_CACHE": str(ctr_dir / "region.cache"),
        "FAKE_HIP_TOTAL_MEM": str(288 << 30),
    })
    out = subprocess.run(
        [str(CSRC / "test" / "hip_consumer"),
         "meminfo", "alloc", str(300 * 1024 * MIB), "stats"],
        env=run_env, capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    lines = [json.loads(l) for l in out.stdout.splitlines()]
    assert lines[0]["total"] == 409600 * MIB    # virtual capacity visible
    assert lines[1]["err"] == 0                 # 300 GiB > physical succeeds
    assert lines[2]["managed"] >= 1             # went through hipMallocManaged
    plugin.stop()


def test_cpx_partition_lifecycle_end_to_end(tmp_path):
    """CPX node end to end: 8 XCD partitions register as typed devices,
    the scheduler places a whole-partition pod (type-pinned via
    use-gputype), and Allocate hands out ONLY that partition's render
    node — hard isolation flowing through every protocol layer."""
    from test_cpx import make_cpx_tree

    from k8s_device_plugin_amd.device.amd import GPU_IN_USE_ANNO

    client = FakeKubeClient()
    client.add_node(NodeInfo(name="n1"))

    topo, pci, drm = make_cpx_tree(tmp_path)
    gpus = enumerate_gpus(str(topo), str(pci), str(drm))
    rm = ResourceManager(gpus, split_count=2)
    cfg = PluginConfig(
        node_name="n1",
        hook_path=str(tmp_path / "hook"),
        plugin_socket_dir=str(tmp_path),
        kubelet_socket=str(tmp_path / "kubelet.sock"),
    )
    plugin = VGPUDevicePlugin(cfg, rm, client)
    plugin.serve()
    register_once(client, "n1", rm)

    sched = Scheduler(client)
    sched.register_from_node_annotations_once()
    # all 8 partitions visible to the scheduler with the CPX type
    info = sched.node_manager.get_node("n1")
    assert info is not None and len(info.devices) == 8
    assert all(d.type.endswith("-CPX") for d in info.devices)

    pod_obj = {
        "kind": "Pod",
        "metadata": {"name": "cpx-1", "namespace": "default",
                     "uid": "uid-cpx-1",
                     "annotations": {
                         GPU_IN_USE_ANNO: "AMD-Instinct-MI355X-CPX"}},
        "spec": {"containers": [{
            "name": "main",
            "resources": {"limits": {
                "amd.com/gpu": "1",
                "amd.com/gpucores": "100",   # whole partition
            }},
        }]},
    }
    pod = PodInfo.from_k8s(pod_obj)
    client.add_pod(pod)
    fr = sched.filter(pod, ["n1"])
    assert fr.node_names == ["n1"], fr
    br = sched.bind("cpx-1", "default", "n1")
    assert not br.error

    # which partition did the scheduler pick?
    from k8s_device_plugin_amd.utils.codec import decode_container_devices
    from k8s_device_plugin_amd.utils.types import IN_REQUEST_DEVICES

    anno = client.get_pod("cpx-1").annotations[IN_REQUEST_DEVICES["AMD"]]
    assigned = decode_container_devices(anno.split(";")[0])
    part_uuid = assigned[0].uuid
    part = rm.by_uuid(part_uuid)
    assert part is not None and part.partition_count == 8

    with grpc.insecure_channel(f"unix://{plugin.socket_path}") as ch:
        stub = dp.DevicePluginClient(ch)
        req = dp.AllocateRequest()
        req.container_requests.add(devicesIDs=[f"{part_uuid}-0"])
        resp = stub.Allocate(req)
    ctr = resp.container_responses[0]
    devpaths = {d.container_path for d in ctr.devices}
    want = f"/dev/dri/renderD{128 + part.partition_index}"
    assert want in devpaths
    for k in range(8):
        if k != part.partition_index:
            assert f"/dev/dri/renderD{128 + k}" not in devpaths
    envs = dict(ctr.envs)
    assert envs["ROCR_VISIBLE_DEVICES"] == part_uuid
    # whole partition: the hardware fences it, no mask env needed
    assert "HSA_CU_MASK" not in envs
    assert client.get_pod("cpx-1").annotations[BIND_PHASE_ANNO] == \
        BIND_PHASE_SUCCESS
    plugin.stop()
