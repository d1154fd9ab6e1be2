"""BASELINE config 4: scheduler-extender bin-packing 32 pods by HBM+CU
across an 8x MI355X node — pure control-plane, CPU-only."""
from k8s_device_plugin_amd.scheduler.core import Scheduler
from k8s_device_plugin_amd.utils.codec import decode_pod_devices, encode_node_devices
from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
from k8s_device_plugin_amd.utils.types import (
    SUPPORT_DEVICES,
    ContainerSpec,
    DeviceInfo,
    NodeInfo,
    PodInfo,
)

MI355X_MEM = 294912
TYPE = "AMD-Instinct-MI355X"


def make_cluster(n_nodes=1, gpus_per_node=8):
    client = FakeKubeClient()
    sched = Scheduler(client)
    from k8s_device_plugin_amd.device.amd import HANDSHAKE_ANNO, REGISTER_ANNO

    for n in range(n_nodes):
        devs = [
            DeviceInfo(id=f"node{n}-GPU-{i}", count=10, devmem=MI355X_MEM,
                       devcore=100, type=TYPE, numa=i // 4, health=True)
            for i in range(gpus_per_node)
        ]
        client.add_node(NodeInfo(
            name=f"node{n}",
            annotations={REGISTER_ANNO: encode_node_devices(devs),
                         HANDSHAKE_ANNO: "Reported now"}))
    sched.register_from_node_annotations_once()
    return client, sched


def test_32_pods_pack_onto_8_gpus():
    client, sched = make_cluster()
    placed = {}
    for i in range(32):
        pod = PodInfo(
            name=f"p{i}", uid=f"uid-{i}",
            containers=[ContainerSpec(name="c", limits={
                "amd.com/gpu": 1,
                "amd.com/gpumem-percentage": 25,
                "amd.com/gpucores": 25,
            })])
        client.add_pod(pod)
        res = sched.filter(pod, ["node0"])
        assert res.node_names == ["node0"], f"pod {i}: {res.error} {res.failed_nodes}"
        devs = decode_pod_devices(SUPPORT_DEVICES, client.get_pod(f"p{i}").annotations)
        uuid = devs["AMD"][0][0].uuid
        placed[uuid] = placed.get(uuid, 0) + 1
    # exactly 4 per GPU: 25% mem & 25% cores quarters each card
    assert sorted(placed.values()) == [4] * 8
    # a 33rd quarter-GPU pod must NOT fit (cores exhausted)
    extra = PodInfo(name="p32", uid="uid-32", containers=[ContainerSpec(
        name="c", limits={"amd.com/gpu": 1, "amd.com/gpumem-percentage": 25,
                          "amd.com/gpucores": 25})])
    client.add_pod(extra)
    res = sched.filter(extra, ["node0"])
    assert res.node_names != ["node0"]


def test_multi_node_binpack_fills_before_spreading():
    client, sched = make_cluster(n_nodes=2, gpus_per_node=2)
    nodes_used = set()
    for i in range(4):
        pod = PodInfo(name=f"q{i}", uid=f"quid-{i}",
                      containers=[ContainerSpec(name="c", limits={
                          "amd.com/gpu": 1, "amd.com/gpumem-percentage": 50})])
        client.add_pod(pod)
        res = sched.filter(pod, ["node0", "node1"])
        assert len(res.node_names) == 1
        nodes_used.add(res.node_names[0])
    # binpack: 4 half-GPU pods fit on the 2 GPUs of ONE node
    assert len(nodes_used) == 1


def test_full_node_8gpu_pod():
    client, sched = make_cluster()
    pod = PodInfo(name="big", uid="uid-big", containers=[ContainerSpec(
        name="c", limits={"amd.com/gpu": 8, "amd.com/gpumem-percentage": 100,
                          "amd.com/gpucores": 100})])
    client.add_pod(pod)
    res = sched.filter(pod, ["node0"])
    assert res.node_names == ["node0"]
    devs = decode_pod_devices(SUPPORT_DEVICES, client.get_pod("big").annotations)
    assert len({d.uuid for d in devs["AMD"][0]}) == 8
