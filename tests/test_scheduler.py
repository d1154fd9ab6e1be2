"""Scheduler fit/score/filter/bind tests.

Covers the reference's scheduler_test.go accounting cases plus the subtle
fit rules SURVEY.md §7 flags: exclusive-100, core=0-on-full-card, %-memory,
NUMA binding, binpack-by-highest-score (score.go:86-226).
"""
import pytest

from k8s_device_plugin_amd.device import init_devices
from k8s_device_plugin_amd.device.amd import NUMA_BIND_ANNO, GPU_IN_USE_ANNO, GPU_NO_USE_ANNO
from k8s_device_plugin_amd.scheduler.core import Scheduler
from k8s_device_plugin_amd.scheduler.score import (
    NodeUsage,
    calc_score,
    fit_in_certain_device,
    pod_device_requests,
)
from k8s_device_plugin_amd.utils.codec import encode_node_devices
from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
from k8s_device_plugin_amd.utils.types import (
    ASSIGNED_NODE_ANNO,
    BIND_PHASE_ALLOCATING,
    BIND_PHASE_ANNO,
    BIND_TIME_ANNO,
    IN_REQUEST_DEVICES,
    NODE_LOCK_ANNO,
    ContainerDeviceRequest,
    ContainerSpec,
    DeviceInfo,
    DeviceUsage,
    NodeInfo,
    PodInfo,
)

init_devices()

MI355X_MEM = 294912
TYPE = "AMD-Instinct-MI355X"


def usage(n=8, count=10, totalmem=MI355X_MEM, totalcore=100, numa_split=True):
    devs = []
    for i in range(n):
        devs.append(
            DeviceUsage(
                id=f"GPU-{i}",
                index=i,
                count=count,
                totalmem=totalmem,
                totalcore=totalcore,
                numa=(i // (n // 2 or 1)) if numa_split and n > 1 else 0,
                type=TYPE,
            )
        )
    return NodeUsage(devices=devs)


def req(nums=1, mem=0, memp=101, cores=0):
    return ContainerDeviceRequest(
        nums=nums, type="AMD", memreq=mem, mem_percentage_req=memp, coresreq=cores
    )


class TestFitInCertainDevice:
    def test_basic_fit(self):
        node = usage(1)
        ok, devs = fit_in_certain_device(node, req(1, mem=73728, cores=25), {})
        assert ok
        d = devs["AMD"][0]
        assert d.usedmem == 73728 and d.usedcores == 25

    def test_memory_percentage(self):
        node = usage(1)
        ok, devs = fit_in_certain_device(node, req(1, memp=50), {})
        assert ok
        assert devs["AMD"][0].usedmem == MI355X_MEM // 2

    def test_absolute_mem_wins_over_percentage(self):
        node = usage(1)
        ok, devs = fit_in_certain_device(node, req(1, mem=1000, memp=50), {})
        assert ok
        assert devs["AMD"][0].usedmem == 1000

    def test_insufficient_memory(self):
        node = usage(1)
        node.devices[0].usedmem = MI355X_MEM - 100
        ok, _ = fit_in_certain_device(node, req(1, mem=200), {})
        assert not ok

    def test_insufficient_cores(self):
        node = usage(1)
        node.devices[0].usedcores = 90
        ok, _ = fit_in_certain_device(node, req(1, mem=100, cores=20), {})
        assert not ok

    def test_cores_over_100_rejected(self):
        node = usage(1)
        ok, _ = fit_in_certain_device(node, req(1, mem=100, cores=150), {})
        assert not ok

    def test_exclusive_rejects_used_card(self):
        node = usage(1)
        node.devices[0].used = 1
        ok, _ = fit_in_certain_device(node, req(1, mem=100, cores=100), {})
        assert not ok

    def test_exclusive_ok_on_idle_card(self):
        node = usage(1)
        ok, _ = fit_in_certain_device(node, req(1, mem=100, cores=100), {})
        assert ok

    def test_core0_rejected_on_corefull_card(self):
        node = usage(1)
        node.devices[0].usedcores = 100
        ok, _ = fit_in_certain_device(node, req(1, mem=100, cores=0), {})
        assert not ok

    def test_count_exhausted(self):
        node = usage(1, count=2)
        node.devices[0].used = 2
        ok, _ = fit_in_certain_device(node, req(1, mem=100), {})
        assert not ok

    def test_multi_gpu_request(self):
        node = usage(4)
        ok, devs = fit_in_certain_device(node, req(3, mem=100), {})
        assert ok and len(devs["AMD"]) == 3
        uuids = {d.uuid for d in devs["AMD"]}
        assert len(uuids) == 3

    def test_numa_bind_restarts_on_boundary(self):
        # 4 devices: numa0 = {0,1}, numa1 = {2,3}; device 1 is nearly full so
        # a 2-GPU numa-bound request must land both GPUs in numa1.
        node = usage(4, numa_split=True)
        node.devices[1].usedmem = MI355X_MEM - 10
        annos = {NUMA_BIND_ANNO: "true"}
        ok, devs = fit_in_certain_device(node, req(2, mem=100), annos)
        assert ok
        numas = {node.devices[d.idx].numa for d in devs["AMD"]}
        assert len(numas) == 1

    def test_type_whitelist(self):
        node = usage(1)
        ok, _ = fit_in_certain_device(node, req(1, mem=100), {GPU_IN_USE_ANNO: "MI355X"})
        assert ok
        ok, _ = fit_in_certain_device(node, req(1, mem=100), {GPU_IN_USE_ANNO: "H100"})
        assert not ok

    def test_type_blacklist(self):
        node = usage(1)
        ok, _ = fit_in_certain_device(node, req(1, mem=100), {GPU_NO_USE_ANNO: "MI355X"})
        assert not ok
        ok, _ = fit_in_certain_device(node, req(1, mem=100), {GPU_NO_USE_ANNO: "H100"})
        assert ok


class TestCalcScore:
    def pod(self, gpus=1, mem=73728, cores=0):
        return PodInfo(
            name="p", uid="u1",
            containers=[ContainerSpec(name="c", limits={
                "amd.com/gpu": gpus, "amd.com/gpumem": mem, "amd.com/gpucores": cores})],
        )

    def test_binpack_prefers_busier_node(self):
        # Node A empty, node B has a slice used on every device -> the device
        # B picks has fewer free slots -> total/free is larger -> B scores
        # higher (binpack across nodes; within a node the walk picks the
        # most-free device, score.go:45-50 + 93).
        nodes = {"a": usage(2), "b": usage(2)}
        for d in nodes["b"].devices:
            d.used = 1
            d.usedmem = 1000
        nums = pod_device_requests(self.pod())
        scores = calc_score(nodes, nums, {})
        assert len(scores) == 2
        best = max(scores, key=lambda s: s.score)
        assert best.node_id == "b"

    def test_no_fit_excluded(self):
        nodes = {"a": usage(1)}
        nodes["a"].devices[0].usedmem = MI355X_MEM
        nums = pod_device_requests(self.pod(mem=1000))
        assert calc_score(nodes, nums, {}) == []

    def test_non_gpu_container_keeps_slot(self):
        pod = PodInfo(
            name="p", uid="u1",
            containers=[
                ContainerSpec(name="sidecar"),
                ContainerSpec(name="main", limits={"amd.com/gpu": 1, "amd.com/gpumem": 1000}),
            ],
        )
        nums = pod_device_requests(pod)
        scores = calc_score({"a": usage(2)}, nums, {})
        assert len(scores) == 1
        pd = scores[0].devices["AMD"]
        # container 0 (sidecar) has an empty slot only if a type entry existed
        # before it; container indices of GPU containers must be correct:
        flat = [(i, d) for i, ctr in enumerate(pd) for d in ctr]
        assert all(i == len(pd) - 1 for i, _ in flat)  # GPU devices on last ctr


class TestSchedulerEndToEnd:
    def setup_method(self):
        self.client = FakeKubeClient()
        self.sched = Scheduler(self.client)
        devs = [
            DeviceInfo(id=f"GPU-{i}", count=10, devmem=MI355X_MEM, devcore=100,
                       type=TYPE, numa=i // 4, health=True)
            for i in range(8)
        ]
        from k8s_device_plugin_amd.device.amd import HANDSHAKE_ANNO, REGISTER_ANNO

        self.client.add_node(NodeInfo(
            name="node1",
            annotations={
                REGISTER_ANNO: encode_node_devices(devs),
                HANDSHAKE_ANNO: "Reported 2026-01-01",
            },
        ))
        self.sched.register_from_node_annotations_once()

    def mkpod(self, name, gpus=1, mem=73728, cores=0):
        pod = PodInfo(
            name=name, uid=f"uid-{name}",
            containers=[ContainerSpec(name="c", limits={
                "amd.com/gpu": gpus, "amd.com/gpumem": mem, "amd.com/gpucores": cores})],
        )
        self.client.add_pod(pod)
        return pod

    def test_register_ingests_devices(self):
        node = self.sched.node_manager.get_node("node1")
        assert node is not None and len(node.devices) == 8
        # handshake was re-challenged
        from k8s_device_plugin_amd.device.amd import HANDSHAKE_ANNO

        assert self.client.get_node("node1").annotations[HANDSHAKE_ANNO].startswith(
            "Requesting_")

    def test_filter_assigns_and_patches(self):
        pod = self.mkpod("p1")
        res = self.sched.filter(pod, ["node1"])
        assert res.error == "" and res.node_names == ["node1"]
        stored = self.client.get_pod("p1")
        assert stored.annotations[ASSIGNED_NODE_ANNO] == "node1"
        assert "GPU-" in stored.annotations[IN_REQUEST_DEVICES["AMD"]]

    def test_filter_accounts_usage_across_pods(self):
        # 8 GPUs, each pod takes 50% mem; 17th pod at 50% must still fit
        # (2 per GPU), but a pod wanting 60% of a GPU with all at 50% fails.
        for i in range(16):
            pod = self.mkpod(f"p{i}", mem=MI355X_MEM // 2)
            res = self.sched.filter(pod, ["node1"])
            assert res.node_names == ["node1"], f"pod {i} failed: {res.error}"
        big = self.mkpod("big", mem=int(MI355X_MEM * 0.6))
        res = self.sched.filter(big, ["node1"])
        assert res.node_names != ["node1"]

    def test_bind_locks_and_patches(self):
        pod = self.mkpod("p1")
        self.sched.filter(pod, ["node1"])
        res = self.sched.bind("p1", "default", "node1")
        assert res.error == ""
        stored = self.client.get_pod("p1")
        assert stored.annotations[BIND_PHASE_ANNO] == BIND_PHASE_ALLOCATING
        assert BIND_TIME_ANNO in stored.annotations
        assert NODE_LOCK_ANNO in self.client.get_node("node1").annotations
        assert self.client.bindings == [("default", "p1", "node1")]

    def test_exclusive_pod_excludes_shared_gpu(self):
        shared = self.mkpod("shared", mem=1000, cores=10)
        assert self.sched.filter(shared, ["node1"]).node_names == ["node1"]
        # 8 exclusive pods: only 7 free GPUs remain fully idle
        fits = 0
        for i in range(8):
            p = self.mkpod(f"x{i}", mem=1000, cores=100)
            if self.sched.filter(p, ["node1"]).node_names == ["node1"]:
                fits += 1
        assert fits == 7

    def test_handshake_timeout_evicts(self):
        import time as _time

        from k8s_device_plugin_amd.device.amd import HANDSHAKE_ANNO

        # scheduler already set Requesting_<now>; simulate 61s of silence
        self.sched.register_from_node_annotations_once(now=_time.time() + 61)
        assert self.sched.node_manager.get_node("node1") is None
        assert self.client.get_node("node1").annotations[HANDSHAKE_ANNO].startswith(
            "Deleted_")


class TestExtenderTLS:
    """The webhook endpoint must serve HTTPS (reference scheduler runs the
    extender behind --cert_file/--key_file, cmd/scheduler/main.go:52-56)."""

    def test_filter_over_https(self, tmp_path):
        import ssl
        import subprocess as sp
        import urllib.request

        from k8s_device_plugin_amd.scheduler.core import Scheduler
        from k8s_device_plugin_amd.scheduler.routes import ExtenderServer
        from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient

        cert = tmp_path / "tls.crt"
        key = tmp_path / "tls.key"
        sp.run(["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
                "-keyout", str(key), "-out", str(cert), "-days", "1",
                "-subj", "/CN=127.0.0.1"], check=True, capture_output=True)
        sched = Scheduler(FakeKubeClient())
        server = ExtenderServer(sched, host="127.0.0.1", port=0,
                                cert_file=str(cert), key_file=str(key))
        server.start()
        try:
            ctx = ssl.create_default_context()
            ctx.check_hostname = False
            ctx.verify_mode = ssl.CERT_NONE
            with urllib.request.urlopen(
                    f"https://127.0.0.1:{server.port}/healthz",
                    context=ctx, timeout=10) as resp:
                assert resp.status == 200
        finally:
            server.stop()


class TestPodCacheEviction:
    """rebuild_pod_cache must evict deleted/completed pods (the reference's
    informer onDelPod, scheduler.go:91-110) or their usage leaks forever."""

    def test_deleted_pod_usage_released(self):
        from k8s_device_plugin_amd.device.amd import REGISTER_ANNO, HANDSHAKE_ANNO
        from k8s_device_plugin_amd.scheduler.core import Scheduler
        from k8s_device_plugin_amd.utils.codec import (
            encode_node_devices,
            encode_pod_single_device,
        )
        from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
        from k8s_device_plugin_amd.utils.types import (
            ASSIGNED_NODE_ANNO,
            SUPPORT_DEVICES,
            ContainerDevice,
            DeviceInfo,
            NodeInfo,
            PodInfo,
        )

        client = FakeKubeClient()
        client.add_node(NodeInfo(name="n1", annotations={
            HANDSHAKE_ANNO: "Reported 2026-01-01 00:00:00",
            REGISTER_ANNO: encode_node_devices([DeviceInfo(
                id="GPU-x", count=10, devmem=294912, devcore=100,
                type="AMD-Instinct-MI355X", numa=0, health=True, index=0)]),
        }))
        sched = Scheduler(client)
        sched.register_from_node_annotations_once()
        pod = PodInfo(name="p", uid="uid-p", node_name="n1", annotations={
            ASSIGNED_NODE_ANNO: "n1",
            SUPPORT_DEVICES["AMD"]: encode_pod_single_device(
                [[ContainerDevice(uuid="GPU-x", type="AMD", usedmem=1000,
                                  usedcores=50)]]),
        })
        client.add_pod(pod)
        sched.rebuild_pod_cache()
        usage, _ = sched.get_nodes_usage(["n1"])
        assert usage["n1"].devices[0].usedmem == 1000
        # pod deleted from the API -> next rebuild releases its usage
        client.delete_pod("p")
        sched.rebuild_pod_cache()
        usage, _ = sched.get_nodes_usage(["n1"])
        assert usage["n1"].devices[0].usedmem == 0
        assert usage["n1"].devices[0].used == 0


class TestSchedulerMetrics:
    """All 10 reference metric families (cmd/scheduler/metrics.go:49-190)
    render with data."""

    def test_families_present(self):
        from k8s_device_plugin_amd.device.amd import REGISTER_ANNO, HANDSHAKE_ANNO
        from k8s_device_plugin_amd.scheduler.core import Scheduler
        from k8s_device_plugin_amd.scheduler.metrics import metrics_text
        from k8s_device_plugin_amd.utils.codec import (
            encode_node_devices,
            encode_pod_single_device,
        )
        from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
        from k8s_device_plugin_amd.utils.types import (
            ASSIGNED_NODE_ANNO,
            SUPPORT_DEVICES,
            ContainerDevice,
            DeviceInfo,
            NodeInfo,
            PodInfo,
        )

        client = FakeKubeClient()
        client.add_node(NodeInfo(name="n1", annotations={
            HANDSHAKE_ANNO: "Reported 2026-01-01 00:00:00",
            REGISTER_ANNO: encode_node_devices([DeviceInfo(
                id="GPU-m", count=10, devmem=294912, devcore=100,
                type="AMD-Instinct-MI355X", numa=0, health=True, index=0)]),
        }))
        sched = Scheduler(client)
        sched.register_from_node_annotations_once()
        client.add_pod(PodInfo(name="p", uid="uid-m", node_name="n1", annotations={
            ASSIGNED_NODE_ANNO: "n1",
            SUPPORT_DEVICES["AMD"]: encode_pod_single_device(
                [[ContainerDevice(uuid="GPU-m", type="AMD", usedmem=1024,
                                  usedcores=25)]]),
        }))
        sched.rebuild_pod_cache()
        sched.get_nodes_usage(["n1"])
        text = metrics_text(sched).decode()
        for family in ["GPUDeviceMemoryLimit", "GPUDeviceCoreLimit",
                       "GPUDeviceMemoryAllocated", "GPUDeviceSharedNum",
                       "GPUDeviceCoreAllocated", "nodeGPUOverview",
                       "nodeGPUMemoryPercentage", "vGPUPodsDeviceAllocated",
                       "vGPUMemoryPercentage", "vGPUCorePercentage"]:
            assert family in text, f"missing metric family {family}"


class TestConcurrentFilter:
    """Parallel /filter calls must keep accounting consistent (the
    reference guards its managers with mutexes, nodes.go:50-53)."""

    def test_parallel_filters_never_overcommit(self):
        import threading

        from k8s_device_plugin_amd.device.amd import REGISTER_ANNO, HANDSHAKE_ANNO
        from k8s_device_plugin_amd.scheduler.core import Scheduler
        from k8s_device_plugin_amd.utils.codec import encode_node_devices
        from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
        from k8s_device_plugin_amd.utils.types import (
            ContainerSpec,
            DeviceInfo,
            NodeInfo,
            PodInfo,
        )

        client = FakeKubeClient()
        client.add_node(NodeInfo(name="n1", annotations={
            HANDSHAKE_ANNO: "Reported 2026-01-01 00:00:00",
            REGISTER_ANNO: encode_node_devices([DeviceInfo(
                id="GPU-c", count=4, devmem=294912, devcore=100,
                type="AMD-Instinct-MI355X", numa=0, health=True, index=0)]),
        }))
        sched = Scheduler(client)
        sched.register_from_node_annotations_once()

        results = []

        def one(i):
            pod = PodInfo(
                name=f"p{i}", uid=f"uid-{i}",
                containers=[ContainerSpec(
                    name="c", limits={"amd.com/gpu": 1,
                                      "amd.com/gpumem": 100000})],
            )
            client.add_pod(pod)
            results.append(sched.filter(pod, ["n1"]))

        threads = [threading.Thread(target=one, args=(i,)) for i in range(6)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        # 294912 MiB total, 100000 each -> exactly 2 fit (filter is
        # serialized snapshot->commit)
        placed = [r for r in results if r.node_names]
        assert len(placed) == 2
        usage, _ = sched.get_nodes_usage(["n1"])
        assert usage["n1"].devices[0].usedmem <= 294912


class TestExtenderHTTP:
    """The actual extender wire: POST /filter /bind /webhook with the JSON
    shapes kube-scheduler sends (ExtenderArgs/ExtenderBindingArgs,
    reference routes/route.go:41-134)."""

    def _serve(self):
        import json as j
        import urllib.request

        from k8s_device_plugin_amd.device.amd import (
            HANDSHAKE_ANNO,
            REGISTER_ANNO,
        )
        from k8s_device_plugin_amd.scheduler.core import Scheduler
        from k8s_device_plugin_amd.scheduler.routes import ExtenderServer
        from k8s_device_plugin_amd.utils.codec import encode_node_devices
        from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
        from k8s_device_plugin_amd.utils.types import DeviceInfo, NodeInfo

        client = FakeKubeClient()
        client.add_node(NodeInfo(name="n1", annotations={
            HANDSHAKE_ANNO: "Reported 2026-01-01 00:00:00",
            REGISTER_ANNO: encode_node_devices([DeviceInfo(
                id="GPU-h", count=10, devmem=294912, devcore=100,
                type="AMD-Instinct-MI355X", numa=0, health=True, index=0)]),
        }))
        sched = Scheduler(client)
        sched.register_from_node_annotations_once()
        server = ExtenderServer(sched, host="127.0.0.1", port=0)
        server.start()

        def post(path, obj):
            req = urllib.request.Request(
                f"http://127.0.0.1:{server.port}{path}",
                data=j.dumps(obj).encode(),
                headers={"Content-Type": "application/json"})
            with urllib.request.urlopen(req, timeout=10) as resp:
                return j.loads(resp.read())

        return client, sched, server, post

    def test_filter_bind_webhook_wire(self):
        client, sched, server, post = self._serve()
        try:
            pod_obj = {
                "kind": "Pod",
                "metadata": {"name": "w1", "namespace": "default",
                             "uid": "uid-w1"},
                "spec": {"containers": [{
                    "name": "main",
                    "resources": {"limits": {"amd.com/gpu": "1",
                                             "amd.com/gpumem": "1024"}},
                }]},
            }
            from k8s_device_plugin_amd.utils.types import PodInfo

            client.add_pod(PodInfo.from_k8s(pod_obj))
            out = post("/filter", {"Pod": pod_obj, "NodeNames": ["n1"]})
            assert out["NodeNames"] == ["n1"]
            assert not out.get("Error")

            out = post("/bind", {"PodName": "w1", "PodNamespace": "default",
                                 "Node": "n1"})
            assert not out.get("Error")
            assert client.get_pod("w1").annotations.get("amd.io/bind-phase") == "allocating"

            review = {
                "apiVersion": "admission.k8s.io/v1",
                "kind": "AdmissionReview",
                "request": {"uid": "wh1", "object": pod_obj,
                            "kind": {"kind": "Pod"}},
            }
            out = post("/webhook", review)
            assert out["response"]["allowed"] and out["response"]["uid"] == "wh1"
        finally:
            server.stop()

    def test_filter_no_fit_reports_failed_nodes(self):
        client, sched, server, post = self._serve()
        try:
            pod_obj = {
                "kind": "Pod",
                "metadata": {"name": "w2", "namespace": "default",
                             "uid": "uid-w2"},
                "spec": {"containers": [{
                    "name": "main",
                    "resources": {"limits": {"amd.com/gpu": "99"}},
                }]},
            }
            out = post("/filter", {"Pod": pod_obj, "NodeNames": ["n1"]})
            assert not out.get("NodeNames")
            assert "n1" in (out.get("FailedNodes") or {})
        finally:
            server.stop()
