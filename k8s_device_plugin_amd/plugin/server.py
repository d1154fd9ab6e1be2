"""Device plugin gRPC server: Register / ListAndWatch / Allocate.

Reference behavior: pkg/device-plugin/.../plugin/server.go:122-583 —
- serve the DevicePlugin service on ``<socket_dir>/amd-gpu.sock`` and
  Register with the kubelet;
- ListAndWatch streams the fake-device fan-out, re-sending on health change;
- Allocate matches the pending pod bound to this node (annotation protocol),
  consumes one container's assignment per call, and injects the enforcement
  environment + mounts (server.go:288-411).

MI355X injection differences from the reference's CUDA path (by design):
- ``ROCR_VISIBLE_DEVICES`` (GPU uuids) instead of NVIDIA_VISIBLE_DEVICES —
  ROCr itself hides the other GPUs;
- ``HSA_CU_MASK`` hard CU partition computed by the XCD-aware allocator
  (ops/cumask.py) — the runtime pins queues, something the CUDA hook can't;
- ``HSA_XNACK=1`` in oversubscribe mode so managed memory pages to host;
- DeviceSpecs mount /dev/kfd + the per-GPU /dev/dri nodes.
"""
from __future__ import annotations

import logging
import os
import threading
import time
import uuid as uuidlib
from concurrent import futures
from typing import Dict, List, Optional

import grpc

from ..device import init_devices, pod_allocation_failed, pod_allocation_try_success
from ..device.amd import AMD_DEVICE_TYPE
from ..ops.cumask import CoreMaskAllocator, hsa_cu_mask_env
from ..proto import deviceplugin as dp
from ..utils.kubeclient import KubeClient
from ..utils.pendingpod import (
    PendingPodError,
    erase_next_device_type_from_annotation,
    get_next_device_request,
    get_pending_pod,
)
from ..utils import nodelock
from .config import PluginConfig
from .rm import ResourceManager

log = logging.getLogger(__name__)


class VGPUDevicePlugin:
    """The DevicePlugin servicer + registration client."""

    def __init__(self, cfg: PluginConfig, rm: ResourceManager, client: KubeClient):
        init_devices()
        self.cfg = cfg
        self.rm = rm
        self.client = client
        self.cumask = CoreMaskAllocator()
        self._update = threading.Event()
        self._stop = threading.Event()
        self._server: Optional[grpc.Server] = None
        self.endpoint = "amd-gpu.sock"
        # pod uid -> [(device uuid, mask)] for release on pod death
        self.pod_masks: Dict[str, List] = {}

    # ---- gRPC servicer methods -----------------------------------------
    def GetDevicePluginOptions(self, request, context):
        return dp.DevicePluginOptions(pre_start_required=False,
                                      get_preferred_allocation_available=True)

    def ListAndWatch(self, request, context):
        while not self._stop.is_set():
            devices = []
            for fd in self.rm.fake_devices():
                devices.append(dp.Device(
                    ID=fd.id,
                    health=dp.HEALTHY if fd.healthy else dp.UNHEALTHY,
                    topology=dp.TopologyInfo(nodes=[dp.NUMANode(ID=fd.numa)]),
                ))
            yield dp.ListAndWatchResponse(devices=devices)
            # block until a health change or shutdown
            self._update.wait()
            self._update.clear()

    def notify_update(self):
        self._update.set()

    def PreStartContainer(self, request, context):
        return dp.PreStartContainerResponse()

    def GetPreferredAllocation(self, request, context):
        """xGMI-aligned preferred allocation.

        The reference has aligned (NVLink) / distributed allocators but
        leaves the server verb commented out (rm/allocate.go:44-121,
        server.go:270-285); here it is live: fractional requests pack onto
        the fewest physical GPUs, multi-GPU requests extend along maximal
        xGMI connectivity (parallel/topology.py)."""
        from ..parallel.topology import GPUTopology

        topo = GPUTopology.from_gpus(self.rm.gpus) if self.rm.gpus else None
        uuid_to_idx = {g.uuid: i for i, g in enumerate(self.rm.gpus)}
        resp = dp.PreferredAllocationResponse()
        for creq in request.container_requests:
            chosen = list(creq.must_include_deviceIDs)
            by_uuid: Dict[str, List[str]] = {}
            for fid in creq.available_deviceIDs:
                if fid in chosen:
                    continue
                by_uuid.setdefault(self.rm.uuid_of_fake(fid), []).append(fid)
            need = creq.allocation_size - len(chosen)
            chosen_uuids = {self.rm.uuid_of_fake(f) for f in chosen}
            # 1. pack: drain fakes of GPUs already in the set
            for u in sorted(chosen_uuids):
                while need > 0 and by_uuid.get(u):
                    chosen.append(by_uuid[u].pop(0))
                    need -= 1
            # 2. extend: next GPU = max xGMI edges to the current set
            while need > 0 and by_uuid:
                cands = [u for u, f in by_uuid.items() if f and u in uuid_to_idx]
                if not cands:
                    break
                if topo is not None and chosen_uuids:
                    cur = [uuid_to_idx[u] for u in chosen_uuids
                           if u in uuid_to_idx]

                    def edges(u):
                        return topo.xgmi_degree(cur + [uuid_to_idx[u]])

                    cands.sort(key=lambda u: (-edges(u),
                                              -len(by_uuid[u]), u))
                else:
                    cands.sort(key=lambda u: (-len(by_uuid[u]), u))
                u = cands[0]
                chosen_uuids.add(u)
                while need > 0 and by_uuid[u]:
                    chosen.append(by_uuid[u].pop(0))
                    need -= 1
                if not by_uuid[u]:
                    del by_uuid[u]
            resp.container_responses.add(deviceIDs=chosen)
        return resp

    def Allocate(self, request, context):
        node = self.cfg.node_name
        try:
            pending = get_pending_pod(self.client, node)
        except PendingPodError as e:
            log.error("Allocate: %s", e)
            try:
                nodelock.release_node_lock(self.client, node)
            except Exception:
                pass
            context.abort(grpc.StatusCode.FAILED_PRECONDITION, str(e))
        responses = dp.AllocateResponse()
        for req in request.container_requests:
            try:
                ctr, devreq = get_next_device_request(AMD_DEVICE_TYPE, pending)
            except PendingPodError as e:
                pod_allocation_failed(self.client, node, pending)
                context.abort(grpc.StatusCode.FAILED_PRECONDITION, str(e))
            if len(devreq) != len(req.devicesIDs):
                pod_allocation_failed(self.client, node, pending)
                context.abort(grpc.StatusCode.FAILED_PRECONDITION,
                              "device allocate number not matched")
            resp = self._container_response(pending, ctr, devreq)
            try:
                erase_next_device_type_from_annotation(
                    self.client, AMD_DEVICE_TYPE, pending)
            except Exception as e:
                pod_allocation_failed(self.client, node, pending)
                context.abort(grpc.StatusCode.INTERNAL, str(e))
            responses.container_responses.append(resp)
        pod_allocation_try_success(self.client, node, pending)
        return responses

    # ---- injection ------------------------------------------------------
    def _container_response(self, pod, ctr, devreq):
        cfg = self.cfg
        resp = dp.ContainerAllocateResponse()
        visible = []
        cards = []
        rsmi_indices = []
        mask_assignments = []
        seen_paths = set()
        for i, dev in enumerate(devreq):
            resp.envs[f"VGPU_DEVICE_MEMORY_LIMIT_{i}"] = f"{dev.usedmem}m"
            gpu = self.rm.by_uuid(dev.uuid)
            if gpu is not None:
                visible.append(gpu.uuid)
                cards.append(f"card{gpu.drm_card}")
                rsmi_indices.append(str(gpu.index))
                for path in gpu.device_paths:
                    if path not in seen_paths:
                        seen_paths.add(path)
                        resp.devices.add(container_path=path, host_path=path,
                                         permissions="rw")
            else:
                visible.append(dev.uuid)
            if dev.usedcores and dev.usedcores < 100 and not cfg.disable_core_limit:
                m = self.cumask.alloc(dev.uuid, dev.usedcores)
                if m:
                    mask_assignments.append((i, m))
                    self.pod_masks.setdefault(pod.uid, []).append((dev.uuid, m))
        if not any(p == "/dev/kfd" for p in seen_paths):
            resp.devices.add(container_path="/dev/kfd", host_path="/dev/kfd",
                             permissions="rw")
        cores = devreq[0].usedcores if devreq else 0
        resp.envs["ROCR_VISIBLE_DEVICES"] = ",".join(visible)
        resp.envs["VGPU_DEVICE_UUIDS"] = ",".join(visible)
        resp.envs["VGPU_DEVICE_CU_LIMIT"] = str(cores)
        resp.envs["VGPU_DEVICE_MEMORY_SHARED_CACHE"] = (
            f"{cfg.hook_path}/vgpu/{uuidlib.uuid4()}.cache")
        if cards:
            resp.envs["VGPU_SYSFS_CARDS"] = ",".join(cards)
        if rsmi_indices:
            resp.envs["VGPU_RSMI_INDICES"] = ",".join(rsmi_indices)
        if cfg.context_overhead_mb > 0:
            resp.envs["VGPU_CONTEXT_OVERHEAD"] = f"{cfg.context_overhead_mb}m"
        if cfg.device_memory_scaling > 1:
            resp.envs["VGPU_OVERSUBSCRIBE"] = "true"
            resp.envs["HSA_XNACK"] = "1"
        if cfg.disable_core_limit:
            resp.envs["GPU_CORE_UTILIZATION_POLICY"] = "disable"
        mask_env = hsa_cu_mask_env(mask_assignments)
        if mask_env:
            resp.envs["HSA_CU_MASK"] = mask_env

        cache_dir = f"{cfg.hook_path}/vgpu/containers/{pod.uid}_{ctr.name}"
        try:
            os.makedirs(cache_dir, mode=0o777, exist_ok=True)
            os.chmod(cache_dir, 0o777)
            os.makedirs("/tmp/vgpulock", mode=0o777, exist_ok=True)
            # allocation record for restart reconciliation (the DCU plugin
            # persists vdev conf files the same way, dcu/server.go:415-465)
            import json as _json

            with open(os.path.join(cache_dir, "vgpu.json"), "w") as f:
                _json.dump({
                    "pod_uid": pod.uid,
                    "container": ctr.name,
                    "devices": [
                        {"uuid": d.uuid, "usedmem": d.usedmem,
                         "usedcores": d.usedcores,
                         "cu_mask": f"{m:x}" if m else ""}
                        for d, m in zip(
                            devreq,
                            [next((mm for ii, mm in mask_assignments if ii == i),
                                  0) for i in range(len(devreq))])
                    ],
                }, f)
        except OSError as e:
            log.warning("cannot create hook dirs: %s", e)
        resp.mounts.add(
            container_path=f"{cfg.hook_path}/vgpu/libvgpu-hip.so",
            host_path=f"{cfg.hook_path}/vgpu/libvgpu-hip.so", read_only=True)
        resp.mounts.add(container_path=f"{cfg.hook_path}/vgpu",
                        host_path=cache_dir, read_only=False)
        resp.mounts.add(container_path="/tmp/vgpulock",
                        host_path="/tmp/vgpulock", read_only=False)
        if "VGPU_DISABLE_CONTROL" not in ctr.env:
            resp.mounts.add(container_path="/etc/ld.so.preload",
                            host_path=f"{cfg.hook_path}/vgpu/ld.so.preload",
                            read_only=True)
        if cfg.device_list_strategy == "cdi-annotations":
            from . import cdi as cdimod

            for k, v in cdimod.annotations(visible).items():
                resp.annotations[k] = v
        return resp

    def reconcile(self, live_pod_uids) -> None:
        """Rebuild CU-mask allocator state from the per-container records
        after a plugin restart, and GC records of dead pods (reference DCU
        RefreshContainerDevices, dcu/server.go:274-316)."""
        import json as _json
        import shutil

        root = f"{self.cfg.hook_path}/vgpu/containers"
        if not os.path.isdir(root):
            return
        for name in os.listdir(root):
            d = os.path.join(root, name)
            rec_path = os.path.join(d, "vgpu.json")
            pod_uid, _, _ctr = name.partition("_")
            if pod_uid not in live_pod_uids:
                if pod_uid in self.pod_masks:
                    self.release_pod(pod_uid)
                try:
                    shutil.rmtree(d)
                    log.info("reconcile: removed orphan %s", name)
                except OSError as e:
                    log.warning("reconcile: cannot remove %s: %s", d, e)
                continue
            if pod_uid in self.pod_masks or not os.path.isfile(rec_path):
                continue
            try:
                with open(rec_path) as f:
                    rec = _json.load(f)
            except (OSError, ValueError) as e:
                log.warning("reconcile: bad record %s: %s", rec_path, e)
                continue
            for dev in rec.get("devices", []):
                mask = int(dev.get("cu_mask") or "0", 16)
                if not mask:
                    continue
                if self.cumask.adopt(dev["uuid"], mask):
                    self.pod_masks.setdefault(rec["pod_uid"], []).append(
                        (dev["uuid"], mask))
                else:
                    log.warning("reconcile: mask conflict for pod %s dev %s",
                                rec["pod_uid"], dev["uuid"])

    def release_pod(self, pod_uid: str) -> None:
        for uuid, mask in self.pod_masks.pop(pod_uid, []):
            self.cumask.free(uuid, mask)

    # ---- serving / registration -----------------------------------------
    @property
    def socket_path(self) -> str:
        return os.path.join(self.cfg.plugin_socket_dir, self.endpoint)

    def serve(self) -> None:
        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=8))
        self._server.add_generic_rpc_handlers((dp.device_plugin_service(self),))
        if os.path.exists(self.socket_path):
            os.unlink(self.socket_path)
        self._server.add_insecure_port(f"unix://{self.socket_path}")
        self._server.start()
        log.info("device plugin serving on %s", self.socket_path)

    def register_with_kubelet(self) -> None:
        with grpc.insecure_channel(f"unix://{self.cfg.kubelet_socket}") as ch:
            client = dp.RegistrationClient(ch)
            client.Register(dp.RegisterRequest(
                version=dp.API_VERSION,
                endpoint=self.endpoint,
                resource_name=self.cfg.resource_name,
                options=dp.DevicePluginOptions(),
            ))
        log.info("registered %s with kubelet", self.cfg.resource_name)

    def stop(self) -> None:
        self._stop.set()
        self._update.set()
        if self._server:
            self._server.stop(grace=1)
