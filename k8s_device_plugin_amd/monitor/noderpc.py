"""Node vGPU-info gRPC service (the :9395 endpoint of the monitor).

Reference: cmd/vGPUmonitor/noderpc/noderpc.proto:25-61 + pathmonitor.go:
130-149 — the reference *serves* NodeVGPUInfo but with an Unimplemented
handler; here GetNodeVGPU is implemented for real: it renders every live
container's shared region (limits, CU limits, per-process usage) so node
tooling can query the enforcement state without scraping Prometheus.

Messages are built at import time from a hand-written FileDescriptorProto
(same approach as proto/deviceplugin.py — no grpcio-tools in the image).
"""
from __future__ import annotations

import logging
from concurrent import futures
from typing import Optional

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

log = logging.getLogger(__name__)

PKG = "noderpc"
_F = descriptor_pb2.FieldDescriptorProto


def _build_pool():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "k8s_device_plugin_amd/noderpc.proto"
    fdp.package = PKG
    fdp.syntax = "proto3"

    def msg(name):
        m = fdp.message_type.add()
        m.name = name
        return m

    def field(m, name, number, ftype, label=_F.LABEL_OPTIONAL, type_name=None):
        f = m.field.add()
        f.name, f.number, f.type, f.label = name, number, ftype, label
        if type_name:
            f.type_name = f".{PKG}.{type_name}"

    S, M, U64, I32, U32 = (_F.TYPE_STRING, _F.TYPE_MESSAGE, _F.TYPE_UINT64,
                           _F.TYPE_INT32, _F.TYPE_UINT32)
    REP = _F.LABEL_REPEATED

    m = msg("ProcSlot")
    field(m, "pid", 1, I32)
    field(m, "hostpid", 2, I32)
    field(m, "used", 3, U64, REP)
    field(m, "monitor_used", 4, U64, REP)

    m = msg("SharedRegion")
    field(m, "num_devices", 1, U32)
    field(m, "uuids", 2, S, REP)
    field(m, "limit", 3, U64, REP)
    field(m, "sm_limit", 4, U64, REP)
    field(m, "procs", 5, M, REP, type_name="ProcSlot")
    field(m, "utilization_switch", 6, I32)
    field(m, "recent_kernel", 7, I32)
    field(m, "priority", 8, I32)
    field(m, "oversubscribe", 9, U32)

    m = msg("PodUsage")
    field(m, "poduuid", 1, S)
    field(m, "container", 2, S)
    field(m, "podvgpuinfo", 3, M, type_name="SharedRegion")

    m = msg("GetNodeVGPURequest")
    field(m, "ctruuid", 1, S)  # optional filter: pod uid or "<uid>_<ctr>"

    m = msg("GetNodeVGPUReply")
    field(m, "nodeid", 1, S)
    field(m, "nodevgpuinfo", 2, M, REP, type_name="PodUsage")

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    return pool


_POOL = _build_pool()


def _cls(name):
    return message_factory.GetMessageClass(_POOL.FindMessageTypeByName(f"{PKG}.{name}"))


ProcSlot = _cls("ProcSlot")
SharedRegionMsg = _cls("SharedRegion")
PodUsage = _cls("PodUsage")
GetNodeVGPURequest = _cls("GetNodeVGPURequest")
GetNodeVGPUReply = _cls("GetNodeVGPUReply")


class NodeVGPUServicer:
    """Backed by the PathMonitor's live regions."""

    def __init__(self, pathmon, node_name: str = ""):
        self.pathmon = pathmon
        self.node_name = node_name

    def GetNodeVGPU(self, request, context):
        reply = GetNodeVGPUReply(nodeid=self.node_name)
        want = request.ctruuid
        for entry in self.pathmon.live_regions():
            if want and want not in (entry.pod_uid, entry.key):
                continue
            try:
                snap = entry.region.snapshot()
            except Exception as e:
                log.warning("snapshot %s failed: %s", entry.key, e)
                continue
            sr = SharedRegionMsg(
                num_devices=snap.num_devices,
                uuids=snap.uuids,
                limit=snap.limit,
                sm_limit=snap.sm_limit,
                utilization_switch=snap.utilization_switch,
                recent_kernel=snap.recent_kernel,
                priority=snap.priority,
                oversubscribe=snap.oversubscribe,
            )
            for p in snap.procs:
                sr.procs.append(ProcSlot(pid=p.pid, hostpid=p.host_pid,
                                         used=p.used_bytes,
                                         monitor_used=p.monitor_used))
            reply.nodevgpuinfo.append(
                PodUsage(poduuid=entry.pod_uid, container=entry.container,
                         podvgpuinfo=sr))
        return reply


def node_vgpu_service(servicer) -> grpc.GenericRpcHandler:
    return grpc.method_handlers_generic_handler(
        f"{PKG}.NodeVGPUInfo",
        {
            "GetNodeVGPU": grpc.unary_unary_rpc_method_handler(
                servicer.GetNodeVGPU,
                request_deserializer=GetNodeVGPURequest.FromString,
                response_serializer=GetNodeVGPUReply.SerializeToString),
        },
    )


def serve(pathmon, node_name: str = "", bind: str = "0.0.0.0:9395",
          max_workers: int = 4):
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    server.add_generic_rpc_handlers((node_vgpu_service(
        NodeVGPUServicer(pathmon, node_name)),))
    port = server.add_insecure_port(bind)
    server.start()
    return server, port


class NodeVGPUClient:
    def __init__(self, channel: grpc.Channel):
        self._get = channel.unary_unary(
            f"/{PKG}.NodeVGPUInfo/GetNodeVGPU",
            request_serializer=GetNodeVGPURequest.SerializeToString,
            response_deserializer=GetNodeVGPUReply.FromString)

    def get_node_vgpu(self, ctruuid: str = "", timeout: float = 5.0):
        return self._get(GetNodeVGPURequest(ctruuid=ctruuid), timeout=timeout)
