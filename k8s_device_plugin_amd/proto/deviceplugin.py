"""Kubelet Device Plugin API v1beta1 — dynamic protobuf + gRPC plumbing.

The image has grpcio + protobuf but no grpcio-tools codegen, so the message
types are built at import time from hand-written FileDescriptorProto specs
(the wire format is defined by k8s's deviceplugin/v1beta1/api.proto; this is
a from-scratch descriptor of that public API, not generated code).

Exports message classes plus server/client helpers:
- ``device_plugin_service(servicer)``: generic handler for the DevicePlugin
  service (used with grpc.server).
- ``registration_service(servicer)``: the kubelet's Registration service
  (used by the stub kubelet in tests — BASELINE config 1).
- ``RegistrationClient`` / ``DevicePluginClient``: typed stubs.
"""
from __future__ import annotations

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

PKG = "v1beta1"
API_VERSION = "v1beta1"
DEVICE_PLUGIN_PATH = "/var/lib/kubelet/device-plugins/"
KUBELET_SOCKET = DEVICE_PLUGIN_PATH + "kubelet.sock"
HEALTHY = "Healthy"
UNHEALTHY = "Unhealthy"

_F = descriptor_pb2.FieldDescriptorProto


def _build_pool():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "k8s_device_plugin_amd/deviceplugin.proto"
    fdp.package = PKG
    fdp.syntax = "proto3"

    def msg(name):
        m = fdp.message_type.add()
        m.name = name
        return m

    def field(m, name, number, ftype, label=_F.LABEL_OPTIONAL, type_name=None):
        f = m.field.add()
        f.name = name
        f.number = number
        f.type = ftype
        f.label = label
        if type_name:
            f.type_name = f".{PKG}.{type_name}"
        return f

    def map_field(m, name, number):
        # a map<string,string> is a repeated nested MapEntry message
        entry = m.nested_type.add()
        entry.name = "".join(p.capitalize() for p in name.split("_")) + "Entry"
        entry.options.map_entry = True
        k = entry.field.add()
        k.name, k.number, k.type, k.label = "key", 1, _F.TYPE_STRING, _F.LABEL_OPTIONAL
        v = entry.field.add()
        v.name, v.number, v.type, v.label = "value", 2, _F.TYPE_STRING, _F.LABEL_OPTIONAL
        f = m.field.add()
        f.name = name
        f.number = number
        f.type = _F.TYPE_MESSAGE
        f.label = _F.LABEL_REPEATED
        f.type_name = f".{PKG}.{m.name}.{entry.name}"

    S, M, B, I64, I32 = (_F.TYPE_STRING, _F.TYPE_MESSAGE, _F.TYPE_BOOL,
                         _F.TYPE_INT64, _F.TYPE_INT32)
    REP = _F.LABEL_REPEATED

    msg("Empty")

    m = msg("DevicePluginOptions")
    field(m, "pre_start_required", 1, B)
    field(m, "get_preferred_allocation_available", 2, B)

    m = msg("RegisterRequest")
    field(m, "version", 1, S)
    field(m, "endpoint", 2, S)
    field(m, "resource_name", 3, S)
    field(m, "options", 4, M, type_name="DevicePluginOptions")

    m = msg("NUMANode")
    field(m, "ID", 1, I64)

    m = msg("TopologyInfo")
    field(m, "nodes", 1, M, REP, type_name="NUMANode")

    m = msg("Device")
    field(m, "ID", 1, S)
    field(m, "health", 2, S)
    field(m, "topology", 3, M, type_name="TopologyInfo")

    m = msg("ListAndWatchResponse")
    field(m, "devices", 1, M, REP, type_name="Device")

    m = msg("ContainerAllocateRequest")
    field(m, "devicesIDs", 1, S, REP)

    m = msg("AllocateRequest")
    field(m, "container_requests", 1, M, REP, type_name="ContainerAllocateRequest")

    m = msg("Mount")
    field(m, "container_path", 1, S)
    field(m, "host_path", 2, S)
    field(m, "read_only", 3, B)

    m = msg("DeviceSpec")
    field(m, "container_path", 1, S)
    field(m, "host_path", 2, S)
    field(m, "permissions", 3, S)

    m = msg("ContainerAllocateResponse")
    map_field(m, "envs", 1)
    field(m, "mounts", 2, M, REP, type_name="Mount")
    field(m, "devices", 3, M, REP, type_name="DeviceSpec")
    map_field(m, "annotations", 4)

    m = msg("AllocateResponse")
    field(m, "container_responses", 1, M, REP, type_name="ContainerAllocateResponse")

    m = msg("PreStartContainerRequest")
    field(m, "devicesIDs", 1, S, REP)

    msg("PreStartContainerResponse")

    m = msg("ContainerPreferredAllocationRequest")
    field(m, "available_deviceIDs", 1, S, REP)
    field(m, "must_include_deviceIDs", 2, S, REP)
    field(m, "allocation_size", 3, I32)

    m = msg("PreferredAllocationRequest")
    field(m, "container_requests", 1, M, REP,
          type_name="ContainerPreferredAllocationRequest")

    m = msg("ContainerPreferredAllocationResponse")
    field(m, "deviceIDs", 1, S, REP)

    m = msg("PreferredAllocationResponse")
    field(m, "container_responses", 1, M, REP,
          type_name="ContainerPreferredAllocationResponse")

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    return pool


_pool = _build_pool()


def _cls(name):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(f"{PKG}.{name}"))


Empty = _cls("Empty")
DevicePluginOptions = _cls("DevicePluginOptions")
RegisterRequest = _cls("RegisterRequest")
NUMANode = _cls("NUMANode")
TopologyInfo = _cls("TopologyInfo")
Device = _cls("Device")
ListAndWatchResponse = _cls("ListAndWatchResponse")
ContainerAllocateRequest = _cls("ContainerAllocateRequest")
AllocateRequest = _cls("AllocateRequest")
Mount = _cls("Mount")
DeviceSpec = _cls("DeviceSpec")
ContainerAllocateResponse = _cls("ContainerAllocateResponse")
AllocateResponse = _cls("AllocateResponse")
PreStartContainerRequest = _cls("PreStartContainerRequest")
PreStartContainerResponse = _cls("PreStartContainerResponse")
PreferredAllocationRequest = _cls("PreferredAllocationRequest")
PreferredAllocationResponse = _cls("PreferredAllocationResponse")
ContainerPreferredAllocationRequest = _cls("ContainerPreferredAllocationRequest")
ContainerPreferredAllocationResponse = _cls("ContainerPreferredAllocationResponse")


def _unary(fn, req_cls, resp_cls):
    return grpc.unary_unary_rpc_method_handler(
        fn, request_deserializer=req_cls.FromString,
        response_serializer=resp_cls.SerializeToString)


def _stream(fn, req_cls, resp_cls):
    return grpc.unary_stream_rpc_method_handler(
        fn, request_deserializer=req_cls.FromString,
        response_serializer=resp_cls.SerializeToString)


def device_plugin_service(servicer) -> grpc.GenericRpcHandler:
    """servicer needs: GetDevicePluginOptions, ListAndWatch (generator),
    Allocate, PreStartContainer, GetPreferredAllocation — all (request,
    context) like generated stubs."""
    return grpc.method_handlers_generic_handler(
        f"{PKG}.DevicePlugin",
        {
            "GetDevicePluginOptions": _unary(
                servicer.GetDevicePluginOptions, Empty, DevicePluginOptions),
            "ListAndWatch": _stream(
                servicer.ListAndWatch, Empty, ListAndWatchResponse),
            "Allocate": _unary(servicer.Allocate, AllocateRequest, AllocateResponse),
            "PreStartContainer": _unary(
                servicer.PreStartContainer, PreStartContainerRequest,
                PreStartContainerResponse),
            "GetPreferredAllocation": _unary(
                servicer.GetPreferredAllocation, PreferredAllocationRequest,
                PreferredAllocationResponse),
        },
    )


def registration_service(servicer) -> grpc.GenericRpcHandler:
    """The kubelet side (our stub kubelet implements this in tests)."""
    return grpc.method_handlers_generic_handler(
        f"{PKG}.Registration",
        {"Register": _unary(servicer.Register, RegisterRequest, Empty)},
    )


class RegistrationClient:
    def __init__(self, channel: grpc.Channel):
        self._register = channel.unary_unary(
            f"/{PKG}.Registration/Register",
            request_serializer=RegisterRequest.SerializeToString,
            response_deserializer=Empty.FromString)

    def Register(self, req, timeout=10):
        return self._register(req, timeout=timeout)


class DevicePluginClient:
    def __init__(self, channel: grpc.Channel):
        p = f"/{PKG}.DevicePlugin/"
        self.GetDevicePluginOptions = channel.unary_unary(
            p + "GetDevicePluginOptions",
            request_serializer=Empty.SerializeToString,
            response_deserializer=DevicePluginOptions.FromString)
        self.ListAndWatch = channel.unary_stream(
            p + "ListAndWatch",
            request_serializer=Empty.SerializeToString,
            response_deserializer=ListAndWatchResponse.FromString)
        self.Allocate = channel.unary_unary(
            p + "Allocate",
            request_serializer=AllocateRequest.SerializeToString,
            response_deserializer=AllocateResponse.FromString)
        self.PreStartContainer = channel.unary_unary(
            p + "PreStartContainer",
            request_serializer=PreStartContainerRequest.SerializeToString,
            response_deserializer=PreStartContainerResponse.FromString)
        self.GetPreferredAllocation = channel.unary_unary(
            p + "GetPreferredAllocation",
            request_serializer=PreferredAllocationRequest.SerializeToString,
            response_deserializer=PreferredAllocationResponse.FromString)
