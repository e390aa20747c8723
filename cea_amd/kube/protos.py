"""Runtime-built protobuf messages for the kubelet device-plugin v1beta1 and
pod-resources APIs.

The reference consumes these APIs through generated Go stubs
(k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1, used from
/root/reference/pkg/gpu/nvidia/beta_plugin.go:39-131 and
/root/reference/pkg/gpu/nvidia/metrics/devices.go:51-101).  This image has the
protobuf + grpcio runtimes but no protoc/grpc_tools, so we build the exact same
wire-compatible message types at import time from hand-written
FileDescriptorProto definitions.  Field numbers/types below mirror the
upstream kubelet api.proto files byte-for-byte on the wire.
"""
from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

F = descriptor_pb2.FieldDescriptorProto

_pool = descriptor_pool.DescriptorPool()


def _msg(name, fields, nested=None):
    """(name, [(fname, number, type, label, type_name_or_None)]) -> DescriptorProto."""
    d = descriptor_pb2.DescriptorProto()
    d.name = name
    for fname, number, ftype, label, type_name in fields:
        f = d.field.add()
        f.name = fname
        f.number = number
        f.type = ftype
        f.label = label
        if type_name:
            f.type_name = type_name
    for n in nested or []:
        d.nested_type.add().CopyFrom(n)
    return d


def _map_entry(name, value_type=F.TYPE_STRING):
    """Synthesize the nested MapEntry message for a map<string, V> field."""
    d = _msg(
        name,
        [
            ("key", 1, F.TYPE_STRING, F.LABEL_OPTIONAL, None),
            ("value", 2, value_type, F.LABEL_OPTIONAL, None),
        ],
    )
    d.options.map_entry = True
    return d


OPT = F.LABEL_OPTIONAL
REP = F.LABEL_REPEATED


def _build_deviceplugin_file():
    fd = descriptor_pb2.FileDescriptorProto()
    fd.name = "deviceplugin_v1beta1.proto"
    fd.package = "v1beta1"
    fd.syntax = "proto3"

    fd.message_type.add().CopyFrom(
        _msg(
            "DevicePluginOptions",
            [
                ("pre_start_required", 1, F.TYPE_BOOL, OPT, None),
                ("get_preferred_allocation_available", 2, F.TYPE_BOOL, OPT, None),
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "RegisterRequest",
            [
                ("version", 1, F.TYPE_STRING, OPT, None),
                ("endpoint", 2, F.TYPE_STRING, OPT, None),
                ("resource_name", 3, F.TYPE_STRING, OPT, None),
                ("options", 4, F.TYPE_MESSAGE, OPT, ".v1beta1.DevicePluginOptions"),
            ],
        )
    )
    fd.message_type.add().CopyFrom(_msg("Empty", []))
    fd.message_type.add().CopyFrom(
        _msg(
            "ListAndWatchResponse",
            [("devices", 1, F.TYPE_MESSAGE, REP, ".v1beta1.Device")],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg("NUMANode", [("ID", 1, F.TYPE_INT64, OPT, None)])
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "TopologyInfo",
            [("nodes", 1, F.TYPE_MESSAGE, REP, ".v1beta1.NUMANode")],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "Device",
            [
                ("ID", 1, F.TYPE_STRING, OPT, None),
                ("health", 2, F.TYPE_STRING, OPT, None),
                ("topology", 3, F.TYPE_MESSAGE, OPT, ".v1beta1.TopologyInfo"),
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "ContainerAllocateRequest",
            [("devices_ids", 1, F.TYPE_STRING, REP, None)],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "AllocateRequest",
            [
                (
                    "container_requests",
                    1,
                    F.TYPE_MESSAGE,
                    REP,
                    ".v1beta1.ContainerAllocateRequest",
                )
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "Mount",
            [
                ("container_path", 1, F.TYPE_STRING, OPT, None),
                ("host_path", 2, F.TYPE_STRING, OPT, None),
                ("read_only", 3, F.TYPE_BOOL, OPT, None),
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "DeviceSpec",
            [
                ("container_path", 1, F.TYPE_STRING, OPT, None),
                ("host_path", 2, F.TYPE_STRING, OPT, None),
                ("permissions", 3, F.TYPE_STRING, OPT, None),
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "CDIDevice",
            [("name", 1, F.TYPE_STRING, OPT, None)],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "ContainerAllocateResponse",
            [
                ("envs", 1, F.TYPE_MESSAGE, REP, ".v1beta1.ContainerAllocateResponse.EnvsEntry"),
                ("mounts", 2, F.TYPE_MESSAGE, REP, ".v1beta1.Mount"),
                ("devices", 3, F.TYPE_MESSAGE, REP, ".v1beta1.DeviceSpec"),
                (
                    "annotations",
                    4,
                    F.TYPE_MESSAGE,
                    REP,
                    ".v1beta1.ContainerAllocateResponse.AnnotationsEntry",
                ),
                ("cdi_devices", 5, F.TYPE_MESSAGE, REP, ".v1beta1.CDIDevice"),
            ],
            nested=[_map_entry("EnvsEntry"), _map_entry("AnnotationsEntry")],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "AllocateResponse",
            [
                (
                    "container_responses",
                    1,
                    F.TYPE_MESSAGE,
                    REP,
                    ".v1beta1.ContainerAllocateResponse",
                )
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "PreStartContainerRequest",
            [("devices_ids", 1, F.TYPE_STRING, REP, None)],
        )
    )
    fd.message_type.add().CopyFrom(_msg("PreStartContainerResponse", []))
    fd.message_type.add().CopyFrom(
        _msg(
            "ContainerPreferredAllocationRequest",
            [
                ("available_deviceIDs", 1, F.TYPE_STRING, REP, None),
                ("must_include_deviceIDs", 2, F.TYPE_STRING, REP, None),
                ("allocation_size", 3, F.TYPE_INT32, OPT, None),
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "PreferredAllocationRequest",
            [
                (
                    "container_requests",
                    1,
                    F.TYPE_MESSAGE,
                    REP,
                    ".v1beta1.ContainerPreferredAllocationRequest",
                )
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "ContainerPreferredAllocationResponse",
            [("device_ids", 1, F.TYPE_STRING, REP, None)],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "PreferredAllocationResponse",
            [
                (
                    "container_responses",
                    1,
                    F.TYPE_MESSAGE,
                    REP,
                    ".v1beta1.ContainerPreferredAllocationResponse",
                )
            ],
        )
    )
    return fd


def _build_podresources_file():
    fd = descriptor_pb2.FileDescriptorProto()
    fd.name = "podresources_v1.proto"
    fd.package = "podresources.v1"
    fd.syntax = "proto3"

    fd.message_type.add().CopyFrom(_msg("ListPodResourcesRequest", []))
    fd.message_type.add().CopyFrom(
        _msg(
            "TopologyInfo",
            [("nodes", 1, F.TYPE_MESSAGE, REP, ".podresources.v1.NUMANode")],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg("NUMANode", [("ID", 1, F.TYPE_INT64, OPT, None)])
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "ContainerDevices",
            [
                ("resource_name", 1, F.TYPE_STRING, OPT, None),
                ("device_ids", 2, F.TYPE_STRING, REP, None),
                ("topology", 3, F.TYPE_MESSAGE, OPT, ".podresources.v1.TopologyInfo"),
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "ContainerResources",
            [
                ("name", 1, F.TYPE_STRING, OPT, None),
                ("devices", 2, F.TYPE_MESSAGE, REP, ".podresources.v1.ContainerDevices"),
                ("cpu_ids", 3, F.TYPE_INT64, REP, None),
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "PodResources",
            [
                ("name", 1, F.TYPE_STRING, OPT, None),
                ("namespace", 2, F.TYPE_STRING, OPT, None),
                ("containers", 3, F.TYPE_MESSAGE, REP, ".podresources.v1.ContainerResources"),
            ],
        )
    )
    fd.message_type.add().CopyFrom(
        _msg(
            "ListPodResourcesResponse",
            [("pod_resources", 1, F.TYPE_MESSAGE, REP, ".podresources.v1.PodResources")],
        )
    )
    return fd


_pool.Add(_build_deviceplugin_file())
_pool.Add(_build_podresources_file())


def _cls(full_name):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(full_name))


# --- device plugin v1beta1 -------------------------------------------------
DEVICE_PLUGIN_VERSION = "v1beta1"
KUBELET_SOCKET = "kubelet.sock"

DevicePluginOptions = _cls("v1beta1.DevicePluginOptions")
RegisterRequest = _cls("v1beta1.RegisterRequest")
Empty = _cls("v1beta1.Empty")
ListAndWatchResponse = _cls("v1beta1.ListAndWatchResponse")
TopologyInfo = _cls("v1beta1.TopologyInfo")
NUMANode = _cls("v1beta1.NUMANode")
Device = _cls("v1beta1.Device")
AllocateRequest = _cls("v1beta1.AllocateRequest")
ContainerAllocateRequest = _cls("v1beta1.ContainerAllocateRequest")
AllocateResponse = _cls("v1beta1.AllocateResponse")
ContainerAllocateResponse = _cls("v1beta1.ContainerAllocateResponse")
Mount = _cls("v1beta1.Mount")
DeviceSpec = _cls("v1beta1.DeviceSpec")
CDIDevice = _cls("v1beta1.CDIDevice")
PreStartContainerRequest = _cls("v1beta1.PreStartContainerRequest")
PreStartContainerResponse = _cls("v1beta1.PreStartContainerResponse")
PreferredAllocationRequest = _cls("v1beta1.PreferredAllocationRequest")
ContainerPreferredAllocationRequest = _cls("v1beta1.ContainerPreferredAllocationRequest")
PreferredAllocationResponse = _cls("v1beta1.PreferredAllocationResponse")
ContainerPreferredAllocationResponse = _cls("v1beta1.ContainerPreferredAllocationResponse")

# kubelet device health strings (pluginapi.Healthy / pluginapi.Unhealthy)
HEALTHY = "Healthy"
UNHEALTHY = "Unhealthy"

# --- pod resources v1 ------------------------------------------------------
ListPodResourcesRequest = _cls("podresources.v1.ListPodResourcesRequest")
ListPodResourcesResponse = _cls("podresources.v1.ListPodResourcesResponse")
PodResources = _cls("podresources.v1.PodResources")
ContainerResources = _cls("podresources.v1.ContainerResources")
ContainerDevices = _cls("podresources.v1.ContainerDevices")

# gRPC method paths (generic stubs — no generated code in this image).
REGISTRATION_REGISTER = "/v1beta1.Registration/Register"
DP_GET_OPTIONS = "/v1beta1.DevicePlugin/GetDevicePluginOptions"
DP_LIST_AND_WATCH = "/v1beta1.DevicePlugin/ListAndWatch"
DP_ALLOCATE = "/v1beta1.DevicePlugin/Allocate"
DP_GET_PREFERRED_ALLOCATION = "/v1beta1.DevicePlugin/GetPreferredAllocation"
DP_PRE_START_CONTAINER = "/v1beta1.DevicePlugin/PreStartContainer"
PODRESOURCES_LIST = "/v1.PodResourcesLister/List"
