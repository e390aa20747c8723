"""Runtime-built protobuf messages for the NRI v1alpha1 API subset the
injector needs.  Field numbers mirror the public containerd NRI schema
(package nri.pkg.api.v1alpha1; verified against the vendored api.proto in
the reference: vendor/github.com/containerd/nri/pkg/api/api.proto)."""
from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

F = descriptor_pb2.FieldDescriptorProto
OPT = F.LABEL_OPTIONAL
REP = F.LABEL_REPEATED

PKG = "nri.pkg.api.v1alpha1"
_pool = descriptor_pool.DescriptorPool()


def _msg(fd, name, fields, nested=None):
    d = fd.message_type.add()
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
        nd = d.nested_type.add()
        nd.CopyFrom(n)
    return d


def _map_entry(name):
    d = descriptor_pb2.DescriptorProto()
    d.name = name
    for fname, num in (("key", 1), ("value", 2)):
        f = d.field.add()
        f.name = fname
        f.number = num
        f.type = F.TYPE_STRING
        f.label = OPT
    d.options.map_entry = True
    return d


fd = descriptor_pb2.FileDescriptorProto()
fd.name = "nri_v1alpha1.proto"
fd.package = PKG
fd.syntax = "proto3"

_msg(fd, "Empty", [])
_msg(fd, "RegisterPluginRequest", [
    ("plugin_name", 1, F.TYPE_STRING, OPT, None),
    ("plugin_idx", 2, F.TYPE_STRING, OPT, None),
])
_msg(fd, "ConfigureRequest", [
    ("config", 1, F.TYPE_STRING, OPT, None),
    ("runtime_name", 2, F.TYPE_STRING, OPT, None),
    ("runtime_version", 3, F.TYPE_STRING, OPT, None),
])
_msg(fd, "ConfigureResponse", [
    ("events", 2, F.TYPE_INT32, OPT, None),
])
_msg(fd, "SynchronizeRequest", [
    ("pods", 1, F.TYPE_MESSAGE, REP, f".{PKG}.PodSandbox"),
    ("containers", 2, F.TYPE_MESSAGE, REP, f".{PKG}.Container"),
])
_msg(fd, "SynchronizeResponse", [])
_msg(fd, "CreateContainerRequest", [
    ("pod", 1, F.TYPE_MESSAGE, OPT, f".{PKG}.PodSandbox"),
    ("container", 2, F.TYPE_MESSAGE, OPT, f".{PKG}.Container"),
])
_msg(fd, "CreateContainerResponse", [
    ("adjust", 1, F.TYPE_MESSAGE, OPT, f".{PKG}.ContainerAdjustment"),
])
_msg(fd, "UpdateContainerRequest", [
    ("pod", 1, F.TYPE_MESSAGE, OPT, f".{PKG}.PodSandbox"),
    ("container", 2, F.TYPE_MESSAGE, OPT, f".{PKG}.Container"),
])
_msg(fd, "UpdateContainerResponse", [])
_msg(fd, "StopContainerRequest", [
    ("pod", 1, F.TYPE_MESSAGE, OPT, f".{PKG}.PodSandbox"),
    ("container", 2, F.TYPE_MESSAGE, OPT, f".{PKG}.Container"),
])
_msg(fd, "StopContainerResponse", [])
_msg(fd, "StateChangeEvent", [
    ("event", 1, F.TYPE_INT32, OPT, None),
])
_msg(fd, "PodSandbox", [
    ("id", 1, F.TYPE_STRING, OPT, None),
    ("name", 2, F.TYPE_STRING, OPT, None),
    ("uid", 3, F.TYPE_STRING, OPT, None),
    ("namespace", 4, F.TYPE_STRING, OPT, None),
    ("labels", 5, F.TYPE_MESSAGE, REP, f".{PKG}.PodSandbox.LabelsEntry"),
    ("annotations", 6, F.TYPE_MESSAGE, REP, f".{PKG}.PodSandbox.AnnotationsEntry"),
], nested=[_map_entry("LabelsEntry"), _map_entry("AnnotationsEntry")])
_msg(fd, "Container", [
    ("id", 1, F.TYPE_STRING, OPT, None),
    ("pod_sandbox_id", 2, F.TYPE_STRING, OPT, None),
    ("name", 3, F.TYPE_STRING, OPT, None),
])
_msg(fd, "ContainerAdjustment", [
    ("annotations", 2, F.TYPE_MESSAGE, REP,
     f".{PKG}.ContainerAdjustment.AnnotationsEntry"),
    ("linux", 6, F.TYPE_MESSAGE, OPT, f".{PKG}.LinuxContainerAdjustment"),
], nested=[_map_entry("AnnotationsEntry")])
_msg(fd, "LinuxContainerAdjustment", [
    ("devices", 1, F.TYPE_MESSAGE, REP, f".{PKG}.LinuxDevice"),
])
_msg(fd, "OptionalFileMode", [("value", 1, F.TYPE_UINT32, OPT, None)])
_msg(fd, "OptionalUInt32", [("value", 1, F.TYPE_UINT32, OPT, None)])
_msg(fd, "LinuxDevice", [
    ("path", 1, F.TYPE_STRING, OPT, None),
    ("type", 2, F.TYPE_STRING, OPT, None),
    ("major", 3, F.TYPE_INT64, OPT, None),
    ("minor", 4, F.TYPE_INT64, OPT, None),
    ("file_mode", 5, F.TYPE_MESSAGE, OPT, f".{PKG}.OptionalFileMode"),
    ("uid", 6, F.TYPE_MESSAGE, OPT, f".{PKG}.OptionalUInt32"),
    ("gid", 7, F.TYPE_MESSAGE, OPT, f".{PKG}.OptionalUInt32"),
])

_pool.Add(fd)


def _cls(name):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(f"{PKG}.{name}"))


Empty = _cls("Empty")
RegisterPluginRequest = _cls("RegisterPluginRequest")
ConfigureRequest = _cls("ConfigureRequest")
ConfigureResponse = _cls("ConfigureResponse")
SynchronizeRequest = _cls("SynchronizeRequest")
SynchronizeResponse = _cls("SynchronizeResponse")
CreateContainerRequest = _cls("CreateContainerRequest")
CreateContainerResponse = _cls("CreateContainerResponse")
UpdateContainerRequest = _cls("UpdateContainerRequest")
UpdateContainerResponse = _cls("UpdateContainerResponse")
StopContainerRequest = _cls("StopContainerRequest")
StopContainerResponse = _cls("StopContainerResponse")
StateChangeEvent = _cls("StateChangeEvent")
PodSandbox = _cls("PodSandbox")
Container = _cls("Container")
ContainerAdjustment = _cls("ContainerAdjustment")
LinuxContainerAdjustment = _cls("LinuxContainerAdjustment")
LinuxDevice = _cls("LinuxDevice")
OptionalFileMode = _cls("OptionalFileMode")
OptionalUInt32 = _cls("OptionalUInt32")

RUNTIME_SERVICE = f"{PKG}.Runtime"
PLUGIN_SERVICE = f"{PKG}.Plugin"
DEFAULT_SOCKET_PATH = "/var/run/nri/nri.sock"

# Event enum values (api.proto enum Event); ConfigureResponse.events is a
# bitmask with bit (event-1) set.
EVENT_CREATE_CONTAINER = 4


def event_mask(*events: int) -> int:
    m = 0
    for e in events:
        m |= 1 << (e - 1)
    return m
