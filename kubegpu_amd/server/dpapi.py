"""Kubernetes device-plugin API (v1beta1) — runtime-built protobuf.

The image has the protobuf runtime but no protoc, so the kubelet
device-plugin messages are constructed at runtime from
FileDescriptorProto.  Message/field names and numbers follow
k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto (the stable
v1beta1 surface).  In-repo tests exercise our own client+server over a
unix socket, so the wire format is self-consistent end to end.

This is the *stock-kubelet* serving path the reference never had (it
plugs into the custom KubeDevice core instead, which kubegpu_amd.core
re-owns); GetPreferredAllocation is where the xGMI subset scorer meets
vanilla Kubernetes.
"""

from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_PKG = "v1beta1"

# kubelet contract constants
DEVICE_PLUGIN_PATH = "/var/lib/kubelet/device-plugins"
KUBELET_SOCKET = DEVICE_PLUGIN_PATH + "/kubelet.sock"
VERSION = "v1beta1"
HEALTHY = "Healthy"
UNHEALTHY = "Unhealthy"

_T = descriptor_pb2.FieldDescriptorProto


def _msg(fdp, name, fields, maps=(), nested=None):
    """fields: (name, number, type, label, type_name)"""
    m = fdp.message_type.add()
    m.name = name
    for fname, num, ftype, label, type_name in fields:
        f = m.field.add()
        f.name = fname
        f.number = num
        f.type = ftype
        f.label = label
        if type_name:
            f.type_name = type_name
    for fname, num in maps:
        # map<string,string> == repeated nested XEntry {key,value} with
        # map_entry option
        entry = m.nested_type.add()
        entry.name = "".join(p.capitalize() for p in fname.split("_")) + "Entry"
        entry.options.map_entry = True
        k = entry.field.add()
        k.name = "key"; k.number = 1; k.type = _T.TYPE_STRING; k.label = _T.LABEL_OPTIONAL
        v = entry.field.add()
        v.name = "value"; v.number = 2; v.type = _T.TYPE_STRING; v.label = _T.LABEL_OPTIONAL
        f = m.field.add()
        f.name = fname
        f.number = num
        f.type = _T.TYPE_MESSAGE
        f.label = _T.LABEL_REPEATED
        f.type_name = f".{_PKG}.{name}.{entry.name}"
    return m


def _build():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "deviceplugin_v1beta1.proto"
    fdp.package = _PKG
    fdp.syntax = "proto3"
    S, M, I32, I64, B = (_T.TYPE_STRING, _T.TYPE_MESSAGE, _T.TYPE_INT32,
                         _T.TYPE_INT64, _T.TYPE_BOOL)
    OPT, REP = _T.LABEL_OPTIONAL, _T.LABEL_REPEATED
    ref = lambda n: f".{_PKG}.{n}"

    _msg(fdp, "Empty", [])
    _msg(fdp, "DevicePluginOptions", [
        ("pre_start_required", 1, B, OPT, None),
        ("get_preferred_allocation_available", 2, B, OPT, None),
    ])
    _msg(fdp, "RegisterRequest", [
        ("version", 1, S, OPT, None),
        ("endpoint", 2, S, OPT, None),
        ("resource_name", 3, S, OPT, None),
        ("options", 4, M, OPT, ref("DevicePluginOptions")),
    ])
    _msg(fdp, "NUMANode", [("ID", 1, I64, OPT, None)])
    _msg(fdp, "TopologyInfo", [("nodes", 1, M, REP, ref("NUMANode"))])
    _msg(fdp, "Device", [
        ("ID", 1, S, OPT, None),
        ("health", 2, S, OPT, None),
        ("topology", 3, M, OPT, ref("TopologyInfo")),
    ])
    _msg(fdp, "ListAndWatchResponse", [("devices", 1, M, REP, ref("Device"))])
    _msg(fdp, "ContainerPreferredAllocationRequest", [
        ("available_deviceIDs", 1, S, REP, None),
        ("must_include_deviceIDs", 2, S, REP, None),
        ("allocation_size", 3, I32, OPT, None),
    ])
    _msg(fdp, "PreferredAllocationRequest", [
        ("container_requests", 1, M, REP, ref("ContainerPreferredAllocationRequest")),
    ])
    _msg(fdp, "ContainerPreferredAllocationResponse", [
        ("deviceIDs", 1, S, REP, None),
    ])
    _msg(fdp, "PreferredAllocationResponse", [
        ("container_responses", 1, M, REP, ref("ContainerPreferredAllocationResponse")),
    ])
    _msg(fdp, "ContainerAllocateRequest", [("devicesIDs", 1, S, REP, None)])
    _msg(fdp, "AllocateRequest", [
        ("container_requests", 1, M, REP, ref("ContainerAllocateRequest")),
    ])
    _msg(fdp, "Mount", [
        ("container_path", 1, S, OPT, None),
        ("host_path", 2, S, OPT, None),
        ("read_only", 3, B, OPT, None),
    ])
    _msg(fdp, "DeviceSpec", [
        ("container_path", 1, S, OPT, None),
        ("host_path", 2, S, OPT, None),
        ("permissions", 3, S, OPT, None),
    ])
    _msg(fdp, "ContainerAllocateResponse", [
        ("mounts", 2, M, REP, ref("Mount")),
        ("devices", 3, M, REP, ref("DeviceSpec")),
    ], maps=[("envs", 1), ("annotations", 4)])
    _msg(fdp, "AllocateResponse", [
        ("container_responses", 1, M, REP, ref("ContainerAllocateResponse")),
    ])
    _msg(fdp, "PreStartContainerRequest", [("devicesIDs", 1, S, REP, None)])
    _msg(fdp, "PreStartContainerResponse", [])

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)

    out = {}
    for mt in fdp.message_type:
        out[mt.name] = message_factory.GetMessageClass(
            pool.FindMessageTypeByName(f"{_PKG}.{mt.name}")
        )
    return out


_MESSAGES = _build()

Empty = _MESSAGES["Empty"]
DevicePluginOptions = _MESSAGES["DevicePluginOptions"]
RegisterRequest = _MESSAGES["RegisterRequest"]
NUMANode = _MESSAGES["NUMANode"]
TopologyInfo = _MESSAGES["TopologyInfo"]
Device = _MESSAGES["Device"]
ListAndWatchResponse = _MESSAGES["ListAndWatchResponse"]
ContainerPreferredAllocationRequest = _MESSAGES["ContainerPreferredAllocationRequest"]
PreferredAllocationRequest = _MESSAGES["PreferredAllocationRequest"]
ContainerPreferredAllocationResponse = _MESSAGES["ContainerPreferredAllocationResponse"]
PreferredAllocationResponse = _MESSAGES["PreferredAllocationResponse"]
ContainerAllocateRequest = _MESSAGES["ContainerAllocateRequest"]
AllocateRequest = _MESSAGES["AllocateRequest"]
Mount = _MESSAGES["Mount"]
DeviceSpec = _MESSAGES["DeviceSpec"]
ContainerAllocateResponse = _MESSAGES["ContainerAllocateResponse"]
AllocateResponse = _MESSAGES["AllocateResponse"]
PreStartContainerRequest = _MESSAGES["PreStartContainerRequest"]
PreStartContainerResponse = _MESSAGES["PreStartContainerResponse"]

DEVICE_PLUGIN_SERVICE = f"{_PKG}.DevicePlugin"
REGISTRATION_SERVICE = f"{_PKG}.Registration"
