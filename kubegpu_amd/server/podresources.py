"""Kubelet pod-resources API (v1) client — in_use reconciliation.

The v1beta1 device-plugin API has no deallocate RPC, so a plugin can
never observe pod teardown through it (which is why the kubelet
Allocate handler does not touch GpuInfo.in_use).  Kubelet DOES expose
the truth elsewhere: the pod-resources API
(/var/lib/kubelet/pod-resources/kubelet.sock, service `v1.PodResources`)
lists every live pod's allocated devices.  This module is a minimal
client for its `List` RPC plus a reconciler that makes the manager's
in_use flags track kubelet's allocation state — set for devices in a
live pod, cleared for devices no longer listed.

Messages follow k8s.io/kubelet/pkg/apis/podresources/v1/api.proto
(runtime-built like dpapi: the image has no protoc).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional, Set

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

from ..api import utils
from ..deviceplugin.manager import AMDGPUManager
from ..plugintypes import RESOURCE_GPU

_PKG = "v1"
_T = descriptor_pb2.FieldDescriptorProto

POD_RESOURCES_PATH = "/var/lib/kubelet/pod-resources"
POD_RESOURCES_SOCKET = POD_RESOURCES_PATH + "/kubelet.sock"
POD_RESOURCES_SERVICE = f"{_PKG}.PodResources"


def _build():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "podresources_v1.proto"
    fdp.package = _PKG
    fdp.syntax = "proto3"
    S, M, I64 = _T.TYPE_STRING, _T.TYPE_MESSAGE, _T.TYPE_INT64
    OPT, REP = _T.LABEL_OPTIONAL, _T.LABEL_REPEATED
    ref = lambda n: f".{_PKG}.{n}"

    def msg(name, fields):
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

    msg("ListPodResourcesRequest", [])
    msg("NUMANode", [("ID", 1, I64, OPT, None)])
    msg("TopologyInfo", [("nodes", 1, M, REP, ref("NUMANode"))])
    msg("ContainerDevices", [
        ("resource_name", 1, S, OPT, None),
        ("device_ids", 2, S, REP, None),
        ("topology", 3, M, OPT, ref("TopologyInfo")),
    ])
    msg("ContainerResources", [
        ("name", 1, S, OPT, None),
        ("devices", 2, M, REP, ref("ContainerDevices")),
    ])
    msg("PodResources", [
        ("name", 1, S, OPT, None),
        ("namespace", 2, S, OPT, None),
        ("containers", 3, M, REP, ref("ContainerResources")),
    ])
    msg("ListPodResourcesResponse", [
        ("pod_resources", 1, M, REP, ref("PodResources")),
    ])
    msg("AllocatableResourcesRequest", [])
    msg("AllocatableResourcesResponse", [
        ("devices", 1, M, REP, ref("ContainerDevices")),
    ])

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    out = {}
    for mt in fdp.message_type:
        out[mt.name] = message_factory.GetMessageClass(
            pool.FindMessageTypeByName(f"{_PKG}.{mt.name}")
        )
    return out


_MESSAGES = _build()

ListPodResourcesRequest = _MESSAGES["ListPodResourcesRequest"]
ListPodResourcesResponse = _MESSAGES["ListPodResourcesResponse"]
AllocatableResourcesRequest = _MESSAGES["AllocatableResourcesRequest"]
AllocatableResourcesResponse = _MESSAGES["AllocatableResourcesResponse"]
ContainerDevices = _MESSAGES["ContainerDevices"]
ContainerResources = _MESSAGES["ContainerResources"]
PodResources = _MESSAGES["PodResources"]
PRNUMANode = _MESSAGES["NUMANode"]
PRTopologyInfo = _MESSAGES["TopologyInfo"]


class PodResourcesClient:
    """Client for kubelet's v1 PodResources List RPC."""

    def __init__(self, socket_path: str = POD_RESOURCES_SOCKET):
        self.socket_path = socket_path

    def available(self) -> bool:
        return os.path.exists(self.socket_path)

    def list(self, timeout_s: float = 10.0):
        channel = grpc.insecure_channel(f"unix://{self.socket_path}")
        try:
            rpc = channel.unary_unary(
                f"/{POD_RESOURCES_SERVICE}/List",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=ListPodResourcesResponse.FromString,
            )
            return rpc(ListPodResourcesRequest(), timeout=timeout_s)
        finally:
            channel.close()

    def get_allocatable_resources(self, timeout_s: float = 10.0):
        """kubelet's view of allocatable devices (v1
        GetAllocatableResources) — lets a monitor cross-check that the
        devices we advertise are the ones kubelet accounts."""
        channel = grpc.insecure_channel(f"unix://{self.socket_path}")
        try:
            rpc = channel.unary_unary(
                f"/{POD_RESOURCES_SERVICE}/GetAllocatableResources",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=AllocatableResourcesResponse.FromString,
            )
            return rpc(AllocatableResourcesRequest(), timeout=timeout_s)
        finally:
            channel.close()

    def allocated_device_ids(
        self, resource_name: str = RESOURCE_GPU, timeout_s: float = 10.0
    ) -> Set[str]:
        """Device ids of *resource_name* held by any live pod."""
        out: Set[str] = set()
        resp = self.list(timeout_s=timeout_s)
        for pod in resp.pod_resources:
            for cont in pod.containers:
                for dev in cont.devices:
                    if dev.resource_name == resource_name:
                        out.update(dev.device_ids)
        return out


def reconcile_in_use(
    manager: AMDGPUManager,
    client: Optional[PodResourcesClient] = None,
    resource_name: str = RESOURCE_GPU,
) -> Optional[Dict[str, List[str]]]:
    """Make manager in_use flags track kubelet's live allocations.

    Returns {"set": [...], "cleared": [...]} of uuids changed, or None
    when the pod-resources socket is unavailable (not an error: the API
    is optional and the KubeDevice path maintains in_use itself via
    allocate/release).
    """
    client = client or PodResourcesClient()
    if not client.available():
        return None
    try:
        held = client.allocated_device_ids(resource_name)
    except grpc.RpcError as e:
        utils.logf(2, "pod-resources List failed (skipping reconcile): %s", e)
        return None
    changed = {"set": [], "cleared": []}
    with manager._lock:
        for uuid, gpu in manager.gpus.items():
            want = uuid in held
            if gpu.in_use != want:
                gpu.in_use = want
                changed["set" if want else "cleared"].append(uuid)
    if changed["set"] or changed["cleared"]:
        utils.logf(
            2, "pod-resources reconcile: +%d in_use, -%d released",
            len(changed["set"]), len(changed["cleared"]),
        )
    return changed
