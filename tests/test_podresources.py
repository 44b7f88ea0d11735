"""Pod-resources (v1) reconciliation tests: kubelet is the allocation
source of truth on the stock path, and the manager's in_use flags must
track it — set while a pod holds a GPU, cleared after teardown (the
deallocate signal v1beta1 never delivers)."""

import os
import threading
from concurrent import futures

import grpc
import pytest

from kubegpu_amd.deviceplugin import create_device_plugin
from kubegpu_amd.discovery import FakeBackend, fixtures
from kubegpu_amd.server import podresources as pr


class _FakePodResources:
    """Minimal kubelet pod-resources v1 endpoint."""

    def __init__(self, sock_path):
        self.sock_path = sock_path
        self.pods = []  # [(pod, namespace, [(container, resource, [ids])])]
        self.server = None
        self.calls = 0

    def start(self):
        def handler(request, context):
            self.calls += 1
            resp = pr.ListPodResourcesResponse()
            for name, ns, containers in self.pods:
                p = resp.pod_resources.add()
                p.name = name
                p.namespace = ns
                for cname, resource, ids in containers:
                    c = p.containers.add()
                    c.name = cname
                    d = c.devices.add()
                    d.resource_name = resource
                    d.device_ids.extend(ids)
            return resp

        if os.path.exists(self.sock_path):
            os.unlink(self.sock_path)
        self.server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
        self.server.add_generic_rpc_handlers((
            grpc.method_handlers_generic_handler(
                pr.POD_RESOURCES_SERVICE,
                {"List": grpc.unary_unary_rpc_method_handler(
                    handler,
                    request_deserializer=pr.ListPodResourcesRequest.FromString,
                    response_serializer=lambda m: m.SerializeToString(),
                )},
            ),
        ))
        self.server.add_insecure_port(f"unix://{self.sock_path}")
        self.server.start()

    def stop(self):
        if self.server is not None:
            self.server.stop(grace=0.2)
            self.server = None
        if os.path.exists(self.sock_path):
            os.unlink(self.sock_path)


@pytest.fixture
def kubelet_pr(tmp_path):
    srv = _FakePodResources(str(tmp_path / "podres.sock"))
    srv.start()
    yield srv
    srv.stop()


def _mgr():
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_2hive_8gpu()))
    mgr.start()
    return mgr


def test_reconcile_sets_and_clears(kubelet_pr):
    mgr = _mgr()
    client = pr.PodResourcesClient(kubelet_pr.sock_path)

    kubelet_pr.pods = [
        ("train", "default", [("main", "amd.com/gpu",
                               ["GPU-mi355x-02", "GPU-mi355x-03"])]),
        ("other", "default", [("c", "nvidia.com/gpu", ["GPU-foreign"])]),
    ]
    changed = pr.reconcile_in_use(mgr, client)
    assert sorted(changed["set"]) == ["GPU-mi355x-02", "GPU-mi355x-03"]
    assert mgr.in_use_uuids() == ["GPU-mi355x-02", "GPU-mi355x-03"]

    # pod teardown: kubelet stops listing the devices -> flags clear
    kubelet_pr.pods = []
    changed = pr.reconcile_in_use(mgr, client)
    assert sorted(changed["cleared"]) == ["GPU-mi355x-02", "GPU-mi355x-03"]
    assert mgr.in_use_uuids() == []


def test_reconcile_ignores_other_resources(kubelet_pr):
    """Foreign resource names and unknown device ids never flip flags."""
    mgr = _mgr()
    client = pr.PodResourcesClient(kubelet_pr.sock_path)
    kubelet_pr.pods = [
        ("p", "ns", [("c", "nvidia.com/gpu", ["GPU-mi355x-00"]),
                     ("c2", "amd.com/gpu", ["GPU-not-ours"])]),
    ]
    changed = pr.reconcile_in_use(mgr, client)
    assert changed == {"set": [], "cleared": []}
    assert mgr.in_use_uuids() == []


def test_reconcile_absent_socket_is_noop(tmp_path):
    mgr = _mgr()
    client = pr.PodResourcesClient(str(tmp_path / "missing.sock"))
    assert pr.reconcile_in_use(mgr, client) is None
    assert mgr.in_use_uuids() == []


def test_reconcile_survives_rpc_failure(tmp_path):
    """Socket exists but nothing serves it: reconcile logs and no-ops."""
    mgr = _mgr()
    sock = tmp_path / "dead.sock"
    sock.touch()
    mgr.gpus["GPU-mi355x-01"].in_use = True
    client = pr.PodResourcesClient(str(sock))
    assert pr.reconcile_in_use(mgr, client) is None
    assert mgr.in_use_uuids() == ["GPU-mi355x-01"]  # untouched


def test_idle_preference_follows_reconciled_state(kubelet_pr, tmp_path):
    """End to end: kubelet says hive-1 GPUs are held by a pod; the next
    GetPreferredAllocation prefers the idle hive."""
    from kubegpu_amd.server import KubeletDevicePlugin, dpapi

    mgr = _mgr()
    client = pr.PodResourcesClient(kubelet_pr.sock_path)
    kubelet_pr.pods = [
        ("busy", "ns", [("c", "amd.com/gpu",
                         ["GPU-mi355x-04", "GPU-mi355x-05"])]),
    ]
    pr.reconcile_in_use(mgr, client)

    p = KubeletDevicePlugin(mgr, socket_path=str(tmp_path / "dp.sock"))
    p.start()
    try:
        ch = grpc.insecure_channel(f"unix://{p.socket_path}")
        resp = ch.unary_unary(
            f"/{dpapi.DEVICE_PLUGIN_SERVICE}/GetPreferredAllocation",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=dpapi.PreferredAllocationResponse.FromString,
        )(dpapi.PreferredAllocationRequest(container_requests=[
            dpapi.ContainerPreferredAllocationRequest(
                available_deviceIDs=[f"GPU-mi355x-{i:02d}" for i in range(8)],
                allocation_size=4)
        ]), timeout=10)
        got = set(resp.container_responses[0].deviceIDs)
        assert got == {f"GPU-mi355x-{i:02d}" for i in range(4)}  # idle hive 0
        ch.close()
    finally:
        p.stop()


def test_get_allocatable_resources_roundtrip(kubelet_pr):
    """GetAllocatableResources client against a fake kubelet endpoint."""
    # extend the fake with the second RPC
    srv = kubelet_pr

    def alloc_handler(request, context):
        resp = pr.AllocatableResourcesResponse()
        d = resp.devices.add()
        d.resource_name = "amd.com/gpu"
        d.device_ids.extend([f"GPU-mi355x-{i:02d}" for i in range(8)])
        return resp

    srv.server.add_generic_rpc_handlers((
        grpc.method_handlers_generic_handler(
            pr.POD_RESOURCES_SERVICE,
            {"GetAllocatableResources": grpc.unary_unary_rpc_method_handler(
                alloc_handler,
                request_deserializer=pr.AllocatableResourcesRequest.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            )},
        ),
    ))
    client = pr.PodResourcesClient(srv.sock_path)
    resp = client.get_allocatable_resources()
    assert len(resp.devices) == 1
    assert len(resp.devices[0].device_ids) == 8
