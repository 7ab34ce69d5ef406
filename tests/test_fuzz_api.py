"""Adversarial/fuzz inputs against the live gRPC surface: the server
must reject garbage cleanly and keep serving."""

import threading

import grpc
import pytest

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    build_kubelet_devices,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.plugin import (
    GenericDevicePlugin,
)
from tests.fixtures import StubKubelet, dial_plugin

WEIRD_IDS = [
    "", " ", "\x00", "../../etc/passwd", "0000:0c:00.0/../40",
    "ффф:аа:00.0", "0000:0c:00.0" * 50, "a" * 4096, "\n0000:0c:00.0",
    "0000:0C:00.0",  # wrong case
]


@pytest.fixture
def live(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    plugin = GenericDevicePlugin(
        "INSTINCT_MI355X",
        build_kubelet_devices(reg.device_map["75a3"]), reg, config=cfg)
    stop = threading.Event()
    plugin.start(stop)
    ch, stub = dial_plugin(plugin.socket_path)
    yield stub
    ch.close()
    stop.set()
    plugin.stop()
    kubelet.stop()


def test_allocate_weird_ids_rejected_cleanly(live):
    for wid in WEIRD_IDS:
        with pytest.raises(grpc.RpcError) as exc:
            live.Allocate(dpapi.AllocateRequest(
                container_requests=[dpapi.ContainerAllocateRequest(
                    devices_ids=[wid])]))
        assert exc.value.code() == grpc.StatusCode.INVALID_ARGUMENT, wid
    # server still healthy afterwards
    resp = live.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest(
            devices_ids=["0000:0c:00.0"])]))
    assert resp.container_responses[0].envs


def test_allocate_empty_and_oversized_requests(live):
    # empty request → empty response, no crash
    resp = live.Allocate(dpapi.AllocateRequest())
    assert len(resp.container_responses) == 0
    # empty container request → response with no devices
    resp = live.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest()]))
    assert len(resp.container_responses) == 1
    assert len(resp.container_responses[0].devices) == 0
    # many container requests at once
    resp = live.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest(
            devices_ids=["0000:0c:00.0"])] * 64))
    assert len(resp.container_responses) == 64


def test_preferred_allocation_weird_inputs(live):
    # unknown ids: falls through to kubelet order, no crash
    resp = live.GetPreferredAllocation(dpapi.PreferredAllocationRequest(
        container_requests=[dpapi.ContainerPreferredAllocationRequest(
            available_deviceIDs=WEIRD_IDS, allocation_size=3)]))
    assert len(resp.container_responses[0].deviceIDs) == 3
    # zero size
    resp = live.GetPreferredAllocation(dpapi.PreferredAllocationRequest(
        container_requests=[dpapi.ContainerPreferredAllocationRequest(
            available_deviceIDs=["0000:0c:00.0"], allocation_size=0)]))
    assert len(resp.container_responses[0].deviceIDs) == 0
    # negative size: treat as nothing to allocate, not a crash
    resp = live.GetPreferredAllocation(dpapi.PreferredAllocationRequest(
        container_requests=[dpapi.ContainerPreferredAllocationRequest(
            available_deviceIDs=["0000:0c:00.0"],
            allocation_size=-3)]))
    assert len(resp.container_responses[0].deviceIDs) == 0


def test_raw_garbage_bytes_on_socket(live, synthetic_host):
    """Non-gRPC bytes on the unix socket must not wedge the server."""
    import socket as pysocket
    h = synthetic_host
    path = h.kubelet_dir + "/kubevirt-INSTINCT_MI355X.sock"
    s = pysocket.socket(pysocket.AF_UNIX, pysocket.SOCK_STREAM)
    s.connect(path)
    s.sendall(b"\x00\xff" * 512 + b"GET / HTTP/1.1\r\n\r\n")
    s.close()
    # still serving gRPC
    assert live.GetDevicePluginOptions(
        dpapi.Empty()).get_preferred_allocation_available
