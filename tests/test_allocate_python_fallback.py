"""Allocate semantics with the native _sysfs extension unavailable —
the pure-Python fallback must behave identically."""

import threading

import grpc
import pytest

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from kubevirt_gpu_device_plugin_amd.device_plugin import plugin as plugin_mod
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    build_kubelet_devices,
)
from tests.fixtures import StubKubelet, dial_plugin


@pytest.fixture
def fallback_rig(synthetic_host, monkeypatch):
    monkeypatch.setattr(plugin_mod, "_sysfs", None)
    created = {}

    def build():
        h = synthetic_host
        cfg = h.config()
        kubelet = StubKubelet(cfg.kubelet_socket)
        reg = discovery.discover(base_path=h.pci, use_native=False)
        plugin = plugin_mod.GenericDevicePlugin(
            "INSTINCT_MI355X",
            build_kubelet_devices(reg.device_map["75a3"]), reg,
            config=cfg)
        stop = threading.Event()
        plugin.start(stop)
        created.update(plugin=plugin, kubelet=kubelet, stop=stop)
        return h, plugin

    yield build
    if created:
        created["stop"].set()
        created["plugin"].stop()
        created["kubelet"].stop()


def test_fallback_happy_path(fallback_rig, synthetic_host):
    synthetic_host.add_gpu("0000:0c:00.0", iommu_group="40")
    h, plugin = fallback_rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest(
            devices_ids=["0000:0c:00.0"])]))
    c = resp.container_responses[0]
    assert [d.host_path for d in c.devices] == [
        h.vfio_dir + "/vfio", h.vfio_dir + "/40"]
    ch.close()


def test_fallback_iommufd_spec_order(fallback_rig, synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40", vfio_dev="vfio3")
    h.enable_iommufd()
    _, plugin = fallback_rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest(
            devices_ids=["0000:0c:00.0"])]))
    assert [d.host_path for d in resp.container_responses[0].devices] \
        == [h.vfio_dir + "/devices/vfio3", h.vfio_dir + "/vfio",
            h.vfio_dir + "/40", h.iommu_dev]
    ch.close()


def test_fallback_toctou_rejected(fallback_rig, synthetic_host):
    import os
    h = synthetic_host
    d = h.add_gpu("0000:0c:00.0", iommu_group="40")
    _, plugin = fallback_rig()
    with open(os.path.join(d, "vendor"), "w") as f:
        f.write("0x10de\n")
    ch, stub = dial_plugin(plugin.socket_path)
    with pytest.raises(grpc.RpcError) as exc:
        stub.Allocate(dpapi.AllocateRequest(
            container_requests=[dpapi.ContainerAllocateRequest(
                devices_ids=["0000:0c:00.0"])]))
    assert exc.value.code() == grpc.StatusCode.INVALID_ARGUMENT
    ch.close()


def test_fallback_missing_cdev_internal(fallback_rig, synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")  # no vfio-dev
    h.enable_iommufd()
    _, plugin = fallback_rig()
    ch, stub = dial_plugin(plugin.socket_path)
    with pytest.raises(grpc.RpcError) as exc:
        stub.Allocate(dpapi.AllocateRequest(
            container_requests=[dpapi.ContainerAllocateRequest(
                devices_ids=["0000:0c:00.0"])]))
    assert exc.value.code() == grpc.StatusCode.INTERNAL
    ch.close()
