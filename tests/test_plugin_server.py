"""End-to-end plugin server tests over real unix-socket gRPC.

Covers the behavioral pins of the reference suite
(generic_device_plugin_test.go): Allocate spec ordering, iommufd flow,
sibling suppression, unknown-BDF rejection, ListAndWatch health flips —
plus registration against the stub kubelet, which the reference never
unit-tested."""

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
from tests.fixtures import StubKubelet, dial_plugin, eventually


@pytest.fixture
def rig(synthetic_host):
    """A started plugin + stub kubelet; yields (host, plugin, stub)."""
    created = {}

    def build(island_of=None):
        h = synthetic_host
        cfg = h.config()
        kubelet = StubKubelet(cfg.kubelet_socket)
        reg = discovery.discover(base_path=h.pci)
        device_id, devs = next(iter(reg.device_map.items()))
        plugin = GenericDevicePlugin(
            "INSTINCT_MI355X", build_kubelet_devices(devs), reg,
            config=cfg, island_of=island_of)
        stop = threading.Event()
        plugin.start(stop)
        created.update(plugin=plugin, kubelet=kubelet, stop=stop)
        return h, plugin, kubelet

    yield build
    if created:
        created["stop"].set()
        created["plugin"].stop()
        created["kubelet"].stop()


def test_register_and_options(rig, synthetic_host):
    synthetic_host.add_gpu("0000:0c:00.0", iommu_group="40")
    h, plugin, kubelet = rig()
    req = kubelet.wait_register()
    assert req.version == "v1beta1"
    assert req.resource_name == "amd.com/INSTINCT_MI355X"
    assert req.endpoint == "kubevirt-INSTINCT_MI355X.sock"

    ch, stub = dial_plugin(plugin.socket_path)
    opts = stub.GetDevicePluginOptions(dpapi.Empty())
    assert opts.get_preferred_allocation_available
    assert not opts.pre_start_required
    ch.close()


def test_listandwatch_initial_list(rig, synthetic_host):
    synthetic_host.add_gpu("0000:0c:00.0", iommu_group="40", numa=1)
    synthetic_host.add_gpu("0000:2f:00.0", iommu_group="41", numa=0)
    h, plugin, kubelet = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    stream = stub.ListAndWatch(dpapi.Empty())
    first = next(stream)
    got = {d.ID: (d.health, d.topology.nodes[0].ID) for d in first.devices}
    assert got == {"0000:0c:00.0": ("Healthy", 1),
                   "0000:2f:00.0": ("Healthy", 0)}
    ch.close()


def test_allocate_happy_path_spec_order(rig, synthetic_host):
    """DeviceSpec order: /dev/vfio/vfio then /dev/vfio/<group>
    (reference: generic_device_plugin_test.go:180-196)."""
    synthetic_host.add_gpu("0000:0c:00.0", iommu_group="40")
    h, plugin, kubelet = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest(
            devices_ids=["0000:0c:00.0"])]))
    c = resp.container_responses[0]
    assert dict(c.envs) == {
        "PCI_RESOURCE_AMD_COM_INSTINCT_MI355X": "0000:0c:00.0"}
    assert [d.host_path for d in c.devices] == [
        h.vfio_dir + "/vfio", h.vfio_dir + "/40"]
    assert all(d.permissions == "mrw" for d in c.devices)
    assert all(d.container_path == d.host_path for d in c.devices)
    ch.close()


def test_allocate_iommufd_spec_order(rig, synthetic_host):
    """iommufd: vfio cdev specs first, then container node, group node,
    /dev/iommu (reference: generic_device_plugin_test.go:301-328)."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40", vfio_dev="vfio3")
    h.enable_iommufd()
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest(
            devices_ids=["0000:0c:00.0"])]))
    c = resp.container_responses[0]
    assert [d.host_path for d in c.devices] == [
        h.vfio_dir + "/devices/vfio3",
        h.vfio_dir + "/vfio",
        h.vfio_dir + "/40",
        h.iommu_dev,
    ]
    ch.close()


def test_allocate_sibling_env_suppression(rig, synthetic_host):
    """Multi-function IOMMU group: env only lists the requested BDF but
    the group's vfio node covers both
    (reference: generic_device_plugin_test.go:273-299)."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    h.add_gpu("0000:0c:00.1", iommu_group="40")
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest(
            devices_ids=["0000:0c:00.0"])]))
    c = resp.container_responses[0]
    assert dict(c.envs) == {
        "PCI_RESOURCE_AMD_COM_INSTINCT_MI355X": "0000:0c:00.0"}
    assert [d.host_path for d in c.devices] == [
        h.vfio_dir + "/vfio", h.vfio_dir + "/40"]
    ch.close()


def test_allocate_unknown_bdf_rejected(rig, synthetic_host):
    """(reference: generic_device_plugin_test.go:350-359)"""
    synthetic_host.add_gpu("0000:0c:00.0", iommu_group="40")
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    with pytest.raises(grpc.RpcError) as exc:
        stub.Allocate(dpapi.AllocateRequest(
            container_requests=[dpapi.ContainerAllocateRequest(
                devices_ids=["0000:ff:00.0"])]))
    assert exc.value.code() == grpc.StatusCode.INVALID_ARGUMENT
    assert "unknown device: 0000:ff:00.0" in exc.value.details()
    ch.close()


def test_allocate_toctou_vendor_change_rejected(rig, synthetic_host):
    """Vendor rewritten after discovery ⇒ reject at Allocate time
    (reference: generic_device_plugin.go:394-398)."""
    import os
    h = synthetic_host
    d = h.add_gpu("0000:0c:00.0", iommu_group="40")
    _, plugin, _ = rig()
    with open(os.path.join(d, "vendor"), "w") as f:
        f.write("0x10de\n")
    ch, stub = dial_plugin(plugin.socket_path)
    with pytest.raises(grpc.RpcError) as exc:
        stub.Allocate(dpapi.AllocateRequest(
            container_requests=[dpapi.ContainerAllocateRequest(
                devices_ids=["0000:0c:00.0"])]))
    assert exc.value.code() == grpc.StatusCode.INVALID_ARGUMENT
    ch.close()


def test_allocate_toctou_iommu_change_rejected(rig, synthetic_host):
    import os
    h = synthetic_host
    d = h.add_gpu("0000:0c:00.0", iommu_group="40")
    _, plugin, _ = rig()
    os.remove(os.path.join(d, "iommu_group"))
    os.symlink("../../../kernel/iommu_groups/99",
               os.path.join(d, "iommu_group"))
    ch, stub = dial_plugin(plugin.socket_path)
    with pytest.raises(grpc.RpcError) as exc:
        stub.Allocate(dpapi.AllocateRequest(
            container_requests=[dpapi.ContainerAllocateRequest(
                devices_ids=["0000:0c:00.0"])]))
    assert exc.value.code() == grpc.StatusCode.INVALID_ARGUMENT
    ch.close()


def test_allocate_multi_device(rig, synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    h.add_gpu("0000:2f:00.0", iommu_group="41")
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest(
            devices_ids=["0000:0c:00.0", "0000:2f:00.0"])]))
    c = resp.container_responses[0]
    assert dict(c.envs) == {"PCI_RESOURCE_AMD_COM_INSTINCT_MI355X":
                            "0000:0c:00.0,0000:2f:00.0"}
    assert [d.host_path for d in c.devices] == [
        h.vfio_dir + "/vfio", h.vfio_dir + "/40", h.vfio_dir + "/41"]
    ch.close()


def test_allocate_env_scoped_per_container(rig, synthetic_host):
    """Deviation from the reference (its env map leaks across container
    requests, generic_device_plugin.go:362-441)."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    h.add_gpu("0000:2f:00.0", iommu_group="41")
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.Allocate(dpapi.AllocateRequest(container_requests=[
        dpapi.ContainerAllocateRequest(devices_ids=["0000:0c:00.0"]),
        dpapi.ContainerAllocateRequest(devices_ids=["0000:2f:00.0"]),
    ]))
    envs = [dict(c.envs) for c in resp.container_responses]
    assert envs[0] == {"PCI_RESOURCE_AMD_COM_INSTINCT_MI355X":
                       "0000:0c:00.0"}
    assert envs[1] == {"PCI_RESOURCE_AMD_COM_INSTINCT_MI355X":
                       "0000:2f:00.0"}
    ch.close()


def test_health_flip_on_vfio_node_removal(rig, synthetic_host):
    """Node removal → Unhealthy streamed; re-creation → Healthy
    (reference: generic_device_plugin_test.go:361-399; recovery via
    directory watch is an improvement — the reference's per-file watch
    cannot see re-creation)."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    stream = stub.ListAndWatch(dpapi.Empty())
    first = next(stream)
    assert first.devices[0].health == "Healthy"

    h.remove_vfio_node("40")
    upd = next(stream)
    assert {d.ID: d.health for d in upd.devices} == {
        "0000:0c:00.0": "Unhealthy"}

    h.add_vfio_node("40")
    upd = next(stream)
    assert {d.ID: d.health for d in upd.devices} == {
        "0000:0c:00.0": "Healthy"}
    ch.close()


def test_kubelet_restart_reregisters(rig, synthetic_host):
    """Socket removal → plugin restarts and re-registers
    (reference: generic_device_plugin.go:685-695)."""
    import os
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    _, plugin, kubelet = rig()
    kubelet.wait_register()
    os.remove(plugin.socket_path)
    req = kubelet.wait_register(timeout=10.0)
    assert req.resource_name == "amd.com/INSTINCT_MI355X"
    # server is serving again on a fresh socket
    eventually(lambda: os.path.exists(plugin.socket_path))
    ch, stub = dial_plugin(plugin.socket_path)
    opts = stub.GetDevicePluginOptions(dpapi.Empty())
    assert opts.get_preferred_allocation_available
    ch.close()


def test_get_preferred_allocation_rpc(rig, synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40", numa=0)
    h.add_gpu("0000:2f:00.0", iommu_group="41", numa=1)
    h.add_gpu("0000:30:00.0", iommu_group="42", numa=0)
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.GetPreferredAllocation(dpapi.PreferredAllocationRequest(
        container_requests=[dpapi.ContainerPreferredAllocationRequest(
            available_deviceIDs=["0000:0c:00.0", "0000:2f:00.0",
                                 "0000:30:00.0"],
            allocation_size=2)]))
    ids = list(resp.container_responses[0].deviceIDs)
    assert ids == ["0000:0c:00.0", "0000:30:00.0"]  # NUMA 0 pair
    ch.close()


def test_get_preferred_allocation_must_include_too_big(rig,
                                                       synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    h.add_gpu("0000:2f:00.0", iommu_group="41")
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    with pytest.raises(grpc.RpcError) as exc:
        stub.GetPreferredAllocation(dpapi.PreferredAllocationRequest(
            container_requests=[dpapi.ContainerPreferredAllocationRequest(
                available_deviceIDs=["0000:0c:00.0", "0000:2f:00.0"],
                must_include_deviceIDs=["0000:0c:00.0", "0000:2f:00.0"],
                allocation_size=1)]))
    assert exc.value.code() == grpc.StatusCode.INVALID_ARGUMENT
    ch.close()


def test_allocate_iommufd_missing_cdev_internal_error(rig,
                                                      synthetic_host):
    """iommufd host but device has no vfio-dev cdev → INTERNAL
    (reference: generic_device_plugin.go:403-409)."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")  # no vfio_dev dir
    h.enable_iommufd()
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    with pytest.raises(grpc.RpcError) as exc:
        stub.Allocate(dpapi.AllocateRequest(
            container_requests=[dpapi.ContainerAllocateRequest(
                devices_ids=["0000:0c:00.0"])]))
    assert exc.value.code() == grpc.StatusCode.INTERNAL
    assert "iommufd" in exc.value.details()
    ch.close()


def test_pre_start_container(rig, synthetic_host):
    synthetic_host.add_gpu("0000:0c:00.0", iommu_group="40")
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.PreStartContainer(dpapi.PreStartContainerRequest(
        devices_ids=["0000:0c:00.0"]))
    assert resp == dpapi.PreStartContainerResponse()
    ch.close()


def test_preferred_allocation_avoids_shared_group(rig, synthetic_host):
    """Two co-grouped functions + one exclusive: a 1-device request
    prefers the exclusive one even though kubelet listed it last."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    h.add_gpu("0000:0c:00.1", iommu_group="40")
    h.add_gpu("0000:2f:00.0", iommu_group="41")
    _, plugin, _ = rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.GetPreferredAllocation(dpapi.PreferredAllocationRequest(
        container_requests=[dpapi.ContainerPreferredAllocationRequest(
            available_deviceIDs=["0000:0c:00.0", "0000:0c:00.1",
                                 "0000:2f:00.0"],
            allocation_size=1)]))
    assert list(resp.container_responses[0].deviceIDs) == ["0000:2f:00.0"]
    ch.close()


def test_resync_health_from_node_state(rig, synthetic_host):
    """After an inotify overflow the health map is re-derived from the
    actual /dev/vfio node state (exercised directly — overflow itself
    needs a kernel-queue-full storm)."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    h.add_gpu("0000:2f:00.0", iommu_group="41")
    _, plugin, _ = rig()
    import os
    os.remove(h.vfio_dir + "/41")
    plugin._resync_health(plugin._group_to_ids())
    assert {d.ID: d.health for d in plugin.devices_snapshot()} == {
        "0000:0c:00.0": "Healthy", "0000:2f:00.0": "Unhealthy"}
    h.add_vfio_node("41")
    plugin._resync_health(plugin._group_to_ids())
    assert all(d.health == "Healthy"
               for d in plugin.devices_snapshot())
