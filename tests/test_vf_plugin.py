"""SR-IOV VF plugin tests: PCI-style allocation + AMD-SMI health fan-out
(reference analogues: generic_vgpu_device_plugin_test.go:43-192)."""

import threading

import pytest

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.amdsmi import (
    EVT_GPU_POST_RESET, EVT_GPU_PRE_RESET, EVT_VMFAULT,
)
from kubevirt_gpu_device_plugin_amd.amdsmi.events import (
    AmdSmiEventWatcher, SharedSmiWatcher,
)
from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    build_kubelet_devices,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.vf_plugin import (
    VfDevicePlugin,
)
from tests.fixtures import FakeSmi, StubKubelet, dial_plugin, eventually


def make_vf_host(h, n_vfs=4):
    pf = "0000:0c:00.0"
    h.add_gpu(pf, driver="gim", iommu_group="40")
    for i in range(n_vfs):
        h.add_vf("0000:0c:02.%d" % i, pf_bdf=pf,
                 iommu_group=str(50 + i))
    return pf


@pytest.fixture
def vf_rig(synthetic_host):
    created = {}

    def build(n_vfs=4, smi=None):
        h = synthetic_host
        pf = make_vf_host(h, n_vfs)
        cfg = h.config()
        kubelet = StubKubelet(cfg.kubelet_socket)
        reg = discovery.discover(base_path=h.pci)
        if smi is not None:
            kw = dict(smi_watcher=SharedSmiWatcher(smi=smi, poll_ms=50))
        else:
            kw = dict(event_watcher_factory=lambda: None)
        plugin = VfDevicePlugin(
            "INSTINCT_MI355X_VF",
            build_kubelet_devices(reg.vf_map["75b3"]), reg, config=cfg,
            **kw)
        stop = threading.Event()
        plugin.start(stop)
        created.update(plugin=plugin, kubelet=kubelet, stop=stop)
        return h, pf, plugin, kubelet

    yield build
    if created:
        created["stop"].set()
        created["plugin"].stop()
        created["kubelet"].stop()


def test_vf_register_and_allocate(vf_rig):
    h, pf, plugin, kubelet = vf_rig()
    req = kubelet.wait_register()
    assert req.resource_name == "amd.com/INSTINCT_MI355X_VF"

    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.Allocate(dpapi.AllocateRequest(
        container_requests=[dpapi.ContainerAllocateRequest(
            devices_ids=["0000:0c:02.1"])]))
    c = resp.container_responses[0]
    # VFs are PCI vfio devices: per-group node wiring, not the
    # reference's single shared /dev/vfio spec for mdevs
    assert dict(c.envs) == {
        "PCI_RESOURCE_AMD_COM_INSTINCT_MI355X_VF": "0000:0c:02.1"}
    assert [d.host_path for d in c.devices] == [
        h.vfio_dir + "/vfio", h.vfio_dir + "/51"]
    ch.close()


def test_vf_preferred_allocation_implemented(vf_rig):
    """The reference's vGPU GetPreferredAllocation is a nil stub
    (generic_vgpu_device_plugin.go:270-278); ours works."""
    h, pf, plugin, kubelet = vf_rig()
    ch, stub = dial_plugin(plugin.socket_path)
    resp = stub.GetPreferredAllocation(dpapi.PreferredAllocationRequest(
        container_requests=[dpapi.ContainerPreferredAllocationRequest(
            available_deviceIDs=["0000:0c:02.0", "0000:0c:02.1"],
            allocation_size=1)]))
    assert list(resp.container_responses[0].deviceIDs) == ["0000:0c:02.0"]
    ch.close()


def test_smi_event_marks_vfs_unhealthy_then_recovers(vf_rig):
    """PF reset event → all child VFs unhealthy; post-reset → healthy
    (reference fan-out: generic_vgpu_device_plugin.go:335-340)."""
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "u0"}])
    h, pf, plugin, kubelet = vf_rig(n_vfs=2, smi=smi)
    eventually(lambda: smi.event_inited == {0})

    smi.push(0, EVT_GPU_PRE_RESET, "reset incoming")
    eventually(lambda: all(
        d.health == dpapi.UNHEALTHY for d in plugin.devices_snapshot()))

    smi.push(0, EVT_GPU_POST_RESET, "reset done")
    eventually(lambda: all(
        d.health == dpapi.HEALTHY for d in plugin.devices_snapshot()))


def test_smi_vmfault_is_not_critical(vf_rig):
    """VMFAULT is an application-level event — skip, like the
    reference skips XIDs 31/43/45 (generic_vgpu_device_plugin.go:416)."""
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "u0"}])
    h, pf, plugin, kubelet = vf_rig(n_vfs=2, smi=smi)
    eventually(lambda: smi.event_inited == {0})
    smi.push(0, EVT_VMFAULT, "guest page fault")
    import time
    time.sleep(0.3)
    assert all(d.health == dpapi.HEALTHY
               for d in plugin.devices_snapshot())


def test_smi_absent_graceful(vf_rig):
    """No AMD-SMI ⇒ plugin still serves (reference:
    generic_vgpu_device_plugin.go:290-297)."""
    h, pf, plugin, kubelet = vf_rig(smi=None)
    ch, stub = dial_plugin(plugin.socket_path)
    opts = stub.GetDevicePluginOptions(dpapi.Empty())
    assert opts.get_preferred_allocation_available
    ch.close()


def test_vf_vfio_node_health(vf_rig):
    h, pf, plugin, kubelet = vf_rig()
    ch, stub = dial_plugin(plugin.socket_path)
    stream = stub.ListAndWatch(dpapi.Empty())
    next(stream)
    h.remove_vfio_node("52")
    upd = next(stream)
    health = {d.ID: d.health for d in upd.devices}
    assert health["0000:0c:02.2"] == "Unhealthy"
    assert health["0000:0c:02.0"] == "Healthy"
    ch.close()


def test_smi_watcher_survives_restart_one_thread(vf_rig):
    """Across a kubelet-restart cycle there is exactly ONE shared SMI
    watcher thread, and events still reach the restarted server."""
    import threading as _threading
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "u0"}])
    h, pf, plugin, kubelet = vf_rig(n_vfs=2, smi=smi)
    eventually(lambda: smi.event_inited == {0}, timeout=10.0)
    plugin.restart()
    eventually(lambda: smi.event_inited == {0}, timeout=10.0)  # re-armed

    # exactly one live thread owned by THIS rig's watcher (other tests
    # may run their own shared watchers in the same process)
    watcher = plugin._smi_watcher
    eventually(lambda: watcher._thread is not None
               and watcher._thread.is_alive(), timeout=10.0)
    assert _threading.active_count() > 0  # sanity

    smi.push(0, EVT_GPU_PRE_RESET, "after restart")
    # generous window: a restart's gRPC handshakes plus an
    # oversubscribed runner can stretch delivery well past 5 s
    eventually(lambda: all(
        d.health == dpapi.UNHEALTHY
        for d in plugin.devices_snapshot()), timeout=20.0)


def test_two_vf_plugins_share_one_watcher(synthetic_host):
    """Two VF resource types on one node: a single shared watcher
    dispatches a PF fault only to the plugin owning that PF."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", driver="gim", iommu_group="40")
    h.add_vf("0000:0c:02.0", pf_bdf="0000:0c:00.0", iommu_group="50")
    h.add_gpu("0000:0d:00.0", driver="gim", iommu_group="41")
    h.add_vf("0000:0d:02.0", pf_bdf="0000:0d:00.0", device_id="75b0",
             iommu_group="51")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "a"},
                   {"index": 1, "bdf": "0000:0d:00.0", "uuid": "b"}])
    shared = SharedSmiWatcher(smi=smi, poll_ms=50)
    stop = threading.Event()
    plugins = []
    try:
        for name, dev_id in (("INSTINCT_MI355X_VF", "75b3"),
                             ("INSTINCT_MI350X_VF", "75b0")):
            p = VfDevicePlugin(name,
                               build_kubelet_devices(reg.vf_map[dev_id]),
                               reg, config=cfg, smi_watcher=shared)
            p.start(stop)
            plugins.append(p)
        eventually(lambda: smi.event_inited == {0, 1})
        assert shared._thread is not None and shared._thread.is_alive()
        smi.push(1, EVT_GPU_PRE_RESET, "pf 0d down")
        eventually(lambda: plugins[1].devices_snapshot()[0].health
                   == dpapi.UNHEALTHY)
        assert plugins[0].devices_snapshot()[0].health == dpapi.HEALTHY
    finally:
        stop.set()
        for p in plugins:
            p.stop()
        kubelet.stop()


def test_smi_becomes_available_after_rescan(vf_rig):
    """libamd_smi absent at start (subscribe returns None) must not be
    permanent: a rescan retries the subscription, and once SMI is
    available PF fault events flow to VF health (regression for the
    lost-resubscription gap; reference degradation contract:
    generic_vgpu_device_plugin.go:290-297)."""
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "u0"}])
    avail = [False]
    smi.available = lambda: avail[0]
    h, pf, plugin, kubelet = vf_rig(smi=smi)
    assert plugin._smi_sub is None  # degraded to sysfs-only health

    avail[0] = True
    reg = discovery.discover(base_path=h.pci)
    plugin.update_registry(reg,
                           build_kubelet_devices(reg.vf_map["75b3"]))
    assert plugin._smi_sub is not None

    smi.push(0, EVT_GPU_PRE_RESET)
    eventually(lambda: all(d.health == dpapi.UNHEALTHY
                           for d in plugin.devices_snapshot()))
    smi.push(0, EVT_GPU_POST_RESET)
    eventually(lambda: all(d.health == dpapi.HEALTHY
                           for d in plugin.devices_snapshot()))
