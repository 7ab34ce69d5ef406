"""BASELINE.json configs 1-5, each exercised end-to-end by name.

1. Mock sysfs + stub kubelet, 1 fake vfio device (plumbing)
2. 1×MI355X full passthrough
3. 8×MI355X full passthrough
4. SR-IOV: 8 VFs/GPU × 8 = 64 VF devices
5. Mixed node: 4 passthrough + 4 SR-IOV, VF hot-unbind fault injection
"""

import threading

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    Controller,
)
from tests.fixtures import StubKubelet, dial_plugin, eventually


def start_controller(h, expect_resources):
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    ctrl = Controller(config=cfg, kfd_nodes_dir=h.kfd_nodes,
                      vf_event_watcher_factory=lambda: None)
    ctrl.create_plugins()
    stop = threading.Event()
    started = ctrl.start(stop)
    names = {kubelet.wait_register(10).resource_name
             for _ in range(len(started))}
    assert names == expect_resources, names
    return ctrl, kubelet, stop, started


def stream_counts(plugin):
    ch, stub = dial_plugin(plugin.socket_path)
    devs = next(stub.ListAndWatch(dpapi.Empty())).devices
    ch.close()
    return len(devs)


def test_config1_single_fake_device_plumbing(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    ctrl, kubelet, stop, started = start_controller(
        h, {"amd.com/INSTINCT_MI355X"})
    try:
        p = started[0]
        assert stream_counts(p) == 1
        ch, stub = dial_plugin(p.socket_path)
        resp = stub.Allocate(dpapi.AllocateRequest(
            container_requests=[dpapi.ContainerAllocateRequest(
                devices_ids=["0000:0c:00.0"])]))
        c = resp.container_responses[0]
        assert c.envs["PCI_RESOURCE_AMD_COM_INSTINCT_MI355X"] \
            == "0000:0c:00.0"
        assert [d.host_path for d in c.devices] == [
            h.vfio_dir + "/vfio", h.vfio_dir + "/40"]
        ch.close()
    finally:
        stop.set()
        ctrl.stop()
        kubelet.stop()


def test_config3_eight_gpu_passthrough(synthetic_host):
    h = synthetic_host
    for g in range(8):
        h.add_gpu("0000:%02x:00.0" % (0x10 + g),
                  iommu_group=str(100 + g), numa=g // 4)
    ctrl, kubelet, stop, started = start_controller(
        h, {"amd.com/INSTINCT_MI355X"})
    try:
        assert stream_counts(started[0]) == 8
        # 8 concurrent single-GPU "VMIs"
        ch, stub = dial_plugin(started[0].socket_path)
        for g in range(8):
            resp = stub.Allocate(dpapi.AllocateRequest(
                container_requests=[dpapi.ContainerAllocateRequest(
                    devices_ids=["0000:%02x:00.0" % (0x10 + g)])]))
            assert len(resp.container_responses[0].devices) == 2
        ch.close()
    finally:
        stop.set()
        ctrl.stop()
        kubelet.stop()


def test_config4_sriov_64_vfs(synthetic_host):
    h = synthetic_host
    for g in range(8):
        pf = "0000:%02x:00.0" % (0x10 + g)
        h.add_gpu(pf, driver="gim", iommu_group=str(100 + g))
        for v in range(8):
            h.add_vf("0000:%02x:02.%d" % (0x10 + g, v), pf_bdf=pf,
                     iommu_group=str(200 + g * 8 + v), numa=g // 4)
    ctrl, kubelet, stop, started = start_controller(
        h, {"amd.com/INSTINCT_MI355X_VF"})
    try:
        assert stream_counts(started[0]) == 64
    finally:
        stop.set()
        ctrl.stop()
        kubelet.stop()


def test_config5_mixed_node_vf_hot_unbind(synthetic_host):
    """4 passthrough + 4 SR-IOV GPUs; hot-unbind one VF (its vfio group
    node disappears) → that VF alone goes Unhealthy, passthrough
    untouched; re-bind → Healthy."""
    h = synthetic_host
    for g in range(4):
        h.add_gpu("0000:%02x:00.0" % (0x10 + g),
                  iommu_group=str(100 + g))
    for g in range(4):
        pf = "0000:%02x:00.0" % (0x20 + g)
        h.add_gpu(pf, driver="gim", iommu_group=str(110 + g))
        for v in range(8):
            h.add_vf("0000:%02x:02.%d" % (0x20 + g, v), pf_bdf=pf,
                     iommu_group=str(300 + g * 8 + v))
    ctrl, kubelet, stop, started = start_controller(
        h, {"amd.com/INSTINCT_MI355X", "amd.com/INSTINCT_MI355X_VF"})
    try:
        by_name = {p.device_name: p for p in started}
        gpu_p = by_name["INSTINCT_MI355X"]
        vf_p = by_name["INSTINCT_MI355X_VF"]
        assert stream_counts(gpu_p) == 4
        assert stream_counts(vf_p) == 32

        h.remove_vfio_node("305")  # VF 0000:20:02.5's group

        def vf_state():
            health = {d.ID: d.health for d in vf_p.devices_snapshot()}
            return health["0000:20:02.5"] == "Unhealthy" and all(
                v == "Healthy" for k, v in health.items()
                if k != "0000:20:02.5")
        eventually(vf_state)
        assert all(d.health == "Healthy"
                   for d in gpu_p.devices_snapshot())

        h.add_vfio_node("305")  # re-bind
        eventually(lambda: all(d.health == "Healthy"
                               for d in vf_p.devices_snapshot()))
    finally:
        stop.set()
        ctrl.stop()
        kubelet.stop()
