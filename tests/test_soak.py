"""Health-flip storm: rapid node removal/creation must coalesce into
consistent ListAndWatch updates with the final state correct."""

import threading

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    build_kubelet_devices,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.plugin import (
    GenericDevicePlugin,
)
from tests.fixtures import StubKubelet, dial_plugin, eventually


def test_health_flip_storm(synthetic_host):
    h = synthetic_host
    for g in range(4):
        h.add_gpu("0000:%02x:00.0" % (0x10 + g),
                  iommu_group=str(100 + g))
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    plugin = GenericDevicePlugin(
        "INSTINCT_MI355X",
        build_kubelet_devices(reg.device_map["75a3"]), reg, config=cfg)
    stop = threading.Event()
    plugin.start(stop)
    try:
        ch, stub = dial_plugin(plugin.socket_path)
        stream = stub.ListAndWatch(dpapi.Empty())
        next(stream)
        # storm: 50 remove/create cycles on two groups, end Unhealthy on
        # group 101 only
        for _ in range(50):
            h.remove_vfio_node("100")
            h.add_vfio_node("100")
            h.remove_vfio_node("101")
            h.add_vfio_node("101")
        h.remove_vfio_node("101")

        def final_state():
            return {d.ID: d.health
                    for d in plugin.devices_snapshot()} == {
                "0000:10:00.0": "Healthy",
                "0000:11:00.0": "Unhealthy",
                "0000:12:00.0": "Healthy",
                "0000:13:00.0": "Healthy"}
        eventually(final_state, timeout=10.0)
        # the stream also converges to that state
        def drain():
            upd = next(stream)
            return {d.ID: d.health for d in upd.devices}[
                "0000:11:00.0"] == "Unhealthy"
        eventually(drain, timeout=10.0)
        ch.close()
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()
