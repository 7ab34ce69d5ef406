"""Ground-truth health fallback when the /dev/vfio watch cannot be
armed, and cdev-mode (iommufd) resync semantics.

The reference goes fully blind when its fsnotify setup fails — devices
stay Healthy forever with zero checks
(generic_device_plugin.go:626-637); here a periodic resync pass takes
over, and re-arms the watch when the directory appears later.
"""

import os
import threading

import pytest

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.device_plugin import (
    discovery, inotify,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    build_kubelet_devices,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.plugin import (
    GenericDevicePlugin,
)
from tests.fixtures import StubKubelet, eventually


def _start_plugin(h, cfg):
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    _, devs = next(iter(reg.device_map.items()))
    plugin = GenericDevicePlugin(
        "INSTINCT_MI355X", build_kubelet_devices(devs), reg, config=cfg)
    stop = threading.Event()
    plugin.start(stop)
    return plugin, kubelet, stop


def _health_of(plugin, bdf):
    return {d.ID: d.health for d in plugin.devices_snapshot()}[bdf]


def test_unwatchable_vfio_dir_falls_back_to_resync(synthetic_host,
                                                   monkeypatch):
    """With the vfio dir unwatchable for the daemon's whole life, node
    removal must still flip the device Unhealthy within one resync
    period (and recovery must flip it back)."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    cfg = h.config()
    cfg.health_resync_s = 0.3

    orig = inotify.Watcher.add_watch

    def refuse_vfio(self, path):
        if path == cfg.vfio_dir:
            raise OSError(13, "unwatchable in this test")
        return orig(self, path)

    monkeypatch.setattr(inotify.Watcher, "add_watch", refuse_vfio)
    plugin, kubelet, stop = _start_plugin(h, cfg)
    try:
        assert _health_of(plugin, "0000:0c:00.0") == dpapi.HEALTHY
        h.remove_vfio_node("40")
        eventually(lambda:
                   _health_of(plugin, "0000:0c:00.0") == dpapi.UNHEALTHY)
        h.add_vfio_node("40")
        eventually(lambda:
                   _health_of(plugin, "0000:0c:00.0") == dpapi.HEALTHY)
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()


def test_vfio_dir_appearing_later_rearms_watch(synthetic_host):
    """vfio_dir missing at start (module not yet loaded): the resync
    pass marks devices Unhealthy, then re-arms the watch once the dir
    exists and event-driven health resumes."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    cfg = h.config()
    cfg.vfio_dir = h.vfio_dir + "-late"
    cfg.health_resync_s = 0.3
    plugin, kubelet, stop = _start_plugin(h, cfg)
    try:
        # no nodes exist under the (absent) dir → ground truth says down
        eventually(lambda:
                   _health_of(plugin, "0000:0c:00.0") == dpapi.UNHEALTHY)
        os.makedirs(cfg.vfio_dir)
        with open(os.path.join(cfg.vfio_dir, "40"), "w"):
            pass
        eventually(lambda:
                   _health_of(plugin, "0000:0c:00.0") == dpapi.HEALTHY)
        # watch is armed now: removal must flip fast, event-driven
        os.remove(os.path.join(cfg.vfio_dir, "40"))
        eventually(lambda:
                   _health_of(plugin, "0000:0c:00.0") == dpapi.UNHEALTHY,
                   timeout=2.0)
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()


def test_resync_cdev_mode_uses_per_device_nodes(synthetic_host):
    """Pure iommufd cdev mode: no /dev/vfio/<group> nodes ever exist;
    the resync pass must judge health by /dev/vfio/devices/vfioN
    instead of declaring every device dead."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40", vfio_node=False,
              vfio_dev="vfio3")
    h.enable_iommufd()
    cdev_dir = os.path.join(h.vfio_dir, "devices")
    os.makedirs(cdev_dir)
    with open(os.path.join(cdev_dir, "vfio3"), "w"):
        pass
    cfg = h.config()
    cfg.health_resync_s = 0.2
    plugin, kubelet, stop = _start_plugin(h, cfg)
    try:
        # several resync periods pass; the cdev node keeps it healthy
        import time
        time.sleep(0.6)
        assert _health_of(plugin, "0000:0c:00.0") == dpapi.HEALTHY
        os.remove(os.path.join(cdev_dir, "vfio3"))
        eventually(lambda:
                   _health_of(plugin, "0000:0c:00.0") == dpapi.UNHEALTHY)
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()
