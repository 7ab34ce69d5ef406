"""Daemon-level lifecycle: subprocess start, serve, SIGTERM shutdown;
ListAndWatch stream reconnects."""

import json
import os
import signal
import subprocess
import sys
import threading

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    build_kubelet_devices,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.plugin import (
    GenericDevicePlugin,
)
from tests.fixtures import StubKubelet, dial_plugin

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_daemon_sigterm_clean_exit(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    proc = subprocess.Popen(
        [sys.executable,
         os.path.join(REPO, "bench_harness", "plugin_proc.py"),
         json.dumps({
             "device_plugin_dir": cfg.device_plugin_dir,
             "kubelet_socket": cfg.kubelet_socket,
             "vfio_dir": cfg.vfio_dir,
             "iommu_dev": cfg.iommu_dev,
             "pci_base": cfg.pci_base,
             "kfd_nodes_dir": h.kfd_nodes,
         })], cwd=REPO)
    try:
        req = kubelet.wait_register(timeout=30)
        assert req.resource_name == "amd.com/INSTINCT_MI355X"
        proc.send_signal(signal.SIGTERM)
        assert proc.wait(timeout=15) == 0
        # socket cleaned up on shutdown
        assert not os.path.exists(
            os.path.join(cfg.device_plugin_dir, req.endpoint))
    finally:
        if proc.poll() is None:
            proc.kill()
            proc.wait()
        kubelet.stop()


def test_listandwatch_reconnect(synthetic_host):
    """kubelet reconnect: a second stream gets the full current list
    and subsequent health flips; the abandoned first stream dies
    quietly."""
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
    try:
        ch1, stub1 = dial_plugin(plugin.socket_path)
        s1 = stub1.ListAndWatch(dpapi.Empty())
        next(s1)
        ch1.close()  # kubelet went away

        ch2, stub2 = dial_plugin(plugin.socket_path)
        s2 = stub2.ListAndWatch(dpapi.Empty())
        first = next(s2)
        assert [d.ID for d in first.devices] == ["0000:0c:00.0"]
        h.remove_vfio_node("40")
        upd = next(s2)
        assert upd.devices[0].health == "Unhealthy"
        ch2.close()
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()
