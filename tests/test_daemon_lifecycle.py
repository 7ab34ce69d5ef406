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


def test_restart_survives_long_kubelet_outage(synthetic_host):
    """Chaos: kubelet stays down for far more re-register attempts than
    the old 30-attempt cap, then comes back — the plugin must still be
    retrying (capped backoff, never abandoned) and register.
    (Reference behavior to beat: ONE attempt then dead,
    generic_device_plugin.go:688-692.)"""
    import time

    from tests.fixtures import eventually

    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    cfg = h.config()
    cfg.connect_timeout_s = 0.05
    cfg.restart_backoff_initial_s = 0.01
    cfg.restart_backoff_max_s = 0.02
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    _, devs = next(iter(reg.device_map.items()))
    plugin = GenericDevicePlugin(
        "INSTINCT_MI355X", build_kubelet_devices(devs), reg, config=cfg)
    stop = threading.Event()
    plugin.start(stop)
    assert kubelet.wait_register()
    kubelet.stop()  # kubelet goes away
    os.remove(cfg.kubelet_socket) if os.path.exists(
        cfg.kubelet_socket) else None

    t = threading.Thread(target=plugin.restart, daemon=True)
    t.start()
    # outage spans well over 30 failed attempts (~0.07s per attempt)
    time.sleep(3.0)
    assert t.is_alive(), "restart loop gave up during the outage"

    kubelet2 = StubKubelet(cfg.kubelet_socket)
    try:
        assert kubelet2.wait_register(timeout=10.0)
        eventually(lambda: not t.is_alive())
        assert plugin._server is not None
    finally:
        stop.set()
        plugin.stop()
        kubelet2.stop()


def test_concurrent_stop_aborts_restart_retry_loop(synthetic_host):
    """A rescan retiring this resource calls stop() while the restart
    loop is mid-outage: stop() must not be blocked behind the unbounded
    retry — the loop aborts promptly and the plugin stays stopped."""
    import time

    from tests.fixtures import eventually

    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    cfg = h.config()
    cfg.connect_timeout_s = 0.2
    cfg.restart_backoff_initial_s = 0.2
    cfg.restart_backoff_max_s = 0.4
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    _, devs = next(iter(reg.device_map.items()))
    plugin = GenericDevicePlugin(
        "INSTINCT_MI355X", build_kubelet_devices(devs), reg, config=cfg)
    stop = threading.Event()
    plugin.start(stop)
    assert kubelet.wait_register()
    kubelet.stop()

    t = threading.Thread(target=plugin.restart, daemon=True)
    t.start()
    time.sleep(0.5)  # let the retry loop begin failing attempts
    assert t.is_alive()

    t0 = time.monotonic()
    plugin.stop()
    stopped_in = time.monotonic() - t0
    assert stopped_in < 2.0, (
        "stop() blocked %.1fs behind the restart retry loop"
        % stopped_in)
    eventually(lambda: not t.is_alive())
    assert plugin._server is None
    assert not os.path.exists(plugin.socket_path)
