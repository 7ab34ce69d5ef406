"""Failure-path behavior: kubelet absent, discovery of empty node."""

import os
import threading

import grpc
import pytest

from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    Controller,
)

from tests.fixtures import StubKubelet, eventually


def test_no_kubelet_all_starts_fail_cleanly(synthetic_host):
    """No kubelet socket: every plugin start fails with a logged error,
    nothing crashes, stop() is a no-op (reference start-failure
    tolerance: device_plugin.go:131-136)."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    cfg = h.config()
    cfg.connect_timeout_s = 0.3  # don't wait 5s for the absent kubelet
    ctrl = Controller(config=cfg, kfd_nodes_dir=h.kfd_nodes)
    ctrl.create_plugins()
    started = ctrl.start(threading.Event())
    assert started == []
    ctrl.stop()


def test_failed_start_leaves_no_zombie_server(synthetic_host):
    """A start() that fails at kubelet registration must tear the whole
    generation down — no bound socket, no live gRPC listener, no health
    thread — and the same plugin must be startable again once kubelet
    exists (regression: the FutureTimeoutError path used to leak a live
    server whose health thread could re-register with a stale registry).
    """
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    cfg = h.config()
    cfg.connect_timeout_s = 0.3
    ctrl = Controller(config=cfg, kfd_nodes_dir=h.kfd_nodes)
    plugin = ctrl.create_plugins()[0]

    with pytest.raises((grpc.RpcError, grpc.FutureTimeoutError)):
        plugin.start(threading.Event())

    assert plugin._server is None
    assert not os.path.exists(plugin.socket_path)
    eventually(lambda: not plugin._health_thread.is_alive())
    # the socket path must be free and the plugin reusable: with a
    # kubelet now present, the same object starts and registers
    kubelet = StubKubelet(cfg.kubelet_socket)
    try:
        plugin.start(threading.Event())
        assert kubelet.wait_register()
    finally:
        plugin.stop()
        kubelet.stop()


def test_empty_node_discovers_nothing(synthetic_host):
    reg = discovery.discover(base_path=synthetic_host.pci)
    assert not reg.bdf_to_iommu
    ctrl = Controller(config=synthetic_host.config(),
                      kfd_nodes_dir=synthetic_host.kfd_nodes)
    assert ctrl.create_plugins() == []


def test_missing_base_path():
    reg = discovery.discover(base_path="/nonexistent/path")
    assert not reg.bdf_to_iommu
