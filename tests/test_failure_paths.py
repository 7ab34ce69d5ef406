"""Failure-path behavior: kubelet absent, discovery of empty node."""

import threading

from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    Controller,
)


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


def test_empty_node_discovers_nothing(synthetic_host):
    reg = discovery.discover(base_path=synthetic_host.pci)
    assert not reg.bdf_to_iommu
    ctrl = Controller(config=synthetic_host.config(),
                      kfd_nodes_dir=synthetic_host.kfd_nodes)
    assert ctrl.create_plugins() == []


def test_missing_base_path():
    reg = discovery.discover(base_path="/nonexistent/path")
    assert not reg.bdf_to_iommu
