"""Controller end-to-end: discovery → plugins → registration
(reference: device_plugin_test.go:102-130 uses fake plugin starters; we
go further and run the real servers against the stub kubelet)."""

import threading

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    Controller, build_kubelet_devices, resolve_name,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.discovery import (
    AmdGpuDevice,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.pciids import (
    BUILTIN_IDS_PATH,
)
from tests.fixtures import StubKubelet, dial_plugin


def test_build_kubelet_devices():
    devs = build_kubelet_devices([
        AmdGpuDevice(addr="0000:0c:00.0", numa_node=1, device_id="75a3",
                     iommu_group="40")])
    assert devs[0].ID == "0000:0c:00.0"
    assert devs[0].health == dpapi.HEALTHY
    assert devs[0].topology.nodes[0].ID == 1


def test_resolve_name_fallback_to_raw_id():
    assert resolve_name("beef", BUILTIN_IDS_PATH) == "beef"
    assert resolve_name("75a3", BUILTIN_IDS_PATH) == "INSTINCT_MI355X"


def test_mixed_node_controller(synthetic_host):
    """BASELINE config 5 shape: passthrough GPUs + SR-IOV VFs on one
    node → one plugin per resource type, all registered."""
    h = synthetic_host
    # 2 passthrough MI355X
    h.add_gpu("0000:10:00.0", iommu_group="100", numa=0)
    h.add_gpu("0000:11:00.0", iommu_group="101", numa=0)
    # 1 gim PF with 2 VFs
    h.add_gpu("0000:20:00.0", driver="gim", iommu_group="110")
    h.add_vf("0000:20:02.0", pf_bdf="0000:20:00.0", iommu_group="120")
    h.add_vf("0000:20:02.1", pf_bdf="0000:20:00.0", iommu_group="121")

    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    ctrl = Controller(config=cfg, kfd_nodes_dir=h.kfd_nodes,
                      vf_event_watcher_factory=lambda: None)
    plugins = ctrl.create_plugins()
    assert sorted(p.device_name for p in plugins) == [
        "INSTINCT_MI355X", "INSTINCT_MI355X_VF"]

    stop = threading.Event()
    try:
        started = ctrl.start(stop)
        assert len(started) == 2
        names = {kubelet.wait_register().resource_name for _ in range(2)}
        assert names == {"amd.com/INSTINCT_MI355X",
                         "amd.com/INSTINCT_MI355X_VF"}
        # both sockets serve
        for p in started:
            ch, stub = dial_plugin(p.socket_path)
            stream = stub.ListAndWatch(dpapi.Empty())
            n = len(next(stream).devices)
            assert n == 2
            ch.close()
    finally:
        stop.set()
        ctrl.stop()
        kubelet.stop()


def test_start_failure_tolerated(synthetic_host):
    """A type that fails to start is dropped; others continue
    (reference: device_plugin.go:131-136)."""
    h = synthetic_host
    h.add_gpu("0000:10:00.0", iommu_group="100")
    h.add_gpu("0000:20:00.0", iommu_group="101", device_id="74a1")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    ctrl = Controller(config=cfg, kfd_nodes_dir=h.kfd_nodes)
    plugins = ctrl.create_plugins()
    assert len(plugins) == 2
    # sabotage one plugin's socket path
    plugins[0].socket_path = "/nonexistent-dir/x.sock"
    stop = threading.Event()
    try:
        started = ctrl.start(stop)
        assert len(started) == 1
    finally:
        stop.set()
        ctrl.stop()
        kubelet.stop()


def test_mixed_device_ids_two_resource_types(synthetic_host):
    """MI300X and MI355X on one node → two distinct resource names from
    pci.ids (reference: one plugin per device id, device_plugin.go:108)."""
    h = synthetic_host
    h.add_gpu("0000:10:00.0", device_id="75a3", iommu_group="100")
    h.add_gpu("0000:11:00.0", device_id="74a1", iommu_group="101")
    ctrl = Controller(config=h.config(), kfd_nodes_dir=h.kfd_nodes)
    plugins = ctrl.create_plugins()
    assert sorted(p.device_name for p in plugins) == [
        "AQUA_VANJARAM_INSTINCT_MI300X", "INSTINCT_MI355X"]
    assert all(len(p._devs) == 1 for p in plugins)


def test_unknown_device_id_uses_raw_hex(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:10:00.0", device_id="beef", iommu_group="100")
    ctrl = Controller(config=h.config(), kfd_nodes_dir=h.kfd_nodes)
    plugins = ctrl.create_plugins()
    assert [p.device_name for p in plugins] == ["beef"]
    assert plugins[0].resource_name == "amd.com/beef"


def test_vf_partition_logging(synthetic_host, caplog):
    import logging
    from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
        log_vf_partitioning,
    )
    from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
    h = synthetic_host
    h.add_gpu("0000:20:00.0", driver="gim", iommu_group="110")
    for v in range(8):
        h.add_vf("0000:20:02.%d" % v, pf_bdf="0000:20:00.0",
                 iommu_group=str(120 + v))
    reg = discovery.discover(base_path=h.pci)
    with caplog.at_level(logging.INFO):
        log_vf_partitioning(reg)
    assert any("8 VFs" in r.message and "36.0 GiB" in r.message
               for r in caplog.records)
