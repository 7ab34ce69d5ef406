"""Rescan (hotplug) flow and concurrency/restart stress tests."""

import threading
from concurrent.futures import ThreadPoolExecutor

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
    build_kubelet_devices, initiate_device_plugin,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.plugin import (
    GenericDevicePlugin,
)
from tests.fixtures import StubKubelet, dial_plugin, eventually


def test_rescan_picks_up_new_vfs(synthetic_host):
    """gim creates VFs after daemon start → SIGHUP analogue (rescan
    event) makes them allocatable without a process restart (the
    reference cannot: SURVEY.md §5 'no hotplug re-scan')."""
    h = synthetic_host
    h.add_gpu("0000:10:00.0", iommu_group="100")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    stop = threading.Event()
    rescan = threading.Event()
    t = threading.Thread(
        target=initiate_device_plugin,
        kwargs=dict(stop_event=stop, rescan_event=rescan, config=cfg,
                    kfd_nodes_dir=h.kfd_nodes,
                    vf_event_watcher_factory=lambda: None),
        daemon=True)
    t.start()
    try:
        assert kubelet.wait_register(10).resource_name \
            == "amd.com/INSTINCT_MI355X"
        # VFs appear later (echo 8 > sriov_numvfs on a gim PF)
        h.add_gpu("0000:20:00.0", driver="gim", iommu_group="110")
        h.add_vf("0000:20:02.0", pf_bdf="0000:20:00.0",
                 iommu_group="120")
        rescan.set()
        # diff-rescan: only the NEW type registers; the existing
        # passthrough resource keeps its socket and registration
        req = kubelet.wait_register(10)
        assert req.resource_name == "amd.com/INSTINCT_MI355X_VF"
    finally:
        stop.set()
        t.join(timeout=10)
        kubelet.stop()


def test_concurrent_allocates(synthetic_host):
    """Parallel Allocate RPCs (kubelet admits pods concurrently) must
    not corrupt each other's specs/envs."""
    h = synthetic_host
    for g in range(8):
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
        bdfs = sorted(reg.bdf_to_iommu)

        def one(i):
            bdf = bdfs[i % 8]
            resp = stub.Allocate(dpapi.AllocateRequest(
                container_requests=[dpapi.ContainerAllocateRequest(
                    devices_ids=[bdf])]))
            c = resp.container_responses[0]
            assert dict(c.envs) == {
                "PCI_RESOURCE_AMD_COM_INSTINCT_MI355X": bdf}
            assert [d.host_path for d in c.devices] == [
                h.vfio_dir + "/vfio",
                h.vfio_dir + "/" + reg.bdf_to_iommu[bdf]]
            return True

        with ThreadPoolExecutor(8) as ex:
            assert all(ex.map(one, range(64)))
        ch.close()
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()


def test_repeated_kubelet_restarts(synthetic_host):
    """Several kubelet restarts in a row: plugin re-registers each time
    and keeps serving."""
    import os
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
        kubelet.wait_register(5)
        for _ in range(3):
            os.remove(plugin.socket_path)
            req = kubelet.wait_register(10)
            assert req.resource_name == "amd.com/INSTINCT_MI355X"
            eventually(lambda: os.path.exists(plugin.socket_path))
        ch, stub = dial_plugin(plugin.socket_path)
        assert stub.GetDevicePluginOptions(
            dpapi.Empty()).get_preferred_allocation_available
        ch.close()
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()


def test_vf_type_validation(synthetic_host):
    """A passthrough BDF sent to the VF plugin is rejected loudly."""
    import grpc
    import pytest
    from kubevirt_gpu_device_plugin_amd.device_plugin.vf_plugin import (
        VfDevicePlugin,
    )
    h = synthetic_host
    h.add_gpu("0000:10:00.0", iommu_group="100")
    h.add_gpu("0000:20:00.0", driver="gim", iommu_group="110")
    h.add_vf("0000:20:02.0", pf_bdf="0000:20:00.0", iommu_group="120")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    plugin = VfDevicePlugin(
        "INSTINCT_MI355X_VF",
        build_kubelet_devices(reg.vf_map["75b3"]), reg, config=cfg,
        event_watcher_factory=lambda: None)
    stop = threading.Event()
    plugin.start(stop)
    try:
        ch, stub = dial_plugin(plugin.socket_path)
        with pytest.raises(grpc.RpcError) as exc:
            stub.Allocate(dpapi.AllocateRequest(
                container_requests=[dpapi.ContainerAllocateRequest(
                    devices_ids=["0000:10:00.0"])]))
        assert exc.value.code() == grpc.StatusCode.INVALID_ARGUMENT
        assert "is not a INSTINCT_MI355X_VF" in exc.value.details()
        ch.close()
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()


def test_no_fd_leak_across_restarts(synthetic_host):
    """Each restart churns a gRPC server, an inotify fd and a kubelet
    channel; fd count must stay stable."""
    import os
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
        kubelet.wait_register(5)
        plugin.restart()
        kubelet.wait_register(5)
        before = len(os.listdir("/proc/self/fd"))
        for _ in range(5):
            plugin.restart()
            kubelet.wait_register(5)
        after = len(os.listdir("/proc/self/fd"))
        assert after - before <= 4, (before, after)
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()


def test_vf_plugin_with_real_smi_binding_degrades(synthetic_host):
    """The REAL _amdsmi binding on a driverless host: init fails inside
    the watcher thread, is logged, and the plugin keeps serving
    (reference: generic_vgpu_device_plugin.go:290-297)."""
    import pytest
    from kubevirt_gpu_device_plugin_amd import amdsmi
    from kubevirt_gpu_device_plugin_amd.device_plugin.vf_plugin import (
        VfDevicePlugin,
    )
    if not amdsmi.is_available():
        pytest.skip("_amdsmi extension not built")
    h = synthetic_host
    h.add_gpu("0000:20:00.0", driver="gim", iommu_group="110")
    h.add_vf("0000:20:02.0", pf_bdf="0000:20:00.0", iommu_group="120")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    plugin = VfDevicePlugin(
        "INSTINCT_MI355X_VF",
        build_kubelet_devices(reg.vf_map["75b3"]), reg, config=cfg)
    stop = threading.Event()
    plugin.start(stop)  # default watcher factory → real binding
    try:
        ch, stub = dial_plugin(plugin.socket_path)
        resp = stub.Allocate(dpapi.AllocateRequest(
            container_requests=[dpapi.ContainerAllocateRequest(
                devices_ids=["0000:20:02.0"])]))
        assert resp.container_responses[0].envs
        ch.close()
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()


def test_restart_retries_until_kubelet_returns(synthetic_host):
    """kubelet removes the plugin socket before its Registration
    service is back: the restart path must retry instead of dying
    (improvement over the reference, which gives up after one try)."""
    import os
    import time
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    plugin = GenericDevicePlugin(
        "INSTINCT_MI355X",
        build_kubelet_devices(reg.device_map["75a3"]), reg, config=cfg)
    plugin.config.connect_timeout_s = 1.0
    stop = threading.Event()
    plugin.start(stop)
    try:
        kubelet.wait_register(5)
        # kubelet "restarts": its socket disappears AND the plugin's
        # socket is removed
        kubelet.stop()  # grpc removes its own unix socket file
        os.remove(plugin.socket_path)
        time.sleep(2.5)  # let a first re-register attempt fail
        new_kubelet = StubKubelet(cfg.kubelet_socket)
        try:
            req = new_kubelet.wait_register(20)
            assert req.resource_name == "amd.com/INSTINCT_MI355X"
        finally:
            stop.set()
            plugin.stop()
            new_kubelet.stop()
    except BaseException:
        stop.set()
        plugin.stop()
        kubelet.stop()
        raise


def test_rescan_updates_existing_type_in_place(synthetic_host):
    """VF count change on an existing resource type: the live
    ListAndWatch stream gets the new inventory — same socket, no
    re-registration."""
    from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
        Controller,
    )
    h = synthetic_host
    h.add_gpu("0000:20:00.0", driver="gim", iommu_group="110")
    h.add_vf("0000:20:02.0", pf_bdf="0000:20:00.0", iommu_group="120")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    ctrl = Controller(config=cfg, kfd_nodes_dir=h.kfd_nodes,
                      vf_event_watcher_factory=lambda: None)
    ctrl.create_plugins()
    stop = threading.Event()
    try:
        started = ctrl.start(stop)
        kubelet.wait_register(10)
        plugin = started[0]
        ch, stub = dial_plugin(plugin.socket_path)
        stream = stub.ListAndWatch(dpapi.Empty())
        assert len(next(stream).devices) == 1

        # operator raises sriov_numvfs: 3 more VFs appear
        for i in (1, 2, 3):
            h.add_vf("0000:20:02.%d" % i, pf_bdf="0000:20:00.0",
                     iommu_group=str(120 + i))
        ctrl.rescan(stop)
        upd = next(stream)  # SAME stream — no socket churn
        assert len(upd.devices) == 4
        # no second registration happened
        assert kubelet.requests.empty()
        # new VF is allocatable immediately
        resp = stub.Allocate(dpapi.AllocateRequest(
            container_requests=[dpapi.ContainerAllocateRequest(
                devices_ids=["0000:20:02.3"])]))
        assert resp.container_responses[0].envs
        # health watch covers the new VF too
        h.remove_vfio_node("123")
        upd = next(stream)
        assert {d.ID: d.health for d in upd.devices}[
            "0000:20:02.3"] == "Unhealthy"
        ch.close()
    finally:
        stop.set()
        ctrl.stop()
        kubelet.stop()


def test_rescan_adds_and_removes_types(synthetic_host):
    """A vanished type's server stops; a new type registers."""
    from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
        Controller,
    )
    import os
    import shutil
    h = synthetic_host
    h.add_gpu("0000:10:00.0", iommu_group="100")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    ctrl = Controller(config=cfg, kfd_nodes_dir=h.kfd_nodes,
                      vf_event_watcher_factory=lambda: None)
    ctrl.create_plugins()
    stop = threading.Event()
    try:
        started = ctrl.start(stop)
        kubelet.wait_register(10)
        gpu_sock = started[0].socket_path
        # the GPU is unbound; a different-type device appears
        shutil.rmtree(os.path.join(h.pci, "0000:10:00.0"))
        h.add_gpu("0000:11:00.0", device_id="74a1", iommu_group="101")
        ctrl.rescan(stop)
        req = kubelet.wait_register(10)
        assert req.resource_name == "amd.com/AQUA_VANJARAM_INSTINCT_MI300X"
        assert not os.path.exists(gpu_sock)  # vanished type stopped
        assert [p.device_name for p in ctrl.plugins] == [
            "AQUA_VANJARAM_INSTINCT_MI300X"]
    finally:
        stop.set()
        ctrl.stop()
        kubelet.stop()


def test_rescan_preserves_health_state(synthetic_host):
    """An Unhealthy device must stay Unhealthy through a rescan."""
    from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
        Controller,
    )
    h = synthetic_host
    h.add_gpu("0000:10:00.0", iommu_group="100")
    h.add_gpu("0000:11:00.0", iommu_group="101")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    ctrl = Controller(config=cfg, kfd_nodes_dir=h.kfd_nodes,
                      vf_event_watcher_factory=lambda: None)
    ctrl.create_plugins()
    stop = threading.Event()
    try:
        started = ctrl.start(stop)
        kubelet.wait_register(10)
        plugin = started[0]
        h.remove_vfio_node("100")
        eventually(lambda: {d.ID: d.health
                            for d in plugin.devices_snapshot()}[
            "0000:10:00.0"] == "Unhealthy")
        h.add_gpu("0000:12:00.0", iommu_group="102")  # hotplug
        ctrl.rescan(stop)
        health = {d.ID: d.health for d in plugin.devices_snapshot()}
        assert health == {"0000:10:00.0": "Unhealthy",
                          "0000:11:00.0": "Healthy",
                          "0000:12:00.0": "Healthy"}
    finally:
        stop.set()
        ctrl.stop()
        kubelet.stop()


def test_rescan_resubscribes_smi_for_new_pf(synthetic_host):
    """A new gim PF (with VFs of an existing type) appears at rescan:
    the SMI subscription must cover it — a fault on the NEW PF flips
    its VFs."""
    from kubevirt_gpu_device_plugin_amd.amdsmi import EVT_GPU_PRE_RESET
    from kubevirt_gpu_device_plugin_amd.amdsmi.events import (
        SharedSmiWatcher,
    )
    from kubevirt_gpu_device_plugin_amd.device_plugin.vf_plugin import (
        VfDevicePlugin,
    )
    from kubevirt_gpu_device_plugin_amd import dpapi as _dpapi
    from tests.fixtures import FakeSmi
    h = synthetic_host
    h.add_gpu("0000:20:00.0", driver="gim", iommu_group="110")
    h.add_vf("0000:20:02.0", pf_bdf="0000:20:00.0", iommu_group="120")
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    smi = FakeSmi([{"index": 0, "bdf": "0000:20:00.0", "uuid": "a"},
                   {"index": 1, "bdf": "0000:21:00.0", "uuid": "b"}])
    shared = SharedSmiWatcher(smi=smi, poll_ms=50)
    reg = discovery.discover(base_path=h.pci)
    plugin = VfDevicePlugin(
        "INSTINCT_MI355X_VF",
        build_kubelet_devices(reg.vf_map["75b3"]), reg, config=cfg,
        smi_watcher=shared)
    stop = threading.Event()
    plugin.start(stop)
    try:
        eventually(lambda: smi.event_inited == {0})
        # second gim PF + VF appears
        h.add_gpu("0000:21:00.0", driver="gim", iommu_group="111")
        h.add_vf("0000:21:02.0", pf_bdf="0000:21:00.0",
                 iommu_group="121")
        reg2 = discovery.discover(base_path=h.pci)
        plugin.update_registry(
            reg2, build_kubelet_devices(reg2.vf_map["75b3"]))
        eventually(lambda: smi.event_inited == {0, 1})
        smi.push(1, EVT_GPU_PRE_RESET, "new pf fault")
        eventually(lambda: {d.ID: d.health
                            for d in plugin.devices_snapshot()} == {
            "0000:20:02.0": _dpapi.HEALTHY,
            "0000:21:02.0": _dpapi.UNHEALTHY})
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()
