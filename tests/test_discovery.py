"""Discovery walk tests (reference: device_plugin_test.go:279-323)."""

from kubevirt_gpu_device_plugin_amd.device_plugin import consts, discovery


def test_basic_discovery(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40", numa=0)
    h.add_gpu("0000:2f:00.0", iommu_group="41", numa=1)
    reg = discovery.discover(base_path=h.pci)
    assert set(reg.device_map) == {"75a3"}
    assert sorted(d.addr for d in reg.device_map["75a3"]) == [
        "0000:0c:00.0", "0000:2f:00.0"]
    assert reg.bdf_to_iommu == {"0000:0c:00.0": "40",
                                "0000:2f:00.0": "41"}
    assert [d.addr for d in reg.iommu_map["40"]] == ["0000:0c:00.0"]
    assert reg.device_map["75a3"][1].numa_node == 1
    assert not reg.vf_map and not reg.pf_vf_map


def test_non_amd_vendor_filtered(synthetic_host):
    h = synthetic_host
    h.add_pci_device("0000:01:00.0", vendor="10de", device_id="2331")
    h.add_pci_device("0000:02:00.0", vendor="8086", device_id="1521")
    h.add_gpu("0000:0c:00.0")
    reg = discovery.discover(base_path=h.pci)
    assert sorted(reg.bdf_to_iommu) == ["0000:0c:00.0"]


def test_unsupported_driver_filtered(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", driver="amdgpu")
    h.add_gpu("0000:2f:00.0", driver="vfio-pci")
    reg = discovery.discover(base_path=h.pci)
    assert sorted(reg.bdf_to_iommu) == ["0000:2f:00.0"]


def test_missing_driver_skipped(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", driver=None)
    reg = discovery.discover(base_path=h.pci)
    assert not reg.bdf_to_iommu


def test_missing_iommu_group_skipped(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group=None)
    reg = discovery.discover(base_path=h.pci)
    assert not reg.bdf_to_iommu


def test_multifunction_iommu_group(synthetic_host):
    """Two functions co-grouped: both land in iommu_map under one group
    (the Allocate group-expansion depends on this,
    reference: device_plugin.go:221-243)."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")
    h.add_gpu("0000:0c:00.1", iommu_group="40")
    reg = discovery.discover(base_path=h.pci)
    assert sorted(d.addr for d in reg.iommu_map["40"]) == [
        "0000:0c:00.0", "0000:0c:00.1"]


def test_vf_discovery(synthetic_host):
    """gim SR-IOV VFs: classified by physfn, mapped PF→VF for health
    fan-out (replaces reference's mdev walk, device_plugin.go:255-291)."""
    h = synthetic_host
    # PF is bound to the gim host driver → NOT allocatable
    h.add_gpu("0000:0c:00.0", driver="gim", iommu_group="40")
    for i in range(1, 9):
        h.add_vf("0000:0c:02.%d" % (i % 8), pf_bdf="0000:0c:00.0",
                 iommu_group=str(50 + i), numa=0)
    reg = discovery.discover(base_path=h.pci)
    assert not reg.device_map  # PF on gim driver is filtered out
    assert len(reg.vf_map["75b3"]) == 8
    assert len(reg.pf_vf_map["0000:0c:00.0"]) == 8
    assert all(d.is_vf for d in reg.vf_map["75b3"])


def test_numa_error_defaults_to_zero(synthetic_host):
    import os
    h = synthetic_host
    d = h.add_gpu("0000:0c:00.0")
    os.remove(os.path.join(d, "numa_node"))
    reg = discovery.discover(base_path=h.pci)
    assert reg.device_map["75a3"][0].numa_node == 0


def test_mi355x_node_scale(synthetic_host):
    """BASELINE config 4: 8 GPUs × 8 VFs = 64 VFs + nothing passthrough."""
    h = synthetic_host
    for g in range(8):
        pf = "0000:%02x:00.0" % (0x10 + g)
        h.add_gpu(pf, driver="gim", iommu_group=str(100 + g))
        for v in range(8):
            h.add_vf("0000:%02x:02.%d" % (0x10 + g, v), pf_bdf=pf,
                     iommu_group=str(200 + g * 8 + v),
                     numa=g // 4)
    reg = discovery.discover(base_path=h.pci)
    assert len(reg.vf_map["75b3"]) == 64
    assert len(reg.pf_vf_map) == 8
    assert all(len(v) == 8 for v in reg.pf_vf_map.values())


def test_native_python_parity(synthetic_host):
    """The C++ scanner and the Python walk must build identical
    registries."""
    import pytest
    h = synthetic_host
    h.add_gpu("0000:10:00.0", iommu_group="100", numa=1)
    h.add_gpu("0000:11:00.0", iommu_group="100", numa=1)  # co-grouped
    h.add_gpu("0000:20:00.0", driver="gim", iommu_group="110")
    h.add_vf("0000:20:02.0", pf_bdf="0000:20:00.0", iommu_group="120")
    h.add_pci_device("0000:30:00.0", vendor="10de")
    try:
        reg_native = discovery.discover(base_path=h.pci, use_native=True)
    except RuntimeError:
        pytest.skip("native _sysfs extension not built")
    reg_py = discovery.discover(base_path=h.pci, use_native=False)
    assert reg_native.bdf_to_iommu == reg_py.bdf_to_iommu
    assert {k: [d.addr for d in v] for k, v in reg_native.iommu_map.items()} \
        == {k: [d.addr for d in v] for k, v in reg_py.iommu_map.items()}
    assert {k: [(d.addr, d.numa_node) for d in v]
            for k, v in reg_native.device_map.items()} \
        == {k: [(d.addr, d.numa_node) for d in v]
            for k, v in reg_py.device_map.items()}
    assert reg_native.pf_vf_map == reg_py.pf_vf_map


def test_garbage_numa_defaults_to_zero(synthetic_host):
    import os
    h = synthetic_host
    d = h.add_gpu("0000:0c:00.0")
    with open(os.path.join(d, "numa_node"), "w") as f:
        f.write("not-a-number\n")
    reg = discovery.discover(base_path=h.pci, use_native=False)
    assert reg.device_map["75a3"][0].numa_node == 0


def test_second_vfio_driver_single_extension_point(synthetic_host):
    """A future vfio-pci-variant driver (the AMD analogue of the
    reference's second accepted driver, nvgrace_gpu_vfio_pci,
    device_plugin.go:75-78) is enabled by adding ONE name to
    consts.SUPPORTED_VFIO_DRIVERS: the same set gates both the Python
    walk and the native-scan records."""
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", iommu_group="40")                # vfio-pci
    h.add_gpu("0000:2f:00.0", iommu_group="41",
              driver="vfio-pci-amdfuture")                     # variant

    for use_native in (False, None):
        reg = discovery.discover(base_path=h.pci,
                                 use_native=use_native)
        assert [d.addr for d in reg.device_map["75a3"]] == \
            ["0000:0c:00.0"], "unknown driver must be skipped"

        extended = consts.SUPPORTED_VFIO_DRIVERS | \
            {"vfio-pci-amdfuture"}
        reg = discovery.discover(base_path=h.pci,
                                 supported_drivers=extended,
                                 use_native=use_native)
        assert [d.addr for d in reg.device_map["75a3"]] == \
            ["0000:0c:00.0", "0000:2f:00.0"], \
            "one-place extension must cover this scan path"
