"""sysfs reader tests (reference: device_plugin_test.go:137-219)."""

import os

import pytest

from kubevirt_gpu_device_plugin_amd.device_plugin import sysfs


def test_read_id_strips_0x(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", vendor="1002", device_id="75a3")
    assert sysfs.read_id_from_file(h.pci, "0000:0c:00.0", "vendor") \
        == "1002"
    assert sysfs.read_id_from_file(h.pci, "0000:0c:00.0", "device") \
        == "75a3"


def test_read_id_missing_file_raises(synthetic_host):
    with pytest.raises(OSError):
        sysfs.read_id_from_file(synthetic_host.pci, "nope", "vendor")


def test_numa_node_negative_clamps_to_zero(synthetic_host):
    h = synthetic_host
    d = h.add_gpu("0000:0c:00.0", numa=0)
    with open(os.path.join(d, "numa_node"), "w") as f:
        f.write("-1\n")
    assert sysfs.read_numa_node(h.pci, "0000:0c:00.0") == 0


def test_numa_node_value(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", numa=3)
    assert sysfs.read_numa_node(h.pci, "0000:0c:00.0") == 3


def test_read_link_basename(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", driver="vfio-pci", iommu_group="42")
    assert sysfs.read_link_basename(h.pci, "0000:0c:00.0", "driver") \
        == "vfio-pci"
    assert sysfs.read_link_basename(
        h.pci, "0000:0c:00.0", "iommu_group") == "42"


def test_physfn(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0")
    h.add_vf("0000:0c:02.0", pf_bdf="0000:0c:00.0")
    assert sysfs.read_physfn_addr(h.pci, "0000:0c:02.0") \
        == "0000:0c:00.0"
    assert sysfs.read_physfn_addr(h.pci, "0000:0c:00.0") is None


def test_read_vfio_dev(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", vfio_dev="vfio7")
    assert sysfs.read_vfio_dev(h.pci, "0000:0c:00.0") == "vfio7"


def test_read_vfio_dev_missing(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0")
    with pytest.raises(OSError):
        sysfs.read_vfio_dev(h.pci, "0000:0c:00.0")


def test_read_sriov_numvfs(synthetic_host):
    import os
    h = synthetic_host
    d = h.add_gpu("0000:0c:00.0")
    assert sysfs.read_sriov_numvfs(h.pci, "0000:0c:00.0") == 0
    with open(os.path.join(d, "sriov_numvfs"), "w") as f:
        f.write("8\n")
    assert sysfs.read_sriov_numvfs(h.pci, "0000:0c:00.0") == 8


def test_supports_iommufd_via_cdev_dir(tmp_path):
    """Containerized detection: /dev/iommu not mounted, but the
    /dev/vfio/devices cdev dir (under the mounted /dev/vfio) implies
    iommufd support on the host."""
    from kubevirt_gpu_device_plugin_amd.device_plugin import sysfs

    iommu = str(tmp_path / "iommu")
    vfio = tmp_path / "vfio"
    vfio.mkdir()
    assert not sysfs.supports_iommufd(iommu, vfio_dir=str(vfio))
    (vfio / "devices").mkdir()
    assert sysfs.supports_iommufd(iommu, vfio_dir=str(vfio))
    # direct /dev/iommu presence still wins on its own
    with open(iommu, "w"):
        pass
    assert sysfs.supports_iommufd(iommu, vfio_dir=None)
