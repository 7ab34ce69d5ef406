"""KFD topology parsing tests."""

from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from kubevirt_gpu_device_plugin_amd.topology import (
    build_island_lookup, island_map_from_kfd,
)
from kubevirt_gpu_device_plugin_amd.topology.kfd import bdf_from_location


def test_bdf_from_location():
    assert bdf_from_location(0, (0x0C << 8) | (0x00 << 3) | 0) \
        == "0000:0c:00.0"
    assert bdf_from_location(0x1, (0xC3 << 8) | (0x1F << 3) | 7) \
        == "0001:c3:1f.7"


def test_island_map_from_kfd(synthetic_host):
    h = synthetic_host
    h.add_kfd_node(0, bdf=None, simd_count=0)  # CPU node, skipped
    h.add_kfd_node(1, bdf="0000:0c:00.0", hive_id=0xABCD)
    h.add_kfd_node(2, bdf="0000:2f:00.0", hive_id=0xABCD)
    h.add_kfd_node(3, bdf="0000:aa:00.0", hive_id=0)  # no hive
    m = island_map_from_kfd(h.kfd_nodes)
    assert m["0000:0c:00.0"] == 0xABCD
    assert m["0000:2f:00.0"] == 0xABCD
    assert m["0000:aa:00.0"] == -1


def test_island_lookup_resolves_vf_through_pf(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0", driver="gim", iommu_group="40")
    h.add_vf("0000:0c:02.0", pf_bdf="0000:0c:00.0", iommu_group="50")
    h.add_kfd_node(1, bdf="0000:0c:00.0", hive_id=0x77)
    reg = discovery.discover(base_path=h.pci)
    island_of = build_island_lookup(reg, nodes_dir=h.kfd_nodes,
                                    use_amdsmi=False)
    assert island_of("0000:0c:02.0") == 0x77  # VF inherits PF island
    assert island_of("0000:0c:00.0") == 0x77
    assert island_of("0000:ff:00.0") == -1


def test_island_lookup_without_kfd(synthetic_host):
    h = synthetic_host
    h.add_gpu("0000:0c:00.0")
    reg = discovery.discover(base_path=h.pci)
    island_of = build_island_lookup(reg, nodes_dir=h.kfd_nodes,
                                    use_amdsmi=False)
    assert island_of("0000:0c:00.0") == -1
