"""Property-based discovery: random synthetic PCI trees must produce
identical native/Python registries with consistent invariants."""

import os

from hypothesis import given, settings, strategies as st

from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
from tests.fixtures import SyntheticHost

bdf_part = st.integers(min_value=0, max_value=0xFF)


@st.composite
def node_config(draw):
    """A random node: up to 10 functions with random vendor/driver/
    grouping/VF-ness."""
    n = draw(st.integers(min_value=0, max_value=10))
    devices = []
    for i in range(n):
        devices.append({
            "bus": draw(bdf_part),
            "fn": draw(st.integers(min_value=0, max_value=7)),
            "vendor": draw(st.sampled_from(["1002", "10de", "8086"])),
            "device_id": draw(st.sampled_from(
                ["75a3", "75b3", "74a1", "beef"])),
            "driver": draw(st.sampled_from(
                ["vfio-pci", "amdgpu", "gim", None])),
            "group": str(draw(st.integers(min_value=0, max_value=5))),
            "numa": draw(st.integers(min_value=-1, max_value=3)),
            "vf_of": draw(st.integers(min_value=-1, max_value=n - 1))
            if n else -1,
        })
    return devices


@settings(max_examples=60, deadline=None)
@given(node_config())
def test_native_python_registry_parity_random(tmp_path_factory, cfg):
    tmp = tmp_path_factory.mktemp("h")
    h = SyntheticHost(tmp)
    try:
        addrs = []
        for i, d in enumerate(cfg):
            addr = "0000:%02x:%02x.%d" % (d["bus"], i, d["fn"])
            pf = addrs[d["vf_of"]] if 0 <= d["vf_of"] < len(addrs) \
                else None
            h.add_pci_device(addr, vendor=d["vendor"],
                             device_id=d["device_id"],
                             driver=d["driver"],
                             iommu_group=d["group"], numa=max(d["numa"],
                                                              -1),
                             physfn=pf)
            # overwrite numa with the raw (possibly negative) value
            with open(os.path.join(h.pci, addr, "numa_node"), "w") as f:
                f.write("%d\n" % d["numa"])
            addrs.append(addr)

        reg_n = discovery.discover(base_path=h.pci, use_native=True)
        reg_p = discovery.discover(base_path=h.pci, use_native=False)

        def canon(reg):
            return (
                sorted(reg.bdf_to_iommu.items()),
                {g: sorted((d.addr, d.numa_node, d.device_id,
                            d.parent_pf) for d in v)
                 for g, v in reg.iommu_map.items()},
                {k: sorted(d.addr for d in v)
                 for k, v in reg.device_map.items()},
                {k: sorted(d.addr for d in v)
                 for k, v in reg.vf_map.items()},
                {k: sorted(v) for k, v in reg.pf_vf_map.items()},
            )

        assert canon(reg_n) == canon(reg_p)

        # invariants
        for reg in (reg_n, reg_p):
            for bdf, grp in reg.bdf_to_iommu.items():
                assert any(d.addr == bdf for d in reg.iommu_map[grp])
            pf_types = set(reg.device_map) | set(reg.vf_map)
            for devs in list(reg.device_map.values()) + list(
                    reg.vf_map.values()):
                for d in devs:
                    assert d.numa_node >= 0
                    assert d.device_id in pf_types
            for vfs in reg.pf_vf_map.values():
                assert vfs  # no empty fan-out entries
    finally:
        h.cleanup()
