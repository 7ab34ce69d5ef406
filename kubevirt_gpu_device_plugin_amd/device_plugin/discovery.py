"""Discovery of vfio-bound AMD GPUs and SR-IOV VFs from sysfs.

One PCI walk feeds both the passthrough and the SR-IOV resource sets
(the reference needs two walks because NVIDIA vGPUs are mdevs under a
separate bus; MxGPU/gim VFs are ordinary PCI functions, so a single
walk + ``physfn`` classification is the idiomatic AMD design —
reference walks: device_plugin.go:187-291).

Semantics preserved from the reference walk (device_plugin.go:192-247):
  * only vendor ``1002`` functions are considered,
  * only functions bound to a supported vfio driver are allocatable,
  * a device missing vendor/driver/iommu_group info is skipped, not fatal,
  * NUMA read errors default the device to node 0,
  * every discovered function lands in ``iommu_map``/``bdf_to_iommu``
    (Allocate expands requests to full IOMMU groups).
"""

import logging
import os
from dataclasses import dataclass, field

from . import consts, sysfs

log = logging.getLogger(__name__)


@dataclass(frozen=True)
class AmdGpuDevice:
    """One allocatable PCI function (GPU or VF).

    ``addr`` is the full BDF directory name under /sys/bus/pci/devices
    (e.g. ``0000:0c:00.0``) — also the device ID advertised to kubelet
    (reference: NvidiaGpuDevice, device_plugin.go:50-53 and
    device_plugin.go:109-123).
    """
    addr: str
    numa_node: int
    device_id: str
    iommu_group: str
    parent_pf: str = ""   # non-empty for SR-IOV VFs

    @property
    def is_vf(self):
        return bool(self.parent_pf)


@dataclass
class DeviceRegistry:
    """All discovery output (replaces the reference's five package-global
    maps, device_plugin.go:56-68, with one immutable-after-discovery
    object so Allocate readers never race a rescan)."""
    # iommu group id -> all member devices (any vendor-1002 vfio function)
    iommu_map: dict = field(default_factory=dict)
    # PCI device id -> passthrough (non-VF) devices of that type
    device_map: dict = field(default_factory=dict)
    # PCI device id -> VF devices of that type (replaces vGpuMap)
    vf_map: dict = field(default_factory=dict)
    # parent PF BDF -> [VF BDFs]  (replaces gpuVgpuMap for health fan-out)
    pf_vf_map: dict = field(default_factory=dict)
    # BDF -> iommu group id
    bdf_to_iommu: dict = field(default_factory=dict)

    def all_devices(self):
        for devs in self.device_map.values():
            yield from devs
        for devs in self.vf_map.values():
            yield from devs


def _native_scan(base_path):
    """One-pass scan via the C++ _sysfs extension; None when the
    extension is unavailable (pure-Python walk takes over)."""
    try:
        from kubevirt_gpu_device_plugin_amd import _sysfs
    except ImportError:
        return None
    try:
        return _sysfs.scan_pci(base_path, consts.AMD_VENDOR_ID)
    except RuntimeError as e:
        log.error("native scan of %s failed: %s", base_path, e)
        return None


def discover(base_path=consts.PCI_DEVICES_PATH,
             supported_drivers=consts.SUPPORTED_VFIO_DRIVERS,
             use_native=None):
    """Walk the PCI bus once and build the :class:`DeviceRegistry`.

    ``use_native``: True forces the C++ scanner, False the Python walk,
    None picks native when built.  Both produce identical registries
    (pinned by tests/test_discovery.py::test_native_python_parity).
    """
    reg = DeviceRegistry()
    records = _native_scan(base_path) if use_native in (None, True) \
        else None
    if use_native is True and records is None:
        raise RuntimeError("native _sysfs extension required but "
                           "unavailable")
    if records is not None:
        for r in records:
            if r["driver"] not in supported_drivers:
                log.info("skipping %s: driver %s is not a supported "
                         "vfio driver", r["addr"], r["driver"])
                continue
            dev = AmdGpuDevice(addr=r["addr"], numa_node=r["numa_node"],
                               device_id=r["device"],
                               iommu_group=r["iommu_group"],
                               parent_pf=r["physfn"])
            _register(reg, dev)
        return reg

    try:
        entries = sorted(os.listdir(base_path))
    except OSError as e:
        log.error("cannot list %s: %s", base_path, e)
        return reg

    for addr in entries:
        try:
            vendor = sysfs.read_id_from_file(base_path, addr, "vendor")
        except OSError:
            log.info("could not get vendor id for device %s", addr)
            continue
        if vendor != consts.AMD_VENDOR_ID:
            continue
        try:
            driver = sysfs.read_link_basename(base_path, addr, "driver")
        except OSError:
            log.info("could not get driver for device %s", addr)
            continue
        if driver not in supported_drivers:
            log.info("skipping %s: driver %s is not a supported "
                     "vfio driver", addr, driver)
            continue
        try:
            iommu_group = sysfs.read_link_basename(
                base_path, addr, "iommu_group")
        except OSError:
            log.info("could not get IOMMU group for device %s", addr)
            continue
        try:
            numa_node = sysfs.read_numa_node(base_path, addr)
        except (OSError, ValueError) as e:
            log.info("could not get NUMA node for %s: %s; defaulting "
                     "to node 0", addr, e)
            numa_node = 0
        try:
            device_id = sysfs.read_id_from_file(base_path, addr, "device")
        except OSError:
            log.info("could not get device id for %s", addr)
            continue

        parent_pf = sysfs.read_physfn_addr(base_path, addr) or ""
        dev = AmdGpuDevice(addr=addr, numa_node=numa_node,
                           device_id=device_id, iommu_group=iommu_group,
                           parent_pf=parent_pf)
        _register(reg, dev)
    return reg


def warn_shared_groups(reg):
    """Surface IOMMU groups holding more than one allocatable function:
    vfio passes whole groups, so allocating one member binds the others
    into the same VM (island IOMMU grouping must be
    honored).  GetPreferredAllocation deprioritises these
    (allocation.py group_size_of)."""
    for group, devs in reg.iommu_map.items():
        if len(devs) > 1:
            log.warning(
                "IOMMU group %s holds %d allocatable functions (%s): "
                "they can only be passed through to the SAME VM", group,
                len(devs), ", ".join(d.addr for d in devs))


def _register(reg, dev):
    reg.iommu_map.setdefault(dev.iommu_group, []).append(dev)
    reg.bdf_to_iommu[dev.addr] = dev.iommu_group
    if dev.is_vf:
        reg.vf_map.setdefault(dev.device_id, []).append(dev)
        reg.pf_vf_map.setdefault(dev.parent_pf, []).append(dev.addr)
        log.info("discovered AMD VF %s (type %s, PF %s, iommu %s, "
                 "numa %d)", dev.addr, dev.device_id, dev.parent_pf,
                 dev.iommu_group, dev.numa_node)
    else:
        reg.device_map.setdefault(dev.device_id, []).append(dev)
        log.info("discovered AMD GPU %s (type %s, iommu %s, numa %d)",
                 dev.addr, dev.device_id, dev.iommu_group, dev.numa_node)
