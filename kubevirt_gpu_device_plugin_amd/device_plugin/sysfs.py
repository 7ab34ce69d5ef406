"""sysfs reader helpers.

Python equivalents of the reference's injectable reader funcs
(reference: device_plugin.go:293-357, generic_device_plugin.go:700-724).
Each helper raises ``OSError``/``ValueError`` on failure; callers decide
the policy (skip device, default NUMA 0, …) exactly as the reference does.

The hot Allocate path can swap these for the C++ ``_sysfs`` extension
(see csrc/sysfs_scan.cpp) — same semantics, one readdir+openat pass.
"""

import os


def read_id_from_file(base_path, device_address, prop):
    """Read a ``0x1002``-style sysfs id file, returning ``1002``.

    Mirrors the reference's ``data[2:]`` slice + newline trim
    (device_plugin.go:293-302): the leading ``0x`` is dropped
    positionally, so a malformed short file raises instead of guessing.
    """
    with open(os.path.join(base_path, device_address, prop), "rb") as f:
        data = f.read()
    return data[2:].decode("ascii").strip("\n")


def read_numa_node(base_path, device_address):
    """Parse ``numa_node``; negative (no NUMA) clamps to 0
    (reference: device_plugin.go:304-320)."""
    with open(os.path.join(base_path, device_address, "numa_node")) as f:
        node = int(f.read().strip())
    return node if node >= 0 else 0


def read_link_basename(base_path, device_address, link):
    """Basename of a sysfs symlink target, e.g. ``driver`` → ``vfio-pci``,
    ``iommu_group`` → ``42`` (reference: device_plugin.go:323-331)."""
    target = os.readlink(os.path.join(base_path, device_address, link))
    return os.path.basename(target)


def read_physfn_addr(base_path, device_address):
    """BDF of the parent physical function of an SR-IOV VF, or ``None``
    when the device is not a VF.  gim-created VFs carry a ``physfn``
    symlink to the PF (this replaces the reference's mdev parent-GPU
    derivation, device_plugin.go:347-357)."""
    try:
        target = os.readlink(
            os.path.join(base_path, device_address, "physfn"))
    except OSError:
        return None
    return os.path.basename(target)


def read_sriov_numvfs(base_path, device_address):
    """Number of VFs currently instantiated on a PF (0 when the file is
    absent, i.e. the function has no SR-IOV capability)."""
    try:
        with open(os.path.join(
                base_path, device_address, "sriov_numvfs")) as f:
            return int(f.read().strip())
    except (OSError, ValueError):
        return 0


def read_vfio_dev(base_path, device_address):
    """iommufd cdev name ``vfioN`` under the device's ``vfio-dev/`` dir
    (reference: readVFIODev, generic_device_plugin.go:710-724)."""
    vfio_dir = os.path.join(base_path, device_address, "vfio-dev")
    for entry in sorted(os.listdir(vfio_dir)):
        if entry.startswith("vfio") and os.path.isdir(
                os.path.join(vfio_dir, entry)):
            return entry
    raise FileNotFoundError("no iommufd device found for %s"
                            % device_address)


def supports_iommufd(iommu_dev_path="/dev/iommu", vfio_dir=None):
    """Host supports iommufd when /dev/iommu exists
    (reference: supportsIOMMUFD, generic_device_plugin.go:700-709) —
    or, containerized with only /dev/vfio mounted, when the cdev
    directory /dev/vfio/devices exists (CONFIG_VFIO_DEVICE_CDEV depends
    on IOMMUFD, so its presence implies /dev/iommu on the host even
    when that node is not mounted into this pod)."""
    if os.path.exists(iommu_dev_path):
        return True
    return vfio_dir is not None and os.path.isdir(
        os.path.join(vfio_dir, "devices"))
