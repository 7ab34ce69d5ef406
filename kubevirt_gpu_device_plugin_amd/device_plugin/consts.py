"""Compile-time constants of the AMD device plugin.

The reference keeps a zero-config surface — no flags, no env vars
(reference: cmd/main.go:33-35) — all knobs are constants.  We keep the
same UX with the AMD values.  The resource-name format
``<namespace>/<PCI-NAME>`` and socket naming ``kubevirt-<NAME>.sock``
stay structurally identical to the reference for KubeVirt compatibility
(reference: generic_device_plugin.go:52,88,300).
"""

# PCI vendor id of AMD (reference uses NVIDIA "10de", device_plugin.go:46).
AMD_VENDOR_ID = "1002"

# Resource namespace: resources are advertised as amd.com/<NAME>
# (reference: DeviceNamespace "nvidia.com", generic_device_plugin.go:52).
DEVICE_NAMESPACE = "amd.com"

# Env prefix consumed by KubeVirt's virt-launcher for PCI host devices.
# virt-launcher derives PCI_RESOURCE_<SANITIZED_RESOURCE_NAME>; for
# amd.com/<NAME> that is PCI_RESOURCE_AMD_COM_<NAME>
# (reference: gpuPrefix, generic_device_plugin.go:58).
GPU_ENV_PREFIX = "PCI_RESOURCE_AMD_COM"

# SR-IOV VFs are *PCI* devices bound to vfio-pci (unlike the reference's
# mdevs), so KubeVirt claims them as pciHostDevices and reads the same
# PCI_RESOURCE_* env.  They get their own resource types via the VF PCI
# device id (reference mdev analogue: vgpuPrefix MDEV_PCI_RESOURCE_…,
# generic_vgpu_device_plugin.go:59; see SURVEY.md §7.4 for why the mdev
# walker is replaced by VF discovery in the PCI walk).
VF_ENV_PREFIX = GPU_ENV_PREFIX

# Host paths (reference: device_plugin.go:70-79,
# generic_device_plugin.go:54-57).
PCI_DEVICES_PATH = "/sys/bus/pci/devices"
VFIO_DEVICE_PATH = "/dev/vfio"
IOMMU_DEVICE_PATH = "/dev/iommu"
KFD_TOPOLOGY_PATH = "/sys/class/kfd/kfd/topology/nodes"
PCI_IDS_FILE_PATH = "/usr/pci.ids"

# Drivers a passthrough-ready function may be bound to
# (reference: supportedVfioDrivers, device_plugin.go:75-78; on MI355X
# only plain vfio-pci exists — gim-created VFs are bound to vfio-pci for
# guest passthrough, the PF stays on the gim host driver and is NOT
# allocatable).
SUPPORTED_VFIO_DRIVERS = frozenset({"vfio-pci"})

# gRPC connection timeout (reference: connectionTimeout,
# generic_device_plugin.go:53).
CONNECTION_TIMEOUT_S = 5.0

# Period of the ground-truth health resync in the health loop: the only
# health source while the /dev/vfio inotify watch cannot be established,
# and a safety net against missed events otherwise (the reference goes
# fully blind when its fsnotify setup fails,
# generic_device_plugin.go:626-637).
HEALTH_RESYNC_S = 30.0

# Kubelet-restart re-registration backoff: doubled per failed attempt
# up to the cap, retried forever while the daemon lives (the reference
# abandons the resource after a single failed re-register,
# generic_device_plugin.go:688-692).
RESTART_BACKOFF_INITIAL_S = 0.5
RESTART_BACKOFF_MAX_S = 30.0

# Device node permissions requested from kubelet.
DEVICE_PERMISSIONS = "mrw"

# HBM3E capacity per MI355X, used only for logging/validation of VF
# partitioning (288 GB / 8 VFs = 36 GB per VF).
MI355X_HBM_BYTES = 288 * 1024**3
