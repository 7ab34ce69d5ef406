"""kubevirt-gpu-device-plugin-amd — MI355X-native KubeVirt GPU device plugin.

A brand-new Kubernetes device plugin, built from scratch for AMD Instinct
MI355X nodes, with the same kubelet-facing capabilities as
NVIDIA/kubevirt-gpu-device-plugin (the reference):

  * discovers AMD GPUs (PCI vendor 0x1002) bound to vfio-pci for full
    passthrough, keyed by IOMMU group and NUMA node
    (reference: pkg/device_plugin/device_plugin.go:187-252),
  * discovers MxGPU/gim SR-IOV virtual functions (``virtfn*``/``physfn``
    sysfs links) in place of the reference's mdev vGPU walk
    (reference: pkg/device_plugin/device_plugin.go:255-291),
  * serves the kubelet DevicePlugin v1beta1 gRPC API — ListAndWatch,
    Allocate, GetPreferredAllocation, PreStartContainer — one server per
    resource type on ``kubevirt-<NAME>.sock``
    (reference: pkg/device_plugin/generic_device_plugin.go),
  * watches device health via inotify on ``/dev/vfio/<group>`` nodes and
    AMD-SMI GPU events (reset / RAS — the CDNA analogue of NVML XID
    critical events, reference: generic_vgpu_device_plugin.go:388-434),
  * prefers xGMI-island + NUMA-local device sets in
    GetPreferredAllocation (reference is NUMA-only:
    generic_device_plugin.go:478-616).

Language note: the reference is Go + one cgo (C) binding over
libnvidia-ml.  This implementation is Python (daemon, gRPC) + C++ native
components: ``_amdsmi`` (dlopen binding over libamd_smi.so, mirroring the
reference's nvml_dl.go lazy-binding pattern), ``_sysfs`` (hot-path PCI
sysfs scanner) and a HIP (gfx950) GPU health probe.
"""

__version__ = "0.1.0"
