"""SR-IOV VF device plugin server (MxGPU/gim).

Takes the role of the reference's GenericVGpuDevicePlugin
(reference: generic_vgpu_device_plugin.go).  Differences rooted in the
hardware model, not ported around:

  * NVIDIA vGPUs are mdevs (UUID devices under /sys/bus/mdev) with a
    single shared /dev/vfio spec; MxGPU VFs are *PCI functions* bound to
    vfio-pci, so a VF allocation uses the exact same vfio group wiring,
    TOCTOU revalidation and iommufd flow as passthrough — the plugin
    subclasses GenericDevicePlugin rather than duplicating the
    mdev-shaped Allocate (generic_vgpu_device_plugin.go:209-247).
  * Health: on top of the /dev/vfio node watch, a PF-level AMD-SMI event
    watcher (GPU reset / RAS — the CDNA analogue of the NVML XID
    critical watcher, generic_vgpu_device_plugin.go:388-434) marks every
    child VF of a faulting PF unhealthy via the pf_vf_map
    (replaces gpuVgpuMap fan-out, generic_vgpu_device_plugin.go:335-340).
  * GetPreferredAllocation is fully implemented (NUMA + xGMI island);
    the reference's vGPU variant is a nil stub
    (generic_vgpu_device_plugin.go:270-278).

Degradation: when libamd_smi is unavailable (not installed, or the PF is
not host-driver-resident) the watcher logs and the plugin continues with
sysfs-only health, matching the reference's NVML-absent behavior
(generic_vgpu_device_plugin.go:290-297).
"""

import logging

import grpc

from .. import dpapi
from . import consts
from .plugin import GenericDevicePlugin

log = logging.getLogger(__name__)


class VfDevicePlugin(GenericDevicePlugin):
    def __init__(self, device_name, devices, registry, config=None,
                 island_of=None, smi_watcher=None,
                 event_watcher_factory=None):
        super().__init__(device_name, devices, registry, config=config,
                         island_of=island_of,
                         env_prefix=consts.VF_ENV_PREFIX)
        # smi_watcher: a SharedSmiWatcher (None → process-wide default).
        # All VF plugins share ONE watcher thread because the amdsmi
        # event queue is global — independent pollers would steal each
        # other's events (see amdsmi/events.py SharedSmiWatcher).
        # event_watcher_factory=lambda: None disables SMI health
        # entirely (test/bench seam; reference analogue: watchXIDs var,
        # generic_vgpu_device_plugin.go:47).
        self._smi_watcher = smi_watcher
        self._event_watcher_factory = event_watcher_factory
        self._smi_sub = None

    def start(self, stop_event):
        super().start(stop_event)
        self._subscribe_smi(self.registry)

    def stop(self):
        sub, self._smi_sub = self._smi_sub, None
        if sub is not None:
            sub.unsubscribe()
        super().stop()

    def update_registry(self, registry, devices, island_of=None):
        super().update_registry(registry, devices, island_of=island_of)
        # PF set may have changed (new gim PFs) — resubscribe.  Always
        # attempted, not only when a prior subscription exists: if
        # libamd_smi was unavailable at start (subscribe returned None),
        # a rescan is exactly when it may have become available and VF
        # health must not silently stay sysfs-only for the daemon's
        # lifetime.
        self._subscribe_smi(registry)

    def _smi_disabled(self):
        """event_watcher_factory=lambda: None turns SMI health off
        entirely (test/bench seam)."""
        return (self._event_watcher_factory is not None
                and self._event_watcher_factory() is None)

    def _subscribe_smi(self, registry):
        """(Re)subscribe the shared watcher for the current PF set;
        degrades to sysfs-only health when AMD-SMI is unavailable
        (reference contract: generic_vgpu_device_plugin.go:290-297)."""
        if self._smi_disabled():
            log.info("[%s] SMI health disabled by configuration",
                     self.device_name)
            return
        sub, self._smi_sub = self._smi_sub, None
        if sub is not None:
            sub.unsubscribe()
        watcher = self._smi_watcher
        if watcher is None:
            from ..amdsmi import events as smi_events
            watcher = smi_events.shared_watcher()
        pf_bdfs = sorted({d.parent_pf
                          for devs in registry.vf_map.values()
                          for d in devs if d.parent_pf})
        self._smi_sub = watcher.subscribe(
            pf_bdfs,
            on_unhealthy=lambda pf: self.set_health(
                self._vfs_of_pf(pf), dpapi.UNHEALTHY),
            on_healthy=lambda pf: self.set_health(
                self._vfs_of_pf(pf), dpapi.HEALTHY))
        if self._smi_sub is None:
            log.warning("[%s] AMD-SMI unavailable; VF health relies on "
                        "vfio node watching only", self.device_name)

    def Allocate(self, request, context):  # noqa: N802
        """VF allocation = passthrough allocation, plus resource-type
        validation: every requested BDF must be a VF of *this* type.

        Deviation from the reference: its vGPU Allocate silently skips
        ids whose mdev type mismatches, returning an empty env
        (generic_vgpu_device_plugin.go:218-223); a loud INVALID_ARGUMENT
        beats a VM that boots without its GPU.
        """
        mine = {d.ID for d in self._devs}
        for req in request.container_requests:
            for bdf in req.devices_ids:
                if bdf not in mine:
                    context.abort(
                        grpc.StatusCode.INVALID_ARGUMENT,
                        "invalid allocation request: %s is not a %s "
                        "device" % (bdf, self.device_name))
        return super().Allocate(request, context)

    def _vfs_of_pf(self, pf_bdf):
        """Child VFs of a PF that belong to *this* resource type."""
        mine = {d.ID for d in self._devs}
        return [vf for vf in self.registry.pf_vf_map.get(pf_bdf, [])
                if vf in mine]

