"""PF-level GPU fault watcher over AMD-SMI event notifications.

The CDNA analogue of the reference's NVML XID critical-event watcher
(reference: watchXIDsFunc, generic_vgpu_device_plugin.go:388-434):

  * NVML ``XidCriticalError`` → amdsmi ``GPU_PRE_RESET`` (device going
    away for reset) and ``THERMAL_THROTTLE``/RAS conditions;
  * the reference skips application-level XIDs 31/43/45
    (generic_vgpu_device_plugin.go:416) → we skip ``VMFAULT`` (a guest
    page fault is an application error, not a device fault) and treat
    ``THERMAL_THROTTLE`` as log-only;
  * ``GPU_POST_RESET`` maps back to healthy — the reference has no
    recovery path for vGPUs; sysfs node re-creation covers it for VFs,
    and POST_RESET covers it here.

The 5000 ms poll matches the reference's ``WaitForEvent(…, 5000)``
(generic_vgpu_device_plugin.go:406).
"""

import logging

from . import (EVT_GPU_POST_RESET, EVT_GPU_PRE_RESET,
               EVT_THERMAL_THROTTLE, EVT_VMFAULT, event_mask, is_available,
               ext)

log = logging.getLogger(__name__)

POLL_MS = 5000

CRITICAL_EVENTS = (EVT_GPU_PRE_RESET,)
RECOVERY_EVENTS = (EVT_GPU_POST_RESET,)
WATCH_MASK = event_mask(EVT_VMFAULT, EVT_THERMAL_THROTTLE,
                        EVT_GPU_PRE_RESET, EVT_GPU_POST_RESET)


class AmdSmiEventWatcher:
    """Blocking watch loop; run on a daemon thread by VfDevicePlugin.

    Two fault sources, both fanned out PF → child VFs:
      * event notifications (reset/thermal/vmfault), handled per the
        criticality table above;
      * polled RAS totals: a growing *uncorrectable* ECC count marks
        the GPU unhealthy (amdsmi has no RAS entry in the notification
        enum, amdsmi.h:1336-1352, so this must be polled — the AMD
        analogue of memory-related critical XIDs).
    """

    def __init__(self, smi=None, poll_ms=POLL_MS):
        self._smi = smi or ext()
        self._poll_ms = poll_ms

    def _check_ecc(self, smi, by_index, baseline, on_unhealthy):
        for idx, bdf in by_index.items():
            try:
                ec = smi.ecc_count(idx)
            except (RuntimeError, AttributeError):
                continue  # RAS not supported here
            bad = ec.get("uncorrectable", 0)
            if idx not in baseline:
                baseline[idx] = bad
            elif bad > baseline[idx]:
                log.warning("uncorrectable ECC errors on %s: %d "
                            "(was %d)", bdf, bad, baseline[idx])
                baseline[idx] = bad
                on_unhealthy(bdf)

    def watch(self, pf_bdfs, on_unhealthy, on_healthy, should_stop):
        smi = self._smi
        smi.init()
        try:
            devices = smi.get_devices()  # [{'index','bdf','uuid'}]
            by_index = {}
            wanted = {b.lower() for b in pf_bdfs}
            for d in devices:
                if d["bdf"].lower() in wanted:
                    by_index[d["index"]] = d["bdf"]
            if not by_index:
                log.info("no AMD-SMI-visible PFs among %s (vfio-bound "
                         "PFs are invisible to the host driver)", pf_bdfs)
                return
            for idx in by_index:
                smi.event_init(idx)
                smi.event_mask(idx, WATCH_MASK)
            ecc_baseline = {}
            self._check_ecc(smi, by_index, ecc_baseline, on_unhealthy)
            try:
                while not should_stop():
                    for idx, etype, msg in smi.get_events(self._poll_ms):
                        bdf = by_index.get(idx)
                        if bdf is None:
                            continue
                        if etype in CRITICAL_EVENTS:
                            log.warning("critical GPU event %d on %s: %s",
                                        etype, bdf, msg)
                            on_unhealthy(bdf)
                        elif etype in RECOVERY_EVENTS:
                            log.info("GPU %s recovered (event %d)",
                                     bdf, etype)
                            on_healthy(bdf)
                        else:
                            log.info("GPU event %d on %s: %s",
                                     etype, bdf, msg)
                    self._check_ecc(smi, by_index, ecc_baseline,
                                    on_unhealthy)
            finally:
                for idx in by_index:
                    smi.event_stop(idx)
        finally:
            smi.shutdown()


def default_watcher():
    """Factory used by VfDevicePlugin; None ⇒ AMD-SMI unavailable."""
    if not is_available():
        return None
    return AmdSmiEventWatcher()
