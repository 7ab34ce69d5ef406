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
import threading

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


class SharedSmiWatcher:
    """Process-wide multiplexer over ONE AmdSmiEventWatcher thread.

    ``amdsmi_get_gpu_event_notification`` drains a *global* event queue
    (amdsmi.h:6063-6104): two independent watcher loops would steal
    each other's events and race init/shut_down.  Plugins therefore
    subscribe here; one thread watches the union of their PFs and
    dispatches by BDF.  The thread restarts (generation bump) when the
    subscriber set changes and exits when it empties.
    """

    def __init__(self, smi=None, poll_ms=POLL_MS):
        self._smi = smi
        self._poll_ms = poll_ms
        self._lock = threading.Lock()    # protects _subs/_generation
        self._mgmt = threading.RLock()   # serializes (un)subscribe
        self._subs = {}       # handle -> (frozenset(pfs), on_un, on_ok)
        self._next_handle = 0
        self._generation = 0
        self._thread = None

    class Subscription:
        def __init__(self, owner, handle):
            self._owner = owner
            self._handle = handle

        def unsubscribe(self):
            self._owner._unsubscribe(self._handle)

    def _available(self):
        if self._smi is not None:
            return self._smi.available()
        return is_available()

    def subscribe(self, pf_bdfs, on_unhealthy, on_healthy):
        """Returns a Subscription, or None when AMD-SMI is unavailable
        on this host (callers degrade to sysfs-only health).  Checked
        per call, not cached: a library that appears after daemon start
        is picked up by the next (re)subscription attempt."""
        if not self._available():
            return None
        with self._mgmt:
            with self._lock:
                handle = self._next_handle
                self._next_handle += 1
                self._subs[handle] = (
                    frozenset(b.lower() for b in pf_bdfs),
                    on_unhealthy, on_healthy)
            self._restart()
        return self.Subscription(self, handle)

    def _unsubscribe(self, handle):
        with self._mgmt:
            with self._lock:
                self._subs.pop(handle, None)
            self._restart()

    def _restart(self):
        """Called under _mgmt: retire the running generation, join it,
        start a new one for the current subscriber union."""
        with self._lock:
            self._generation += 1
            gen = self._generation
            old = self._thread
            subs = list(self._subs.values())
        if old is not None and old.is_alive() \
                and old is not threading.current_thread():
            # the old generation notices the bump within one poll;
            # join so two loops never touch the amdsmi event API at once
            old.join(timeout=30.0)
        self._thread = None
        if not subs:
            return
        pfs = sorted(set().union(*(s[0] for s in subs)))
        t = threading.Thread(target=self._run, args=(gen, pfs),
                             name="smi-events-shared", daemon=True)
        self._thread = t
        t.start()

    def _dispatch(self, which, pf):
        pf = pf.lower()
        with self._lock:
            subs = list(self._subs.values())
        for pfs, on_un, on_ok in subs:
            if pf in pfs:
                (on_un if which == "unhealthy" else on_ok)(pf)

    def _run(self, gen, pfs):
        watcher = AmdSmiEventWatcher(smi=self._smi,
                                     poll_ms=self._poll_ms)
        try:
            watcher.watch(
                pfs,
                on_unhealthy=lambda pf: self._dispatch("unhealthy", pf),
                on_healthy=lambda pf: self._dispatch("healthy", pf),
                should_stop=lambda: gen != self._generation)
        except Exception:
            log.exception("shared AMD-SMI watcher failed; subscribers "
                          "fall back to sysfs-only health")


_shared = None
_shared_lock = threading.Lock()


def shared_watcher():
    """The module-level SharedSmiWatcher used by VF plugins."""
    global _shared
    with _shared_lock:
        if _shared is None:
            _shared = SharedSmiWatcher()
        return _shared
