"""AMD-SMI access with graceful degradation.

The native binding lives in ``kubevirt_gpu_device_plugin_amd._amdsmi``
(csrc/amdsmi_binding.cpp) — a C++ extension that dlopens
``libamd_smi.so`` lazily at call time, mirroring the reference's NVML
cgo binding pattern (reference: nvml_dl.go:29-36, bindings.go:20-21) so
the daemon runs on driverless nodes.  This wrapper degrades to
"unavailable" when either the extension or the library is missing
(reference behavior: log and continue without NVML,
generic_vgpu_device_plugin.go:290-297).
"""

import logging

log = logging.getLogger(__name__)

try:
    from kubevirt_gpu_device_plugin_amd import _amdsmi as _ext
except ImportError as _e:  # extension not built on this host
    _ext = None
    log.info("native _amdsmi extension not importable: %s", _e)

# Event types (amdsmi_evt_notification_type_t, amdsmi.h:1336-1352).
EVT_VMFAULT = 1
EVT_THERMAL_THROTTLE = 2
EVT_GPU_PRE_RESET = 3
EVT_GPU_POST_RESET = 4


def event_mask(*event_types):
    """amdsmi masks use bit (value-1)
    (AMDSMI_EVENT_MASK_FROM_INDEX, amdsmi.h:1360)."""
    m = 0
    for t in event_types:
        m |= 1 << (t - 1)
    return m


def is_available():
    return _ext is not None and _ext.available()


def ext():
    """The raw extension module (None when not built)."""
    return _ext
