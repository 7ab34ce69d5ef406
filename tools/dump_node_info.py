#!/usr/bin/env python3
"""Operator diagnostic: dump everything the plugin's discovery and
topology layers see on this node (PCI walk, KFD topology, AMD-SMI)."""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubevirt_gpu_device_plugin_amd import amdsmi  # noqa: E402
from kubevirt_gpu_device_plugin_amd.device_plugin import consts  # noqa: E402
from kubevirt_gpu_device_plugin_amd.topology import (  # noqa: E402
    island_map_from_kfd,
)


def pci_walk():
    try:
        from kubevirt_gpu_device_plugin_amd import _sysfs
        return _sysfs.scan_pci(consts.PCI_DEVICES_PATH,
                               consts.AMD_VENDOR_ID)
    except Exception as e:
        return {"error": str(e)}


def smi_dump():
    if not amdsmi.is_available():
        return {"available": False}
    smi = amdsmi.ext()
    out = {"available": True}
    try:
        smi.init()
    except RuntimeError as e:
        out["init_error"] = str(e)
        return out
    try:
        devs = smi.get_devices()
        out["devices"] = devs
        for d in devs:
            i = d["index"]
            for name, fn in (("xgmi", smi.xgmi_info),
                             ("ecc", smi.ecc_count)):
                try:
                    d[name] = fn(i)
                except RuntimeError as e:
                    d[name] = {"error": str(e)}
    finally:
        smi.shutdown()
    return out


def partition_modes():
    """MI300-family compute/memory partitioning (CPX/SPX, NPS1/NPS4)
    per host-driver-resident GPU — affects how many KFD nodes one PF
    exposes, so it belongs in any topology diagnosis."""
    import glob
    out = {}
    for card in sorted(glob.glob("/sys/class/drm/card*/device")):
        mode = {}
        for key in ("current_compute_partition",
                    "current_memory_partition"):
            try:
                with open(os.path.join(card, key)) as f:
                    mode[key] = f.read().strip()
            except OSError:
                pass
        if mode:
            try:
                bdf = os.path.basename(os.readlink(card))
            except OSError:
                bdf = card
            out[bdf] = mode
    return out


def main():
    print(json.dumps({
        "pci_vendor_1002": pci_walk(),
        "kfd_islands": island_map_from_kfd(),
        "partition_modes": partition_modes(),
        "amdsmi": smi_dump(),
    }, indent=2, default=str))


if __name__ == "__main__":
    main()
