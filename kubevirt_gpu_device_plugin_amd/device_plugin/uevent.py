"""Kernel uevent (netlink) listener for PCI driver bind/unbind.

Automatic hotplug: when a vendor-1002 function is bound to or unbound
from a driver (gim creating VFs, an operator running driverctl), the
kernel broadcasts a uevent on NETLINK_KOBJECT_UEVENT; the controller
rescans.  The reference discovers exactly once and needs a process
restart (SURVEY.md §5 "no hotplug re-scan"); SIGHUP remains as the
manual trigger and sole mechanism when the netlink socket is
unavailable (e.g. every capability dropped AND non-root).

Kernel uevent wire format: "ACTION@DEVPATH\\0KEY=VALUE\\0..." — PCI
events carry PCI_ID ("1002:75B3"), PCI_SLOT_NAME (the BDF) and
DRIVER.  udevd's repeats use a different magic ("libudev") and are
filtered out.
"""

import logging
import os
import select
import socket
import threading

from . import consts

log = logging.getLogger(__name__)

NETLINK_KOBJECT_UEVENT = 15
UEVENT_GROUP_KERNEL = 1

# actions that change what discovery would find
RESCAN_ACTIONS = frozenset({"bind", "unbind", "add", "remove"})


def parse_uevent(data):
    """Parse one uevent datagram into a dict (None for non-kernel
    messages, e.g. udevd's libudev-tagged repeats)."""
    if data.startswith(b"libudev") or b"@" not in data.split(b"\0", 1)[0]:
        return None
    parts = data.split(b"\0")
    header = parts[0].decode("utf-8", "replace")
    action, _, devpath = header.partition("@")
    ev = {"ACTION": action, "DEVPATH": devpath}
    for p in parts[1:]:
        if b"=" in p:
            k, _, v = p.partition(b"=")
            ev[k.decode("utf-8", "replace")] = v.decode(
                "utf-8", "replace")
    return ev


def is_amd_pci_driver_event(ev, vendor=consts.AMD_VENDOR_ID):
    """True when the event is a driver/bus change of a vendor-1002 PCI
    function — the kind that changes discovery output."""
    if ev is None or ev.get("ACTION") not in RESCAN_ACTIONS:
        return False
    if ev.get("SUBSYSTEM") != "pci":
        return False
    return ev.get("PCI_ID", "").lower().startswith(vendor.lower() + ":")


class UeventListener:
    """Background thread: netlink uevents → rescan_event.set().

    Construction raises OSError when the netlink socket cannot be
    opened (insufficient privileges); callers degrade to SIGHUP-only.
    """

    def __init__(self, rescan_event, vendor=consts.AMD_VENDOR_ID):
        self._rescan_event = rescan_event
        self._vendor = vendor
        self._sock = socket.socket(socket.AF_NETLINK, socket.SOCK_DGRAM,
                                   NETLINK_KOBJECT_UEVENT)
        try:
            self._sock.bind((os.getpid(), UEVENT_GROUP_KERNEL))
        except OSError:
            self._sock.close()
            raise
        self._thread = None

    def start(self, should_stop):
        self._thread = threading.Thread(
            target=self._run, args=(should_stop,),
            name="uevent-listener", daemon=True)
        self._thread.start()
        return self._thread

    def _run(self, should_stop):
        try:
            while not should_stop():
                r, _, _ = select.select([self._sock], [], [], 0.5)
                if not r:
                    continue
                try:
                    data = self._sock.recv(16384)
                except OSError:
                    return
                ev = parse_uevent(data)
                if is_amd_pci_driver_event(ev, self._vendor):
                    log.info("uevent %s %s (%s): scheduling rescan",
                             ev.get("ACTION"), ev.get("PCI_SLOT_NAME",
                                                      ev.get("DEVPATH")),
                             ev.get("DRIVER", "?"))
                    self._rescan_event.set()
        finally:
            self._sock.close()


def start_listener(rescan_event, should_stop,
                   vendor=consts.AMD_VENDOR_ID):
    """Best-effort: returns the listener or None (logged) when netlink
    is unavailable."""
    try:
        listener = UeventListener(rescan_event, vendor=vendor)
    except OSError as e:
        log.warning("uevent netlink unavailable (%s); hotplug rescan "
                    "via SIGHUP only", e)
        return None
    listener.start(should_stop)
    log.info("listening for PCI uevents (auto-rescan on vendor %s "
             "driver changes)", vendor)
    return listener
