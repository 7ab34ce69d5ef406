"""uevent netlink listener tests."""

import threading

import pytest

from kubevirt_gpu_device_plugin_amd.device_plugin import uevent


def test_parse_kernel_uevent():
    data = (b"bind@/devices/pci0000:00/0000:0c:02.0\0"
            b"ACTION=bind\0DEVPATH=/devices/pci0000:00/0000:0c:02.0\0"
            b"SUBSYSTEM=pci\0DRIVER=vfio-pci\0"
            b"PCI_ID=1002:75B3\0PCI_SLOT_NAME=0000:0c:02.0\0SEQNUM=42\0")
    ev = uevent.parse_uevent(data)
    assert ev["ACTION"] == "bind"
    assert ev["SUBSYSTEM"] == "pci"
    assert ev["PCI_ID"] == "1002:75B3"
    assert ev["PCI_SLOT_NAME"] == "0000:0c:02.0"


def test_parse_rejects_libudev():
    assert uevent.parse_uevent(b"libudev\0\1\2\3whatever") is None
    assert uevent.parse_uevent(b"garbage-without-at-sign\0x=y\0") is None


def test_amd_pci_filter():
    base = {"ACTION": "bind", "SUBSYSTEM": "pci", "PCI_ID": "1002:75B3"}
    assert uevent.is_amd_pci_driver_event(dict(base))
    assert uevent.is_amd_pci_driver_event(
        dict(base, ACTION="unbind"))
    assert uevent.is_amd_pci_driver_event(
        dict(base, ACTION="remove"))
    assert not uevent.is_amd_pci_driver_event(
        dict(base, ACTION="change"))
    assert not uevent.is_amd_pci_driver_event(
        dict(base, PCI_ID="10DE:2331"))
    assert not uevent.is_amd_pci_driver_event(
        dict(base, SUBSYSTEM="usb"))
    assert not uevent.is_amd_pci_driver_event(None)
    assert not uevent.is_amd_pci_driver_event(
        {"ACTION": "bind", "SUBSYSTEM": "pci"})  # no PCI_ID


def test_listener_socket_lifecycle():
    """Bind the real netlink socket when the environment allows; the
    thread must start and stop cleanly."""
    rescan = threading.Event()
    stop = threading.Event()
    try:
        listener = uevent.UeventListener(rescan)
    except OSError:
        pytest.skip("netlink uevent socket unavailable here")
    t = listener.start(stop.is_set)
    assert t.is_alive()
    stop.set()
    t.join(timeout=5)
    assert not t.is_alive()


def test_start_listener_degrades(monkeypatch):
    """When the socket can't be created, start_listener returns None
    and the daemon continues (SIGHUP-only rescan)."""
    def boom(*a, **k):
        raise OSError(1, "no netlink for you")
    monkeypatch.setattr(uevent, "UeventListener", boom)
    rescan = threading.Event()
    assert uevent.start_listener(rescan, lambda: False) is None
