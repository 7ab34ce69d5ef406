"""Test rig: synthetic sysfs trees and a stub kubelet.

The reference achieves hardware independence with monkey-patched reader
funcs + tempdir sysfs trees (reference: device_plugin_test.go:54-165);
here the paths are injectable through PluginConfig so the same synthetic
trees work without monkey-patching.  The stub kubelet (a tiny
Registration gRPC server) is new scope the reference never had
(SURVEY.md §4 note) — it lets tests and the latency bench drive the full
register → dial-back → Allocate flow.
"""

import os
import queue
from concurrent import futures

import grpc

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.device_plugin.plugin_base import (
    PluginConfig,
)


class SyntheticHost:
    """Builds a fake host filesystem under a tempdir."""

    def __init__(self, root):
        import tempfile
        self.root = str(root)
        self.pci = os.path.join(self.root, "sys", "bus", "pci", "devices")
        self.vfio_dir = os.path.join(self.root, "dev", "vfio")
        self.iommu_dev = os.path.join(self.root, "dev", "iommu")
        # unix sockets live here — must stay SHORT (AF_UNIX caps paths
        # at ~107 bytes and pytest/xdist tmp paths exceed it)
        self.kubelet_dir = tempfile.mkdtemp(prefix="dpk-")
        self.kfd_nodes = os.path.join(self.root, "kfd", "nodes")
        for d in (self.pci, self.vfio_dir, self.kubelet_dir,
                  self.kfd_nodes):
            os.makedirs(d, exist_ok=True)
        # the vfio container node always exists on a vfio host
        self._touch(os.path.join(self.vfio_dir, "vfio"))

    def cleanup(self):
        import shutil
        shutil.rmtree(self.kubelet_dir, ignore_errors=True)

    @staticmethod
    def _touch(path):
        with open(path, "w"):
            pass

    def config(self):
        return PluginConfig(
            device_plugin_dir=self.kubelet_dir,
            kubelet_socket=os.path.join(self.kubelet_dir, "kubelet.sock"),
            vfio_dir=self.vfio_dir,
            iommu_dev=self.iommu_dev,
            pci_base=self.pci,
            connect_timeout_s=5.0,
        )

    # ---- devices --------------------------------------------------------

    def add_pci_device(self, bdf, vendor="1002", device_id="75a3",
                       driver="vfio-pci", iommu_group="42", numa=0,
                       physfn=None, vfio_dev=None, vfio_node=True):
        d = os.path.join(self.pci, bdf)
        os.makedirs(d, exist_ok=True)
        with open(os.path.join(d, "vendor"), "w") as f:
            f.write("0x%s\n" % vendor)
        with open(os.path.join(d, "device"), "w") as f:
            f.write("0x%s\n" % device_id)
        with open(os.path.join(d, "numa_node"), "w") as f:
            f.write("%d\n" % numa)
        if driver is not None:
            os.symlink("../../../bus/pci/drivers/%s" % driver,
                       os.path.join(d, "driver"))
        if iommu_group is not None:
            os.symlink("../../../kernel/iommu_groups/%s" % iommu_group,
                       os.path.join(d, "iommu_group"))
            if vfio_node:
                self.add_vfio_node(iommu_group)
        if physfn is not None:
            os.symlink("../%s" % physfn, os.path.join(d, "physfn"))
        if vfio_dev is not None:
            os.makedirs(os.path.join(d, "vfio-dev", vfio_dev),
                        exist_ok=True)
        return d

    def add_gpu(self, bdf, **kw):
        return self.add_pci_device(bdf, **kw)

    def add_vf(self, bdf, pf_bdf, device_id="75b3", **kw):
        return self.add_pci_device(bdf, device_id=device_id,
                                   physfn=pf_bdf, **kw)

    def add_vfio_node(self, group):
        self._touch(os.path.join(self.vfio_dir, str(group)))

    def remove_vfio_node(self, group):
        os.remove(os.path.join(self.vfio_dir, str(group)))

    def enable_iommufd(self):
        self._touch(self.iommu_dev)

    def add_kfd_node(self, index, bdf=None, hive_id=0, simd_count=256,
                     extra=None):
        d = os.path.join(self.kfd_nodes, str(index))
        os.makedirs(d, exist_ok=True)
        props = dict(extra or {})
        props["simd_count"] = simd_count
        props["hive_id"] = hive_id
        if bdf is not None:
            domain, bus, devfn = bdf.split(":")
            dev, fn = devfn.split(".")
            props["domain"] = int(domain, 16)
            props["location_id"] = (int(bus, 16) << 8) | \
                (int(dev, 16) << 3) | int(fn, 16)
        with open(os.path.join(d, "properties"), "w") as f:
            for k, v in props.items():
                f.write("%s %d\n" % (k, v))


class StubKubelet:
    """Registration gRPC server standing in for kubelet."""

    def __init__(self, socket_path):
        self.socket_path = socket_path
        self.requests = queue.Queue()
        outer = self

        class _Reg(dpapi.RegistrationServicer):
            def Register(self, request, context):  # noqa: N802
                outer.requests.put(request)
                return dpapi.Empty()

        self._server = grpc.server(futures.ThreadPoolExecutor(4))
        dpapi.add_registration_servicer(_Reg(), self._server)
        self._server.add_insecure_port("unix:" + socket_path)
        self._server.start()

    def wait_register(self, timeout=5.0):
        return self.requests.get(timeout=timeout)

    def stop(self):
        self._server.stop(grace=None)


def dial_plugin(socket_path, timeout=5.0):
    ch = grpc.insecure_channel("unix:" + socket_path)
    grpc.channel_ready_future(ch).result(timeout=timeout)
    return ch, dpapi.DevicePluginStub(ch)


class FakeSmi:
    """In-process fake of the _amdsmi extension for event-flow tests
    (reference analogue: fakeNvml* seams,
    generic_vgpu_device_plugin_test.go:43-74)."""

    def __init__(self, devices, events=()):
        # devices: [{'index':0,'bdf':'0000:0c:00.0','uuid':'...'}]
        self._devices = devices
        self._events = queue.Queue()
        for e in events:
            self._events.put(e)
        self.inited = False
        self.event_inited = set()
        self.masks = {}

    def push(self, index, etype, msg=""):
        self._events.put((index, etype, msg))

    def available(self):
        return True

    def init(self):
        self.inited = True

    def shutdown(self):
        self.inited = False

    def get_devices(self):
        return list(self._devices)

    def event_init(self, idx):
        self.event_inited.add(idx)

    def event_mask(self, idx, mask):
        self.masks[idx] = mask

    def event_stop(self, idx):
        self.event_inited.discard(idx)

    def get_events(self, timeout_ms):
        try:
            return [self._events.get(timeout=min(timeout_ms, 50) / 1000.0)]
        except queue.Empty:
            return []


def eventually(fn, timeout=5.0, interval=0.02):
    """Poll until fn() is truthy (replaces the reference tests'
    time.Sleep sync, which SURVEY.md §4 calls out as flaky)."""
    import time
    deadline = time.time() + timeout
    last = None
    while time.time() < deadline:
        last = fn()
        if last:
            return last
        time.sleep(interval)
    raise AssertionError("condition not met within %.1fs (last=%r)"
                         % (timeout, last))
