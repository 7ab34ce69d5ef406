"""Benchmark rig: synthetic node + plugin daemon + stub kubelet client.

Implements the self-measured baseline SURVEY.md §6 / BASELINE.md call
for: a synthetic sysfs tree with N GPUs (optionally +VFs), the plugin
running as a *separate process* (a real unix-socket RPC boundary, as
kubelet would see), and a kubelet-side client that measures Allocate.
"""

import json
import os
import subprocess
import sys
import tempfile
import time

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO_ROOT)

from kubevirt_gpu_device_plugin_amd import dpapi  # noqa: E402
from tests.fixtures import StubKubelet, SyntheticHost, dial_plugin  # noqa: E402


def build_node(tmp, n_gpus, vfs_per_gpu=0, iommufd=False):
    """N passthrough MI355X (one per IOMMU group, NUMA split in half)
    and optionally gim PFs with VFs."""
    h = SyntheticHost(tmp)
    for g in range(n_gpus):
        bdf = "0000:%02x:00.0" % (0x10 + g)
        if vfs_per_gpu:
            h.add_gpu(bdf, driver="gim", iommu_group=str(100 + g),
                      numa=g * 2 // max(n_gpus, 2))
            for v in range(vfs_per_gpu):
                h.add_vf("0000:%02x:02.%d" % (0x10 + g, v), pf_bdf=bdf,
                         iommu_group=str(200 + g * 8 + v),
                         numa=g * 2 // max(n_gpus, 2),
                         vfio_dev="vfio%d" % (g * 8 + v)
                         if iommufd else None)
        else:
            h.add_gpu(bdf, iommu_group=str(100 + g),
                      numa=g * 2 // max(n_gpus, 2),
                      vfio_dev="vfio%d" % g if iommufd else None)
    if iommufd:
        h.enable_iommufd()
    return h


class PluginProcess:
    """The daemon in its own process, wired to the synthetic host."""

    def __init__(self, host):
        self.host = host
        cfg = host.config()
        self.proc = subprocess.Popen(
            [sys.executable, os.path.join(
                os.path.dirname(os.path.abspath(__file__)),
                "plugin_proc.py"),
             json.dumps({
                 "device_plugin_dir": cfg.device_plugin_dir,
                 "kubelet_socket": cfg.kubelet_socket,
                 "vfio_dir": cfg.vfio_dir,
                 "iommu_dev": cfg.iommu_dev,
                 "pci_base": cfg.pci_base,
                 "kfd_nodes_dir": host.kfd_nodes,
             })],
            cwd=REPO_ROOT)

    def stop(self):
        self.proc.terminate()
        try:
            self.proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            self.proc.kill()
            self.proc.wait()


def measure_allocate(n_gpus, steps, warmup, iommufd=False,
                     allocate_all=True, vfs_per_gpu=0):
    """Start the rig, wait for registration + full device list, then
    time `steps` Allocate RPCs.  Returns (latencies_s, n_advertised).

    ``vfs_per_gpu`` > 0 switches to the SR-IOV config (BASELINE config
    4): n_gpus gim PFs × vfs_per_gpu VFs; the VF resource is the one
    advertised, and each RPC allocates one PF's worth of VFs."""
    with tempfile.TemporaryDirectory() as tmp:
        host = build_node(tmp, n_gpus, vfs_per_gpu=vfs_per_gpu,
                          iommufd=iommufd)
        cfg = host.config()
        kubelet = StubKubelet(cfg.kubelet_socket)
        t_boot = time.perf_counter()
        plugin = PluginProcess(host)
        try:
            req = kubelet.wait_register(timeout=30.0)
            sock = os.path.join(cfg.device_plugin_dir, req.endpoint)
            ch, stub = dial_plugin(sock, timeout=10.0)
            stream = stub.ListAndWatch(dpapi.Empty())
            devices = [d.ID for d in next(stream).devices]
            # "time to allocatable": daemon exec → discovery → server up
            # → registered → first full device list streamed
            measure_allocate.last_startup_s = \
                time.perf_counter() - t_boot
            expected = n_gpus * vfs_per_gpu if vfs_per_gpu else n_gpus
            assert len(devices) == expected, \
                "advertised %d != %d" % (len(devices), expected)

            if vfs_per_gpu:
                # one full GPU's VF set per RPC, round-robin over PFs
                by_pf = {}
                for d in devices:
                    by_pf.setdefault(d.rsplit(".", 1)[0][:7], []) \
                        .append(d)
                reqs = [dpapi.AllocateRequest(container_requests=[
                    dpapi.ContainerAllocateRequest(devices_ids=vfs)])
                    for vfs in by_pf.values()]
            elif allocate_all:
                reqs = [dpapi.AllocateRequest(container_requests=[
                    dpapi.ContainerAllocateRequest(devices_ids=devices)])]
            else:
                reqs = [dpapi.AllocateRequest(container_requests=[
                    dpapi.ContainerAllocateRequest(devices_ids=[d])])
                    for d in devices]

            for i in range(warmup):
                stub.Allocate(reqs[i % len(reqs)])
            lat = []
            for i in range(steps):
                t0 = time.perf_counter()
                resp = stub.Allocate(reqs[i % len(reqs)])
                lat.append(time.perf_counter() - t0)
                assert resp.container_responses
            ch.close()
            return lat, len(devices)
        finally:
            plugin.stop()
            kubelet.stop()
            host.cleanup()
