"""Passthrough (vfio-pci) device plugin server.

AMD equivalent of the reference's GenericDevicePlugin
(reference: generic_device_plugin.go).  One instance per PCI device id
(resource type); advertises BDFs, serves Allocate with full-IOMMU-group
vfio wiring, TOCTOU revalidation and sibling env suppression, and
topology-aware GetPreferredAllocation (NUMA + xGMI island).

EGM note: the reference injects NVIDIA Grace Extended GPU Memory nodes
(/dev/egmN) on coherent CPU-GPU systems (generic_device_plugin.go:62-185).
MI355X is a discrete-HBM OAM part — there is no host-memory-extension
device node to forward, so that subsystem intentionally has no AMD
equivalent here.
"""

import logging
import os

import grpc

from .. import dpapi
from . import consts, sysfs
from .allocation import preferred_allocation
from .plugin_base import DevicePluginBase

try:  # hot-path revalidation in C++ (csrc/sysfs_scan.cpp)
    from kubevirt_gpu_device_plugin_amd import _sysfs
except ImportError:
    _sysfs = None

log = logging.getLogger(__name__)


class GenericDevicePlugin(DevicePluginBase):
    def __init__(self, device_name, devices, registry, config=None,
                 island_of=None, env_prefix=consts.GPU_ENV_PREFIX):
        super().__init__(device_name, devices, config=config)
        self.registry = registry
        self.island_of = island_of or (lambda bdf: -1)
        self.env_prefix = env_prefix
        # Allocate hot-path caches: the env key and every DeviceSpec
        # host path are deterministic per plugin, so build them once
        # (protobuf message construction dominates the handler
        # otherwise; kubelet copies on extend(), reuse is safe).
        self._env_key = "%s_%s" % (env_prefix, device_name.upper())
        self._vfio_prefix = self.config.vfio_dir.rstrip("/") + "/"
        self._cdev_prefix = self._vfio_prefix + "devices/"
        self._spec_cache = {}

    def _spec(self, host_path):
        s = self._spec_cache.get(host_path)
        if s is None:
            s = dpapi.DeviceSpec(host_path=host_path,
                                 container_path=host_path,
                                 permissions=consts.DEVICE_PERMISSIONS)
            self._spec_cache[host_path] = s
        return s

    def update_registry(self, registry, devices, island_of=None):
        """Swap in a fresh post-rescan registry + device list.  The
        registry reference swap is atomic (Allocate readers pick up
        either the old or the new immutable registry)."""
        self.registry = registry
        if island_of is not None:
            self.island_of = island_of
        self.update_devices(devices)

    # ---- health wiring --------------------------------------------------

    def _group_to_ids(self):
        """/dev/vfio/<group> node name → advertised BDFs sharing it
        (reference: pathDeviceMap, generic_device_plugin.go:645-665)."""
        mapping = {}
        for dev in self._devs:
            group = self.registry.bdf_to_iommu.get(dev.ID)
            if group is None:
                log.warning("[%s] no IOMMU group known for %s",
                            self.device_name, dev.ID)
                continue
            mapping.setdefault(group, []).append(dev.ID)
        return mapping

    def _resync_health(self, group_to_ids):
        """Ground-truth health pass, cdev-aware: a device is healthy
        when its /dev/vfio/<group> node exists, or — on hosts running
        vfio in pure iommufd cdev mode, where group nodes are never
        created — when its per-device /dev/vfio/devices/vfioN node
        exists (vfioN resolved from the sysfs vfio-dev entry)."""
        for group, ids in group_to_ids.items():
            if os.path.exists(self._vfio_prefix + group):
                self.set_health(ids, dpapi.HEALTHY)
                continue
            for bdf in ids:
                try:
                    vfiodev = sysfs.read_vfio_dev(
                        self.config.pci_base, bdf)
                    ok = os.path.exists(self._cdev_prefix + vfiodev)
                except OSError:
                    ok = False  # no vfio-dev entry → not cdev-bound
                self.set_health([bdf], dpapi.HEALTHY if ok
                                else dpapi.UNHEALTHY)

    # ---- Allocate -------------------------------------------------------

    def Allocate(self, request, context):  # noqa: N802
        """Validate and wire up vfio for every requested BDF
        (reference: Allocate, generic_device_plugin.go:353-451).

        Exact semantics preserved:
          * request expands to the full IOMMU group; every member is
            revalidated against live sysfs (group link unchanged, vendor
            still AMD) before anything is handed out — TOCTOU guard;
          * env lists only the *requested* BDF, never group siblings
            (KubeVirt assigns env addresses positionally; a sibling would
            steal another device's slot);
          * DeviceSpec order per request: iommufd cdevs of group members
            first (when /dev/iommu exists), then /dev/vfio/vfio, the
            /dev/vfio/<group> node, and /dev/iommu — deduplicated,
            first-add order kept.

        Deviation: env accumulation is per container request (the
        reference shares one env map across all container requests,
        generic_device_plugin.go:362-441, leaking container A's BDFs
        into container B's env).
        """
        base = self.config.pci_base
        # /dev/iommu present ⇒ iommufd cdev flow
        iommufd = sysfs.supports_iommufd(self.config.iommu_dev,
                                         vfio_dir=self.config.vfio_dir)

        response = dpapi.AllocateResponse()
        for req in request.container_requests:
            specs = []
            seen = set()
            env_devices = {}
            spec_of = self._spec

            def add_spec(host_path):
                if host_path not in seen:
                    seen.add(host_path)
                    specs.append(spec_of(host_path))

            # Resolve every requested BDF to its IOMMU group first; the
            # whole request then revalidates in ONE native call (the
            # reference re-reads sysfs per member per request,
            # generic_device_plugin.go:383-410 — that is where its
            # Allocate latency lives).
            plan = []
            for bdf in req.devices_ids:
                group = self.registry.bdf_to_iommu.get(bdf)
                members = self.registry.iommu_map.get(group, [])
                if group is None or not members \
                        or not any(d.addr == bdf for d in members):
                    context.abort(
                        grpc.StatusCode.INVALID_ARGUMENT,
                        "invalid allocation request: unknown device: %s"
                        % bdf)
                plan.append((bdf, group, members))

            if _sysfs is not None:
                flat = [(d.addr, group)
                        for _, group, members in plan for d in members]
                failed, reason, vfio_devs = _sysfs.revalidate(
                    base, flat, consts.AMD_VENDOR_ID, iommufd)
                if failed and reason == "no_cdev":
                    context.abort(
                        grpc.StatusCode.INTERNAL,
                        "could not determine iommufd device for "
                        "device %s" % failed)
                if failed:
                    log.warning("revalidation failed for %s", failed)
                    context.abort(
                        grpc.StatusCode.INVALID_ARGUMENT,
                        "invalid allocation request: unknown "
                        "device: %s" % failed)
                cdevs = iter(vfio_devs)
                for bdf, group, members in plan:
                    if iommufd:
                        for _ in members:
                            add_spec(self._cdev_prefix + next(cdevs))
                    self._finish_bdf(bdf, group, env_devices, add_spec,
                                     iommufd)
                plan = []

            for bdf, group, members in plan:  # pure-Python fallback
                for dev in members:
                    try:
                        live_group = sysfs.read_link_basename(
                            base, dev.addr, "iommu_group")
                    except OSError:
                        live_group = None
                    if live_group != group:
                        log.warning("IOMMU group changed for %s",
                                    dev.addr)
                        context.abort(
                            grpc.StatusCode.INVALID_ARGUMENT,
                            "invalid allocation request: unknown "
                            "device: %s" % dev.addr)
                    try:
                        vendor = sysfs.read_id_from_file(
                            base, dev.addr, "vendor")
                    except OSError:
                        vendor = None
                    if vendor != consts.AMD_VENDOR_ID:
                        log.warning("vendor changed for %s", dev.addr)
                        context.abort(
                            grpc.StatusCode.INVALID_ARGUMENT,
                            "invalid allocation request: unknown "
                            "device: %s" % dev.addr)
                    if iommufd:
                        try:
                            vfiodev = sysfs.read_vfio_dev(base, dev.addr)
                        except OSError:
                            context.abort(
                                grpc.StatusCode.INTERNAL,
                                "could not determine iommufd device "
                                "for device %s" % dev.addr)
                        add_spec(self._cdev_prefix + vfiodev)
                self._finish_bdf(bdf, group, env_devices, add_spec,
                                 iommufd)

            container = response.container_responses.add()
            for key, bdfs in env_devices.items():
                container.envs[key] = ",".join(bdfs)
            container.devices.extend(specs)
            log.info("[%s] allocated: envs=%s specs=%d", self.device_name,
                     dict(container.envs), len(specs))
        return response

    def _finish_bdf(self, bdf, group, env_devices, add_spec, iommufd):
        """Env entry (requested BDF only) + the per-group device specs,
        in reference order (generic_device_plugin.go:414-432)."""
        env_devices.setdefault(self._env_key, []).append(bdf)
        add_spec(self._vfio_prefix + "vfio")
        add_spec(self._vfio_prefix + group)
        if iommufd:
            add_spec(self.config.iommu_dev)

    # ---- GetPreferredAllocation ----------------------------------------

    def GetPreferredAllocation(self, request, context):  # noqa: N802
        """NUMA + xGMI-island preferred sets (reference is NUMA-only,
        generic_device_plugin.go:478-616; see allocation.py)."""
        with self._lock:
            numa = {d.ID: (d.topology.nodes[0].ID if d.topology.nodes
                           else -1)
                    for d in self._devs}
        bdf_to_iommu = self.registry.bdf_to_iommu
        iommu_map = self.registry.iommu_map

        def group_size_of(bdf):
            return len(iommu_map.get(bdf_to_iommu.get(bdf), ()))

        response = dpapi.PreferredAllocationResponse()
        for req in request.container_requests:
            try:
                ids = preferred_allocation(
                    list(req.available_deviceIDs),
                    list(req.must_include_deviceIDs),
                    int(req.allocation_size),
                    numa_of=lambda i: numa.get(i, -1),
                    island_of=self.island_of,
                    group_size_of=group_size_of)
            except ValueError as e:
                context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(e))
            response.container_responses.add(deviceIDs=ids)
        return response
