"""Controller: discovery → plugin creation → lifetime.

AMD equivalent of the reference controller
(reference: InitiateDevicePlugin/createDevicePlugins,
device_plugin.go:89-176): one passthrough plugin per GPU PCI device id,
one VF plugin per VF PCI device id, each started independently (a
per-type start failure is logged and the others continue,
device_plugin.go:131-136), then block until stopped.
"""

import logging
import threading

from .. import dpapi
from ..topology import build_island_lookup
from . import consts, discovery, pciids
from .plugin import GenericDevicePlugin
from .plugin_base import PluginConfig
from .vf_plugin import VfDevicePlugin

log = logging.getLogger(__name__)


def build_kubelet_devices(devs):
    """pluginapi.Device list with NUMA topology
    (reference: device_plugin.go:109-123)."""
    out = []
    for d in devs:
        out.append(dpapi.Device(
            ID=d.addr, health=dpapi.HEALTHY,
            topology=dpapi.TopologyInfo(
                nodes=[dpapi.NUMANode(ID=d.numa_node)])))
    return out


def resolve_name(device_id, pci_ids_path=None):
    """Marketing name from pci.ids, raw hex id as fallback
    (reference: device_plugin.go:124-128)."""
    name = pciids.get_device_name(device_id, pci_ids_path=pci_ids_path)
    if not name:
        log.error("could not find device name for device id %s; using "
                  "raw id", device_id)
        return device_id
    return name


def log_vf_partitioning(registry):
    """Surface the HBM split per VF (VF partitioning sized
    against 288 GB HBM3E per MI355X — 8 VFs ⇒ 36 GB each)."""
    mi355x_vf_ids = {"75b0", "75b3"}
    for pf, vfs in sorted(registry.pf_vf_map.items()):
        vf_types = {d.device_id for devs in registry.vf_map.values()
                    for d in devs if d.parent_pf == pf}
        if vf_types & mi355x_vf_ids:
            per_vf_gib = consts.MI355X_HBM_BYTES / len(vfs) / 1024**3
            log.info("PF %s: %d VFs → %.1f GiB HBM3E per VF",
                     pf, len(vfs), per_vf_gib)
        else:
            log.info("PF %s: %d VFs", pf, len(vfs))


class Controller:
    """Owns discovery output and the per-resource plugin servers."""

    def __init__(self, config=None, pci_ids_path=None,
                 kfd_nodes_dir=consts.KFD_TOPOLOGY_PATH,
                 vf_event_watcher_factory=None):
        self.config = config or PluginConfig()
        self.pci_ids_path = pci_ids_path
        self.kfd_nodes_dir = kfd_nodes_dir
        self.vf_event_watcher_factory = vf_event_watcher_factory
        self.registry = None
        self.plugins = []

    def create_plugins(self):
        """Run discovery and instantiate (but not start) all plugin
        servers.  Returns the plugin list."""
        self.registry = discovery.discover(base_path=self.config.pci_base)
        discovery.warn_shared_groups(self.registry)
        island_of = build_island_lookup(
            self.registry, nodes_dir=self.kfd_nodes_dir)
        log.info("iommu map: %s",
                 {g: [d.addr for d in v]
                  for g, v in self.registry.iommu_map.items()})
        log.info("device map: %s",
                 {k: [d.addr for d in v]
                  for k, v in self.registry.device_map.items()})
        log.info("vf map: %s",
                 {k: [d.addr for d in v]
                  for k, v in self.registry.vf_map.items()})
        log.info("pf→vf map: %s", self.registry.pf_vf_map)
        log_vf_partitioning(self.registry)

        self.plugins = [
            self._make_plugin(name, kind, devs, island_of)
            for name, (kind, devs) in self._desired_resources().items()
        ]
        return self.plugins

    def _make_plugin(self, name, kind, devs, island_of):
        if kind == "gpu":
            return GenericDevicePlugin(
                name, build_kubelet_devices(devs), self.registry,
                config=self.config, island_of=island_of)
        return VfDevicePlugin(
            name, build_kubelet_devices(devs), self.registry,
            config=self.config, island_of=island_of,
            event_watcher_factory=self.vf_event_watcher_factory)

    def start(self, stop_event):
        started = []
        for p in self.plugins:
            try:
                p.start(stop_event)
                started.append(p)
            except Exception as e:
                log.error("error starting %s device plugin: %s",
                          p.device_name, e)
        self.plugins = started
        return started

    def stop(self):
        for p in self.plugins:
            try:
                p.stop()
            except Exception:
                log.exception("error stopping %s", p.device_name)

    def _desired_resources(self):
        """name -> (kind, discovery devices) for the current registry."""
        desired = {}
        for device_id, devs in sorted(self.registry.device_map.items()):
            desired[resolve_name(device_id, self.pci_ids_path)] = \
                ("gpu", devs)
        for device_id, devs in sorted(self.registry.vf_map.items()):
            desired[resolve_name(device_id, self.pci_ids_path)] = \
                ("vf", devs)
        return desired

    def rescan(self, stop_event):
        """Diff-based re-discovery (hotplug): resource types that still
        exist get an in-place device-list update over their live
        ListAndWatch stream — no socket churn, no re-registration;
        vanished types stop; new types start and register.  The common
        case (gim VF count change on an existing type) is therefore
        zero-disruption.  The reference has no rescan at all
        (SURVEY.md §5)."""
        self.registry = discovery.discover(base_path=self.config.pci_base)
        discovery.warn_shared_groups(self.registry)
        island_of = build_island_lookup(
            self.registry, nodes_dir=self.kfd_nodes_dir)
        desired = self._desired_resources()
        kept = []
        for p in self.plugins:
            if p.device_name in desired:
                _, devs = desired.pop(p.device_name)
                p.update_registry(self.registry,
                                  build_kubelet_devices(devs),
                                  island_of=island_of)
                kept.append(p)
            else:
                log.info("resource %s vanished; stopping its server",
                         p.device_name)
                try:
                    p.stop()
                except Exception:
                    log.exception("error stopping %s", p.device_name)
        for name, (kind, devs) in desired.items():
            p = self._make_plugin(name, kind, devs, island_of)
            try:
                p.start(stop_event)
                kept.append(p)
            except Exception as e:
                log.error("error starting %s device plugin: %s",
                          name, e)
        self.plugins = kept
        return kept


def initiate_device_plugin(stop_event=None, rescan_event=None,
                           uevent_autoscan=True, **kwargs):
    """Blocking entry point (reference: InitiateDevicePlugin,
    device_plugin.go:89-96).

    ``rescan_event`` triggers a diff-based re-discovery (see
    Controller.rescan); it is set by SIGHUP (cmd/main.py) and — when
    ``uevent_autoscan`` and the netlink socket are available — by
    kernel PCI uevents for vendor-1002 driver bind/unbind (gim creating
    VFs, driverctl overrides).  The reference discovers exactly once
    and needs a process restart to pick up new devices (SURVEY.md §5
    "no hotplug re-scan").
    """
    stop_event = stop_event or threading.Event()
    rescan_event = rescan_event if rescan_event is not None \
        else threading.Event()
    if uevent_autoscan:
        from . import uevent
        uevent.start_listener(rescan_event, stop_event.is_set)
    controller = Controller(**kwargs)
    controller.create_plugins()
    controller.start(stop_event)
    # Tail-latency: park the startup object graph in the permanent
    # generation and relax gen-0 collection so steady-state Allocate
    # churn (which is acyclic) cannot trigger a cyclic-GC pause in the
    # middle of an RPC on the pod-admission critical path.
    import gc
    gc.collect()
    gc.freeze()
    gc.set_threshold(50000, 50, 50)
    while not stop_event.is_set():
        # woken immediately by a rescan request; the 0.5 s timeout only
        # bounds shutdown latency (stop_event has no waiter-wakeup hook
        # we own — callers hand us an arbitrary Event)
        rescan_event.wait(0.5)
        if stop_event.is_set():
            break
        if rescan_event.is_set():
            # debounce: a gim VF burst emits one uevent per function
            stop_event.wait(0.5)
            rescan_event.clear()
            log.info("rescan requested: re-running discovery")
            controller.rescan(stop_event)
    log.info("shutting down device plugin controller")
    controller.stop()
