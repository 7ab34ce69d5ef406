"""xGMI island discovery from KFD sysfs topology.

The reference has no collective/topology layer at all; on MI355X the
8-GPU xGMI island (7 p2p links per GPU) is the physical domain a
multi-GPU VMI should stay inside, so discovery builds an island map
consumed by GetPreferredAllocation (SURVEY.md §5 "Distributed
communication backend" and §3.5).

Sources, in order:
  1. ``/sys/class/kfd/kfd/topology/nodes/<n>/properties`` — ``hive_id``
     (all GPUs of one xGMI hive share it) plus ``domain``/``location_id``
     to recover the PCI BDF.  Available whenever the PF is owned by the
     host amdgpu/gim driver (the SR-IOV case).
  2. AMD-SMI ``amdsmi_xgmi_info_t.xgmi_hive_id`` (amdsmi.h:672-677) via
     the native binding, same availability.

A GPU bound to vfio-pci is invisible to both (no host driver attached) —
its island resolves to -1 and preferred allocation degrades to the
reference's NUMA-only behavior.  VFs inherit the island of their parent
PF via ``physfn``.
"""

import logging
import os

from ..device_plugin import consts

log = logging.getLogger(__name__)


def _parse_properties(path):
    props = {}
    try:
        with open(path) as f:
            for line in f:
                parts = line.split()
                if len(parts) == 2:
                    props[parts[0]] = int(parts[1])
    except (OSError, ValueError) as e:
        log.debug("unreadable KFD properties %s: %s", path, e)
    return props


def bdf_from_location(domain, location_id):
    """KFD encodes the PCI location as (bus<<8 | dev<<3 | func)."""
    bus = (location_id >> 8) & 0xFF
    dev = (location_id >> 3) & 0x1F
    fn = location_id & 0x7
    return "%04x:%02x:%02x.%x" % (domain & 0xFFFF, bus, dev, fn)


def island_map_from_kfd(nodes_dir=consts.KFD_TOPOLOGY_PATH):
    """Return {pci_bdf: island_id}; island_id is the xGMI hive id
    (-1 = not in any hive)."""
    islands = {}
    try:
        nodes = sorted(os.listdir(nodes_dir))
    except OSError:
        return islands
    for node in nodes:
        props = _parse_properties(
            os.path.join(nodes_dir, node, "properties"))
        if props.get("simd_count", 0) <= 0:
            continue  # CPU node
        if "location_id" not in props:
            continue
        bdf = bdf_from_location(props.get("domain", 0),
                                props["location_id"])
        hive = props.get("hive_id", 0)
        islands[bdf] = hive if hive != 0 else -1
    return islands


def island_map_from_amdsmi():
    """Fallback source: xgmi hive id via the native AMD-SMI binding."""
    from .. import amdsmi
    if not amdsmi.is_available():
        return {}
    smi = amdsmi.ext()
    islands = {}
    try:
        smi.init()
    except RuntimeError as e:
        log.info("amdsmi init failed: %s", e)
        return islands
    try:
        for d in smi.get_devices():
            try:
                info = smi.xgmi_info(d["index"])
            except RuntimeError:
                continue
            hive = info.get("hive_id", 0)
            islands[d["bdf"].lower()] = hive if hive != 0 else -1
    finally:
        smi.shutdown()
    return islands


def build_island_lookup(registry, nodes_dir=consts.KFD_TOPOLOGY_PATH,
                        use_amdsmi=True):
    """Callable bdf -> island id for GetPreferredAllocation.

    VFs resolve through their parent PF.  Unknown devices → -1.
    """
    islands = island_map_from_kfd(nodes_dir)
    if not islands and use_amdsmi:
        islands = island_map_from_amdsmi()
    islands = {k.lower(): v for k, v in islands.items()}
    parent = {}
    for devs in registry.vf_map.values():
        for d in devs:
            parent[d.addr.lower()] = d.parent_pf.lower()
    if islands:
        log.info("xGMI island map: %s", islands)

    def island_of(bdf):
        b = bdf.lower()
        b = parent.get(b, b)
        return islands.get(b, -1)

    return island_of
