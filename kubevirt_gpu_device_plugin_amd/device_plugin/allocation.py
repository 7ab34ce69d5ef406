"""Topology-aware preferred-allocation selection.

Extends the reference's NUMA-only packing
(reference: GetPreferredAllocation, generic_device_plugin.go:478-616)
with xGMI-island awareness: on an 8×MI355X node every GPU has 7
point-to-point xGMI links to its island peers, so a multi-GPU VMI gets
full-bandwidth p2p only when all its GPUs share an island.  Selection
order:

  1. must-include devices first (error if they exceed the request size,
     matching the reference);
  2. try to satisfy the whole request from a single xGMI island,
     preferring devices of one NUMA node inside that island;
  3. else the reference behavior: a single NUMA node;
  4. else fall back to the kubelet-provided order.

When no island data is available every device maps to island -1 and
step 2 degenerates into step 3, i.e. exact reference behavior.
"""


def preferred_allocation(available_ids, must_include_ids, size,
                         numa_of, island_of=None, group_size_of=None):
    """Return the preferred device list (len == ``size`` when possible).

    ``numa_of``/``island_of``: callables id -> int (-1 when unknown).
    ``group_size_of``: callable id -> member count of the device's
    IOMMU group.  When given, devices in singleton groups are preferred
    (stable) over co-grouped ones: allocating one member of a shared
    group binds every sibling's vfio group into the VM, so siblings
    handed to another pod later would be broken — the reference leaves
    this hazard entirely to the operator (it only suppresses siblings
    from env, generic_device_plugin.go:414-420).
    Raises ``ValueError`` when must-include exceeds ``size``.
    """
    size = max(0, int(size))  # a nonsensical negative size means "none"
    if island_of is None:
        island_of = lambda _id: -1  # noqa: E731
    if group_size_of is not None:
        # stable: kubelet order is preserved within each group-size tier
        available_ids = sorted(available_ids, key=group_size_of)

    preferred = []
    chosen = set()

    def add(dev_id):
        if dev_id not in chosen:
            chosen.add(dev_id)
            preferred.append(dev_id)

    for dev_id in must_include_ids:
        add(dev_id)
    if len(preferred) > size:
        raise ValueError(
            "number of MustIncludeDeviceIDs (%d) exceeds allocation "
            "size (%d)" % (len(preferred), size))

    # Group available devices, preserving kubelet order
    # (reference: generic_device_plugin.go:501-512).
    def group_by(key):
        groups, order = {}, []
        for dev_id in available_ids:
            k = key(dev_id)
            if k not in groups:
                groups[k] = []
                order.append(k)
            groups[k].append(dev_id)
        return groups, order

    def selected_count(key, val):
        return sum(1 for d in preferred if key(d) == val)

    def try_fill_from(group_ids, inner_key=None):
        """Fill the remaining slots from ``group_ids`` only; with
        ``inner_key``, prefer completing one inner group first."""
        if inner_key is not None:
            inner, inner_order = {}, []
            for dev_id in group_ids:
                k = inner_key(dev_id)
                if k not in inner:
                    inner[k] = []
                    inner_order.append(k)
                inner[k].append(dev_id)
            # Inner groups that can hold the whole remainder come first.
            need = size - len(preferred)
            inner_order.sort(
                key=lambda k: 0 if len(
                    [d for d in inner[k] if d not in chosen]) >= need
                else 1)
            ordered = [d for k in inner_order for d in inner[k]]
        else:
            ordered = group_ids
        for dev_id in ordered:
            if len(preferred) >= size:
                break
            add(dev_id)

    def satisfiable(groups, order, key):
        """First group (prioritising groups already holding selected
        devices) whose selected+free total reaches ``size``
        (reference: generic_device_plugin.go:548-584).  The unknown
        group (-1) is never packable — the reference likewise treats
        "no NUMA node found" as falling through to kubelet order
        (targetNode stays -1, generic_device_plugin.go:549,576)."""
        selected_first = sorted(
            order, key=lambda k: 0 if selected_count(key, k) else 1)
        for k in selected_first:
            if k == -1:
                continue
            free = sum(1 for d in groups[k] if d not in chosen)
            if selected_count(key, k) + free >= size:
                return k
        return None

    if len(preferred) < size:
        # Step 2: one xGMI island, NUMA-packed inside.
        islands, island_order = group_by(island_of)
        k = satisfiable(islands, island_order, island_of)
        if k is not None:
            try_fill_from(islands[k], inner_key=numa_of)

    if len(preferred) < size:
        # Step 3: one NUMA node (reference behavior).
        numas, numa_order = group_by(numa_of)
        k = satisfiable(numas, numa_order, numa_of)
        if k is not None:
            try_fill_from(numas[k])

    if len(preferred) < size:
        # Step 4: kubelet-provided order
        # (reference: generic_device_plugin.go:586-596).
        try_fill_from(list(available_ids))

    return preferred
