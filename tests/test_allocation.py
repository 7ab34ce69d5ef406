"""Preferred-allocation policy tests
(reference scenarios: device_plugin_test.go:438-533, extended with xGMI
islands)."""

import pytest

from kubevirt_gpu_device_plugin_amd.device_plugin.allocation import (
    preferred_allocation,
)


def numa_map(m):
    return lambda i: m.get(i, -1)


def test_single_numa_preferred():
    numa = numa_map({"a": 0, "b": 1, "c": 0, "d": 1})
    got = preferred_allocation(["a", "b", "c", "d"], [], 2, numa_of=numa)
    assert got == ["a", "c"]


def test_must_include_first():
    numa = numa_map({"a": 0, "b": 1, "c": 0, "d": 1})
    got = preferred_allocation(["a", "b", "c", "d"], ["b"], 2,
                               numa_of=numa)
    assert got[0] == "b"
    assert got == ["b", "d"]  # completes from b's NUMA node


def test_must_include_exceeds_size_errors():
    with pytest.raises(ValueError):
        preferred_allocation(["a", "b"], ["a", "b"], 1,
                             numa_of=numa_map({}))


def test_fallback_to_kubelet_order():
    numa = numa_map({"a": 0, "b": 1, "c": 2})
    got = preferred_allocation(["a", "b", "c"], [], 3, numa_of=numa)
    assert got == ["a", "b", "c"]


def test_no_topology_info_behaves_like_kubelet_order():
    got = preferred_allocation(["a", "b", "c"], [], 2,
                               numa_of=lambda i: -1)
    assert got == ["a", "b"]


def test_island_preferred_over_numa():
    """4-GPU request on a 2-island node: stay inside one xGMI island even
    when NUMA nodes are split inside it."""
    numa = numa_map({"a": 0, "b": 0, "c": 1, "d": 1,
                     "e": 0, "f": 0, "g": 1, "h": 1})
    island = lambda i: 1 if i in "abcd" else 2  # noqa: E731
    got = preferred_allocation(list("aebfcgdh"), [], 4, numa_of=numa,
                               island_of=island)
    assert set(got) <= {"a", "b", "c", "d"} or set(got) <= {
        "e", "f", "g", "h"}


def test_island_numa_packing_inside_island():
    """Inside the chosen island, one NUMA node is preferred first."""
    numa = numa_map({"a": 0, "b": 1, "c": 0, "d": 1})
    island = lambda i: 7  # noqa: E731
    got = preferred_allocation(["a", "b", "c", "d"], [], 2,
                               numa_of=numa, island_of=island)
    assert set(got) == {"a", "c"}


def test_island_with_must_include():
    numa = numa_map({"a": 0, "b": 0, "c": 0, "d": 0})
    island = lambda i: 1 if i in "ab" else 2  # noqa: E731
    got = preferred_allocation(["a", "b", "c", "d"], ["c"], 2,
                               numa_of=numa, island_of=island)
    assert got == ["c", "d"]  # island 2 holds the must-include


def test_unknown_island_degrades_to_numa_only():
    numa = numa_map({"a": 0, "b": 1, "c": 0, "d": 1})
    got = preferred_allocation(["a", "b", "c", "d"], [], 2,
                               numa_of=numa, island_of=lambda i: -1)
    assert got == ["a", "c"]


def test_full_node_8gpu():
    numa = numa_map({chr(97 + i): i // 4 for i in range(8)})
    island = lambda i: 42  # noqa: E731
    avail = [chr(97 + i) for i in range(8)]
    got = preferred_allocation(avail, [], 8, numa_of=numa,
                               island_of=island)
    assert sorted(got) == avail


def test_singleton_groups_preferred_over_shared():
    """Devices in exclusive IOMMU groups win over co-grouped ones
    (allocating one member of a shared group drags its siblings)."""
    numa = numa_map({"a": 0, "b": 0, "c": 0})
    gsize = {"a": 2, "b": 1, "c": 2}.__getitem__
    got = preferred_allocation(["a", "b", "c"], [], 1, numa_of=numa,
                               group_size_of=gsize)
    assert got == ["b"]


def test_group_size_tiebreak_is_stable():
    numa = numa_map({"a": 0, "b": 0, "c": 0})
    gsize = {"a": 1, "b": 1, "c": 1}.__getitem__
    got = preferred_allocation(["a", "b", "c"], [], 2, numa_of=numa,
                               group_size_of=gsize)
    assert got == ["a", "b"]  # kubelet order preserved
