"""Property-based invariants (hypothesis) for discovery and
preferred-allocation."""

import string

from hypothesis import given, settings, strategies as st

from kubevirt_gpu_device_plugin_amd.device_plugin.allocation import (
    preferred_allocation,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.pciids import (
    sanitize_name,
)

ids = st.lists(
    st.text(alphabet=string.ascii_lowercase, min_size=1, max_size=4),
    min_size=1, max_size=24, unique=True)


@st.composite
def alloc_case(draw):
    avail = draw(ids)
    numa = {i: draw(st.integers(min_value=-1, max_value=3))
            for i in avail}
    island = {i: draw(st.integers(min_value=-1, max_value=2))
              for i in avail}
    must = draw(st.lists(st.sampled_from(avail), max_size=4,
                         unique=True))
    size = draw(st.integers(min_value=max(1, len(must)),
                            max_value=len(avail)))
    return avail, must, size, numa, island


@settings(max_examples=300, deadline=None)
@given(alloc_case())
def test_preferred_allocation_invariants(case):
    avail, must, size, numa, island = case
    got = preferred_allocation(avail, must, size,
                               numa_of=lambda i: numa.get(i, -1),
                               island_of=lambda i: island.get(i, -1))
    # exactly `size` devices (enough are always available here)
    assert len(got) == size
    # no duplicates
    assert len(set(got)) == len(got)
    # every must-include present, and first, in order
    assert got[:len(must)] == must
    # everything selected was available (or explicitly must-included)
    assert set(got) <= set(avail) | set(must)


@settings(max_examples=300, deadline=None)
@given(alloc_case())
def test_preferred_allocation_island_packing(case):
    """If some single island could satisfy the whole request (and there
    are no must-includes), the result stays inside one island."""
    avail, _must, size, numa, island = case
    feasible = [k for k in set(island.values()) if k != -1 and sum(
        1 for i in avail if island[i] == k) >= size]
    got = preferred_allocation(avail, [], size,
                               numa_of=lambda i: numa.get(i, -1),
                               island_of=lambda i: island.get(i, -1))
    if feasible:
        used = {island[i] for i in got}
        assert len(used) == 1 and used <= set(feasible)


@settings(max_examples=200, deadline=None)
@given(st.text(max_size=64))
def test_sanitize_name_charset(name):
    out = sanitize_name(name)
    assert all(c.isalnum() and not c.islower() or c in "_." or
               c.isdigit() for c in out) or out == ""
    # idempotent
    assert sanitize_name(out) == out
