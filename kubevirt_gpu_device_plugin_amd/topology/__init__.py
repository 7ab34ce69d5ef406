"""xGMI / NUMA topology sources for allocation affinity."""

from .kfd import island_map_from_kfd, build_island_lookup  # noqa: F401
