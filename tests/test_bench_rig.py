"""The latency-bench rig itself is part of the deliverable
(BASELINE.md self-measured baseline) — keep it covered on CPU."""

from bench_harness.rig import measure_allocate


def test_measure_allocate_small():
    lat, n = measure_allocate(2, steps=5, warmup=1)
    assert n == 2
    assert len(lat) == 5
    assert all(0 < t < 5.0 for t in lat)


def test_measure_allocate_iommufd():
    lat, n = measure_allocate(1, steps=3, warmup=1, iommufd=True)
    assert n == 1 and len(lat) == 3
