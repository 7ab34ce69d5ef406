#!/usr/bin/env python3
"""Flagship benchmark: p50 Allocate() RPC latency of the device plugin.

BASELINE.json metric: "allocatable GPUs+VFs per node; p50 Allocate()
RPC latency at 1/2/4/8 GPU".  The reference publishes no numbers
(BASELINE.md), so the baseline is self-measured on this rig: a
synthetic sysfs node with N advertised MI355X passthrough devices, the
plugin in its own process, a stub kubelet driving Allocate over the
real unix-socket gRPC boundary.  Each step is one Allocate RPC claiming
all N devices (full-node VMI — the hardest, scaling-relevant case,
including per-call TOCTOU sysfs revalidation of every IOMMU group).

Contract: ``python bench.py --gpus N --steps K --warmup W`` — W untimed
warmup steps, K timed steps bracketed by barrier+synchronize, MAX over
ranks, rank 0 prints one JSON line.  Under torch.distributed.run, rank 0
runs the rig; other ranks only join the barriers (this workload is a
per-node control-plane daemon — there is no per-GPU compute to shard).
"""

import argparse
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1,
                    help="advertised GPU count (devices on the "
                         "synthetic node)")
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=20)
    ap.add_argument("--iommufd", action="store_true",
                    help="bench the iommufd cdev flow")
    ap.add_argument("--vfs-per-gpu", type=int, default=0,
                    help="SR-IOV mode: N gim PFs x this many VFs; each "
                         "RPC allocates one PF's VF set")
    ap.add_argument("--single", action="store_true",
                    help="allocate 1 device per RPC round-robin "
                         "(pod-per-GPU pattern) instead of all N")
    ap.add_argument("--vf-check", action="store_true",
                    help="also verify the 64-VF SR-IOV config counts")
    args = ap.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    dist = None
    device = None
    if world_size > 1:
        import torch
        import torch.distributed as tdist
        dist = tdist
        from datetime import timedelta

        # RCCL only when every rank actually has its own GPU
        backend = ("nccl" if torch.cuda.is_available()
                   and torch.cuda.device_count() >= world_size
                   else "gloo")
        if backend == "nccl":
            local_rank = int(os.environ.get("LOCAL_RANK", rank))
            device = torch.device("cuda", local_rank)
            torch.cuda.set_device(device)
        # bounded init: a dead rank must fail the run fast, not hang it
        dist.init_process_group(backend=backend,
                                timeout=timedelta(seconds=300))

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if device is not None:
            import torch
            torch.cuda.synchronize(device)

    lat, n_devices = [], args.gpus
    barrier_sync()
    t_start = time.perf_counter()
    if rank == 0:
        from bench_harness.rig import measure_allocate
        lat, n_devices = measure_allocate(
            args.gpus, args.steps, args.warmup, iommufd=args.iommufd,
            vfs_per_gpu=args.vfs_per_gpu,
            allocate_all=not args.single)
    elapsed = time.perf_counter() - t_start
    barrier_sync()

    if dist is not None:
        import torch
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device is not None else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        p50_us = statistics.median(lat) * 1e6
        p99_us = sorted(lat)[max(0, int(len(lat) * 0.99) - 1)] * 1e6
        vf_counts = None
        if args.vf_check:
            import tempfile
            from bench_harness.rig import build_node
            from kubevirt_gpu_device_plugin_amd.device_plugin import (
                discovery,
            )
            with tempfile.TemporaryDirectory() as tmp:
                h = build_node(tmp, 8, vfs_per_gpu=8)
                reg = discovery.discover(base_path=h.pci)
                vf_counts = sum(len(v) for v in reg.vf_map.values())
        result = {
            "metric": "p50 Allocate() RPC latency",
            "value": round(p50_us, 1),
            "unit": "us",
            "n_gpus": args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(sum(lat) / len(lat) * 1e3, 4),
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "n/a",
            "data": "synthetic sysfs node + stub kubelet (BASELINE.md "
                    "self-measured rig)",
            "config": {
                "model": "kubevirt-gpu-device-plugin-amd",
                "allocatable_gpus": n_devices,
                "allocate_request": (
                    "%d VFs (one PF set) per RPC" % args.vfs_per_gpu
                    if args.vfs_per_gpu else
                    "1 device per RPC (round-robin)" if args.single
                    else "all %d devices per RPC" % n_devices),
                "vfs_per_gpu": args.vfs_per_gpu,
                "iommufd": bool(args.iommufd),
                "p99_us": round(p99_us, 1),
                "startup_to_allocatable_ms": round(
                    getattr(measure_allocate, "last_startup_s", 0)
                    * 1e3, 1),
                "vf_config_allocatable": vf_counts,
                "parallelism": "1 plugin process per node",
                "global_batch": None,
                "seq_len": None,
            },
        }
        print(json.dumps(result), flush=True)

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
