#!/usr/bin/env python3
"""Build the native extensions in-tree.

Outputs (git-ignored, but they travel with the gpurun snapshot):
  kubevirt_gpu_device_plugin_amd/_amdsmi.so       (g++, dlopen binding)
  kubevirt_gpu_device_plugin_amd/_sysfs.so        (g++, sysfs scanner)
  kubevirt_gpu_device_plugin_amd/_healthprobe.so  (hipcc, gfx950 kernel)

hipcc cross-compiles gfx950 without a GPU, so this runs on CPU-only
hosts too.  ``python build_native.py`` or ``make build``.
"""

import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.abspath(__file__))
PKG = os.path.join(ROOT, "kubevirt_gpu_device_plugin_amd")
CSRC = os.path.join(ROOT, "csrc")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def pybind11_includes():
    out = subprocess.check_output(
        [sys.executable, "-m", "pybind11", "--includes"], text=True)
    return out.split()


def newer(src, dst):
    return (not os.path.exists(dst)
            or os.path.getmtime(src) > os.path.getmtime(dst))


def build(force=False):
    incs = pybind11_includes()
    jobs = []

    src = os.path.join(CSRC, "amdsmi_binding.cpp")
    dst = os.path.join(PKG, "_amdsmi.so")
    if force or newer(src, dst):
        jobs.append(["g++", "-O2", "-std=c++17", "-shared", "-fPIC",
                     "-I%s/include" % ROCM, *incs, src, "-ldl",
                     "-o", dst])

    src = os.path.join(CSRC, "sysfs_scan.cpp")
    dst = os.path.join(PKG, "_sysfs.so")
    if force or newer(src, dst):
        jobs.append(["g++", "-O2", "-std=c++17", "-shared", "-fPIC",
                     *incs, src, "-o", dst])

    src = os.path.join(CSRC, "health_probe.hip")
    dst = os.path.join(PKG, "_healthprobe.so")
    if force or newer(src, dst):
        jobs.append(["%s/bin/hipcc" % ROCM,
                     "--offload-arch=%s" % GFX_ARCH, "-O3",
                     "-std=c++17", "-shared", "-fPIC", *incs, src,
                     "-o", dst])

    procs = [(cmd, subprocess.Popen(cmd)) for cmd in jobs]
    failed = []
    for cmd, p in procs:
        print("+", " ".join(cmd), flush=True)
        if p.wait() != 0:
            failed.append(cmd)
    if failed:
        raise SystemExit("build failed: %r" % failed)
    print("native extensions built")


if __name__ == "__main__":
    build(force="--force" in sys.argv)
