#!/usr/bin/env python3
"""GPU burn-in: loop the gfx950 health probe (HBM pattern + bandwidth,
LDS banks, cross-XCD atomics, MFMA) across all visible devices.

Usage: python tools/gpu_burnin.py [--seconds 60] [--mib 2048]
Exit code 0 = every pass clean on every GPU.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=60.0)
    ap.add_argument("--mib", type=int, default=2048)
    args = ap.parse_args()

    from kubevirt_gpu_device_plugin_amd import _healthprobe

    n = _healthprobe.device_count()
    if n == 0:
        print("no HIP devices visible")
        return 2
    print("burn-in: %d device(s), %.0fs, %d MiB buffer"
          % (n, args.seconds, args.mib))
    deadline = time.time() + args.seconds
    passes, failures = 0, 0
    while time.time() < deadline:
        for dev in range(n):
            r = _healthprobe.probe(dev, args.mib)
            passes += 1
            if not r["ok"]:
                failures += 1
                print("FAIL dev%d: %r" % (dev, r))
            else:
                print("pass %4d dev%d: write %.2f TB/s read %.2f TB/s "
                      "lds=%d atomics=%s mfma=%s"
                      % (passes, dev, r["write_gbps"] / 1e3,
                         r["read_gbps"] / 1e3, r["lds_errors"],
                         r["atomics_ok"], r["mfma_ok"]), flush=True)
    print("burn-in done: %d passes, %d failures" % (passes, failures))
    return 1 if failures else 0


if __name__ == "__main__":
    sys.exit(main())
