#!/usr/bin/env python3
"""Daemon soak: run the plugin for N seconds under continuous Allocate
traffic, health flips, and periodic SIGHUP rescans; track RSS/fd growth.

Usage: python tools/daemon_soak.py [--seconds 120]
Exit 0 = no errors and bounded resource growth.
"""

import argparse
import os
import random
import signal
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bench_harness.rig import PluginProcess, build_node  # noqa: E402
from kubevirt_gpu_device_plugin_amd import dpapi  # noqa: E402
from tests.fixtures import StubKubelet, dial_plugin  # noqa: E402


def proc_stats(pid):
    with open("/proc/%d/status" % pid) as f:
        rss = next(int(l.split()[1]) for l in f if l.startswith("VmRSS"))
    fds = len(os.listdir("/proc/%d/fd" % pid))
    return rss, fds


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=120.0)
    ap.add_argument("--clients", type=int, default=1,
                    help="parallel Allocate clients (kubelet admits "
                         "pods concurrently)")
    ap.add_argument("--kubelet-restarts", action="store_true",
                    help="chaos: periodically kill the stub kubelet, "
                         "wipe the plugin socket (what a restarting "
                         "kubelet does) and bring it back after a "
                         "random outage — exercises the socket-removal "
                         "restart + capped-backoff re-registration "
                         "path under live Allocate traffic")
    args = ap.parse_args()

    with tempfile.TemporaryDirectory() as tmp:
        host = build_node(tmp, 8)
        cfg = host.config()
        kubelet = StubKubelet(cfg.kubelet_socket)
        plugin = PluginProcess(host)
        rc = 1
        try:
            req = kubelet.wait_register(30)
            sock = os.path.join(cfg.device_plugin_dir, req.endpoint)
            ch, stub = dial_plugin(sock, timeout=10)
            devices = [d.ID for d in next(
                stub.ListAndWatch(dpapi.Empty())).devices]
            assert len(devices) == 8
            pid = plugin.proc.pid
            time.sleep(1.0)
            rss0, fds0 = proc_stats(pid)
            deadline = time.time() + args.seconds
            rng = random.Random(42)
            counters = {"allocs": 0, "errors": 0}
            import threading
            clock = threading.Lock()

            def client(seed):
                crng = random.Random(seed)
                cch, cstub = dial_plugin(sock, timeout=10)
                a = e = 0
                while time.time() < deadline:
                    try:
                        bdf = crng.choice(devices)
                        cstub.Allocate(dpapi.AllocateRequest(
                            container_requests=[
                                dpapi.ContainerAllocateRequest(
                                    devices_ids=[bdf])]), timeout=5)
                        a += 1
                    except Exception:
                        e += 1
                        time.sleep(0.1)
                cch.close()
                with clock:
                    counters["allocs"] += a
                    counters["errors"] += e

            threads = [threading.Thread(target=client, args=(i,),
                                        daemon=True)
                       for i in range(args.clients)]
            for t in threads:
                t.start()

            rescans = flips = kubelet_restarts = 0
            last_hup = last_kr = time.time()
            while time.time() < deadline:
                g = rng.choice([str(100 + i) for i in range(8)])
                host.remove_vfio_node(g)
                host.add_vfio_node(g)
                flips += 1
                if time.time() - last_hup > 10:
                    plugin.proc.send_signal(signal.SIGHUP)
                    rescans += 1
                    last_hup = time.time()
                    # diff-rescan: same types ⇒ same socket, no
                    # re-registration — traffic just continues
                if args.kubelet_restarts \
                        and time.time() - last_kr > 15:
                    kubelet.stop()
                    try:  # a restarting kubelet wipes plugin sockets
                        os.remove(sock)
                    except FileNotFoundError:
                        pass
                    time.sleep(rng.uniform(0.5, 3.0))  # outage window
                    kubelet = StubKubelet(cfg.kubelet_socket)
                    assert kubelet.wait_register(30), \
                        "plugin did not re-register after kubelet " \
                        "restart"
                    kubelet_restarts += 1
                    last_kr = time.time()
                time.sleep(0.02)
            for t in threads:
                t.join(timeout=30)
            allocs, errors = counters["allocs"], counters["errors"]
            rss1, fds1 = proc_stats(pid)
            ch.close()
            print("soak %.0fs: %d allocs, %d transient errors, "
                  "%d rescans, %d health flips, %d kubelet restarts" %
                  (args.seconds, allocs, errors, rescans, flips,
                   kubelet_restarts))
            print("daemon RSS %d→%d kB (Δ%+d), fds %d→%d" %
                  (rss0, rss1, rss1 - rss0, fds0, fds1))
            grew = rss1 - rss0 > 20_000 or fds1 - fds0 > 10
            rc = 1 if grew else 0
            print("RESULT:", "FAIL (resource growth)" if grew else "OK")
        finally:
            plugin.stop()
            kubelet.stop()
            host.cleanup()
        return rc


if __name__ == "__main__":
    sys.exit(main())
