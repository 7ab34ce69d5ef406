"""The real zero-config entry point: starts against live host paths
(discovers nothing on a non-vfio test host), and shuts down cleanly on
SIGTERM; SIGHUP triggers a rescan without crashing."""

import os
import signal
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _spawn():
    return subprocess.Popen(
        [sys.executable, "-m", "kubevirt_gpu_device_plugin_amd"],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)


def test_main_sigterm_clean_exit():
    proc = _spawn()
    try:
        time.sleep(2.0)
        assert proc.poll() is None, proc.stdout.read()
        proc.send_signal(signal.SIGTERM)
        assert proc.wait(timeout=15) == 0
        out = proc.stdout.read()
        assert "starting AMD KubeVirt GPU device plugin" in out
        assert "shutting down" in out
    finally:
        if proc.poll() is None:
            proc.kill()
            proc.wait()


def test_main_sighup_rescan_no_crash():
    proc = _spawn()
    try:
        time.sleep(1.5)
        proc.send_signal(signal.SIGHUP)
        time.sleep(1.5)
        assert proc.poll() is None
        proc.send_signal(signal.SIGTERM)
        assert proc.wait(timeout=15) == 0
        assert "rescan requested" in proc.stdout.read()
    finally:
        if proc.poll() is None:
            proc.kill()
            proc.wait()
