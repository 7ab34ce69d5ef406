"""Daemon entry point — zero flags, zero env config
(reference: cmd/main.go:33-35).

Run as ``python -m kubevirt_gpu_device_plugin_amd.cmd.main``.
"""

import logging
import signal
import threading

from ..device_plugin.controller import initiate_device_plugin


def main():
    logging.basicConfig(
        level=logging.INFO,
        format="%(asctime)s %(levelname).1s %(name)s: %(message)s")
    logging.info("starting AMD KubeVirt GPU device plugin")
    stop_event = threading.Event()
    rescan_event = threading.Event()

    def _on_signal(signum, frame):
        logging.info("received signal %d; shutting down", signum)
        stop_event.set()

    signal.signal(signal.SIGTERM, _on_signal)
    signal.signal(signal.SIGINT, _on_signal)
    # SIGHUP = re-discover (e.g. after `echo 8 > sriov_numvfs`)
    signal.signal(signal.SIGHUP, lambda *a: rescan_event.set())
    initiate_device_plugin(stop_event=stop_event,
                           rescan_event=rescan_event)


if __name__ == "__main__":
    main()
