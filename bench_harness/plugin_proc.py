"""Run the device-plugin daemon against injected host paths.

Used by the bench rig and integration tests to put the plugin in its
own process (real RPC boundary).  Takes one argv: a JSON dict of
PluginConfig fields + kfd_nodes_dir.
"""

import json
import logging
import os
import signal
import sys
import threading

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (  # noqa: E402
    initiate_device_plugin,
)
from kubevirt_gpu_device_plugin_amd.device_plugin.plugin_base import (  # noqa: E402
    PluginConfig,
)


def main():
    logging.basicConfig(level=logging.WARNING)
    spec = json.loads(sys.argv[1])
    kfd = spec.pop("kfd_nodes_dir")
    cfg = PluginConfig(**spec)
    stop = threading.Event()
    rescan = threading.Event()
    signal.signal(signal.SIGTERM, lambda *a: stop.set())
    signal.signal(signal.SIGINT, lambda *a: stop.set())
    signal.signal(signal.SIGHUP, lambda *a: rescan.set())
    initiate_device_plugin(stop_event=stop, rescan_event=rescan,
                           uevent_autoscan=False,  # deterministic rig
                           config=cfg, kfd_nodes_dir=kfd,
                           vf_event_watcher_factory=lambda: None)


if __name__ == "__main__":
    main()
