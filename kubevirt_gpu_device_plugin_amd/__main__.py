"""``python -m kubevirt_gpu_device_plugin_amd`` runs the daemon."""

from .cmd.main import main

main()
