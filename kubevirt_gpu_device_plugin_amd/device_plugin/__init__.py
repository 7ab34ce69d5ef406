"""Device-plugin core: discovery, plugin servers, controller."""

from .discovery import AmdGpuDevice, DeviceRegistry, discover  # noqa: F401
from .plugin import GenericDevicePlugin  # noqa: F401
from .plugin_base import PluginConfig  # noqa: F401
from .vf_plugin import VfDevicePlugin  # noqa: F401
from .controller import Controller, initiate_device_plugin  # noqa: F401
