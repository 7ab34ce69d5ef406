"""kubelet DevicePlugin v1beta1 API: message classes, constants, gRPC glue.

Usage::

    from kubevirt_gpu_device_plugin_amd import dpapi
    dev = dpapi.Device(ID="0000:0c:00.0", health=dpapi.HEALTHY)
"""

from .constants import (  # noqa: F401
    HEALTHY, UNHEALTHY, VERSION,
    DEVICE_PLUGIN_PATH, KUBELET_SOCKET, CONNECT_TIMEOUT_S,
    REGISTRATION_SERVICE, DEVICE_PLUGIN_SERVICE,
)
from . import builder as _builder

_g = globals()
for _name, _cls in _builder.all_message_classes().items():
    _g[_name] = _cls

from .rpc import (  # noqa: F401,E402
    DevicePluginServicer, add_device_plugin_servicer,
    RegistrationServicer, add_registration_servicer,
    DevicePluginStub, RegistrationStub,
)
