"""Kubelet DevicePlugin v1beta1 protocol constants.

Mirrors the upstream Kubernetes constants
(reference: vendor/k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/constants.go:20-32).
These values are protocol facts — kubelet expects them verbatim.
"""

# Healthy means that the device is healthy.
HEALTHY = "Healthy"
# Unhealthy means that the device is unhealthy.
UNHEALTHY = "Unhealthy"

# Current version of the API supported by kubelet.
VERSION = "v1beta1"

# DEVICE_PLUGIN_PATH is the folder the device plugin is expecting sockets
# to be on.
DEVICE_PLUGIN_PATH = "/var/lib/kubelet/device-plugins/"
# KUBELET_SOCKET is the path of the kubelet registry socket.
KUBELET_SOCKET = DEVICE_PLUGIN_PATH + "kubelet.sock"

# Timeout of the gRPC session (the reference uses a 5s connection timeout,
# generic_device_plugin.go:53).
CONNECT_TIMEOUT_S = 5.0

# Fully-qualified gRPC service names.
REGISTRATION_SERVICE = "v1beta1.Registration"
DEVICE_PLUGIN_SERVICE = "v1beta1.DevicePlugin"
