"""gRPC servicers and client stubs for the v1beta1 API.

Hand-rolled equivalents of what ``grpcio-tools`` would generate
(reference service definitions:
vendor/k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto:13-66).
"""

import grpc

from . import builder
from .constants import REGISTRATION_SERVICE, DEVICE_PLUGIN_SERVICE

_c = builder.message_class


class RegistrationServicer:
    """kubelet's Registration service — implemented here only by the test /
    bench stub kubelet (the real one lives in kubelet)."""

    def Register(self, request, context):  # noqa: N802
        raise NotImplementedError


def add_registration_servicer(servicer, server):
    handlers = {
        "Register": grpc.unary_unary_rpc_method_handler(
            servicer.Register,
            request_deserializer=_c("RegisterRequest").FromString,
            response_serializer=lambda m: m.SerializeToString(),
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(REGISTRATION_SERVICE, handlers),)
    )


class DevicePluginServicer:
    """Base class for DevicePlugin service implementations."""

    def GetDevicePluginOptions(self, request, context):  # noqa: N802
        raise NotImplementedError

    def ListAndWatch(self, request, context):  # noqa: N802
        raise NotImplementedError

    def GetPreferredAllocation(self, request, context):  # noqa: N802
        raise NotImplementedError

    def Allocate(self, request, context):  # noqa: N802
        raise NotImplementedError

    def PreStartContainer(self, request, context):  # noqa: N802
        raise NotImplementedError


def add_device_plugin_servicer(servicer, server):
    ser = lambda m: m.SerializeToString()  # noqa: E731
    handlers = {
        "GetDevicePluginOptions": grpc.unary_unary_rpc_method_handler(
            servicer.GetDevicePluginOptions,
            request_deserializer=_c("Empty").FromString,
            response_serializer=ser,
        ),
        "ListAndWatch": grpc.unary_stream_rpc_method_handler(
            servicer.ListAndWatch,
            request_deserializer=_c("Empty").FromString,
            response_serializer=ser,
        ),
        "GetPreferredAllocation": grpc.unary_unary_rpc_method_handler(
            servicer.GetPreferredAllocation,
            request_deserializer=_c("PreferredAllocationRequest").FromString,
            response_serializer=ser,
        ),
        "Allocate": grpc.unary_unary_rpc_method_handler(
            servicer.Allocate,
            request_deserializer=_c("AllocateRequest").FromString,
            response_serializer=ser,
        ),
        "PreStartContainer": grpc.unary_unary_rpc_method_handler(
            servicer.PreStartContainer,
            request_deserializer=_c("PreStartContainerRequest").FromString,
            response_serializer=ser,
        ),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(DEVICE_PLUGIN_SERVICE, handlers),)
    )


class RegistrationStub:
    """Client used by the plugin to self-register with kubelet
    (reference: generic_device_plugin.go:289-310)."""

    def __init__(self, channel):
        self.Register = channel.unary_unary(
            "/%s/Register" % REGISTRATION_SERVICE,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=_c("Empty").FromString,
        )


class DevicePluginStub:
    """Client used by the stub kubelet (tests/bench) and by the plugin's own
    readiness self-dial (reference: generic_device_plugin.go:187-214)."""

    def __init__(self, channel):
        ser = lambda m: m.SerializeToString()  # noqa: E731
        p = "/%s/" % DEVICE_PLUGIN_SERVICE
        self.GetDevicePluginOptions = channel.unary_unary(
            p + "GetDevicePluginOptions", request_serializer=ser,
            response_deserializer=_c("DevicePluginOptions").FromString)
        self.ListAndWatch = channel.unary_stream(
            p + "ListAndWatch", request_serializer=ser,
            response_deserializer=_c("ListAndWatchResponse").FromString)
        self.GetPreferredAllocation = channel.unary_unary(
            p + "GetPreferredAllocation", request_serializer=ser,
            response_deserializer=_c("PreferredAllocationResponse").FromString)
        self.Allocate = channel.unary_unary(
            p + "Allocate", request_serializer=ser,
            response_deserializer=_c("AllocateResponse").FromString)
        self.PreStartContainer = channel.unary_unary(
            p + "PreStartContainer", request_serializer=ser,
            response_deserializer=_c("PreStartContainerResponse").FromString)
