"""Programmatic construction of the kubelet DevicePlugin v1beta1 protobuf API.

The image has no ``protoc``/``grpcio-tools``, so instead of shipping
generated ``*_pb2.py`` files we build the ``FileDescriptorProto`` for the
v1beta1 API in code and ask the installed protobuf runtime (upb) for
message classes.  The message/field names, numbers, and types below are
protocol facts of the upstream Kubernetes API
(reference: vendor/k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto:40-218)
— they must match exactly for wire compatibility with kubelet.
"""

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto

_PKG = "v1beta1"
_FILE_NAME = "kubevirt_gpu_device_plugin_amd/dpapi/api.proto"


def _field(name, number, ftype, label=_F.LABEL_OPTIONAL, type_name=None,
           json_name=None):
    f = _F()
    f.name = name
    f.number = number
    f.type = ftype
    f.label = label
    if type_name is not None:
        f.type_name = type_name
    if json_name is not None:
        f.json_name = json_name
    return f


def _message(fdp, name, fields, nested=()):
    m = fdp.message_type.add()
    m.name = name
    for f in fields:
        m.field.append(f)
    for n in nested:
        m.nested_type.append(n)
    return m


def _map_entry(name):
    """A map<string,string> synthesises a nested repeated MapEntry message."""
    m = descriptor_pb2.DescriptorProto()
    m.name = name
    m.options.map_entry = True
    m.field.append(_field("key", 1, _F.TYPE_STRING))
    m.field.append(_field("value", 2, _F.TYPE_STRING))
    return m


def build_file_descriptor_proto():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = _FILE_NAME
    fdp.package = _PKG
    fdp.syntax = "proto3"

    S, M, R = _F.TYPE_STRING, _F.TYPE_MESSAGE, _F.LABEL_REPEATED
    B, I32, I64 = _F.TYPE_BOOL, _F.TYPE_INT32, _F.TYPE_INT64

    def t(n):
        return ".%s.%s" % (_PKG, n)

    _message(fdp, "DevicePluginOptions", [
        _field("pre_start_required", 1, B),
        _field("get_preferred_allocation_available", 2, B),
    ])
    _message(fdp, "RegisterRequest", [
        _field("version", 1, S),
        _field("endpoint", 2, S),
        _field("resource_name", 3, S),
        _field("options", 4, M, type_name=t("DevicePluginOptions")),
    ])
    _message(fdp, "Empty", [])
    _message(fdp, "ListAndWatchResponse", [
        _field("devices", 1, M, R, t("Device")),
    ])
    _message(fdp, "TopologyInfo", [
        _field("nodes", 1, M, R, t("NUMANode")),
    ])
    _message(fdp, "NUMANode", [
        _field("ID", 1, I64, json_name="ID"),
    ])
    _message(fdp, "Device", [
        _field("ID", 1, S, json_name="ID"),
        _field("health", 2, S),
        _field("topology", 3, M, type_name=t("TopologyInfo")),
    ])
    _message(fdp, "PreferredAllocationRequest", [
        _field("container_requests", 1, M, R,
               t("ContainerPreferredAllocationRequest")),
    ])
    _message(fdp, "ContainerPreferredAllocationRequest", [
        _field("available_deviceIDs", 1, S, R, json_name="availableDeviceIDs"),
        _field("must_include_deviceIDs", 2, S, R,
               json_name="mustIncludeDeviceIDs"),
        _field("allocation_size", 3, I32),
    ])
    _message(fdp, "PreferredAllocationResponse", [
        _field("container_responses", 1, M, R,
               t("ContainerPreferredAllocationResponse")),
    ])
    _message(fdp, "ContainerPreferredAllocationResponse", [
        _field("deviceIDs", 1, S, R, json_name="deviceIDs"),
    ])
    _message(fdp, "PreStartContainerRequest", [
        _field("devices_ids", 1, S, R),
    ])
    _message(fdp, "PreStartContainerResponse", [])
    _message(fdp, "AllocateRequest", [
        _field("container_requests", 1, M, R, t("ContainerAllocateRequest")),
    ])
    _message(fdp, "ContainerAllocateRequest", [
        _field("devices_ids", 1, S, R),
    ])
    _message(fdp, "AllocateResponse", [
        _field("container_responses", 1, M, R, t("ContainerAllocateResponse")),
    ])
    _message(
        fdp, "ContainerAllocateResponse",
        [
            _field("envs", 1, M, R, t("ContainerAllocateResponse.EnvsEntry")),
            _field("mounts", 2, M, R, t("Mount")),
            _field("devices", 3, M, R, t("DeviceSpec")),
            _field("annotations", 4, M, R,
                   t("ContainerAllocateResponse.AnnotationsEntry")),
            _field("cdi_devices", 5, M, R, t("CDIDevice"),
                   json_name="cdiDevices"),
        ],
        nested=[_map_entry("EnvsEntry"), _map_entry("AnnotationsEntry")],
    )
    _message(fdp, "Mount", [
        _field("container_path", 1, S),
        _field("host_path", 2, S),
        _field("read_only", 3, B),
    ])
    _message(fdp, "DeviceSpec", [
        _field("container_path", 1, S),
        _field("host_path", 2, S),
        _field("permissions", 3, S),
    ])
    _message(fdp, "CDIDevice", [
        _field("name", 1, S),
    ])

    reg = fdp.service.add()
    reg.name = "Registration"
    m = reg.method.add()
    m.name = "Register"
    m.input_type = t("RegisterRequest")
    m.output_type = t("Empty")

    dp = fdp.service.add()
    dp.name = "DevicePlugin"
    for name, inp, out, streaming in [
        ("GetDevicePluginOptions", "Empty", "DevicePluginOptions", False),
        ("ListAndWatch", "Empty", "ListAndWatchResponse", True),
        ("GetPreferredAllocation", "PreferredAllocationRequest",
         "PreferredAllocationResponse", False),
        ("Allocate", "AllocateRequest", "AllocateResponse", False),
        ("PreStartContainer", "PreStartContainerRequest",
         "PreStartContainerResponse", False),
    ]:
        m = dp.method.add()
        m.name = name
        m.input_type = t(inp)
        m.output_type = t(out)
        m.server_streaming = streaming

    return fdp


_pool = descriptor_pool.DescriptorPool()
_file_desc = _pool.Add(build_file_descriptor_proto())

_MESSAGE_NAMES = [
    "DevicePluginOptions", "RegisterRequest", "Empty",
    "ListAndWatchResponse", "TopologyInfo", "NUMANode", "Device",
    "PreferredAllocationRequest", "ContainerPreferredAllocationRequest",
    "PreferredAllocationResponse", "ContainerPreferredAllocationResponse",
    "PreStartContainerRequest", "PreStartContainerResponse",
    "AllocateRequest", "ContainerAllocateRequest",
    "AllocateResponse", "ContainerAllocateResponse",
    "Mount", "DeviceSpec", "CDIDevice",
]

_classes = {
    name: message_factory.GetMessageClass(
        _pool.FindMessageTypeByName("%s.%s" % (_PKG, name)))
    for name in _MESSAGE_NAMES
}


def message_class(name):
    return _classes[name]


def all_message_classes():
    return dict(_classes)
