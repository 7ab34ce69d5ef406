"""Wire-format and API-surface tests for the hand-built v1beta1 API."""

from kubevirt_gpu_device_plugin_amd import dpapi
from kubevirt_gpu_device_plugin_amd.dpapi import builder


def test_constants():
    assert dpapi.VERSION == "v1beta1"
    assert dpapi.DEVICE_PLUGIN_PATH == "/var/lib/kubelet/device-plugins/"
    assert dpapi.KUBELET_SOCKET.endswith("kubelet.sock")
    assert dpapi.HEALTHY == "Healthy"
    assert dpapi.UNHEALTHY == "Unhealthy"


def test_field_numbers_match_upstream():
    """Field numbers are the wire contract with kubelet — pin them."""
    fdp = builder.build_file_descriptor_proto()
    msgs = {m.name: {f.name: (f.number, f.type) for f in m.field}
            for m in fdp.message_type}
    F = builder._F
    assert msgs["RegisterRequest"]["version"] == (1, F.TYPE_STRING)
    assert msgs["RegisterRequest"]["endpoint"] == (2, F.TYPE_STRING)
    assert msgs["RegisterRequest"]["resource_name"] == (3, F.TYPE_STRING)
    assert msgs["Device"]["ID"] == (1, F.TYPE_STRING)
    assert msgs["Device"]["health"] == (2, F.TYPE_STRING)
    assert msgs["Device"]["topology"] == (3, F.TYPE_MESSAGE)
    assert msgs["DeviceSpec"]["container_path"] == (1, F.TYPE_STRING)
    assert msgs["DeviceSpec"]["host_path"] == (2, F.TYPE_STRING)
    assert msgs["DeviceSpec"]["permissions"] == (3, F.TYPE_STRING)
    assert msgs["ContainerAllocateResponse"]["envs"][0] == 1
    assert msgs["ContainerAllocateResponse"]["devices"][0] == 3
    assert msgs["ContainerPreferredAllocationRequest"][
        "available_deviceIDs"][0] == 1
    assert msgs["ContainerPreferredAllocationRequest"][
        "allocation_size"] == (3, F.TYPE_INT32)
    services = {s.name: [m.name for m in s.method] for s in fdp.service}
    assert services["Registration"] == ["Register"]
    assert services["DevicePlugin"] == [
        "GetDevicePluginOptions", "ListAndWatch",
        "GetPreferredAllocation", "Allocate", "PreStartContainer"]


def test_listandwatch_is_server_streaming():
    fdp = builder.build_file_descriptor_proto()
    dp = [s for s in fdp.service if s.name == "DevicePlugin"][0]
    by_name = {m.name: m for m in dp.method}
    assert by_name["ListAndWatch"].server_streaming
    assert not by_name["Allocate"].server_streaming


def test_roundtrip_allocate_response():
    r = dpapi.ContainerAllocateResponse()
    r.envs["PCI_RESOURCE_AMD_COM_INSTINCT_MI355X"] = \
        "0000:0c:00.0,0000:2f:00.0"
    r.devices.add(container_path="/dev/vfio/vfio",
                  host_path="/dev/vfio/vfio", permissions="mrw")
    r.devices.add(container_path="/dev/vfio/42",
                  host_path="/dev/vfio/42", permissions="mrw")
    r2 = dpapi.ContainerAllocateResponse.FromString(r.SerializeToString())
    assert list(r2.envs.values()) == ["0000:0c:00.0,0000:2f:00.0"]
    assert [d.host_path for d in r2.devices] == [
        "/dev/vfio/vfio", "/dev/vfio/42"]


def test_roundtrip_device_topology():
    d = dpapi.Device(ID="0000:0c:00.0", health=dpapi.HEALTHY,
                     topology=dpapi.TopologyInfo(
                         nodes=[dpapi.NUMANode(ID=3)]))
    d2 = dpapi.Device.FromString(d.SerializeToString())
    assert d2.topology.nodes[0].ID == 3
    # proto3 default: zero-valued NUMA id survives via presence of node msg
    d3 = dpapi.Device(ID="x", topology=dpapi.TopologyInfo(
        nodes=[dpapi.NUMANode(ID=0)]))
    d4 = dpapi.Device.FromString(d3.SerializeToString())
    assert len(d4.topology.nodes) == 1 and d4.topology.nodes[0].ID == 0


def test_golden_wire_bytes():
    """Pin exact wire encodings (hand-verified against the proto3 wire
    format): field 1 LEN "v1beta1", etc.  Any drift in field numbers or
    types breaks kubelet compatibility and must fail loudly here."""
    r = dpapi.RegisterRequest(version="v1beta1",
                              endpoint="kubevirt-X.sock",
                              resource_name="amd.com/X")
    assert r.SerializeToString().hex() == (
        "0a0776316265746131120f6b756265766972742d582e736f636b"
        "1a09616d642e636f6d2f58")
    d = dpapi.Device(ID="0000:0c:00.0", health="Healthy",
                     topology=dpapi.TopologyInfo(
                         nodes=[dpapi.NUMANode(ID=1)]))
    assert d.SerializeToString().hex() == (
        "0a0c303030303a30633a30302e3012074865616c7468791a040a020801")
    a = dpapi.ContainerAllocateResponse()
    a.envs["K"] = "V"
    a.devices.add(container_path="/dev/vfio/vfio",
                  host_path="/dev/vfio/vfio", permissions="mrw")
    assert a.SerializeToString().hex() == (
        "0a060a014b1201561a250a0e2f6465762f7666696f2f7666696f"
        "120e2f6465762f7666696f2f7666696f1a036d7277")
