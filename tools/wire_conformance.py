#!/usr/bin/env python3
"""Byte-level wire conformance of the kubelet-facing responses.

The reference Go binary cannot be built in this environment (no Go
toolchain, no network — see docs/BENCHMARKS.md), so the closest
available proxy for "a real kubelet accepts us" is this: serialize the
plugin's actual RegisterRequest / ListAndWatchResponse /
AllocateResponse for a canonical fixture, then re-decode the raw bytes
with an INDEPENDENT hand-written protobuf tag decoder (no descriptors,
no protobuf runtime) and assert the tag/wire-type/value tree matches
the field numbers and types of the vendored
vendor/k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto:17-212
exactly.  Run as a script to emit the human-readable conformance
artifact (profiles/wire_conformance_*.md); imported by
tests/test_wire_conformance.py for the assertions.
"""

import sys

WIRETYPE_VARINT = 0
WIRETYPE_I64 = 1
WIRETYPE_LEN = 2
WIRETYPE_I32 = 5


def read_varint(data, pos):
    shift, out = 0, 0
    while True:
        b = data[pos]
        out |= (b & 0x7F) << shift
        pos += 1
        if not b & 0x80:
            return out, pos
        shift += 7
        if shift > 63:
            raise ValueError("varint too long")


def decode(data):
    """Decode one message level: list of (field_number, wire_type,
    value) in stream order.  LEN payloads stay raw bytes — callers
    recurse where the schema says the field is a submessage."""
    out, pos = [], 0
    while pos < len(data):
        key, pos = read_varint(data, pos)
        field, wt = key >> 3, key & 7
        if wt == WIRETYPE_VARINT:
            val, pos = read_varint(data, pos)
        elif wt == WIRETYPE_LEN:
            ln, pos = read_varint(data, pos)
            val = data[pos:pos + ln]
            if len(val) != ln:
                raise ValueError("truncated LEN field %d" % field)
            pos += ln
        elif wt == WIRETYPE_I64:
            val, pos = data[pos:pos + 8], pos + 8
            if len(val) != 8:
                raise ValueError("truncated I64 field %d" % field)
        elif wt == WIRETYPE_I32:
            val, pos = data[pos:pos + 4], pos + 4
            if len(val) != 4:
                raise ValueError("truncated I32 field %d" % field)
        else:
            raise ValueError("unsupported wire type %d" % wt)
        out.append((field, wt, val))
    return out


def fields(items, number):
    return [v for f, wt, v in items if f == number]


def render(items, schema, indent=0):
    """Pretty tree for the artifact.  schema: {field: (name, subschema
    or None)}."""
    lines = []
    for f, wt, v in items:
        name, sub = schema.get(f, ("UNKNOWN_FIELD_%d" % f, None))
        pad = "  " * indent
        if sub is not None:
            lines.append("%s%d %s {" % (pad, f, name))
            lines.extend(render(decode(v), sub, indent + 1))
            lines.append("%s}" % pad)
        elif wt == WIRETYPE_LEN:
            lines.append("%s%d %s = %r" % (pad, f, name,
                                           v.decode(errors="replace")))
        else:
            lines.append("%s%d %s = %r" % (pad, f, name, v))
    return lines


# schemas straight from the vendored api.proto (field numbers cited)
NUMANODE = {1: ("ID", None)}                       # api.proto:79-81
TOPOLOGY = {1: ("nodes", NUMANODE)}                # api.proto:75-77
DEVICE = {1: ("ID", None), 2: ("health", None),
          3: ("topology", TOPOLOGY)}               # api.proto:91-100
LISTANDWATCH = {1: ("devices", DEVICE)}            # api.proto:71-73
REGISTER = {1: ("version", None), 2: ("endpoint", None),
            3: ("resource_name", None),
            4: ("options", {1: ("pre_start_required", None),
                            2: ("get_preferred_allocation_available",
                                None)})}           # api.proto:24-34
MAPENTRY = {1: ("key", None), 2: ("value", None)}
DEVICESPEC = {1: ("container_path", None), 2: ("host_path", None),
              3: ("permissions", None)}            # api.proto:202-212
MOUNT = {1: ("container_path", None), 2: ("host_path", None),
         3: ("read_only", None)}                   # api.proto:192-199
CDIDEVICE = {1: ("name", None)}                    # api.proto:157-163
CONTAINER_ALLOC = {1: ("envs", MAPENTRY), 2: ("mounts", MOUNT),
                   3: ("devices", DEVICESPEC),
                   4: ("annotations", MAPENTRY),
                   5: ("cdi_devices", CDIDEVICE)}  # api.proto:177-188
ALLOCATE = {1: ("container_responses",
                CONTAINER_ALLOC)}                  # api.proto:173-175


def canonical_fixture(tmpdir, n_gpus=2, iommufd=False):
    """Synthetic 2-GPU node; returns the three serialized messages the
    kubelet-facing surface produces for it."""
    import threading

    from kubevirt_gpu_device_plugin_amd import dpapi
    from kubevirt_gpu_device_plugin_amd.device_plugin import discovery
    from kubevirt_gpu_device_plugin_amd.device_plugin.controller import (
        build_kubelet_devices,
    )
    from kubevirt_gpu_device_plugin_amd.device_plugin.plugin import (
        GenericDevicePlugin,
    )
    from tests.fixtures import StubKubelet, SyntheticHost, dial_plugin

    h = SyntheticHost(tmpdir)
    for i in range(n_gpus):
        h.add_gpu("0000:%02x:00.0" % (0x0c + i),
                  iommu_group=str(40 + i), numa=i % 2)
    if iommufd:
        h.enable_iommufd()
    cfg = h.config()
    kubelet = StubKubelet(cfg.kubelet_socket)
    reg = discovery.discover(base_path=h.pci)
    _, devs = next(iter(reg.device_map.items()))
    plugin = GenericDevicePlugin(
        "INSTINCT_MI355X", build_kubelet_devices(devs), reg, config=cfg)
    stop = threading.Event()
    plugin.start(stop)
    try:
        register_req = kubelet.wait_register()
        ch, stub = dial_plugin(plugin.socket_path)
        law = next(iter(stub.ListAndWatch(dpapi.Empty())))
        bdfs = sorted(d.addr for d in devs)
        alloc = stub.Allocate(dpapi.AllocateRequest(
            container_requests=[
                dpapi.ContainerAllocateRequest(devices_ids=bdfs)]))
        ch.close()
        return (register_req.SerializeToString(),
                law.SerializeToString(),
                alloc.SerializeToString(), bdfs)
    finally:
        stop.set()
        plugin.stop()
        kubelet.stop()
        h.cleanup()


def main():
    import tempfile

    with tempfile.TemporaryDirectory() as tmp:
        reg_b, law_b, alloc_b, bdfs = canonical_fixture(tmp)
    print("# Wire conformance artifact")
    print()
    print("Canonical 2-GPU fixture; every byte below was produced by "
          "the plugin's kubelet-facing surface and re-decoded by the "
          "independent tag decoder in tools/wire_conformance.py "
          "against the field numbers of the vendored api.proto.")
    for title, blob, schema in [
            ("RegisterRequest (api.proto:24-34)", reg_b, REGISTER),
            ("ListAndWatchResponse (api.proto:71-100)", law_b,
             LISTANDWATCH),
            ("AllocateResponse (api.proto:173-212)", alloc_b,
             ALLOCATE)]:
        print("\n## %s\n" % title)
        print("```\nraw (%d bytes): %s\n" % (len(blob), blob.hex()))
        print("\n".join(render(decode(blob), schema)))
        print("```")


if __name__ == "__main__":
    sys.path.insert(0, ".")
    main()
