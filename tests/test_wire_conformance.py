"""Byte-level conformance of kubelet-facing messages, decoded by an
independent hand-written tag decoder (no protobuf runtime) against the
field numbers/types of the vendored api.proto — the closest available
proxy for "a real kubelet accepts us" given the reference binary cannot
be built here (no Go toolchain; docs/BENCHMARKS.md)."""

import importlib.util
import os

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

spec = importlib.util.spec_from_file_location(
    "wire_conformance", os.path.join(REPO, "tools",
                                     "wire_conformance.py"))
wc = importlib.util.module_from_spec(spec)
spec.loader.exec_module(wc)

LEN, VARINT = wc.WIRETYPE_LEN, wc.WIRETYPE_VARINT


@pytest.fixture(scope="module")
def blobs(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("wire")
    return wc.canonical_fixture(str(tmp))


def test_register_request_wire(blobs):
    reg_b, _, _, _ = blobs
    items = wc.decode(reg_b)
    by = {f: (wt, v) for f, wt, v in items}
    # api.proto:24-34: 1=version 2=endpoint 3=resource_name, all LEN
    assert by[1] == (LEN, b"v1beta1")
    assert by[2] == (LEN, b"kubevirt-INSTINCT_MI355X.sock")
    assert by[3] == (LEN, b"amd.com/INSTINCT_MI355X")
    assert set(by) <= {1, 2, 3, 4}


def test_listandwatch_wire(blobs):
    _, law_b, _, bdfs = blobs
    devices = wc.fields(wc.decode(law_b), 1)
    assert len(devices) == len(bdfs)
    seen = {}
    for raw in devices:
        d = wc.decode(raw)
        by = {f: (wt, v) for f, wt, v in d}
        # Device (api.proto:91-100): 1=ID str, 2=health str, 3=topology
        assert by[2] == (LEN, b"Healthy")
        topo = wc.decode(by[3][1])
        nodes = wc.fields(topo, 1)
        assert len(nodes) == 1
        decoded = wc.decode(nodes[0])
        if decoded:
            (numa_field, numa_wt, numa_val), = decoded
            # NUMANode.ID (api.proto:79-81): field 1, int64 => VARINT
            assert (numa_field, numa_wt) == (1, VARINT)
        else:
            numa_val = 0  # proto3 default: ID=0 serializes to empty
        seen[by[1][1].decode()] = numa_val
    assert sorted(seen) == bdfs
    assert seen["0000:0c:00.0"] == 0
    assert seen["0000:0d:00.0"] == 1


def test_allocate_response_wire(blobs):
    _, _, alloc_b, bdfs = blobs
    containers = wc.fields(wc.decode(alloc_b), 1)
    assert len(containers) == 1
    items = wc.decode(containers[0])

    # envs (field 1, map<string,string>): one entry, the requested
    # BDFs comma-joined under the resource env key
    envs = [wc.decode(e) for e in wc.fields(items, 1)]
    assert len(envs) == 1
    env_by = {f: v for f, wt, v in envs[0]}
    assert env_by[1] == b"PCI_RESOURCE_AMD_COM_INSTINCT_MI355X"
    assert env_by[2].decode() == ",".join(bdfs)

    # devices (field 3, DeviceSpec): /dev/vfio/vfio then the two group
    # nodes, each with container_path=host_path and permissions "mrw"
    specs = [wc.decode(s) for s in wc.fields(items, 3)]
    host_paths = []
    for s in specs:
        by = {f: v for f, wt, v in s}
        assert by[1] == by[2], "container_path must equal host_path"
        assert by[3] == b"mrw"
        host_paths.append(by[2].decode())
    # the fixture's vfio dir is tempdir-rooted; order and node names
    # are the contract (vfio container first, then the group nodes)
    assert [os.path.basename(p) for p in host_paths] == \
        ["vfio", "40", "41"]
    assert len({os.path.dirname(p) for p in host_paths}) == 1
    assert os.path.dirname(host_paths[0]).endswith("/dev/vfio")

    # no mounts/annotations/cdi fields are emitted for passthrough
    assert not wc.fields(items, 2)
    assert not wc.fields(items, 4)
    assert not wc.fields(items, 5)


def test_decoder_roundtrip_sanity():
    """The independent decoder agrees with the protobuf runtime on a
    crafted message (guards the decoder itself)."""
    from kubevirt_gpu_device_plugin_amd import dpapi

    m = dpapi.Device(ID="x", health="Healthy",
                     topology=dpapi.TopologyInfo(
                         nodes=[dpapi.NUMANode(ID=300)]))
    items = wc.decode(m.SerializeToString())
    by = {f: (wt, v) for f, wt, v in items}
    assert by[1] == (LEN, b"x")
    assert by[2] == (LEN, b"Healthy")
    topo = wc.decode(by[3][1])
    node = wc.decode(wc.fields(topo, 1)[0])
    assert node == [(1, VARINT, 300)]  # multi-byte varint exercised


def test_decoder_fixed_width_and_errors():
    """Decoder edge cases: I64/I32 wire types, truncated LEN, overlong
    varint — none appear in api.proto traffic, but the decoder must
    fail loudly rather than misparse if they ever do."""
    import struct

    import pytest

    # field 1, wire type 1 (I64) = tag 0x09 + 8 bytes
    items = wc.decode(b"\x09" + struct.pack("<d", 2.5))
    assert items == [(1, wc.WIRETYPE_I64, struct.pack("<d", 2.5))]
    # field 2, wire type 5 (I32) = tag 0x15 + 4 bytes
    items = wc.decode(b"\x15" + struct.pack("<f", 1.0))
    assert items == [(2, wc.WIRETYPE_I32, struct.pack("<f", 1.0))]
    # truncated LEN payload
    with pytest.raises(ValueError):
        wc.decode(b"\x0a\x05abc")
    # unsupported wire type (3 = deprecated group start)
    with pytest.raises(ValueError):
        wc.decode(b"\x0b")
    # overlong varint (>10 bytes of continuation)
    with pytest.raises(ValueError):
        wc.decode(b"\x08" + b"\x80" * 11)
    # truncated fixed-width payloads
    with pytest.raises(ValueError):
        wc.decode(b"\x09\x00\x00")
    with pytest.raises(ValueError):
        wc.decode(b"\x15\x00")


def test_decoder_agrees_with_runtime_on_random_messages():
    """Property check: for randomized AllocateResponse trees, the
    independent decoder recovers exactly the fields the protobuf
    runtime serialized (numbers, multiplicity, string payloads)."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from kubevirt_gpu_device_plugin_amd import dpapi

    text = st.text(
        alphabet=st.characters(min_codepoint=32, max_codepoint=126),
        max_size=20)

    specs = st.lists(
        st.tuples(text, text, st.sampled_from(["r", "rw", "mrw"])),
        max_size=4)

    @settings(max_examples=200, deadline=None)
    @given(envs=st.dictionaries(text.filter(bool), text, max_size=4),
           specs=specs)
    def check(envs, specs):
        car = dpapi.ContainerAllocateResponse(
            envs=envs,
            devices=[dpapi.DeviceSpec(container_path=c, host_path=h,
                                      permissions=p)
                     for c, h, p in specs])
        items = wc.decode(car.SerializeToString())
        got_envs = {}
        for e in wc.fields(items, 1):
            kv = {f: v for f, _, v in wc.decode(e)}
            got_envs[kv.get(1, b"").decode()] = kv.get(2, b"").decode()
        assert got_envs == dict(envs)
        got_specs = []
        for s in wc.fields(items, 3):
            kv = {f: v for f, _, v in wc.decode(s)}
            got_specs.append((kv.get(1, b"").decode(),
                              kv.get(2, b"").decode(),
                              kv.get(3, b"").decode()))
        assert got_specs == [tuple(x) for x in specs]

    check()
