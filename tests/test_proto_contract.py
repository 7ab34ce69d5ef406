"""Mechanical wire-contract pin against the vendored kubelet api.proto.

The dpapi package hand-builds the v1beta1 FileDescriptorProto (no protoc
in the image).  A transposed field number or wrong wire type in a message
the golden-byte tests do not cover would break real-kubelet compatibility
silently.  This test parses the authoritative proto source shipped inside
the reference tree
(reference: vendor/k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto:13-212)
and asserts that EVERY message, field name, field number, type, label,
map entry, service and method (including streaming flags) in
``dpapi.builder`` matches — both directions, so an extra field fails too.

Skipped when the reference tree is absent (e.g. on the GPU box snapshot);
the CPU CI suite always runs it.
"""

import os
import re

import pytest

from kubevirt_gpu_device_plugin_amd.dpapi import builder

API_PROTO = ("/root/reference/vendor/k8s.io/kubelet/pkg/apis/"
             "deviceplugin/v1beta1/api.proto")

pytestmark = pytest.mark.skipif(
    not os.path.exists(API_PROTO),
    reason="vendored api.proto not available on this host")

# proto scalar keyword -> FieldDescriptorProto.Type
_F = builder._F
SCALAR_TYPES = {
    "string": _F.TYPE_STRING,
    "bool": _F.TYPE_BOOL,
    "int32": _F.TYPE_INT32,
    "int64": _F.TYPE_INT64,
    "uint32": _F.TYPE_UINT32,
    "uint64": _F.TYPE_UINT64,
    "bytes": _F.TYPE_BYTES,
    "double": _F.TYPE_DOUBLE,
    "float": _F.TYPE_FLOAT,
}


def _strip_comments(text):
    text = re.sub(r"/\*.*?\*/", "", text, flags=re.S)
    text = re.sub(r"//[^\n]*", "", text)
    return text


def parse_proto(path):
    """Parse the subset of proto3 used by api.proto into plain dicts.

    Returns (package, messages, services):
      messages: name -> list of field dicts
                {name, number, label, type, type_name(opt), map(opt)}
      services: name -> list of method dicts
                {name, input, output, client_streaming, server_streaming}
    """
    with open(path) as f:
        text = _strip_comments(f.read())

    pkg = re.search(r"\bpackage\s+([\w.]+)\s*;", text).group(1)

    messages, services = {}, {}
    # rpc declarations carry `{}` bodies inside service blocks, so block
    # extents need brace counting rather than a non-greedy regex.
    blocks = []
    for m in re.finditer(r"\b(message|service)\s+(\w+)\s*\{", text):
        depth, i = 1, m.end()
        while depth:
            if text[i] == "{":
                depth += 1
            elif text[i] == "}":
                depth -= 1
            i += 1
        blocks.append((m.group(1), m.group(2), text[m.end():i - 1]))
    for kind, name, body in blocks:
        if kind == "service":
            methods = []
            for mname, inp, stream_kw, out in re.findall(
                    r"\brpc\s+(\w+)\s*\(\s*(?:stream\s+)?(\w+)\s*\)\s*"
                    r"returns\s*\(\s*(stream\s+)?(\w+)\s*\)", body):
                methods.append({
                    "name": mname, "input": inp, "output": out,
                    "server_streaming": bool(stream_kw.strip()),
                })
            services[name] = methods
            continue
        fields = []
        for stmt in body.split(";"):
            stmt = stmt.strip()
            if not stmt:
                continue
            m = re.match(r"map\s*<\s*(\w+)\s*,\s*(\w+)\s*>\s+(\w+)\s*=\s*"
                         r"(\d+)$", stmt)
            if m:
                fields.append({
                    "name": m.group(3), "number": int(m.group(4)),
                    "map": (m.group(1), m.group(2)),
                })
                continue
            m = re.match(r"(repeated\s+)?([\w.]+)\s+(\w+)\s*=\s*(\d+)$",
                         stmt)
            assert m, "unparsed field statement: %r" % stmt
            fields.append({
                "name": m.group(3), "number": int(m.group(4)),
                "repeated": bool(m.group(1)),
                "type": m.group(2),
            })
        messages[name] = fields
    return pkg, messages, services


def to_json_name(name):
    """protoc's ToJsonName: drop underscores, capitalize the following
    character; other characters (including the first) unchanged."""
    out, cap = [], False
    for ch in name:
        if ch == "_":
            cap = True
            continue
        out.append(ch.upper() if cap else ch)
        cap = False
    return "".join(out)


@pytest.fixture(scope="module")
def parsed():
    return parse_proto(API_PROTO)


@pytest.fixture(scope="module")
def fdp():
    return builder.build_file_descriptor_proto()


def test_package(parsed, fdp):
    pkg, _, _ = parsed
    assert fdp.package == pkg


def test_message_set_identical(parsed, fdp):
    _, messages, _ = parsed
    ours = {m.name for m in fdp.message_type}
    assert ours == set(messages), (
        "message sets differ: only-ours=%s only-proto=%s"
        % (ours - set(messages), set(messages) - ours))


def test_every_field_matches(parsed, fdp):
    """Name, number, label, type and (for message fields) type_name of
    every field in every message, both directions."""
    pkg, messages, _ = parsed
    ours = {m.name: m for m in fdp.message_type}
    for mname, pfields in messages.items():
        built = ours[mname]
        built_by_name = {f.name: f for f in built.field}
        assert set(built_by_name) == {f["name"] for f in pfields}, (
            "%s: field name sets differ" % mname)
        for pf in pfields:
            bf = built_by_name[pf["name"]]
            ctx = "%s.%s" % (mname, pf["name"])
            assert bf.number == pf["number"], ctx
            if "map" in pf:
                # map<k,v> == repeated nested MapEntry message
                assert bf.label == _F.LABEL_REPEATED, ctx
                assert bf.type == _F.TYPE_MESSAGE, ctx
                nested = {n.name: n for n in built.nested_type}
                entry_name = to_json_name(pf["name"])
                entry_name = (entry_name[0].upper() + entry_name[1:]
                              + "Entry")
                assert bf.type_name == ".%s.%s.%s" % (
                    pkg, mname, entry_name), ctx
                entry = nested[entry_name]
                assert entry.options.map_entry, ctx
                kf, vf = entry.field[0], entry.field[1]
                assert (kf.name, kf.number, kf.type) == (
                    "key", 1, SCALAR_TYPES[pf["map"][0]]), ctx
                assert (vf.name, vf.number, vf.type) == (
                    "value", 2, SCALAR_TYPES[pf["map"][1]]), ctx
                continue
            expected_label = (_F.LABEL_REPEATED if pf["repeated"]
                              else _F.LABEL_OPTIONAL)
            assert bf.label == expected_label, ctx
            if pf["type"] in SCALAR_TYPES:
                assert bf.type == SCALAR_TYPES[pf["type"]], ctx
                assert not bf.type_name, ctx
            else:
                assert bf.type == _F.TYPE_MESSAGE, ctx
                assert bf.type_name == ".%s.%s" % (pkg, pf["type"]), ctx


def test_no_extra_nested_types(parsed, fdp):
    """Only map-entry messages may be nested (api.proto declares no
    nested message types)."""
    for m in fdp.message_type:
        for n in m.nested_type:
            assert n.options.map_entry, (
                "%s.%s is not a map entry" % (m.name, n.name))


def test_json_names_match_protoc_derivation(parsed):
    """The runtime descriptors must expose the same json_name protoc
    would generate from the proto field names (a wrong explicit
    json_name in builder.py would diverge in JSON transcodings)."""
    from kubevirt_gpu_device_plugin_amd import dpapi
    _, messages, _ = parsed
    for mname, pfields in messages.items():
        desc = getattr(dpapi, mname).DESCRIPTOR
        for pf in pfields:
            fd = desc.fields_by_name[pf["name"]]
            assert fd.json_name == to_json_name(pf["name"]), (
                "%s.%s json_name %r != protoc-derived %r"
                % (mname, pf["name"], fd.json_name,
                   to_json_name(pf["name"])))


def test_services_and_methods(parsed, fdp):
    pkg, _, services = parsed
    ours = {s.name: s for s in fdp.service}
    assert set(ours) == set(services)
    for sname, methods in services.items():
        built = {m.name: m for m in ours[sname].method}
        assert set(built) == {m["name"] for m in methods}, sname
        for pm in methods:
            bm = built[pm["name"]]
            ctx = "%s.%s" % (sname, pm["name"])
            assert bm.input_type == ".%s.%s" % (pkg, pm["input"]), ctx
            assert bm.output_type == ".%s.%s" % (pkg, pm["output"]), ctx
            assert bm.server_streaming == pm["server_streaming"], ctx
            assert not bm.client_streaming, ctx


def test_parser_sanity(parsed):
    """The parser itself saw the full API (guards against a silent
    regex miss turning every other assertion vacuous)."""
    _, messages, services = parsed
    assert len(messages) == 20
    assert set(services) == {"Registration", "DevicePlugin"}
    assert len(services["DevicePlugin"]) == 5
    # spot-check one deep fact straight from the proto text
    car = {f["name"]: f for f in messages["ContainerAllocateResponse"]}
    assert car["cdi_devices"]["number"] == 5
    assert car["envs"]["map"] == ("string", "string")
