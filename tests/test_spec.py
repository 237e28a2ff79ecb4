"""oim.v0 schema tests: round-trips, oneof semantics, spec.md drift.

Counterpart of the reference's proto-drift meta test (Makefile:85-103,
126-127): docs/spec.md is the literate source; the runtime-built
descriptors must match the protobuf blocks embedded there.
"""

import os
import re

from oim_amd import spec

HERE = os.path.dirname(os.path.abspath(__file__))
SPEC_MD = os.path.join(HERE, "..", "docs", "spec.md")


def test_roundtrip_value():
    msg = spec.SetValueRequest(value=spec.Value(path="ctrl-1/address", value="tcp://x:1"))
    data = msg.SerializeToString()
    parsed = spec.SetValueRequest.FromString(data)
    assert parsed.value.path == "ctrl-1/address"
    assert parsed.value.value == "tcp://x:1"


def test_map_volume_oneof():
    req = spec.MapVolumeRequest(volume_id="vol-1", malloc=spec.MallocParams())
    assert req.WhichOneof("params") == "malloc"
    req2 = spec.MapVolumeRequest.FromString(req.SerializeToString())
    assert req2.WhichOneof("params") == "malloc"
    req2.ceph.CopyFrom(
        spec.CephParams(user_id="admin", secret="s", monitors="m:1", pool="p", image="i")
    )
    assert req2.WhichOneof("params") == "ceph"


def test_pci_address_wire_numbers():
    # Field numbers are the wire contract; check via serialized tags.
    addr = spec.PCIAddress(domain=1, bus=2, device=3, function=4)
    data = addr.SerializeToString()
    # varint fields 1..4 -> tags 0x08, 0x10, 0x18, 0x20
    assert data == bytes([0x08, 1, 0x10, 2, 0x18, 3, 0x20, 4])


def test_provision_request_wire():
    req = spec.ProvisionMallocBDevRequest(bdev_name="b", size=512)
    data = req.SerializeToString()
    # field 1 (len-delimited) tag 0x0A, field 2 (varint) tag 0x10
    assert data == bytes([0x0A, 1, ord("b"), 0x10, 0x80, 0x04])


def _extract_proto_from_spec_md():
    with open(SPEC_MD) as f:
        text = f.read()
    blocks = re.findall(r"```protobuf\n(.*?)```", text, re.DOTALL)
    return "\n".join(blocks)


def test_spec_md_drift():
    """Every message field and service method in docs/spec.md exists in
    the runtime schema with the same number, and vice versa."""
    proto = _extract_proto_from_spec_md()
    # parse message blocks
    md_fields = {}  # message -> {field_name: number}
    for m in re.finditer(r"message (\w+) \{(.*?)\n\}", proto, re.DOTALL):
        name, body = m.group(1), m.group(2)
        fields = {}
        for fm in re.finditer(r"(?:repeated\s+)?[\w.]+\s+(\w+)\s*=\s*(\d+);", body):
            fields[fm.group(1)] = int(fm.group(2))
        md_fields[name] = fields
    from oim_amd.spec import oim_v0

    code_fields = {
        m.name: {f.name: f.number for f in m.fields} for m in oim_v0.MESSAGES
    }
    assert md_fields == code_fields

    md_methods = set()
    for sm in re.finditer(r"service (\w+) \{(.*?)\n\}", proto, re.DOTALL):
        for mm in re.finditer(r"rpc (\w+)\((\w+)\)\s*\n?\s*returns \((\w+)\)", sm.group(2)):
            md_methods.add((sm.group(1),) + mm.groups())
    code_methods = {
        (s.name, name, req, resp)
        for s in oim_v0.SERVICES
        for (name, req, resp) in s.methods
    }
    assert md_methods == code_methods
