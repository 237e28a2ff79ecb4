"""The oim.v0 API schema (authored from docs/spec.md).

Wire-compatible with the reference's pkg/spec/oim/v0/oim.pb.go: same
package, message names, field names/numbers and service methods
(reference spec.md:25-196).  docs/spec.md in this repo is the literate
source; tests/test_spec.py checks this module against it so the two
cannot drift (the reference's Makefile:85-103 "spec.md -> oim.proto"
extraction, done as a consistency test instead of codegen).
"""

from __future__ import annotations

from ._build import Field, Message, Service, build_file

PACKAGE = "oim.v0"

MESSAGES = [
    Message("SetValueRequest", [Field("value", 1, "Value")]),
    Message("Value", [Field("path", 1, "string"), Field("value", 2, "string")]),
    Message("SetValueReply", []),
    Message("GetValuesRequest", [Field("path", 1, "string")]),
    Message("GetValuesReply", [Field("values", 1, "Value", repeated=True)]),
    Message(
        "MapVolumeRequest",
        [
            Field("volume_id", 1, "string"),
            Field("malloc", 2, "MallocParams", oneof="params"),
            Field("ceph", 3, "CephParams", oneof="params"),
        ],
    ),
    Message("MallocParams", []),
    Message(
        "CephParams",
        [
            Field("user_id", 1, "string"),
            Field("secret", 2, "string"),
            Field("monitors", 3, "string"),
            Field("pool", 4, "string"),
            Field("image", 5, "string"),
        ],
    ),
    Message(
        "MapVolumeReply",
        [
            Field("pci_address", 1, "PCIAddress"),
            Field("scsi_disk", 2, "SCSIDisk"),
        ],
    ),
    Message(
        "PCIAddress",
        [
            Field("domain", 1, "uint32"),
            Field("bus", 2, "uint32"),
            Field("device", 3, "uint32"),
            Field("function", 4, "uint32"),
        ],
    ),
    Message("SCSIDisk", [Field("target", 1, "uint32"), Field("lun", 2, "uint32")]),
    Message("UnmapVolumeRequest", [Field("volume_id", 1, "string")]),
    Message("UnmapVolumeReply", []),
    Message(
        "ProvisionMallocBDevRequest",
        [Field("bdev_name", 1, "string"), Field("size", 2, "int64")],
    ),
    Message("ProvisionMallocBDevReply", []),
    Message("CheckMallocBDevRequest", [Field("bdev_name", 1, "string")]),
    Message("CheckMallocBDevReply", []),
    Message("CloneMallocBDevRequest",
            [Field("source", 1, "string"), Field("dest", 2, "string")]),
    Message("CloneMallocBDevReply", []),
    Message("ResizeMallocBDevRequest",
            [Field("bdev_name", 1, "string"), Field("size", 2, "int64")]),
    Message("ResizeMallocBDevReply", []),
    Message("ListMallocBDevsRequest", [Field("prefix", 1, "string")]),
    Message("BDevInfo",
            [Field("name", 1, "string"), Field("size", 2, "int64"),
             Field("block_size", 3, "int64"),
             Field("product_name", 4, "string")]),
    Message("ListMallocBDevsReply",
            [Field("bdevs", 1, "BDevInfo", repeated=True)]),
    Message("GetIOStatsRequest", [Field("bdev_name", 1, "string")]),
    Message("BDevIOStats",
            [Field("name", 1, "string"),
             Field("num_read_ops", 2, "int64"),
             Field("num_write_ops", 3, "int64"),
             Field("num_unmap_ops", 4, "int64"),
             Field("bytes_read", 5, "int64"),
             Field("bytes_written", 6, "int64")]),
    Message("GetIOStatsReply",
            [Field("bdevs", 1, "BDevIOStats", repeated=True)]),
]

SERVICES = [
    Service(
        "Registry",
        [
            ("SetValue", "SetValueRequest", "SetValueReply"),
            ("GetValues", "GetValuesRequest", "GetValuesReply"),
        ],
    ),
    Service(
        "Controller",
        [
            ("MapVolume", "MapVolumeRequest", "MapVolumeReply"),
            ("UnmapVolume", "UnmapVolumeRequest", "UnmapVolumeReply"),
            ("ProvisionMallocBDev", "ProvisionMallocBDevRequest", "ProvisionMallocBDevReply"),
            ("CheckMallocBDev", "CheckMallocBDevRequest", "CheckMallocBDevReply"),
            ("CloneMallocBDev", "CloneMallocBDevRequest", "CloneMallocBDevReply"),
            ("ResizeMallocBDev", "ResizeMallocBDevRequest", "ResizeMallocBDevReply"),
            ("ListMallocBDevs", "ListMallocBDevsRequest", "ListMallocBDevsReply"),
            ("GetIOStats", "GetIOStatsRequest", "GetIOStatsReply"),
        ],
    ),
]

_classes = build_file(
    name="oim_amd/oim.proto", package=PACKAGE, messages=MESSAGES, services=SERVICES
)

SetValueRequest = _classes["SetValueRequest"]
Value = _classes["Value"]
SetValueReply = _classes["SetValueReply"]
GetValuesRequest = _classes["GetValuesRequest"]
GetValuesReply = _classes["GetValuesReply"]
MapVolumeRequest = _classes["MapVolumeRequest"]
MallocParams = _classes["MallocParams"]
CephParams = _classes["CephParams"]
MapVolumeReply = _classes["MapVolumeReply"]
PCIAddress = _classes["PCIAddress"]
SCSIDisk = _classes["SCSIDisk"]
UnmapVolumeRequest = _classes["UnmapVolumeRequest"]
UnmapVolumeReply = _classes["UnmapVolumeReply"]
ProvisionMallocBDevRequest = _classes["ProvisionMallocBDevRequest"]
ProvisionMallocBDevReply = _classes["ProvisionMallocBDevReply"]
CheckMallocBDevRequest = _classes["CheckMallocBDevRequest"]
CheckMallocBDevReply = _classes["CheckMallocBDevReply"]
CloneMallocBDevRequest = _classes["CloneMallocBDevRequest"]
CloneMallocBDevReply = _classes["CloneMallocBDevReply"]
ResizeMallocBDevRequest = _classes["ResizeMallocBDevRequest"]
ResizeMallocBDevReply = _classes["ResizeMallocBDevReply"]
ListMallocBDevsRequest = _classes["ListMallocBDevsRequest"]
BDevInfo = _classes["BDevInfo"]
ListMallocBDevsReply = _classes["ListMallocBDevsReply"]
GetIOStatsRequest = _classes["GetIOStatsRequest"]
BDevIOStats = _classes["BDevIOStats"]
GetIOStatsReply = _classes["GetIOStatsReply"]
