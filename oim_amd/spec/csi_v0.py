"""CSI v0.3 schema subset (Identity/Controller/Node) built at runtime.

The reference carries verbatim vendored CSI 0.3 bindings
(reference pkg/spec/csi/v0/csi.pb.go) so its driver can serve pre-1.0
kubelets and the ceph-csi v0.3 emulation. This is the same wire
surface built the way this repo builds all its protos — runtime
descriptors, no codegen. Field names/numbers follow CSI spec v0.3
(container-storage-interface/spec @ v0.3.0); only the messages the
0.3 personality implements are declared, the rest of the service
surface is registered Unimplemented (rpc_csi0).

v0.3 / v1 wire differences that matter here:
  - package csi.v0; Volume is {capacity_bytes=1, id=2, attributes=3}
    (v1: volume_id/volume_context naming, same numbers for capacity).
  - ValidateVolumeCapabilitiesResponse is {supported=1, message=2}
    (v1 replaced `supported` with the `confirmed` message).
  - Node has NodeGetId (dropped in v1); secrets fields are
    per-call maps named *_secrets; volume_context is `attributes` /
    `volume_attributes`; NodeStage/Publish carry `publish_info`.
"""

from __future__ import annotations

from ._build import Field, Message, Service, build_file

PACKAGE = "csi.v0"

# PluginCapability.Service.Type
PLUGIN_CAPABILITY_UNKNOWN = 0
PLUGIN_CAPABILITY_CONTROLLER_SERVICE = 1
PLUGIN_CAPABILITY_ACCESSIBILITY_CONSTRAINTS = 2

# VolumeCapability.AccessMode.Mode (same values as v1)
ACCESS_MODE_UNKNOWN = 0
ACCESS_MODE_SINGLE_NODE_WRITER = 1
ACCESS_MODE_SINGLE_NODE_READER_ONLY = 2
ACCESS_MODE_MULTI_NODE_READER_ONLY = 3
ACCESS_MODE_MULTI_NODE_SINGLE_WRITER = 4
ACCESS_MODE_MULTI_NODE_MULTI_WRITER = 5

# ControllerServiceCapability.RPC.Type
CTRL_CAP_UNKNOWN = 0
CTRL_CAP_CREATE_DELETE_VOLUME = 1
CTRL_CAP_PUBLISH_UNPUBLISH_VOLUME = 2
CTRL_CAP_LIST_VOLUMES = 3
CTRL_CAP_GET_CAPACITY = 4
CTRL_CAP_CREATE_DELETE_SNAPSHOT = 5
CTRL_CAP_LIST_SNAPSHOTS = 6

# NodeServiceCapability.RPC.Type
NODE_CAP_UNKNOWN = 0
NODE_CAP_STAGE_UNSTAGE_VOLUME = 1

MESSAGES = [
    # Identity
    Message("GetPluginInfoRequest", []),
    Message(
        "GetPluginInfoResponse",
        [Field("name", 1, "string"), Field("vendor_version", 2, "string")],
        map_fields=[("manifest", 3, "string", "string")],
    ),
    Message("GetPluginCapabilitiesRequest", []),
    Message(
        "GetPluginCapabilitiesResponse",
        [Field("capabilities", 1, "PluginCapability", repeated=True)],
    ),
    Message(
        "PluginCapability",
        [Field("service", 1, "PluginCapability.Service", oneof="type")],
    ),
    Message("ProbeRequest", []),
    Message("ProbeResponse", [Field("ready", 1, ".google.protobuf.BoolValue")]),
    # Controller
    Message(
        "CreateVolumeRequest",
        [
            Field("name", 1, "string"),
            Field("capacity_range", 2, "CapacityRange"),
            Field("volume_capabilities", 3, "VolumeCapability",
                  repeated=True),
        ],
        map_fields=[("parameters", 4, "string", "string"),
                    ("controller_create_secrets", 5, "string", "string")],
    ),
    Message("CreateVolumeResponse", [Field("volume", 1, "Volume")]),
    Message(
        "Volume",
        [Field("capacity_bytes", 1, "int64"), Field("id", 2, "string")],
        map_fields=[("attributes", 3, "string", "string")],
    ),
    Message(
        "CapacityRange",
        [Field("required_bytes", 1, "int64"),
         Field("limit_bytes", 2, "int64")],
    ),
    Message(
        "VolumeCapability",
        [
            Field("block", 1, "VolumeCapability.BlockVolume",
                  oneof="access_type"),
            Field("mount", 2, "VolumeCapability.MountVolume",
                  oneof="access_type"),
            Field("access_mode", 3, "VolumeCapability.AccessMode"),
        ],
    ),
    Message(
        "DeleteVolumeRequest",
        [Field("volume_id", 1, "string")],
        map_fields=[("controller_delete_secrets", 2, "string", "string")],
    ),
    Message("DeleteVolumeResponse", []),
    Message(
        "ValidateVolumeCapabilitiesRequest",
        [
            Field("volume_id", 1, "string"),
            Field("volume_capabilities", 2, "VolumeCapability",
                  repeated=True),
        ],
        map_fields=[("volume_attributes", 3, "string", "string")],
    ),
    Message(
        "ValidateVolumeCapabilitiesResponse",
        [Field("supported", 1, "bool"), Field("message", 2, "string")],
    ),
    Message("ControllerGetCapabilitiesRequest", []),
    Message(
        "ControllerGetCapabilitiesResponse",
        [Field("capabilities", 1, "ControllerServiceCapability",
               repeated=True)],
    ),
    Message(
        "ControllerServiceCapability",
        [Field("rpc", 1, "ControllerServiceCapability.RPC", oneof="type")],
    ),
    # Node
    Message(
        "NodeStageVolumeRequest",
        [
            Field("volume_id", 1, "string"),
            Field("staging_target_path", 3, "string"),
            Field("volume_capability", 4, "VolumeCapability"),
        ],
        map_fields=[("publish_info", 2, "string", "string"),
                    ("node_stage_secrets", 5, "string", "string"),
                    ("volume_attributes", 6, "string", "string")],
    ),
    Message("NodeStageVolumeResponse", []),
    Message(
        "NodeUnstageVolumeRequest",
        [Field("volume_id", 1, "string"),
         Field("staging_target_path", 2, "string")],
    ),
    Message("NodeUnstageVolumeResponse", []),
    Message(
        "NodePublishVolumeRequest",
        [
            Field("volume_id", 1, "string"),
            Field("staging_target_path", 3, "string"),
            Field("target_path", 4, "string"),
            Field("volume_capability", 5, "VolumeCapability"),
            Field("readonly", 6, "bool"),
        ],
        map_fields=[("publish_info", 2, "string", "string"),
                    ("node_publish_secrets", 7, "string", "string"),
                    ("volume_attributes", 8, "string", "string")],
    ),
    Message("NodePublishVolumeResponse", []),
    Message(
        "NodeUnpublishVolumeRequest",
        [Field("volume_id", 1, "string"),
         Field("target_path", 2, "string")],
    ),
    Message("NodeUnpublishVolumeResponse", []),
    Message("NodeGetIdRequest", []),
    Message("NodeGetIdResponse", [Field("node_id", 1, "string")]),
    Message("NodeGetInfoRequest", []),
    Message(
        "NodeGetInfoResponse",
        [Field("node_id", 1, "string"),
         Field("max_volumes_per_node", 2, "int64")],
    ),
    Message("NodeGetCapabilitiesRequest", []),
    Message(
        "NodeGetCapabilitiesResponse",
        [Field("capabilities", 1, "NodeServiceCapability", repeated=True)],
    ),
    Message(
        "NodeServiceCapability",
        [Field("rpc", 1, "NodeServiceCapability.RPC", oneof="type")],
    ),
]

# Flatten nested type names (see csi_v1.py: names never travel on the
# wire, only field numbers and method paths do).
_NESTED = {
    "PluginCapability.Service": "PluginCapabilityService",
    "VolumeCapability.BlockVolume": "VolumeCapabilityBlockVolume",
    "VolumeCapability.MountVolume": "VolumeCapabilityMountVolume",
    "VolumeCapability.AccessMode": "VolumeCapabilityAccessMode",
    "ControllerServiceCapability.RPC": "ControllerServiceCapabilityRPC",
    "NodeServiceCapability.RPC": "NodeServiceCapabilityRPC",
}
for _message in MESSAGES:
    for _field in _message.fields:
        if _field.type in _NESTED:
            _field.type = _NESTED[_field.type]

MESSAGES += [
    Message("PluginCapabilityService", [Field("type", 1, "int32")]),
    Message("VolumeCapabilityBlockVolume", []),
    Message(
        "VolumeCapabilityMountVolume",
        [Field("fs_type", 1, "string"),
         Field("mount_flags", 2, "string", repeated=True)],
    ),
    Message("VolumeCapabilityAccessMode", [Field("mode", 1, "int32")]),
    Message("ControllerServiceCapabilityRPC", [Field("type", 1, "int32")]),
    Message("NodeServiceCapabilityRPC", [Field("type", 1, "int32")]),
]

SERVICES = [
    Service(
        "Identity",
        [
            ("GetPluginInfo", "GetPluginInfoRequest",
             "GetPluginInfoResponse"),
            ("GetPluginCapabilities", "GetPluginCapabilitiesRequest",
             "GetPluginCapabilitiesResponse"),
            ("Probe", "ProbeRequest", "ProbeResponse"),
        ],
    ),
    Service(
        "Controller",
        [
            ("CreateVolume", "CreateVolumeRequest", "CreateVolumeResponse"),
            ("DeleteVolume", "DeleteVolumeRequest", "DeleteVolumeResponse"),
            ("ValidateVolumeCapabilities",
             "ValidateVolumeCapabilitiesRequest",
             "ValidateVolumeCapabilitiesResponse"),
            ("ControllerGetCapabilities",
             "ControllerGetCapabilitiesRequest",
             "ControllerGetCapabilitiesResponse"),
        ],
    ),
    Service(
        "Node",
        [
            ("NodeStageVolume", "NodeStageVolumeRequest",
             "NodeStageVolumeResponse"),
            ("NodeUnstageVolume", "NodeUnstageVolumeRequest",
             "NodeUnstageVolumeResponse"),
            ("NodePublishVolume", "NodePublishVolumeRequest",
             "NodePublishVolumeResponse"),
            ("NodeUnpublishVolume", "NodeUnpublishVolumeRequest",
             "NodeUnpublishVolumeResponse"),
            ("NodeGetId", "NodeGetIdRequest", "NodeGetIdResponse"),
            ("NodeGetInfo", "NodeGetInfoRequest", "NodeGetInfoResponse"),
            ("NodeGetCapabilities", "NodeGetCapabilitiesRequest",
             "NodeGetCapabilitiesResponse"),
        ],
    ),
]

# v0.3 methods the 0.3 personality leaves Unimplemented, matching the
# reference twins (controllerserver0.go:93-99,133-160).
UNIMPLEMENTED_CONTROLLER_METHODS = (
    "ControllerPublishVolume",
    "ControllerUnpublishVolume",
    "ListVolumes",
    "GetCapacity",
    "CreateSnapshot",
    "DeleteSnapshot",
    "ListSnapshots",
)

import google.protobuf.wrappers_pb2  # noqa: E402  (registers wrappers.proto)

_classes = build_file(
    name="oim_amd/csi_v0.proto",
    package=PACKAGE,
    messages=MESSAGES,
    services=SERVICES,
    dependencies=["google/protobuf/wrappers.proto"],
)

globals().update(_classes)

__all__ = list(_classes.keys()) + [
    n for n in dir() if n.startswith(("PLUGIN_", "ACCESS_MODE_",
                                      "CTRL_CAP_", "NODE_CAP_",
                                      "UNIMPLEMENTED_"))
]
