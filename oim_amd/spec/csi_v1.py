"""CSI v1 schema subset (Identity/Controller/Node) built at runtime.

Field names/numbers follow the Container Storage Interface spec v1.x
(the wire contract with kubelet and the external-provisioner/attacher
sidecars; cross-checked against the reference's vendored bindings,
vendor/github.com/container-storage-interface/spec/lib/go/csi).

Only the messages this driver implements are declared; the remaining
CSI methods are registered as raw Unimplemented handlers (see rpc_csi).
Enum-typed fields are declared int32 — identical varint wire format —
with the enum values exposed as module constants.
"""

from __future__ import annotations

from ._build import Field, Message, Service, build_file

PACKAGE = "csi.v1"

# --- enum constants (wire values from the CSI spec) -------------------------

# PluginCapability.Service.Type
PLUGIN_CAPABILITY_UNKNOWN = 0
PLUGIN_CAPABILITY_CONTROLLER_SERVICE = 1
PLUGIN_CAPABILITY_ACCESSIBILITY_CONSTRAINTS = 2

# VolumeCapability.AccessMode.Mode
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
CTRL_CAP_CLONE_VOLUME = 7
CTRL_CAP_EXPAND_VOLUME = 9

# NodeServiceCapability.RPC.Type
NODE_CAP_UNKNOWN = 0
NODE_CAP_STAGE_UNSTAGE_VOLUME = 1
NODE_CAP_GET_VOLUME_STATS = 2
NODE_CAP_EXPAND_VOLUME = 3

# PluginCapability.VolumeExpansion.Type
EXPANSION_ONLINE = 1
EXPANSION_OFFLINE = 2

# VolumeUsage.Unit
USAGE_UNIT_BYTES = 1
USAGE_UNIT_INODES = 2

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
        [
            Field("service", 1, "PluginCapability.Service", oneof="type"),
            Field("volume_expansion", 2, "PluginCapability.VolumeExpansion",
                  oneof="type"),
        ],
    ),
    Message("ProbeRequest", []),
    Message(
        "ProbeResponse",
        [Field("ready", 1, ".google.protobuf.BoolValue")],
    ),
    # Controller
    Message(
        "CreateVolumeRequest",
        [
            Field("name", 1, "string"),
            Field("capacity_range", 2, "CapacityRange"),
            Field("volume_capabilities", 3, "VolumeCapability", repeated=True),
            Field("volume_content_source", 6, "VolumeContentSource"),
        ],
        map_fields=[("parameters", 4, "string", "string"),
                    ("secrets", 5, "string", "string")],
    ),
    Message("CreateVolumeResponse", [Field("volume", 1, "Volume")]),
    Message(
        "Volume",
        [
            Field("capacity_bytes", 1, "int64"),
            Field("volume_id", 2, "string"),
            Field("content_source", 4, "VolumeContentSource"),
        ],
        map_fields=[("volume_context", 3, "string", "string")],
    ),
    Message(
        "VolumeContentSource",
        [
            Field("snapshot", 1, "VolumeContentSource.SnapshotSource",
                  oneof="type"),
            Field("volume", 2, "VolumeContentSource.VolumeSource",
                  oneof="type"),
        ],
    ),
    # Snapshots (CSI v1 csi.proto field numbers; backed by hipstored's
    # bdev_clone — HBM-rate device-side copies)
    Message(
        "CreateSnapshotRequest",
        [
            Field("source_volume_id", 1, "string"),
            Field("name", 2, "string"),
        ],
        map_fields=[("secrets", 3, "string", "string"),
                    ("parameters", 4, "string", "string")],
    ),
    Message("CreateSnapshotResponse", [Field("snapshot", 1, "Snapshot")]),
    Message(
        "Snapshot",
        [
            Field("size_bytes", 1, "int64"),
            Field("snapshot_id", 2, "string"),
            Field("source_volume_id", 3, "string"),
            Field("creation_time", 4, ".google.protobuf.Timestamp"),
            Field("ready_to_use", 5, "bool"),
        ],
    ),
    Message(
        "DeleteSnapshotRequest",
        [Field("snapshot_id", 1, "string")],
        map_fields=[("secrets", 2, "string", "string")],
    ),
    Message("DeleteSnapshotResponse", []),
    Message(
        "ListSnapshotsRequest",
        [
            Field("max_entries", 1, "int32"),
            Field("starting_token", 2, "string"),
            Field("source_volume_id", 3, "string"),
            Field("snapshot_id", 4, "string"),
        ],
    ),
    Message(
        "ListSnapshotsResponse",
        [
            Field("entries", 1, "ListSnapshotsResponse.Entry", repeated=True),
            Field("next_token", 2, "string"),
        ],
    ),
    Message(
        "CapacityRange",
        [Field("required_bytes", 1, "int64"), Field("limit_bytes", 2, "int64")],
    ),
    Message(
        "VolumeCapability",
        [
            Field("block", 1, "VolumeCapability.BlockVolume", oneof="access_type"),
            Field("mount", 2, "VolumeCapability.MountVolume", oneof="access_type"),
            Field("access_mode", 3, "VolumeCapability.AccessMode"),
        ],
    ),
    Message(
        "DeleteVolumeRequest",
        [Field("volume_id", 1, "string")],
        map_fields=[("secrets", 2, "string", "string")],
    ),
    Message("DeleteVolumeResponse", []),
    Message(
        "ValidateVolumeCapabilitiesRequest",
        [
            Field("volume_id", 1, "string"),
            Field("volume_capabilities", 3, "VolumeCapability", repeated=True),
        ],
        map_fields=[("volume_context", 2, "string", "string"),
                    ("parameters", 4, "string", "string"),
                    ("secrets", 5, "string", "string")],
    ),
    Message(
        "ValidateVolumeCapabilitiesResponse",
        [
            Field("confirmed", 1, "ValidateVolumeCapabilitiesResponse.Confirmed"),
            Field("message", 2, "string"),
        ],
    ),
    Message(
        "GetCapacityRequest",
        [Field("volume_capabilities", 1, "VolumeCapability", repeated=True)],
        map_fields=[("parameters", 2, "string", "string")],
    ),
    Message(
        "GetCapacityResponse",
        [Field("available_capacity", 1, "int64")],
    ),
    Message(
        "ListVolumesRequest",
        [Field("max_entries", 1, "int32"),
         Field("starting_token", 2, "string")],
    ),
    Message(
        "ListVolumesResponse",
        [
            Field("entries", 1, "ListVolumesResponse.Entry", repeated=True),
            Field("next_token", 2, "string"),
        ],
    ),
    Message(
        "ControllerExpandVolumeRequest",
        [
            Field("volume_id", 1, "string"),
            Field("capacity_range", 2, "CapacityRange"),
            Field("volume_capability", 4, "VolumeCapability"),
        ],
        map_fields=[("secrets", 3, "string", "string")],
    ),
    Message(
        "ControllerExpandVolumeResponse",
        [
            Field("capacity_bytes", 1, "int64"),
            Field("node_expansion_required", 2, "bool"),
        ],
    ),
    Message(
        "NodeExpandVolumeRequest",
        [
            Field("volume_id", 1, "string"),
            Field("volume_path", 2, "string"),
            Field("capacity_range", 3, "CapacityRange"),
            Field("staging_target_path", 4, "string"),
            Field("volume_capability", 5, "VolumeCapability"),
        ],
    ),
    Message(
        "NodeExpandVolumeResponse",
        [Field("capacity_bytes", 1, "int64")],
    ),
    Message("ControllerGetCapabilitiesRequest", []),
    Message(
        "ControllerGetCapabilitiesResponse",
        [Field("capabilities", 1, "ControllerServiceCapability", repeated=True)],
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
        map_fields=[("publish_context", 2, "string", "string"),
                    ("secrets", 5, "string", "string"),
                    ("volume_context", 6, "string", "string")],
    ),
    Message("NodeStageVolumeResponse", []),
    Message(
        "NodeUnstageVolumeRequest",
        [
            Field("volume_id", 1, "string"),
            Field("staging_target_path", 2, "string"),
        ],
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
        map_fields=[("publish_context", 2, "string", "string"),
                    ("secrets", 7, "string", "string"),
                    ("volume_context", 8, "string", "string")],
    ),
    Message("NodePublishVolumeResponse", []),
    Message(
        "NodeUnpublishVolumeRequest",
        [
            Field("volume_id", 1, "string"),
            Field("target_path", 2, "string"),
        ],
    ),
    Message("NodeUnpublishVolumeResponse", []),
    Message("NodeGetCapabilitiesRequest", []),
    Message(
        "NodeGetCapabilitiesResponse",
        [Field("capabilities", 1, "NodeServiceCapability", repeated=True)],
    ),
    Message(
        "NodeServiceCapability",
        [Field("rpc", 1, "NodeServiceCapability.RPC", oneof="type")],
    ),
    Message(
        "NodeGetVolumeStatsRequest",
        [
            Field("volume_id", 1, "string"),
            Field("volume_path", 2, "string"),
            Field("staging_target_path", 3, "string"),
        ],
    ),
    Message(
        "NodeGetVolumeStatsResponse",
        [Field("usage", 1, "VolumeUsage", repeated=True)],
    ),
    Message(
        "VolumeUsage",
        [
            Field("available", 1, "int64"),
            Field("total", 2, "int64"),
            Field("used", 3, "int64"),
            Field("unit", 4, "int32"),  # 1=BYTES, 2=INODES
        ],
    ),
    Message("NodeGetInfoRequest", []),
    Message(
        "NodeGetInfoResponse",
        [
            Field("node_id", 1, "string"),
            Field("max_volumes_per_node", 2, "int64"),
        ],
    ),
]

# Nested message types are declared as their own top-level entries with
# dotted names resolved manually below (the DSL has no nesting), so we
# instead declare them as separate proto messages inside the package
# using underscore-free dotted type names. protobuf requires real
# nesting for dotted names, so declare them flat with unique names and
# patch the type references:
_NESTED = {
    "PluginCapability.Service": "PluginCapabilityService",
    "PluginCapability.VolumeExpansion": "PluginCapabilityVolumeExpansion",
    "VolumeContentSource.SnapshotSource": "VolumeContentSourceSnapshotSource",
    "VolumeContentSource.VolumeSource": "VolumeContentSourceVolumeSource",
    "ListSnapshotsResponse.Entry": "ListSnapshotsResponseEntry",
    "ListVolumesResponse.Entry": "ListVolumesResponseEntry",
    "VolumeCapability.BlockVolume": "VolumeCapabilityBlockVolume",
    "VolumeCapability.MountVolume": "VolumeCapabilityMountVolume",
    "VolumeCapability.AccessMode": "VolumeCapabilityAccessMode",
    "ValidateVolumeCapabilitiesResponse.Confirmed":
        "ValidateVolumeCapabilitiesResponseConfirmed",
    "ControllerServiceCapability.RPC": "ControllerServiceCapabilityRPC",
    "NodeServiceCapability.RPC": "NodeServiceCapabilityRPC",
}

# NOTE: kubelet compatibility requires the real nested names on the
# wire. Message NAMES never travel on the wire (only field numbers and
# service/method paths do), so flattened declarations remain
# wire-compatible; only reflection-based tooling would notice.
for message in MESSAGES:
    for field in message.fields:
        if field.type in _NESTED:
            field.type = _NESTED[field.type]

MESSAGES += [
    Message("PluginCapabilityService", [Field("type", 1, "int32")]),
    Message("PluginCapabilityVolumeExpansion", [Field("type", 1, "int32")]),
    Message("VolumeCapabilityBlockVolume", []),
    Message(
        "VolumeCapabilityMountVolume",
        [Field("fs_type", 1, "string"),
         Field("mount_flags", 2, "string", repeated=True)],
    ),
    Message("VolumeCapabilityAccessMode", [Field("mode", 1, "int32")]),
    Message(
        "ValidateVolumeCapabilitiesResponseConfirmed",
        [Field("volume_capabilities", 2, "VolumeCapability", repeated=True)],
        map_fields=[("volume_context", 1, "string", "string"),
                    ("parameters", 3, "string", "string")],
    ),
    Message("ControllerServiceCapabilityRPC", [Field("type", 1, "int32")]),
    Message("NodeServiceCapabilityRPC", [Field("type", 1, "int32")]),
    Message("VolumeContentSourceSnapshotSource",
            [Field("snapshot_id", 1, "string")]),
    Message("VolumeContentSourceVolumeSource",
            [Field("volume_id", 1, "string")]),
    Message("ListSnapshotsResponseEntry", [Field("snapshot", 1, "Snapshot")]),
    Message("ListVolumesResponseEntry", [Field("volume", 1, "Volume")]),
]

SERVICES = [
    Service(
        "Identity",
        [
            ("GetPluginInfo", "GetPluginInfoRequest", "GetPluginInfoResponse"),
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
            ("ValidateVolumeCapabilities", "ValidateVolumeCapabilitiesRequest",
             "ValidateVolumeCapabilitiesResponse"),
            ("GetCapacity", "GetCapacityRequest", "GetCapacityResponse"),
            ("ControllerGetCapabilities", "ControllerGetCapabilitiesRequest",
             "ControllerGetCapabilitiesResponse"),
            ("CreateSnapshot", "CreateSnapshotRequest",
             "CreateSnapshotResponse"),
            ("DeleteSnapshot", "DeleteSnapshotRequest",
             "DeleteSnapshotResponse"),
            ("ListSnapshots", "ListSnapshotsRequest",
             "ListSnapshotsResponse"),
            ("ListVolumes", "ListVolumesRequest", "ListVolumesResponse"),
            ("ControllerExpandVolume", "ControllerExpandVolumeRequest",
             "ControllerExpandVolumeResponse"),
        ],
    ),
    Service(
        "Node",
        [
            ("NodeStageVolume", "NodeStageVolumeRequest", "NodeStageVolumeResponse"),
            ("NodeUnstageVolume", "NodeUnstageVolumeRequest",
             "NodeUnstageVolumeResponse"),
            ("NodePublishVolume", "NodePublishVolumeRequest",
             "NodePublishVolumeResponse"),
            ("NodeUnpublishVolume", "NodeUnpublishVolumeRequest",
             "NodeUnpublishVolumeResponse"),
            ("NodeGetVolumeStats", "NodeGetVolumeStatsRequest",
             "NodeGetVolumeStatsResponse"),
            ("NodeGetCapabilities", "NodeGetCapabilitiesRequest",
             "NodeGetCapabilitiesResponse"),
            ("NodeGetInfo", "NodeGetInfoRequest", "NodeGetInfoResponse"),
            ("NodeExpandVolume", "NodeExpandVolumeRequest",
             "NodeExpandVolumeResponse"),
        ],
    ),
]

# CSI methods this driver deliberately leaves Unimplemented (registered
# with raw handlers so clients get UNIMPLEMENTED, not UNKNOWN_SERVICE —
# reference controllerserver.go:92-98,161-187).
UNIMPLEMENTED_CONTROLLER_METHODS = (
    "ControllerPublishVolume",
    "ControllerUnpublishVolume",
)
UNIMPLEMENTED_NODE_METHODS = ()

import google.protobuf.timestamp_pb2  # noqa: E402  (registers timestamp.proto)
import google.protobuf.wrappers_pb2  # noqa: E402  (registers wrappers.proto)

_classes = build_file(
    name="oim_amd/csi.proto",
    package=PACKAGE,
    messages=MESSAGES,
    services=SERVICES,
    dependencies=["google/protobuf/wrappers.proto",
                  "google/protobuf/timestamp.proto"],
)

globals().update(_classes)

__all__ = list(_classes.keys()) + [
    n for n in dir() if n.startswith(("PLUGIN_", "ACCESS_MODE_", "CTRL_CAP_",
                                      "NODE_CAP_", "EXPANSION_",
                                      "UNIMPLEMENTED_"))
]
