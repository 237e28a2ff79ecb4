"""gRPC stubs/handlers for the CSI v1 services (no codegen)."""

from __future__ import annotations

import grpc

from . import csi_v1 as csi

IDENTITY_SERVICE = "csi.v1.Identity"
CONTROLLER_SERVICE = "csi.v1.Controller"
NODE_SERVICE = "csi.v1.Node"


def _unary(fn, request_class):
    return grpc.unary_unary_rpc_method_handler(
        fn,
        request_deserializer=request_class.FromString,
        response_serializer=lambda m: m.SerializeToString(),
    )


def _unimplemented_handler(method_name):
    def handler(request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED,
                      f"{method_name} is not implemented")

    return grpc.unary_unary_rpc_method_handler(
        handler,
        request_deserializer=lambda b: b,
        response_serializer=lambda b: b,
    )


class CSIIdentityServicer:
    def GetPluginInfo(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "GetPluginInfo")

    def GetPluginCapabilities(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "GetPluginCapabilities")

    def Probe(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "Probe")


class CSIControllerServicer:
    def CreateVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "CreateVolume")

    def DeleteVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "DeleteVolume")

    def ValidateVolumeCapabilities(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "ValidateVolumeCapabilities")

    def GetCapacity(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "GetCapacity")

    def ControllerGetCapabilities(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "ControllerGetCapabilities")

    def CreateSnapshot(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "CreateSnapshot")

    def DeleteSnapshot(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "DeleteSnapshot")

    def ListSnapshots(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "ListSnapshots")

    def ListVolumes(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "ListVolumes")

    def ControllerExpandVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "ControllerExpandVolume")


class CSINodeServicer:
    def NodeGetVolumeStats(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "NodeGetVolumeStats")

    def NodeStageVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "NodeStageVolume")

    def NodeUnstageVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "NodeUnstageVolume")

    def NodePublishVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "NodePublishVolume")

    def NodeUnpublishVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "NodeUnpublishVolume")

    def NodeGetCapabilities(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "NodeGetCapabilities")

    def NodeGetInfo(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "NodeGetInfo")

    def NodeExpandVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "NodeExpandVolume")


def add_csi_identity_to_server(servicer, server):
    handlers = {
        "GetPluginInfo": _unary(servicer.GetPluginInfo, csi.GetPluginInfoRequest),
        "GetPluginCapabilities": _unary(
            servicer.GetPluginCapabilities, csi.GetPluginCapabilitiesRequest),
        "Probe": _unary(servicer.Probe, csi.ProbeRequest),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(IDENTITY_SERVICE, handlers),))


def add_csi_controller_to_server(servicer, server):
    handlers = {
        "CreateVolume": _unary(servicer.CreateVolume, csi.CreateVolumeRequest),
        "DeleteVolume": _unary(servicer.DeleteVolume, csi.DeleteVolumeRequest),
        "ValidateVolumeCapabilities": _unary(
            servicer.ValidateVolumeCapabilities,
            csi.ValidateVolumeCapabilitiesRequest),
        "GetCapacity": _unary(servicer.GetCapacity, csi.GetCapacityRequest),
        "ControllerGetCapabilities": _unary(
            servicer.ControllerGetCapabilities,
            csi.ControllerGetCapabilitiesRequest),
        "CreateSnapshot": _unary(servicer.CreateSnapshot,
                                 csi.CreateSnapshotRequest),
        "DeleteSnapshot": _unary(servicer.DeleteSnapshot,
                                 csi.DeleteSnapshotRequest),
        "ListSnapshots": _unary(servicer.ListSnapshots,
                                csi.ListSnapshotsRequest),
        "ListVolumes": _unary(servicer.ListVolumes, csi.ListVolumesRequest),
        "ControllerExpandVolume": _unary(servicer.ControllerExpandVolume,
                                         csi.ControllerExpandVolumeRequest),
    }
    for name in csi.UNIMPLEMENTED_CONTROLLER_METHODS:
        handlers[name] = _unimplemented_handler(name)
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(CONTROLLER_SERVICE, handlers),))


def add_csi_node_to_server(servicer, server):
    handlers = {
        "NodeStageVolume": _unary(servicer.NodeStageVolume,
                                  csi.NodeStageVolumeRequest),
        "NodeUnstageVolume": _unary(servicer.NodeUnstageVolume,
                                    csi.NodeUnstageVolumeRequest),
        "NodePublishVolume": _unary(servicer.NodePublishVolume,
                                    csi.NodePublishVolumeRequest),
        "NodeUnpublishVolume": _unary(servicer.NodeUnpublishVolume,
                                      csi.NodeUnpublishVolumeRequest),
        "NodeGetVolumeStats": _unary(servicer.NodeGetVolumeStats,
                                     csi.NodeGetVolumeStatsRequest),
        "NodeGetCapabilities": _unary(servicer.NodeGetCapabilities,
                                      csi.NodeGetCapabilitiesRequest),
        "NodeGetInfo": _unary(servicer.NodeGetInfo, csi.NodeGetInfoRequest),
        "NodeExpandVolume": _unary(servicer.NodeExpandVolume,
                                   csi.NodeExpandVolumeRequest),
    }
    for name in csi.UNIMPLEMENTED_NODE_METHODS:
        handlers[name] = _unimplemented_handler(name)
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(NODE_SERVICE, handlers),))


class CSIIdentityStub:
    def __init__(self, channel):
        def method(name, reply_class):
            return channel.unary_unary(
                f"/{IDENTITY_SERVICE}/{name}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=reply_class.FromString)

        self.GetPluginInfo = method("GetPluginInfo", csi.GetPluginInfoResponse)
        self.GetPluginCapabilities = method(
            "GetPluginCapabilities", csi.GetPluginCapabilitiesResponse)
        self.Probe = method("Probe", csi.ProbeResponse)


class CSIControllerStub:
    def __init__(self, channel):
        def method(name, reply_class):
            return channel.unary_unary(
                f"/{CONTROLLER_SERVICE}/{name}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=reply_class.FromString)

        self.CreateVolume = method("CreateVolume", csi.CreateVolumeResponse)
        self.DeleteVolume = method("DeleteVolume", csi.DeleteVolumeResponse)
        self.ValidateVolumeCapabilities = method(
            "ValidateVolumeCapabilities", csi.ValidateVolumeCapabilitiesResponse)
        self.GetCapacity = method("GetCapacity", csi.GetCapacityResponse)
        self.ControllerGetCapabilities = method(
            "ControllerGetCapabilities", csi.ControllerGetCapabilitiesResponse)
        self.CreateSnapshot = method("CreateSnapshot",
                                     csi.CreateSnapshotResponse)
        self.DeleteSnapshot = method("DeleteSnapshot",
                                     csi.DeleteSnapshotResponse)
        self.ListSnapshots = method("ListSnapshots", csi.ListSnapshotsResponse)
        self.ListVolumes = method("ListVolumes", csi.ListVolumesResponse)
        self.ControllerExpandVolume = method(
            "ControllerExpandVolume", csi.ControllerExpandVolumeResponse)


class CSINodeStub:
    def __init__(self, channel):
        def method(name, reply_class):
            return channel.unary_unary(
                f"/{NODE_SERVICE}/{name}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=reply_class.FromString)

        self.NodeStageVolume = method("NodeStageVolume",
                                      csi.NodeStageVolumeResponse)
        self.NodeUnstageVolume = method("NodeUnstageVolume",
                                        csi.NodeUnstageVolumeResponse)
        self.NodePublishVolume = method("NodePublishVolume",
                                        csi.NodePublishVolumeResponse)
        self.NodeUnpublishVolume = method("NodeUnpublishVolume",
                                          csi.NodeUnpublishVolumeResponse)
        self.NodeGetVolumeStats = method("NodeGetVolumeStats",
                                         csi.NodeGetVolumeStatsResponse)
        self.NodeGetCapabilities = method("NodeGetCapabilities",
                                          csi.NodeGetCapabilitiesResponse)
        self.NodeGetInfo = method("NodeGetInfo", csi.NodeGetInfoResponse)
        self.NodeExpandVolume = method("NodeExpandVolume",
                                       csi.NodeExpandVolumeResponse)
