"""gRPC stubs/handlers for the CSI v0.3 services (no codegen).

Counterpart of rpc_csi for the legacy personality (reference
pkg/spec/csi/v0 generated bindings + the oimDriver03 twins)."""

from __future__ import annotations

import grpc

from . import csi_v0 as csi

IDENTITY_SERVICE = "csi.v0.Identity"
CONTROLLER_SERVICE = "csi.v0.Controller"
NODE_SERVICE = "csi.v0.Node"


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


def add_csi0_identity_to_server(servicer, server):
    handlers = {
        "GetPluginInfo": _unary(servicer.GetPluginInfo,
                                csi.GetPluginInfoRequest),
        "GetPluginCapabilities": _unary(servicer.GetPluginCapabilities,
                                        csi.GetPluginCapabilitiesRequest),
        "Probe": _unary(servicer.Probe, csi.ProbeRequest),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(IDENTITY_SERVICE, handlers),))


def add_csi0_controller_to_server(servicer, server):
    handlers = {
        "CreateVolume": _unary(servicer.CreateVolume,
                               csi.CreateVolumeRequest),
        "DeleteVolume": _unary(servicer.DeleteVolume,
                               csi.DeleteVolumeRequest),
        "ValidateVolumeCapabilities": _unary(
            servicer.ValidateVolumeCapabilities,
            csi.ValidateVolumeCapabilitiesRequest),
        "ControllerGetCapabilities": _unary(
            servicer.ControllerGetCapabilities,
            csi.ControllerGetCapabilitiesRequest),
    }
    for method in csi.UNIMPLEMENTED_CONTROLLER_METHODS:
        handlers[method] = _unimplemented_handler(method)
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(CONTROLLER_SERVICE,
                                              handlers),))


def add_csi0_node_to_server(servicer, server):
    handlers = {
        "NodeStageVolume": _unary(servicer.NodeStageVolume,
                                  csi.NodeStageVolumeRequest),
        "NodeUnstageVolume": _unary(servicer.NodeUnstageVolume,
                                    csi.NodeUnstageVolumeRequest),
        "NodePublishVolume": _unary(servicer.NodePublishVolume,
                                    csi.NodePublishVolumeRequest),
        "NodeUnpublishVolume": _unary(servicer.NodeUnpublishVolume,
                                      csi.NodeUnpublishVolumeRequest),
        "NodeGetId": _unary(servicer.NodeGetId, csi.NodeGetIdRequest),
        "NodeGetInfo": _unary(servicer.NodeGetInfo, csi.NodeGetInfoRequest),
        "NodeGetCapabilities": _unary(servicer.NodeGetCapabilities,
                                      csi.NodeGetCapabilitiesRequest),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(NODE_SERVICE, handlers),))


class _Stub:
    def __init__(self, channel, service, methods):
        for method, (request_class, response_class) in methods.items():
            setattr(self, method, channel.unary_unary(
                f"/{service}/{method}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=response_class.FromString,
            ))


class CSI0IdentityStub(_Stub):
    def __init__(self, channel):
        super().__init__(channel, IDENTITY_SERVICE, {
            "GetPluginInfo": (csi.GetPluginInfoRequest,
                              csi.GetPluginInfoResponse),
            "GetPluginCapabilities": (csi.GetPluginCapabilitiesRequest,
                                      csi.GetPluginCapabilitiesResponse),
            "Probe": (csi.ProbeRequest, csi.ProbeResponse),
        })


class CSI0ControllerStub(_Stub):
    def __init__(self, channel):
        super().__init__(channel, CONTROLLER_SERVICE, {
            "CreateVolume": (csi.CreateVolumeRequest,
                             csi.CreateVolumeResponse),
            "DeleteVolume": (csi.DeleteVolumeRequest,
                             csi.DeleteVolumeResponse),
            "ValidateVolumeCapabilities": (
                csi.ValidateVolumeCapabilitiesRequest,
                csi.ValidateVolumeCapabilitiesResponse),
            "ControllerGetCapabilities": (
                csi.ControllerGetCapabilitiesRequest,
                csi.ControllerGetCapabilitiesResponse),
        })


class CSI0NodeStub(_Stub):
    def __init__(self, channel):
        super().__init__(channel, NODE_SERVICE, {
            "NodeStageVolume": (csi.NodeStageVolumeRequest,
                                csi.NodeStageVolumeResponse),
            "NodeUnstageVolume": (csi.NodeUnstageVolumeRequest,
                                  csi.NodeUnstageVolumeResponse),
            "NodePublishVolume": (csi.NodePublishVolumeRequest,
                                  csi.NodePublishVolumeResponse),
            "NodeUnpublishVolume": (csi.NodeUnpublishVolumeRequest,
                                    csi.NodeUnpublishVolumeResponse),
            "NodeGetId": (csi.NodeGetIdRequest, csi.NodeGetIdResponse),
            "NodeGetInfo": (csi.NodeGetInfoRequest, csi.NodeGetInfoResponse),
            "NodeGetCapabilities": (csi.NodeGetCapabilitiesRequest,
                                    csi.NodeGetCapabilitiesResponse),
        })
