"""gRPC stubs and server registration for oim.v0 (no codegen).

Hand-written equivalents of the generated RegistryClient /
RegisterRegistryServer / ControllerClient / RegisterControllerServer
(reference oim.pb.go:561-744), built on grpcio's generic handler API.
"""

from __future__ import annotations

import grpc

from . import oim_v0 as pb

# Metadata key used by the registry's transparent proxy to select the
# target controller (reference spec.md:65-73).
CONTROLLER_ID_KEY = "controllerid"

REGISTRY_SERVICE = "oim.v0.Registry"
CONTROLLER_SERVICE = "oim.v0.Controller"


class RegistryStub:
    def __init__(self, channel: grpc.Channel):
        self.SetValue = channel.unary_unary(
            f"/{REGISTRY_SERVICE}/SetValue",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=pb.SetValueReply.FromString,
        )
        self.GetValues = channel.unary_unary(
            f"/{REGISTRY_SERVICE}/GetValues",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=pb.GetValuesReply.FromString,
        )


class ControllerStub:
    def __init__(self, channel: grpc.Channel):
        def method(name, reply_class):
            return channel.unary_unary(
                f"/{CONTROLLER_SERVICE}/{name}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=reply_class.FromString,
            )

        self.MapVolume = method("MapVolume", pb.MapVolumeReply)
        self.UnmapVolume = method("UnmapVolume", pb.UnmapVolumeReply)
        self.ProvisionMallocBDev = method(
            "ProvisionMallocBDev", pb.ProvisionMallocBDevReply
        )
        self.CheckMallocBDev = method("CheckMallocBDev", pb.CheckMallocBDevReply)
        self.CloneMallocBDev = method("CloneMallocBDev", pb.CloneMallocBDevReply)
        self.ResizeMallocBDev = method("ResizeMallocBDev", pb.ResizeMallocBDevReply)
        self.ListMallocBDevs = method("ListMallocBDevs", pb.ListMallocBDevsReply)
        self.GetIOStats = method("GetIOStats", pb.GetIOStatsReply)


class RegistryServicer:
    """Interface for the Registry service; subclass and override."""

    def SetValue(self, request: "pb.SetValueRequest", context) -> "pb.SetValueReply":
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "SetValue not implemented")

    def GetValues(self, request: "pb.GetValuesRequest", context) -> "pb.GetValuesReply":
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "GetValues not implemented")


class ControllerServicer:
    """Interface for the Controller service; subclass and override."""

    def MapVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "MapVolume not implemented")

    def UnmapVolume(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "UnmapVolume not implemented")

    def ProvisionMallocBDev(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "ProvisionMallocBDev not implemented")

    def CheckMallocBDev(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "CheckMallocBDev not implemented")

    def CloneMallocBDev(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "CloneMallocBDev not implemented")

    def ResizeMallocBDev(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "ResizeMallocBDev not implemented")

    def ListMallocBDevs(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "ListMallocBDevs not implemented")

    def GetIOStats(self, request, context):
        context.abort(grpc.StatusCode.UNIMPLEMENTED, "GetIOStats not implemented")


def _unary(fn, request_class):
    return grpc.unary_unary_rpc_method_handler(
        fn,
        request_deserializer=request_class.FromString,
        response_serializer=lambda m: m.SerializeToString(),
    )


def add_registry_to_server(servicer: RegistryServicer, server: grpc.Server) -> None:
    handlers = {
        "SetValue": _unary(servicer.SetValue, pb.SetValueRequest),
        "GetValues": _unary(servicer.GetValues, pb.GetValuesRequest),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(REGISTRY_SERVICE, handlers),)
    )


def add_controller_to_server(servicer: ControllerServicer, server: grpc.Server) -> None:
    handlers = {
        "MapVolume": _unary(servicer.MapVolume, pb.MapVolumeRequest),
        "UnmapVolume": _unary(servicer.UnmapVolume, pb.UnmapVolumeRequest),
        "ProvisionMallocBDev": _unary(
            servicer.ProvisionMallocBDev, pb.ProvisionMallocBDevRequest
        ),
        "CheckMallocBDev": _unary(servicer.CheckMallocBDev, pb.CheckMallocBDevRequest),
        "CloneMallocBDev": _unary(servicer.CloneMallocBDev, pb.CloneMallocBDevRequest),
        "ResizeMallocBDev": _unary(servicer.ResizeMallocBDev, pb.ResizeMallocBDevRequest),
        "ListMallocBDevs": _unary(servicer.ListMallocBDevs, pb.ListMallocBDevsRequest),
        "GetIOStats": _unary(servicer.GetIOStats, pb.GetIOStatsRequest),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(CONTROLLER_SERVICE, handlers),)
    )
