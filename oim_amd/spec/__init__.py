"""oim.v0 API: runtime-built protobuf messages + gRPC stubs."""

from .oim_v0 import (  # noqa: F401
    SetValueRequest,
    Value,
    SetValueReply,
    GetValuesRequest,
    GetValuesReply,
    MapVolumeRequest,
    MallocParams,
    CephParams,
    MapVolumeReply,
    PCIAddress,
    SCSIDisk,
    UnmapVolumeRequest,
    UnmapVolumeReply,
    ProvisionMallocBDevRequest,
    ProvisionMallocBDevReply,
    CheckMallocBDevRequest,
    CheckMallocBDevReply,
    CloneMallocBDevRequest,
    CloneMallocBDevReply,
)
from .rpc import (  # noqa: F401
    CONTROLLER_ID_KEY,
    REGISTRY_SERVICE,
    CONTROLLER_SERVICE,
    RegistryStub,
    ControllerStub,
    RegistryServicer,
    ControllerServicer,
    add_registry_to_server,
    add_controller_to_server,
)
