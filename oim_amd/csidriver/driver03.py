"""CSI v0.3 twin personality (reference identityserver0.go /
controllerserver0.go / nodeserver0.go + driver0.go).

The reference keeps a second, generated-bindings copy of the whole
driver so pre-1.0 kubelets (and ceph-csi v0.3 emulation) can talk to
it. Here the twins are thin adapters: each v0 request is translated
onto the v1 request shape and delegated to the SAME servicers (the
business rules live once), with the handful of genuine v0 semantic
differences — `supported`-style ValidateVolumeCapabilities, the
NodeGetId RPC, `attributes`/`*_secrets` field spellings — handled in
the translation layer."""

from __future__ import annotations

import grpc

from ..log import from_context
from ..spec import csi_v0 as csi0
from ..spec import csi_v1 as csi1


def _v1_capability(cap0):
    """csi.v0 VolumeCapability -> csi.v1 (same numbers, new class)."""
    cap1 = csi1.VolumeCapability()
    which = cap0.WhichOneof("access_type")
    if which == "mount":
        cap1.mount.fs_type = cap0.mount.fs_type
        cap1.mount.mount_flags.extend(cap0.mount.mount_flags)
    elif which == "block":
        cap1.block.SetInParent()
    cap1.access_mode.mode = cap0.access_mode.mode
    return cap1


class IdentityServer0:
    def __init__(self, driver_name: str, version: str = "0.3.0"):
        self.driver_name = driver_name
        self.version = version

    def GetPluginInfo(self, request, context):
        return csi0.GetPluginInfoResponse(name=self.driver_name,
                                          vendor_version=self.version)

    def Probe(self, request, context):
        response = csi0.ProbeResponse()
        response.ready.value = True
        return response

    def GetPluginCapabilities(self, request, context):
        response = csi0.GetPluginCapabilitiesResponse()
        cap = response.capabilities.add()
        cap.service.type = csi0.PLUGIN_CAPABILITY_CONTROLLER_SERVICE
        return response


class ControllerServer0:
    """Adapts csi.v0 controller calls onto the v1 ControllerServer."""

    def __init__(self, v1_controller):
        self._v1 = v1_controller

    def CreateVolume(self, request, context):
        req1 = csi1.CreateVolumeRequest(name=request.name)
        req1.capacity_range.required_bytes = \
            request.capacity_range.required_bytes
        req1.capacity_range.limit_bytes = request.capacity_range.limit_bytes
        for cap0 in request.volume_capabilities:
            req1.volume_capabilities.add().CopyFrom(_v1_capability(cap0))
        req1.parameters.update(request.parameters)
        req1.secrets.update(request.controller_create_secrets)
        resp1 = self._v1.CreateVolume(req1, context)
        response = csi0.CreateVolumeResponse()
        response.volume.id = resp1.volume.volume_id
        response.volume.capacity_bytes = resp1.volume.capacity_bytes
        response.volume.attributes.update(resp1.volume.volume_context)
        return response

    def DeleteVolume(self, request, context):
        self._v1.DeleteVolume(
            csi1.DeleteVolumeRequest(volume_id=request.volume_id), context)
        return csi0.DeleteVolumeResponse()

    def ValidateVolumeCapabilities(self, request, context):
        # v0 semantics (controllerserver0.go:102-131): boolean
        # supported/message rather than v1's echoed `confirmed`.
        if not request.volume_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "Volume ID missing in request")
        if not request.volume_capabilities:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "Volume capabilities missing in request")
        req1 = csi1.ValidateVolumeCapabilitiesRequest(
            volume_id=request.volume_id)
        for cap0 in request.volume_capabilities:
            req1.volume_capabilities.add().CopyFrom(_v1_capability(cap0))
        resp1 = self._v1.ValidateVolumeCapabilities(req1, context)
        supported = bool(resp1.confirmed.volume_capabilities)
        return csi0.ValidateVolumeCapabilitiesResponse(
            supported=supported, message=resp1.message)

    def ControllerGetCapabilities(self, request, context):
        response = csi0.ControllerGetCapabilitiesResponse()
        cap = response.capabilities.add()
        cap.rpc.type = csi0.CTRL_CAP_CREATE_DELETE_VOLUME
        return response


class NodeServer0:
    """Adapts csi.v0 node calls onto the v1 NodeServer (same backend,
    same mounter, same keyed mutexes)."""

    def __init__(self, node_id: str, v1_node):
        self.node_id = node_id
        self._v1 = v1_node

    def NodeGetId(self, request, context):
        # v0-only RPC (nodeserver0.go:24-28); dropped in CSI 1.0.
        return csi0.NodeGetIdResponse(node_id=self.node_id)

    def NodeGetInfo(self, request, context):
        return csi0.NodeGetInfoResponse(node_id=self.node_id)

    def NodeGetCapabilities(self, request, context):
        response = csi0.NodeGetCapabilitiesResponse()
        cap = response.capabilities.add()
        cap.rpc.type = csi0.NODE_CAP_STAGE_UNSTAGE_VOLUME
        return response

    def NodeStageVolume(self, request, context):
        req1 = csi1.NodeStageVolumeRequest(
            volume_id=request.volume_id,
            staging_target_path=request.staging_target_path)
        req1.volume_capability.CopyFrom(
            _v1_capability(request.volume_capability))
        # v0 `volume_attributes` / `node_stage_secrets` are v1
        # `volume_context` / `secrets` — the ceph-csi emulation mapper
        # consumes them from there either way (ceph-csi.go:50-157).
        req1.volume_context.update(request.volume_attributes)
        req1.secrets.update(request.node_stage_secrets)
        self._v1.NodeStageVolume(req1, context)
        return csi0.NodeStageVolumeResponse()

    def NodeUnstageVolume(self, request, context):
        self._v1.NodeUnstageVolume(
            csi1.NodeUnstageVolumeRequest(
                volume_id=request.volume_id,
                staging_target_path=request.staging_target_path), context)
        return csi0.NodeUnstageVolumeResponse()

    def NodePublishVolume(self, request, context):
        req1 = csi1.NodePublishVolumeRequest(
            volume_id=request.volume_id,
            staging_target_path=request.staging_target_path,
            target_path=request.target_path,
            readonly=request.readonly)
        req1.volume_capability.CopyFrom(
            _v1_capability(request.volume_capability))
        req1.volume_context.update(request.volume_attributes)
        self._v1.NodePublishVolume(req1, context)
        return csi0.NodePublishVolumeResponse()

    def NodeUnpublishVolume(self, request, context):
        self._v1.NodeUnpublishVolume(
            csi1.NodeUnpublishVolumeRequest(
                volume_id=request.volume_id,
                target_path=request.target_path), context)
        return csi0.NodeUnpublishVolumeResponse()
