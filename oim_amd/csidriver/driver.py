"""The OIM CSI driver: Identity/Controller/Node services over one
endpoint (reference pkg/oim-csi-driver/oim-driver.go, controllerserver.go,
nodeserver.go).

Supported modes (mutually exclusive, oim-driver.go:216-221):
  - local:  --hipstored-socket: drive hipstored directly, NBD device
  - remote: --oim-registry-address + --controller-id: drive a remote
            controller through the registry proxy

CSI version: v1 only. The reference also carried a CSI 0.3 twin for
old ceph-csi compatibility; this rebuild keeps the emulation hook
(--emulate=ceph-csi) on the v1 surface instead (SURVEY.md section 7.3
recommends exactly this simplification).
"""

from __future__ import annotations

import os
import threading
from typing import Dict, Optional

import grpc

from .. import __version__
from ..common.server import NonBlockingGRPCServer
from ..common.tracing import LogServerInterceptor, strip_secrets_formatter
from ..log import from_context
from ..spec import csi_v1 as csi
from ..spec.rpc_csi import (
    CSIControllerServicer,
    CSIIdentityServicer,
    CSINodeServicer,
    add_csi_controller_to_server,
    add_csi_identity_to_server,
    add_csi_node_to_server,
)
from .backend import OIMBackend, VolumeExistsError
from .cephemu import EMULATIONS
from .mount import Mounter


class _KeyedMutex:
    def __init__(self):
        self._guard = threading.Lock()
        self._locks: Dict[str, threading.Lock] = {}

    def get(self, key: str) -> threading.Lock:
        with self._guard:
            return self._locks.setdefault(key, threading.Lock())


class IdentityServer(CSIIdentityServicer):
    def __init__(self, driver_name: str, offline_expansion: bool = False):
        self.driver_name = driver_name
        self.offline_expansion = offline_expansion

    def GetPluginInfo(self, request, context):
        return csi.GetPluginInfoResponse(
            name=self.driver_name, vendor_version=__version__)

    def GetPluginCapabilities(self, request, context):
        response = csi.GetPluginCapabilitiesResponse()
        cap = response.capabilities.add()
        cap.service.type = csi.PLUGIN_CAPABILITY_CONTROLLER_SERVICE
        if self.offline_expansion:
            cap = response.capabilities.add()
            # The backing store moves during resize, so channels must
            # be closed: OFFLINE expansion.
            cap.volume_expansion.type = csi.EXPANSION_OFFLINE
        return response

    def Probe(self, request, context):
        response = csi.ProbeResponse()
        response.ready.value = True
        return response


def _validate_capabilities(caps, context):
    """Reject block volumes and multi-writer modes
    (controllerserver.go:29-47)."""
    for cap in caps:
        if cap.WhichOneof("access_type") == "block":
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "block volumes are not supported")
        if cap.access_mode.mode in (
                csi.ACCESS_MODE_MULTI_NODE_SINGLE_WRITER,
                csi.ACCESS_MODE_MULTI_NODE_MULTI_WRITER):
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "multi-node writers are not supported")


class ControllerServer(CSIControllerServicer):
    def __init__(self, backend: OIMBackend):
        self.backend = backend
        self._name_mutex = _KeyedMutex()

    def CreateVolume(self, request, context):
        if not request.name:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "missing name")
        if not request.volume_capabilities:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume capabilities")
        _validate_capabilities(request.volume_capabilities, context)
        size = request.capacity_range.required_bytes or (1 << 20)
        source = request.volume_content_source
        if source.WhichOneof("type") == "snapshot":
            return self._create_from_snapshot(request, context)
        if source.WhichOneof("type") == "volume":
            return self._create_from_volume(request, context)
        with self._name_mutex.get(request.name):
            try:
                volume_id, volume_context = self.backend.create_volume(
                    request.name, size, dict(request.parameters))
            except VolumeExistsError as exc:
                # Same name, incompatible parameters (CSI spec: 6)
                context.abort(grpc.StatusCode.ALREADY_EXISTS, str(exc))
            except ValueError as exc:
                context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(exc))
            except grpc.RpcError as exc:
                context.abort(exc.code(), exc.details())
        response = csi.CreateVolumeResponse()
        response.volume.volume_id = volume_id
        response.volume.capacity_bytes = size
        for key, value in volume_context.items():
            response.volume.volume_context[key] = value
        return response

    def _create_from_snapshot(self, request, context):
        if not self.backend.supports_snapshots():
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "snapshot restore is not supported in this mode")
        snapshot_id = request.volume_content_source.snapshot.snapshot_id
        if not snapshot_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing snapshot id in content source")
        with self._name_mutex.get(request.name):
            try:
                volume_id, size = self.backend.restore_snapshot(
                    snapshot_id, request.name)
            except LookupError as exc:
                context.abort(grpc.StatusCode.NOT_FOUND, str(exc))
            except RuntimeError as exc:
                context.abort(grpc.StatusCode.INTERNAL, str(exc))
            except grpc.RpcError as exc:
                context.abort(exc.code(), exc.details())
        required = request.capacity_range.required_bytes
        if size and required and required > size:
            context.abort(
                grpc.StatusCode.OUT_OF_RANGE,
                f"snapshot is {size} bytes; cannot satisfy {required}")
        response = csi.CreateVolumeResponse()
        response.volume.volume_id = volume_id
        response.volume.capacity_bytes = size
        response.volume.content_source.snapshot.snapshot_id = snapshot_id
        return response

    def _create_from_volume(self, request, context):
        """CSI volume cloning: CreateVolume with a volume content
        source — a bdev_clone of the source volume."""
        if not self.backend.supports_snapshots():
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "volume cloning is not supported in this mode")
        source_id = request.volume_content_source.volume.volume_id
        if not source_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume id in content source")
        with self._name_mutex.get(request.name):
            try:
                volume_id, size = self.backend.clone_volume(
                    source_id, request.name)
            except LookupError as exc:
                context.abort(grpc.StatusCode.NOT_FOUND, str(exc))
            except RuntimeError as exc:
                context.abort(grpc.StatusCode.INTERNAL, str(exc))
            except grpc.RpcError as exc:
                context.abort(exc.code(), exc.details())
        required = request.capacity_range.required_bytes
        if size and required and required > size:
            context.abort(
                grpc.StatusCode.OUT_OF_RANGE,
                f"source volume is {size} bytes; cannot satisfy {required}")
        response = csi.CreateVolumeResponse()
        response.volume.volume_id = volume_id
        response.volume.capacity_bytes = size
        response.volume.content_source.volume.volume_id = source_id
        return response

    def DeleteVolume(self, request, context):
        if not request.volume_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "missing volume id")
        with self._name_mutex.get(request.volume_id):
            try:
                self.backend.delete_volume(request.volume_id)
            except grpc.RpcError as exc:
                context.abort(exc.code(), exc.details())
        return csi.DeleteVolumeResponse()

    def ValidateVolumeCapabilities(self, request, context):
        if not request.volume_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "missing volume id")
        if not request.volume_capabilities:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume capabilities")
        if not self.backend.check_volume_exists(request.volume_id):
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"volume {request.volume_id} not found")
        response = csi.ValidateVolumeCapabilitiesResponse()
        ok = True
        for cap in request.volume_capabilities:
            if cap.WhichOneof("access_type") == "block":
                ok = False
            if cap.access_mode.mode not in (
                    csi.ACCESS_MODE_SINGLE_NODE_WRITER,
                    csi.ACCESS_MODE_SINGLE_NODE_READER_ONLY,
                    csi.ACCESS_MODE_MULTI_NODE_READER_ONLY):
                ok = False
        if ok:
            for cap in request.volume_capabilities:
                response.confirmed.volume_capabilities.add().CopyFrom(cap)
        else:
            response.message = "unsupported volume capabilities"
        return response

    def GetCapacity(self, request, context):
        capacity = self.backend.get_capacity()
        if capacity is None:
            context.abort(
                grpc.StatusCode.UNIMPLEMENTED,
                "capacity reporting is only available in local mode "
                "(the oim.v0 controller API has no capacity RPC)")
        return csi.GetCapacityResponse(available_capacity=capacity)

    def ControllerGetCapabilities(self, request, context):
        response = csi.ControllerGetCapabilitiesResponse()
        cap = response.capabilities.add()
        cap.rpc.type = csi.CTRL_CAP_CREATE_DELETE_VOLUME
        if self.backend.get_capacity() is not None:
            cap = response.capabilities.add()
            cap.rpc.type = csi.CTRL_CAP_GET_CAPACITY
        if self.backend.supports_snapshots():
            cap = response.capabilities.add()
            cap.rpc.type = csi.CTRL_CAP_CREATE_DELETE_SNAPSHOT
            cap = response.capabilities.add()
            cap.rpc.type = csi.CTRL_CAP_LIST_SNAPSHOTS
            cap = response.capabilities.add()
            cap.rpc.type = csi.CTRL_CAP_CLONE_VOLUME
        if self.backend.list_volumes() is not None:
            cap = response.capabilities.add()
            cap.rpc.type = csi.CTRL_CAP_LIST_VOLUMES
        if self.backend.supports_expansion():
            cap = response.capabilities.add()
            cap.rpc.type = csi.CTRL_CAP_EXPAND_VOLUME
        return response

    def ControllerExpandVolume(self, request, context):
        if not self.backend.supports_expansion():
            context.abort(grpc.StatusCode.UNIMPLEMENTED,
                          "expansion is not supported in this mode")
        if not request.volume_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume id")
        required = request.capacity_range.required_bytes
        if required <= 0:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing capacity range")
        with self._name_mutex.get(request.volume_id):
            try:
                new_size = self.backend.expand_volume(request.volume_id,
                                                      required)
            except LookupError as exc:
                context.abort(grpc.StatusCode.NOT_FOUND, str(exc))
            except RuntimeError as exc:
                # live channels hold the old backing: offline-only
                context.abort(grpc.StatusCode.FAILED_PRECONDITION, str(exc))
            except grpc.RpcError as exc:
                context.abort(exc.code(), exc.details())
        response = csi.ControllerExpandVolumeResponse()
        response.capacity_bytes = new_size
        response.node_expansion_required = True  # fs must grow too
        return response

    def ListVolumes(self, request, context):
        volumes = self.backend.list_volumes()
        if volumes is None:
            context.abort(grpc.StatusCode.UNIMPLEMENTED,
                          "volume listing is not available in this mode")
        if request.max_entries < 0:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "max_entries must be non-negative")
        start = 0
        if request.starting_token:
            try:
                start = int(request.starting_token)
            except ValueError:
                context.abort(grpc.StatusCode.ABORTED,
                              f"bad starting_token "
                              f"{request.starting_token!r}")
            if start < 0 or start > len(volumes):
                context.abort(grpc.StatusCode.ABORTED,
                              "starting_token out of range")
        end = len(volumes)
        if request.max_entries:
            end = min(end, start + request.max_entries)
        response = csi.ListVolumesResponse()
        for volume_id, size in volumes[start:end]:
            entry = response.entries.add()
            entry.volume.volume_id = volume_id
            entry.volume.capacity_bytes = size
        if end < len(volumes):
            response.next_token = str(end)
        return response

    def CreateSnapshot(self, request, context):
        """Snapshot = hipstored bdev_clone (device-side HBM-rate copy);
        instantly ready_to_use (no background upload)."""
        if not self.backend.supports_snapshots():
            context.abort(grpc.StatusCode.UNIMPLEMENTED,
                          "snapshots are not supported in this mode")
        if not request.source_volume_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing source volume id")
        if not request.name:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "missing name")
        with self._name_mutex.get("snap-" + request.name):
            try:
                snapshot_id, size, ctime = self.backend.create_snapshot(
                    request.name, request.source_volume_id)
            except VolumeExistsError as exc:
                context.abort(grpc.StatusCode.ALREADY_EXISTS, str(exc))
            except LookupError as exc:
                context.abort(grpc.StatusCode.NOT_FOUND, str(exc))
            except RuntimeError as exc:
                context.abort(grpc.StatusCode.INTERNAL, str(exc))
            except grpc.RpcError as exc:
                context.abort(exc.code(), exc.details())
        response = csi.CreateSnapshotResponse()
        response.snapshot.snapshot_id = snapshot_id
        response.snapshot.source_volume_id = request.source_volume_id
        response.snapshot.size_bytes = size
        response.snapshot.creation_time.seconds = ctime
        response.snapshot.ready_to_use = True
        return response

    def DeleteSnapshot(self, request, context):
        if not self.backend.supports_snapshots():
            context.abort(grpc.StatusCode.UNIMPLEMENTED,
                          "snapshots are not supported in this mode")
        if not request.snapshot_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing snapshot id")
        with self._name_mutex.get("snap-" + request.snapshot_id):
            try:
                self.backend.delete_snapshot(request.snapshot_id)
            except RuntimeError as exc:
                context.abort(grpc.StatusCode.INTERNAL, str(exc))
            except grpc.RpcError as exc:
                context.abort(exc.code(), exc.details())
        return csi.DeleteSnapshotResponse()

    def ListSnapshots(self, request, context):
        if not self.backend.supports_snapshots():
            context.abort(grpc.StatusCode.UNIMPLEMENTED,
                          "snapshots are not supported in this mode")
        response = csi.ListSnapshotsResponse()
        for snap_id, source, size, ctime in self.backend.list_snapshots():
            if request.snapshot_id and snap_id != request.snapshot_id:
                continue
            if request.source_volume_id and source != request.source_volume_id:
                continue
            entry = response.entries.add()
            entry.snapshot.snapshot_id = snap_id
            entry.snapshot.source_volume_id = source
            entry.snapshot.size_bytes = size
            entry.snapshot.creation_time.seconds = ctime
            entry.snapshot.ready_to_use = True
        return response


class NodeServer(CSINodeServicer):
    def __init__(self, node_id: str, backend: OIMBackend,
                 mounter: Optional[Mounter] = None):
        self.node_id = node_id
        self.backend = backend
        self.mounter = mounter or Mounter()
        self._volume_mutex = _KeyedMutex()

    def NodeStageVolume(self, request, context):
        if not request.volume_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "missing volume id")
        if not request.staging_target_path:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing staging target path")
        if request.volume_capability.WhichOneof("access_type") is None:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume capability")
        if request.volume_capability.WhichOneof("access_type") == "block":
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "block volumes are not supported")
        with self._volume_mutex.get(request.volume_id):
            # Already staged? (nodeserver.go:169-183)
            if self.mounter.is_mount_point(request.staging_target_path):
                return csi.NodeStageVolumeResponse()
            stage_info = {
                "volume_context": dict(request.volume_context),
                "secrets": dict(request.secrets),
                "staging_path": request.staging_target_path,
            }
            try:
                device = self.backend.create_device(request.volume_id,
                                                    stage_info)
            except TimeoutError as exc:
                context.abort(grpc.StatusCode.DEADLINE_EXCEEDED, str(exc))
            except grpc.RpcError as exc:
                context.abort(exc.code(), exc.details())
            except (RuntimeError, ValueError) as exc:
                context.abort(grpc.StatusCode.INTERNAL, str(exc))
            fs_type = request.volume_capability.mount.fs_type or "ext4"
            os.makedirs(request.staging_target_path, exist_ok=True)
            try:
                self.mounter.format_and_mount(
                    device, request.staging_target_path, fs_type,
                    request.volume_capability.mount.mount_flags)
            except Exception as exc:  # noqa: BLE001 - surface as INTERNAL
                context.abort(grpc.StatusCode.INTERNAL,
                              f"format/mount failed: {exc}")
        return csi.NodeStageVolumeResponse()

    def NodeUnstageVolume(self, request, context):
        if not request.volume_id or not request.staging_target_path:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume id or staging path")
        with self._volume_mutex.get(request.volume_id):
            if self.mounter.is_mount_point(request.staging_target_path):
                try:
                    self.mounter.unmount(request.staging_target_path)
                except Exception as exc:  # noqa: BLE001
                    context.abort(grpc.StatusCode.INTERNAL, str(exc))
            try:
                self.backend.delete_device(request.volume_id)
            except grpc.RpcError as exc:
                context.abort(exc.code(), exc.details())
        return csi.NodeUnstageVolumeResponse()

    def NodePublishVolume(self, request, context):
        if (not request.volume_id or not request.staging_target_path
                or not request.target_path):
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume id or paths")
        if request.volume_capability.WhichOneof("access_type") is None:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume capability")
        with self._volume_mutex.get(request.volume_id):
            if self.mounter.is_mount_point(request.target_path):
                return csi.NodePublishVolumeResponse()
            os.makedirs(request.target_path, exist_ok=True)
            try:
                self.mounter.bind_mount(request.staging_target_path,
                                        request.target_path,
                                        readonly=request.readonly)
            except Exception as exc:  # noqa: BLE001
                context.abort(grpc.StatusCode.INTERNAL, str(exc))
        return csi.NodePublishVolumeResponse()

    def NodeUnpublishVolume(self, request, context):
        if not request.volume_id or not request.target_path:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume id or target path")
        with self._volume_mutex.get(request.volume_id):
            if self.mounter.is_mount_point(request.target_path):
                try:
                    self.mounter.unmount(request.target_path)
                except Exception as exc:  # noqa: BLE001
                    context.abort(grpc.StatusCode.INTERNAL, str(exc))
        return csi.NodeUnpublishVolumeResponse()

    def NodeGetVolumeStats(self, request, context):
        path = request.volume_path or request.staging_target_path
        if not request.volume_id or not path:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume id or path")
        if not os.path.exists(path):
            context.abort(grpc.StatusCode.NOT_FOUND, f"no such path: {path}")
        st = os.statvfs(path)
        response = csi.NodeGetVolumeStatsResponse()
        usage = response.usage.add()
        usage.unit = csi.USAGE_UNIT_BYTES
        usage.total = st.f_frsize * st.f_blocks
        usage.available = st.f_frsize * st.f_bavail
        usage.used = st.f_frsize * (st.f_blocks - st.f_bfree)
        inodes = response.usage.add()
        inodes.unit = csi.USAGE_UNIT_INODES
        inodes.total = st.f_files
        inodes.available = st.f_favail
        inodes.used = st.f_files - st.f_ffree
        return response

    def NodeExpandVolume(self, request, context):
        if not request.volume_id or not request.volume_path:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "missing volume id or path")
        fs_type = request.volume_capability.mount.fs_type or "ext4"
        with self._volume_mutex.get(request.volume_id):
            if not os.path.exists(request.volume_path):
                context.abort(grpc.StatusCode.NOT_FOUND,
                              f"no such path: {request.volume_path}")
            try:
                self.mounter.resize_fs(request.volume_path, fs_type)
            except ValueError as exc:
                context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(exc))
            except Exception as exc:  # noqa: BLE001
                context.abort(grpc.StatusCode.INTERNAL,
                              f"filesystem grow failed: {exc}")
        response = csi.NodeExpandVolumeResponse()
        response.capacity_bytes = request.capacity_range.required_bytes
        return response

    def NodeGetCapabilities(self, request, context):
        response = csi.NodeGetCapabilitiesResponse()
        cap = response.capabilities.add()
        cap.rpc.type = csi.NODE_CAP_STAGE_UNSTAGE_VOLUME
        cap = response.capabilities.add()
        cap.rpc.type = csi.NODE_CAP_GET_VOLUME_STATS
        cap = response.capabilities.add()
        cap.rpc.type = csi.NODE_CAP_EXPAND_VOLUME
        return response

    def NodeGetInfo(self, request, context):
        return csi.NodeGetInfoResponse(node_id=self.node_id)


class OIMDriver:
    """Assembles the three services on one endpoint
    (reference Start, oim-driver.go:275-292). `csi_version` selects
    the personality: "1.0" (default) registers the csi.v1 services,
    "0.3" the legacy twins (reference oimDriver03, driver0.go) — same
    servicers underneath, adapted in oim_amd.csidriver.driver03."""

    def __init__(self, driver_name: str, node_id: str, endpoint: str,
                 backend: OIMBackend, mounter: Optional[Mounter] = None,
                 csi_version: str = "1.0"):
        if csi_version not in ("1.0", "0.3"):
            raise ValueError(f"unsupported CSI version {csi_version!r}")
        self.csi_version = csi_version
        self.identity = IdentityServer(
            driver_name, offline_expansion=backend.supports_expansion())
        self.controller = ControllerServer(backend)
        self.node = NodeServer(node_id, backend, mounter)
        if csi_version == "0.3":
            from .driver03 import (ControllerServer0, IdentityServer0,
                                   NodeServer0)
            self.identity0 = IdentityServer0(driver_name)
            self.controller0 = ControllerServer0(self.controller)
            self.node0 = NodeServer0(node_id, self.node)
        # CSI requests carry `secrets` maps (NodeStage, CreateVolume):
        # payload logging must redact them (the reference used
        # protosanitizer StripSecrets the same way).
        self.server = NonBlockingGRPCServer(
            endpoint=endpoint,
            interceptors=[LogServerInterceptor(strip_secrets_formatter)])

    def _register(self, server) -> None:
        if self.csi_version == "0.3":
            from ..spec.rpc_csi0 import (add_csi0_controller_to_server,
                                         add_csi0_identity_to_server,
                                         add_csi0_node_to_server)
            add_csi0_identity_to_server(self.identity0, server)
            add_csi0_controller_to_server(self.controller0, server)
            add_csi0_node_to_server(self.node0, server)
            return
        add_csi_identity_to_server(self.identity, server)
        add_csi_controller_to_server(self.controller, server)
        add_csi_node_to_server(self.node, server)

    def start(self) -> None:
        self.server.start(self._register)
        from_context().info("CSI driver started", endpoint=self.server.addr(),
                            csi_version=self.csi_version)

    def addr(self) -> str:
        return self.server.addr()

    def stop(self) -> None:
        self.server.stop()

    def run(self) -> None:
        self.server.run(self._register)


def make_params_mapper(emulate: str):
    """Resolve --emulate to a params mapper (oim-driver.go:244-269)."""
    from .remote import malloc_params

    if not emulate:
        return malloc_params
    try:
        return EMULATIONS[emulate]
    except KeyError:
        raise ValueError(
            f"unknown emulation {emulate!r}; supported: "
            f"{', '.join(EMULATIONS)}") from None
