"""Remote backend: drive an OIM controller through the registry proxy
(reference pkg/oim-csi-driver/remote.go).

create_device flow (remote.go:116-215): read the card's PCI BDF from
the registry (``<id>/pci``), MapVolume through the proxy, merge the
reply's (possibly partial) PCI address with the registry default, wait
for the matching block device under /sys/dev/block, then mknod a
private /dev node (container-safe). The sysfs root and /dev directory
are injectable so unit tests run against a fake symlink farm
(reference nodeserver_test.go:25-180).
"""

from __future__ import annotations

import os
import re
import stat
import time
from typing import Callable, Dict, Optional, Tuple

import grpc

from .. import spec
from ..common.pci import (
    PCIAddress,
    complete_pci_address,
    parse_bdf_string,
)
from ..common.server import grpc_target
from ..common.tlsutil import (
    TLSConfig,
    channel_options_for_peer,
    load_tls_channel_credentials,
)
from ..log import from_context
from .backend import OIMBackend
from .local import MAX_VOLUME_SIZE, round_to_blocks

# A mapper turns CSI stage info into the oneof params of a
# MapVolumeRequest (the --emulate hook, reference oim-driver.go:244-269).
ParamsMapper = Callable[[str, Dict[str, str], Dict[str, str], str], None]


def malloc_params(request: spec.MapVolumeRequest, volume_id: str,
                  volume_context: Dict[str, str],
                  secrets: Dict[str, str], staging_path: str) -> None:
    request.malloc.CopyFrom(spec.MallocParams())


class RemoteBackend(OIMBackend):
    def __init__(
        self,
        registry_address: str,
        controller_id: str,
        tls: Optional[TLSConfig] = None,
        params_mapper=malloc_params,
        sysfs_block_dir: str = "/sys/dev/block",
        dev_dir: str = "/var/run/oim-csi-driver/dev",
        device_timeout: float = 30.0,
    ):
        self.registry_address = registry_address
        self.controller_id = controller_id
        self.tls = tls
        self.params_mapper = params_mapper
        self.sysfs_block_dir = sysfs_block_dir
        self.dev_dir = dev_dir
        self.device_timeout = device_timeout
        # Provenance metadata for snapshots created through this
        # driver: snap id -> (source volume, ctime). Sizes come from
        # the daemon via the ListMallocBDevs extension.
        self._snap_meta: Dict[str, tuple] = {}

    # --- registry plumbing --------------------------------------------------

    def _dial_registry(self) -> grpc.Channel:
        """Per-operation dial (remote.go:101-114): no held connections.
        Outgoing payloads are logged with secrets stripped, like the
        reference's ChooseDialOpts interceptor chain (grpc.go:56)."""
        from ..common.tracing import LogClientInterceptor

        target = grpc_target(self.registry_address)
        if self.tls is not None:
            creds = load_tls_channel_credentials(self.tls)
            options = channel_options_for_peer("component.registry")
            channel = grpc.secure_channel(target, creds, options=options)
        else:
            channel = grpc.insecure_channel(target)
        return grpc.intercept_channel(channel, LogClientInterceptor())

    def _metadata(self):
        return ((spec.CONTROLLER_ID_KEY, self.controller_id),)

    # --- volume lifecycle ---------------------------------------------------

    def create_volume(self, name: str, size: int,
                      parameters: Dict[str, str] = None
                      ) -> Tuple[str, Dict[str, str]]:
        if (parameters or {}).get("backing", "malloc") != "malloc":
            raise ValueError(
                "remote mode provisions malloc bdevs only (oim.v0 "
                "ProvisionMallocBDev)")
        if size > MAX_VOLUME_SIZE:
            raise ValueError(f"volume too large: {size}")
        size = round_to_blocks(size)
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            stub.ProvisionMallocBDev(
                spec.ProvisionMallocBDevRequest(bdev_name=name, size=size),
                metadata=self._metadata(), timeout=30)
        return name, {}

    def delete_volume(self, volume_id: str) -> None:
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            stub.ProvisionMallocBDev(
                spec.ProvisionMallocBDevRequest(bdev_name=volume_id, size=0),
                metadata=self._metadata(), timeout=30)

    def check_volume_exists(self, volume_id: str) -> bool:
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            try:
                stub.CheckMallocBDev(
                    spec.CheckMallocBDevRequest(bdev_name=volume_id),
                    metadata=self._metadata(), timeout=30)
                return True
            except grpc.RpcError as err:
                if err.code() == grpc.StatusCode.NOT_FOUND:
                    return False
                raise

    def supports_expansion(self) -> bool:
        return True

    def expand_volume(self, volume_id, size):
        size = round_to_blocks(size)
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            try:
                stub.ResizeMallocBDev(
                    spec.ResizeMallocBDevRequest(bdev_name=volume_id,
                                                 size=size),
                    metadata=self._metadata(), timeout=60)
            except grpc.RpcError as err:
                if err.code() == grpc.StatusCode.NOT_FOUND:
                    raise LookupError(
                        f"volume {volume_id} not found") from None
                raise
        return size

    # --- snapshots (CloneMallocBDev extension, docs/spec.md) ----------------

    SNAP_PREFIX = "csi-snap-"

    def supports_snapshots(self) -> bool:
        return True

    def _bdev_size(self, name) -> int:
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            reply = stub.ListMallocBDevs(
                spec.ListMallocBDevsRequest(prefix=name),
                metadata=self._metadata(), timeout=30)
        for info in reply.bdevs:
            if info.name == name:
                return info.size
        return 0

    def create_snapshot(self, name, source_volume_id):
        import time as _time

        from .backend import VolumeExistsError

        snap_id = self.SNAP_PREFIX + name
        existing = self._snap_meta.get(snap_id)
        if existing and existing[0] != source_volume_id:
            raise VolumeExistsError(
                f"snapshot {name} exists for volume {existing[0]}")
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            try:
                stub.CloneMallocBDev(
                    spec.CloneMallocBDevRequest(source=source_volume_id,
                                                dest=snap_id),
                    metadata=self._metadata(), timeout=60)
            except grpc.RpcError as err:
                if err.code() == grpc.StatusCode.NOT_FOUND:
                    raise LookupError(
                        f"volume {source_volume_id} not found") from None
                raise
        size = self._bdev_size(snap_id)
        if existing:
            return snap_id, size, existing[1]
        ctime = int(_time.time())
        self._snap_meta[snap_id] = (source_volume_id, ctime)
        return snap_id, size, ctime

    def delete_snapshot(self, snapshot_id) -> None:
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            stub.ProvisionMallocBDev(
                spec.ProvisionMallocBDevRequest(bdev_name=snapshot_id,
                                                size=0),
                metadata=self._metadata(), timeout=30)
        self._snap_meta.pop(snapshot_id, None)

    def list_snapshots(self):
        """Daemon truth via ListMallocBDevs; source/ctime come from
        this driver's metadata when it created the snapshot (the
        oim.v0 API carries no provenance), else blank/0."""
        out = []
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            reply = stub.ListMallocBDevs(
                spec.ListMallocBDevsRequest(prefix=self.SNAP_PREFIX),
                metadata=self._metadata(), timeout=30)
        for info in reply.bdevs:
            source, ctime = "", 0
            meta = self._snap_meta.get(info.name)
            if meta:
                source, ctime = meta
            out.append((info.name, source, info.size, ctime))
        return out

    def list_volumes(self):
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            reply = stub.ListMallocBDevs(
                spec.ListMallocBDevsRequest(),
                metadata=self._metadata(), timeout=30)
        return [(info.name, info.size) for info in reply.bdevs
                if not info.name.startswith(self.SNAP_PREFIX)]

    def clone_volume(self, source_volume_id, volume_name):
        return self.restore_snapshot(source_volume_id, volume_name)

    def restore_snapshot(self, snapshot_id, volume_name):
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            try:
                stub.CloneMallocBDev(
                    spec.CloneMallocBDevRequest(source=snapshot_id,
                                                dest=volume_name),
                    metadata=self._metadata(), timeout=60)
            except grpc.RpcError as err:
                if err.code() == grpc.StatusCode.NOT_FOUND:
                    raise LookupError(
                        f"snapshot {snapshot_id} not found") from None
                raise
        return volume_name, self._bdev_size(volume_name)

    # --- device lifecycle ---------------------------------------------------

    def _registry_pci(self, stub: spec.RegistryStub) -> PCIAddress:
        """Read <id>/pci (remote.go:128-145); unset when absent."""
        reply = stub.GetValues(
            spec.GetValuesRequest(path=f"{self.controller_id}/pci"), timeout=30)
        for value in reply.values:
            if value.path == f"{self.controller_id}/pci":
                try:
                    return parse_bdf_string(value.value)
                except ValueError:
                    from_context().warn("bad pci registry entry",
                                        value=value.value)
        return PCIAddress()

    def create_device(self, volume_id: str, stage_info: Dict) -> str:
        volume_context = stage_info.get("volume_context", {})
        secrets = stage_info.get("secrets", {})
        staging_path = stage_info.get("staging_path", "")
        with self._dial_registry() as channel:
            registry_stub = spec.RegistryStub(channel)
            default_pci = self._registry_pci(registry_stub)
            request = spec.MapVolumeRequest(volume_id=volume_id)
            self.params_mapper(request, volume_id, volume_context, secrets,
                               staging_path)
            controller_stub = spec.ControllerStub(channel)
            reply = controller_stub.MapVolume(
                request, metadata=self._metadata(), timeout=60)
        reply_pci = PCIAddress(
            domain=reply.pci_address.domain, bus=reply.pci_address.bus,
            device=reply.pci_address.device,
            function=reply.pci_address.function)
        pci = complete_pci_address(reply_pci, default_pci)
        if not pci.is_complete():
            raise RuntimeError(
                f"incomplete PCI address {pci} for {volume_id}: "
                f"set {self.controller_id}/pci in the registry")
        return self.wait_for_device(
            pci, reply.scsi_disk.target, reply.scsi_disk.lun)

    def delete_device(self, volume_id: str) -> None:
        with self._dial_registry() as channel:
            stub = spec.ControllerStub(channel)
            stub.UnmapVolume(
                spec.UnmapVolumeRequest(volume_id=volume_id),
                metadata=self._metadata(), timeout=60)

    # --- sysfs discovery (remote.go:249-373) --------------------------------

    def wait_for_device(self, pci: PCIAddress, target: int, lun: int) -> str:
        """Poll /sys/dev/block for a device on the given PCI function and
        SCSI target:lun; mknod and return a private /dev node.

        The reference watches with fsnotify plus a 5 s re-poll because
        "inotify seems to miss events" (remote.go:281-285); a plain
        250 ms poll keeps the same worst-case latency profile without
        the inotify dependency.
        """
        deadline = time.monotonic() + self.device_timeout
        while True:
            found = self.find_device(pci, target, lun)
            if found:
                major_minor, name = found
                return self._mknod(major_minor, name)
            if time.monotonic() > deadline:
                raise TimeoutError(
                    f"timed out waiting for block device on "
                    f"{pci.domain:04x}:{pci.bus:02x}:{pci.device:02x}."
                    f"{pci.function} target {target} lun {lun}")
            time.sleep(0.25)

    def find_device(self, pci: PCIAddress, target: int,
                    lun: int) -> Optional[Tuple[str, str]]:
        """Scan symlinks (findDev, remote.go:292-373). Returns
        ((major:minor), devname) or None."""
        bdf = (f"{pci.domain:04x}:{pci.bus:02x}:{pci.device:02x}."
               f"{pci.function}")
        scsi_re = re.compile(rf"^\d+:\d+:{target}:{lun}$")
        try:
            entries = os.listdir(self.sysfs_block_dir)
        except FileNotFoundError:
            return None
        for entry in entries:
            link = os.path.join(self.sysfs_block_dir, entry)
            try:
                dest = os.readlink(link)
            except OSError:
                continue
            parts = dest.split("/")
            if f"{bdf}" not in parts:
                continue
            if not any(scsi_re.match(p) for p in parts):
                continue
            return entry, parts[-1]
        return None

    def _mknod(self, major_minor: str, name: str) -> str:
        """Create the block node under our private dev dir
        (remote.go:197-215, makedev :237-243)."""
        major, minor = (int(x) for x in major_minor.split(":"))
        os.makedirs(self.dev_dir, exist_ok=True)
        path = os.path.join(self.dev_dir, name)
        if os.path.exists(path):
            st = os.stat(path)
            if stat.S_ISBLK(st.st_mode) and st.st_rdev == os.makedev(major, minor):
                return path
            os.unlink(path)
        os.mknod(path, stat.S_IFBLK | 0o600, os.makedev(major, minor))
        return path
