"""Local backend: hipstored on this node, device via kernel NBD
(reference pkg/oim-csi-driver/local.go).

Volume sizing rules kept from the reference (local.go:50-84): round up
to 512 bytes, minimum 1 MiB, maximum 1 TiB."""

from __future__ import annotations

import os
from typing import Dict, Tuple

from .. import hipstore
from ..log import from_context
from .backend import OIMBackend, VolumeExistsError

MIN_VOLUME_SIZE = 1 << 20
MAX_VOLUME_SIZE = 1 << 40


def round_to_blocks(size: int) -> int:
    size = max(size, MIN_VOLUME_SIZE)
    return (size + 511) // 512 * 512


class LocalBackend(OIMBackend):
    def __init__(self, hipstored_socket: str, nbd_prefix: str = "/dev/nbd",
                 aio_dir: str = "/var/lib/oim-aio"):
        self.socket = hipstored_socket
        self.nbd_prefix = nbd_prefix
        # Backing directory for `backing: aio` StorageClass volumes
        # (file-backed, data survives daemon restarts).
        self.aio_dir = aio_dir
        # snapshot id -> (source volume, creation time); hipstored owns
        # the data, this is presentation metadata only
        self._snap_meta = {}

    def _client(self) -> hipstore.Client:
        return hipstore.Client(self.socket)

    def create_volume(self, name: str, size: int,
                      parameters: Dict[str, str] = None
                      ) -> Tuple[str, Dict[str, str]]:
        if size > MAX_VOLUME_SIZE:
            raise ValueError(f"volume too large: {size}")
        size = round_to_blocks(size)
        backing = (parameters or {}).get("backing", "malloc")
        if backing not in ("malloc", "aio"):
            raise ValueError(f"unknown backing {backing!r}")
        with self._client() as client:
            try:
                bdevs = hipstore.get_bdevs(client, name)
            except hipstore.RpcError as err:
                if not err.is_not_found():
                    raise
                bdevs = []
            if bdevs:
                if bdevs[0].size_bytes != size:
                    raise VolumeExistsError(
                        f"volume {name} exists with different size")
                return name, {}
            if backing == "aio":
                # File-backed (persists across daemon restarts); the
                # file IS the volume, sized here, removed on delete.
                os.makedirs(self.aio_dir, exist_ok=True)
                path = os.path.join(self.aio_dir, f"{name}.img")
                with open(path, "ab") as f:
                    f.truncate(size)
                client.invoke("construct_aio_bdev",
                              {"name": name, "filename": path,
                               "block_size": 512})
                return name, {"backing": "aio"}
            hipstore.construct_malloc_bdev(
                client, num_blocks=size // 512, block_size=512, name=name)
        return name, {}

    def delete_volume(self, volume_id: str) -> None:
        with self._client() as client:
            try:
                bdevs = hipstore.get_bdevs(client, volume_id)
                is_aio = bdevs and bdevs[0].product_name == "AIO disk"
                hipstore.delete_bdev(client, volume_id)
            except hipstore.RpcError as err:
                if not err.is_not_found():
                    raise
                return
        if is_aio:
            # CSI DeleteVolume destroys the data: remove the backing file
            path = os.path.join(self.aio_dir, f"{volume_id}.img")
            try:
                os.unlink(path)
            except FileNotFoundError:
                pass

    def check_volume_exists(self, volume_id: str) -> bool:
        with self._client() as client:
            try:
                return bool(hipstore.get_bdevs(client, volume_id))
            except hipstore.RpcError as err:
                if err.is_not_found():
                    return False
                raise

    def _find_exported(self, client, volume_id: str) -> str:
        for disk in client.invoke("ublk_get_disks") or []:
            if disk["bdev_name"] == volume_id:
                return disk["device"]
        for disk in hipstore.get_nbd_disks(client):
            if disk.bdev_name == volume_id:
                return disk.nbd_device
        return ""

    def create_device(self, volume_id: str, volume_context) -> str:
        with self._client() as client:
            # Reuse an existing export (local.go:128,208-219).
            device = self._find_exported(client, volume_id)
            if device:
                return device
            # Preferred path: ublk (the GPU pool's kernels ship
            # ublk_drv but no nbd module) — the daemon allocates the
            # device id and mknods /dev/ublkbN itself.
            try:
                disk = client.invoke("ublk_start_disk",
                                     {"bdev_name": volume_id})
                return disk["device"]
            except hipstore.RpcError as err:
                from_context().warn("ublk attach unavailable, trying NBD",
                                    error=str(err))
            in_use = {d.nbd_device for d in hipstore.get_nbd_disks(client)}
            # Probe /dev/nbd0.. for a free device (local.go:139-176; the
            # reference notes the size==0 probe is racy — daemon-side
            # bookkeeping covers our own exports, the probe covers
            # foreign users of the device).
            for i in range(16):
                candidate = f"{self.nbd_prefix}{i}"
                if candidate in in_use or not os.path.exists(candidate):
                    continue
                try:
                    hipstore.start_nbd_disk(client, volume_id, candidate)
                except hipstore.RpcError as err:
                    from_context().warn("NBD attach failed, trying next",
                                        device=candidate, error=str(err))
                    continue
                return candidate
            raise RuntimeError("no free NBD/ublk device found")

    def delete_device(self, volume_id: str) -> None:
        with self._client() as client:
            for disk in client.invoke("ublk_get_disks") or []:
                if disk["bdev_name"] == volume_id:
                    client.invoke("ublk_stop_disk",
                                  {"dev_id": disk["dev_id"]})
                    return
            device = self._find_exported(client, volume_id)
            if device:
                hipstore.stop_nbd_disk(client, device)

    def get_capacity(self):
        """Free HBM on the daemon's GPU (host MemAvailable in CPU
        mode — malloc bdevs consume host RAM there)."""
        with self._client() as client:
            info = client.invoke("get_hbm_info")
        free = int(info.get("free_bytes", 0))
        return free if free > 0 else None

    def supports_expansion(self) -> bool:
        return True

    def expand_volume(self, volume_id, size):
        size = round_to_blocks(size)
        with self._client() as client:
            try:
                current = hipstore.get_bdevs(client, volume_id)[0].size_bytes
            except hipstore.RpcError as err:
                if err.is_not_found():
                    raise LookupError(
                        f"volume {volume_id} not found") from None
                raise RuntimeError(str(err)) from None
            if size <= current:
                return current  # CSI expansion never shrinks
            try:
                client.invoke("resize_malloc_bdev",
                              {"name": volume_id, "size": size})
            except hipstore.RpcError as err:
                raise RuntimeError(str(err)) from None
        return size

    def list_volumes(self):
        out = []
        with self._client() as client:
            for bdev in hipstore.get_bdevs(client):
                if bdev.name.startswith(self.SNAP_PREFIX):
                    continue  # snapshots are listed via ListSnapshots
                out.append((bdev.name, bdev.size_bytes))
        return out

    # --- snapshots (bdev_clone-backed) --------------------------------------

    SNAP_PREFIX = "csi-snap-"

    def supports_snapshots(self) -> bool:
        return True

    def create_snapshot(self, name, source_volume_id):
        import time as _time

        snap_id = self.SNAP_PREFIX + name
        with self._client() as client:
            try:
                existing = hipstore.get_bdevs(client, snap_id)
            except hipstore.RpcError as err:
                if not err.is_not_found():
                    raise
                existing = []
            if existing:
                meta = self._snap_meta.get(snap_id)
                if meta and meta[0] != source_volume_id:
                    raise VolumeExistsError(
                        f"snapshot {name} exists for volume {meta[0]}")
                return (snap_id, existing[0].size_bytes,
                        meta[1] if meta else 0)
            try:
                hipstore.get_bdevs(client, source_volume_id)
            except hipstore.RpcError as err:
                if err.is_not_found():
                    raise LookupError(
                        f"volume {source_volume_id} not found") from None
                raise
            try:
                client.invoke("bdev_clone",
                              {"src": source_volume_id, "name": snap_id})
            except hipstore.RpcError as err:
                raise RuntimeError(str(err)) from None
            bdev = hipstore.get_bdevs(client, snap_id)[0]
        ctime = int(_time.time())
        self._snap_meta[snap_id] = (source_volume_id, ctime)
        return snap_id, bdev.size_bytes, ctime

    def delete_snapshot(self, snapshot_id) -> None:
        with self._client() as client:
            try:
                hipstore.delete_bdev(client, snapshot_id)
            except hipstore.RpcError as err:
                if not err.is_not_found():
                    raise RuntimeError(str(err)) from None
        self._snap_meta.pop(snapshot_id, None)

    def list_snapshots(self):
        out = []
        with self._client() as client:
            for bdev in hipstore.get_bdevs(client):
                if not bdev.name.startswith(self.SNAP_PREFIX):
                    continue
                source, ctime = self._snap_meta.get(bdev.name, ("", 0))
                out.append((bdev.name, source, bdev.size_bytes, ctime))
        return out

    def clone_volume(self, source_volume_id, volume_name):
        return self.restore_snapshot(source_volume_id, volume_name)

    def restore_snapshot(self, snapshot_id, volume_name):
        with self._client() as client:
            try:
                existing = hipstore.get_bdevs(client, volume_name)
            except hipstore.RpcError as err:
                if not err.is_not_found():
                    raise
                existing = []
            if existing:  # idempotent re-create from the same snapshot
                return volume_name, existing[0].size_bytes
            try:
                hipstore.get_bdevs(client, snapshot_id)
            except hipstore.RpcError as err:
                if err.is_not_found():
                    raise LookupError(
                        f"snapshot {snapshot_id} not found") from None
                raise
            try:
                client.invoke("bdev_clone",
                              {"src": snapshot_id, "name": volume_name})
            except hipstore.RpcError as err:
                raise RuntimeError(str(err)) from None
            bdev = hipstore.get_bdevs(client, volume_name)[0]
        return volume_name, bdev.size_bytes
